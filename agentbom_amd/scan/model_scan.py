"""ML model-artifact safety scanning (pickle payload detection).

Reference: src/agent_bom/model_pickle_scan.py + model_files.py — static
disassembly of pickle streams inside .pkl/.pt/.pth/.ckpt/.joblib artifacts
looking for code-execution opcodes (GLOBAL/STACK_GLOBAL + REDUCE reaching
os/subprocess/builtins.eval etc.); safetensors/GGUF are recognized as
safe-by-construction formats.  Never unpickles — pickletools only.
"""

from __future__ import annotations

import pickletools
import zipfile
from dataclasses import dataclass, field
from pathlib import Path
from typing import Optional

from agentbom_amd.models.finding import Asset, Finding, FindingSource, FindingType, stable_id

# module prefixes whose import inside a pickle means code execution
DANGEROUS_GLOBALS = (
    "os", "posix", "nt", "subprocess", "sys", "builtins.eval", "builtins.exec",
    "builtins.compile", "builtins.__import__", "builtins.getattr", "operator.attrgetter",
    "importlib", "runpy", "socket", "shutil.rmtree", "pty", "commands", "pip",
    "webbrowser", "torch.load",
)
# commonly legitimate in model checkpoints
BENIGN_PREFIXES = (
    "torch", "numpy", "collections", "argparse", "__builtin__.set", "builtins.set",
    "_codecs", "copyreg", "sklearn", "joblib", "pandas", "scipy", "xgboost", "lightgbm",
)

SAFE_FORMATS = {".safetensors", ".gguf", ".onnx", ".tflite"}
PICKLE_SUFFIXES = {".pkl", ".pickle", ".pt", ".pth", ".ckpt", ".bin", ".joblib", ".model"}


@dataclass
class ModelScanResult:
    path: str
    format: str  # pickle | zip-pickle | safetensors | gguf | onnx | unknown
    safe_format: bool
    dangerous_imports: list[str] = field(default_factory=list)
    suspicious_imports: list[str] = field(default_factory=list)
    has_reduce: bool = False
    error: Optional[str] = None

    @property
    def verdict(self) -> str:
        if self.safe_format:
            return "safe-format"
        if self.dangerous_imports:
            return "malicious"
        if self.suspicious_imports and self.has_reduce:
            return "suspicious"
        if self.error:
            return "unscannable"
        return "clean"

    def to_dict(self) -> dict:
        return {
            "path": self.path, "format": self.format, "verdict": self.verdict,
            "dangerous_imports": self.dangerous_imports,
            "suspicious_imports": self.suspicious_imports,
            "has_reduce": self.has_reduce, "error": self.error,
        }


def _classify_global(mod: str, name: str) -> Optional[str]:
    full = f"{mod}.{name}"
    for danger in DANGEROUS_GLOBALS:
        if full == danger or full.startswith(danger + ".") or mod == danger:
            return "dangerous"
    for benign in BENIGN_PREFIXES:
        if mod == benign or mod.startswith(benign + "."):
            return None
    return "suspicious"


def scan_pickle_bytes(data: bytes, path: str = "<memory>") -> ModelScanResult:
    result = ModelScanResult(path=path, format="pickle", safe_format=False)
    pending: list[str] = []
    try:
        for opcode, arg, _pos in pickletools.genops(data):
            if opcode.name in ("GLOBAL", "INST"):
                parts = str(arg).split()
                mod, name = (parts + [""])[:2]
                cls = _classify_global(mod, name)
                full = f"{mod}.{name}"
                if cls == "dangerous" and full not in result.dangerous_imports:
                    result.dangerous_imports.append(full)
                elif cls == "suspicious" and full not in result.suspicious_imports:
                    result.suspicious_imports.append(full)
            elif opcode.name == "STACK_GLOBAL":
                # args are the two preceding strings
                if len(pending) >= 2:
                    mod, name = pending[-2], pending[-1]
                    cls = _classify_global(mod, name)
                    full = f"{mod}.{name}"
                    if cls == "dangerous" and full not in result.dangerous_imports:
                        result.dangerous_imports.append(full)
                    elif cls == "suspicious" and full not in result.suspicious_imports:
                        result.suspicious_imports.append(full)
            elif opcode.name in ("REDUCE", "BUILD", "NEWOBJ"):
                result.has_reduce = True
            if opcode.name in ("SHORT_BINUNICODE", "BINUNICODE", "UNICODE", "MEMOIZE"):
                if opcode.name != "MEMOIZE" and isinstance(arg, str):
                    pending.append(arg)
                    pending = pending[-8:]
    except Exception as exc:  # noqa: BLE001 — malformed stream boundary
        result.error = f"pickle parse stopped: {exc}"
    return result


def scan_model_file(path: str | Path) -> ModelScanResult:
    path = Path(path)
    suffix = path.suffix.lower()
    if suffix in SAFE_FORMATS:
        return ModelScanResult(path=str(path), format=suffix.lstrip("."), safe_format=True)
    data = path.read_bytes()
    # torch .pt/.pth are usually zip archives containing data.pkl
    if data[:4] == b"PK\x03\x04":
        try:
            with zipfile.ZipFile(path) as zf:
                pkls = [n for n in zf.namelist() if n.endswith((".pkl", "data.pkl"))]
                merged = ModelScanResult(path=str(path), format="zip-pickle", safe_format=False)
                for n in pkls:
                    sub = scan_pickle_bytes(zf.read(n), path=f"{path}!{n}")
                    merged.dangerous_imports.extend(
                        x for x in sub.dangerous_imports if x not in merged.dangerous_imports)
                    merged.suspicious_imports.extend(
                        x for x in sub.suspicious_imports if x not in merged.suspicious_imports)
                    merged.has_reduce |= sub.has_reduce
                    merged.error = merged.error or sub.error
                return merged
        except zipfile.BadZipFile:
            pass
    if suffix in PICKLE_SUFFIXES or data[:1] == b"\x80":
        return scan_pickle_bytes(data, path=str(path))
    return ModelScanResult(path=str(path), format="unknown", safe_format=False,
                           error="unrecognized model format")


def scan_model_tree(root: str | Path) -> list[ModelScanResult]:
    root = Path(root)
    candidates = [root] if root.is_file() else [
        p for p in root.rglob("*")
        if p.is_file() and p.suffix.lower() in (PICKLE_SUFFIXES | SAFE_FORMATS)
    ]
    return [scan_model_file(p) for p in sorted(candidates)]


def model_result_to_finding(result: ModelScanResult) -> Optional[Finding]:
    if result.verdict in ("clean", "safe-format"):
        return None
    severity = {"malicious": "critical", "suspicious": "medium",
                "unscannable": "low"}[result.verdict]
    ftype = (FindingType.MALICIOUS_MODEL if result.verdict == "malicious"
             else FindingType.MODEL_INTEGRITY)
    return Finding(
        finding_type=ftype,
        source=FindingSource.MODEL_SCAN,
        asset=Asset(name=Path(result.path).name, asset_type="model_file",
                    location=result.path),
        severity=severity,
        title=f"Model artifact {result.verdict}: {Path(result.path).name}",
        description=(
            f"pickle stream imports {', '.join(result.dangerous_imports[:5])}"
            if result.dangerous_imports else
            f"model artifact could not be verified ({result.error or 'suspicious imports'})"
        ),
        evidence=result.to_dict(),
        is_actionable=result.verdict == "malicious",
        id=stable_id("model-scan", result.path, result.verdict),
    )
