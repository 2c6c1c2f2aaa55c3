"""Policy-as-code engine: declarative rules + safe expression conditions.

Reference: src/agent_bom/policy.py — JSON/YAML rule files, two styles:
declarative ANDed fields and a whitelisted expression language
(``epss_score > 0.7 and has_credentials and severity >= HIGH``) evaluated
per BlastRadius; actions fail/warn.  No arbitrary code execution — only
registered fields and comparison/boolean operators.
"""

from __future__ import annotations

import json
import operator
import re
from pathlib import Path
from typing import Any, Callable

SEVERITY_POLICY_ORDER = {"NONE": 0, "UNKNOWN": 0, "LOW": 1, "MEDIUM": 2, "HIGH": 3, "CRITICAL": 4}
RISK_LEVEL_ORDER = {"high": 3, "medium": 2, "low": 1}

_CMP = {">": operator.gt, "<": operator.lt, ">=": operator.ge,
        "<=": operator.le, "==": operator.eq, "!=": operator.ne}

FIELD_GETTERS: dict[str, Callable] = {
    "severity": lambda b: SEVERITY_POLICY_ORDER.get(b.vulnerability.severity.value.upper(), 0),
    "cvss_score": lambda b: b.vulnerability.cvss_score or 0.0,
    "epss_score": lambda b: b.vulnerability.epss_score or 0.0,
    "is_kev": lambda b: bool(b.vulnerability.is_kev),
    "has_fix": lambda b: bool(b.vulnerability.fixed_version),
    "vuln_id": lambda b: b.vulnerability.id,
    "package_name": lambda b: b.package.name,
    "ecosystem": lambda b: b.package.ecosystem,
    "scorecard_score": lambda b: b.package.scorecard_score if b.package.scorecard_score is not None else 0.0,
    "is_malicious": lambda b: bool(b.package.is_malicious),
    "risk_score": lambda b: b.risk_score,
    "agent_count": lambda b: len(b.affected_agents),
    "server_count": lambda b: len(b.affected_servers),
    "tool_count": lambda b: len(b.exposed_tools),
    "credential_count": lambda b: len(b.exposed_credentials),
    "has_credentials": lambda b: bool(b.exposed_credentials),
    "ai_risk": lambda b: bool(b.ai_risk_context),
    "graph_reachable": lambda b: b.graph_reachable,
    "symbol_reachability": lambda b: b.symbol_reachability,
    "hop_depth": lambda b: b.hop_depth,
    "impact_category": lambda b: b.impact_category,
    "reachability": lambda b: b.reachability,
    "owasp_tags": lambda b: b.owasp_tags,
    "owasp_mcp_tags": lambda b: b.owasp_mcp_tags,
    "owasp_agentic_tags": lambda b: b.owasp_agentic_tags,
    "nist_csf_tags": lambda b: b.nist_csf_tags,
    "nist_ai_rmf_tags": lambda b: b.nist_ai_rmf_tags,
    "nist_800_53_tags": lambda b: b.nist_800_53_tags,
    "atlas_tags": lambda b: b.atlas_tags,
    "attack_tags": lambda b: b.attack_tags,
    "iso_27001_tags": lambda b: b.iso_27001_tags,
    "soc2_tags": lambda b: b.soc2_tags,
    "cis_tags": lambda b: b.cis_tags,
    "cmmc_tags": lambda b: b.cmmc_tags,
    "eu_ai_act_tags": lambda b: b.eu_ai_act_tags,
    "fedramp_tags": lambda b: b.fedramp_tags,
    "pci_dss_tags": lambda b: b.pci_dss_tags,
}

_TOKEN_RE = re.compile(
    r"""\s*(?:
        (?P<num>\d+\.\d+|\d+)
      | "(?P<dstr>(?:[^"\\]|\\.)*)"
      | '(?P<sstr>(?:[^'\\]|\\.)*)'
      | (?P<op>>=|<=|!=|==|>|<)
      | (?P<paren>[()])
      | (?P<kw>and\b|or\b|not\b|in\b|true\b|false\b)
      | (?P<ident>[A-Za-z_][A-Za-z0-9_]*)
    )""",
    re.VERBOSE,
)


def _tokenize(expr: str) -> list[tuple[str, Any]]:
    tokens: list[tuple[str, Any]] = []
    pos = 0
    while pos < len(expr):
        m = _TOKEN_RE.match(expr, pos)
        if m is None or m.end() == pos:
            rest = expr[pos:].strip()
            if not rest:
                break
            raise ValueError(f"invalid token at {rest[:20]!r}")
        if m.group("num") is not None:
            tokens.append(("NUM", float(m.group("num"))))
        elif m.group("dstr") is not None:
            tokens.append(("STR", m.group("dstr")))
        elif m.group("sstr") is not None:
            tokens.append(("STR", m.group("sstr")))
        elif m.group("op"):
            tokens.append(("OP", m.group("op")))
        elif m.group("paren"):
            tokens.append(("PAREN", m.group("paren")))
        elif m.group("kw"):
            tokens.append(("KW", m.group("kw")))
        else:
            tokens.append(("IDENT", m.group("ident")))
        pos = m.end()
    return tokens


def _validate_fields(tokens: list[tuple[str, Any]]) -> None:
    for ttype, value in tokens:
        if ttype != "IDENT":
            continue
        if value in FIELD_GETTERS or value.upper() in SEVERITY_POLICY_ORDER:
            continue
        raise ValueError(f"unknown field {value!r} in policy condition")


class _Parser:
    """Recursive descent: or_expr -> and_expr -> not_expr -> cmp -> atom."""

    def __init__(self, tokens: list[tuple[str, Any]], br: Any):
        self.tokens = tokens
        self.pos = 0
        self.br = br

    def _peek(self):
        return self.tokens[self.pos] if self.pos < len(self.tokens) else (None, None)

    def _eat(self):
        tok = self._peek()
        self.pos += 1
        return tok

    def parse(self) -> bool:
        result = self._or()
        if self.pos != len(self.tokens):
            raise ValueError("trailing tokens in condition")
        return bool(result)

    def _or(self):
        left = self._and()
        while self._peek() == ("KW", "or"):
            self._eat()
            right = self._and()
            left = bool(left) or bool(right)
        return left

    def _and(self):
        left = self._not()
        while self._peek() == ("KW", "and"):
            self._eat()
            right = self._not()
            left = bool(left) and bool(right)
        return left

    def _not(self):
        if self._peek() == ("KW", "not"):
            self._eat()
            return not self._not()
        return self._cmp()

    def _cmp(self):
        left = self._atom()
        ttype, value = self._peek()
        if ttype == "OP":
            self._eat()
            right = self._atom()
            try:
                return _CMP[value](left, right)
            except TypeError:
                return False
        if (ttype, value) == ("KW", "in"):
            self._eat()
            right = self._atom()
            try:
                return left in right
            except TypeError:
                return False
        return left

    def _atom(self):
        ttype, value = self._eat()
        if ttype == "NUM":
            return value
        if ttype == "STR":
            return value
        if (ttype, value) == ("PAREN", "("):
            inner = self._or()
            if self._eat() != ("PAREN", ")"):
                raise ValueError("unbalanced parentheses")
            return inner
        if ttype == "KW" and value in ("true", "false"):
            return value == "true"
        if ttype == "IDENT":
            if value in FIELD_GETTERS:
                try:
                    return FIELD_GETTERS[value](self.br)
                except AttributeError as exc:
                    # partial finding objects (dict imports, external scans)
                    # must degrade to a policy error, never an internal crash
                    raise ValueError(
                        f"field {value!r} unavailable on this finding: {exc}")
            if value.upper() in SEVERITY_POLICY_ORDER:
                return SEVERITY_POLICY_ORDER[value.upper()]
            raise ValueError(f"unknown field {value!r}")
        raise ValueError(f"unexpected token {value!r}")


def evaluate_expression(expr: str, br: Any) -> bool:
    """Evaluate a condition expression against one BlastRadius finding."""
    tokens = _tokenize(expr)
    _validate_fields(tokens)
    return _Parser(tokens, br).parse()


# ── policy file ────────────────────────────────────────────────────────────


def load_policy(path: str) -> dict:
    p = Path(path)
    if not p.exists():
        raise FileNotFoundError(f"Policy file not found: {path}")
    text = p.read_text()
    if p.suffix in (".yaml", ".yml"):
        import yaml

        data = yaml.safe_load(text)
    else:
        try:
            data = json.loads(text)
        except json.JSONDecodeError as e:
            raise ValueError(f"Invalid JSON in policy file: {e}") from e
    validate_policy(data)
    return data


def validate_policy(policy: dict) -> None:
    """Fail-fast validation of structure + condition syntax at load time."""
    if not isinstance(policy, dict):
        raise ValueError("Policy must be a JSON object")
    rules = policy.get("rules")
    if not isinstance(rules, list):
        raise ValueError("Policy must have a 'rules' array")
    for i, rule in enumerate(rules):
        rid = rule.get("id") or f"index-{i}"
        if "id" not in rule:
            raise ValueError(f"Rule at index {i} missing 'id'")
        if rule.get("action") not in ("fail", "warn", "jira", None):
            raise ValueError(f"Rule '{rid}' action must be 'fail', 'warn', or 'jira'")
        condition = rule.get("condition")
        if condition and isinstance(condition, str):
            try:
                _validate_fields(_tokenize(condition))
            except ValueError as e:
                raise ValueError(f"Rule '{rid}' has invalid condition syntax: {e}") from e
        if "severity_gte" in rule and rule["severity_gte"].upper() not in SEVERITY_POLICY_ORDER:
            raise ValueError(f"Rule '{rid}' severity_gte {rule['severity_gte']!r} invalid")


def _rule_matches(rule: dict, br: Any) -> bool:
    """Expression mode + declarative fields; all present clauses must match."""
    if "condition" in rule and isinstance(rule["condition"], str):
        if not evaluate_expression(rule["condition"], br):
            return False
    if "severity_gte" in rule:
        need = SEVERITY_POLICY_ORDER[rule["severity_gte"].upper()]
        if SEVERITY_POLICY_ORDER.get(br.vulnerability.severity.value.upper(), 0) < need:
            return False
    if rule.get("is_kev") is not None and "is_kev" in rule:
        if bool(br.vulnerability.is_kev) != rule["is_kev"]:
            return False
    if "max_epss_score" in rule:
        if (br.vulnerability.epss_score or 0.0) <= rule["max_epss_score"]:
            return False
    if "min_scorecard_score" in rule:
        sc = br.package.scorecard_score
        if sc is None or sc >= rule["min_scorecard_score"]:
            return False
    if "min_agents" in rule and len(br.affected_agents) < rule["min_agents"]:
        return False
    if "min_tools" in rule and len(br.exposed_tools) < rule["min_tools"]:
        return False
    if rule.get("has_credentials") is not None and "has_credentials" in rule:
        if bool(br.exposed_credentials) != rule["has_credentials"]:
            return False
    if rule.get("ai_risk") is not None and "ai_risk" in rule:
        if bool(br.ai_risk_context) != rule["ai_risk"]:
            return False
    if "ecosystem" in rule and br.package.ecosystem.lower() != rule["ecosystem"].lower():
        return False
    if "package_name_contains" in rule:
        if rule["package_name_contains"].lower() not in br.package.name.lower():
            return False
    if "owasp_tag" in rule and rule["owasp_tag"] not in br.owasp_tags:
        return False
    if "owasp_mcp_tag" in rule and rule["owasp_mcp_tag"] not in br.owasp_mcp_tags:
        return False
    if rule.get("has_fix") is not None and "has_fix" in rule:
        if bool(br.vulnerability.fixed_version) != rule["has_fix"]:
            return False
    return True


def evaluate_policy(policy: dict, blast_radii: list, dry_run: bool = False) -> dict:
    """Evaluate every rule against every finding.

    Returns {passed, violations[], warnings[], rules_evaluated}."""
    violations: list[dict] = []
    warnings: list[dict] = []
    for rule in policy.get("rules", []):
        action = rule.get("action", "fail")
        for br in blast_radii:
            if not _rule_matches(rule, br):
                continue
            hit = {
                "rule_id": rule["id"],
                "description": rule.get("description", ""),
                "action": action,
                "vulnerability_id": br.vulnerability.id,
                "package": f"{br.package.name}@{br.package.version}",
                "risk_score": br.risk_score,
            }
            if action == "warn":
                warnings.append(hit)
            else:
                violations.append(hit)
    return {
        "policy_name": policy.get("name", "unnamed"),
        "passed": not violations or dry_run,
        "violations": violations,
        "warnings": warnings,
        "rules_evaluated": len(policy.get("rules", [])),
    }
