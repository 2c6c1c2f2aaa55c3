"""Graph-shape contract gate: demo estate must match the committed snapshot.

Any change to builder edges/nodes or risk ranking requires an explicit
rebaseline (scripts/rebaseline_graph_contract.py) — never silent drift.
"""

import json
from pathlib import Path


def test_graph_contract_snapshot():
    import importlib.util

    script = (Path(__file__).resolve().parents[1]
              / "scripts" / "rebaseline_graph_contract.py")
    spec = importlib.util.spec_from_file_location("rebaseline", script)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    live = mod.build_contract()
    committed = json.loads(
        (Path(__file__).resolve().parents[1]
         / "docs" / "graph_contract.json").read_text())
    assert live == committed, (
        "graph shape drifted from docs/graph_contract.json — if intended, "
        "run scripts/rebaseline_graph_contract.py and review the diff")
