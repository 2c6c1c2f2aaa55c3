"""Neptune/Gremlin store: opt-in flag, query shapes, unsupported ops."""

from __future__ import annotations

import json

import httpx
import pytest

from agentbom_amd.graph.container import UnifiedEdge, UnifiedGraph, UnifiedNode
from agentbom_amd.graph.neptune_store import NeptuneGraphStore, NeptuneUnsupported
from agentbom_amd.graph.types import EntityType, RelationshipType
from agentbom_amd.utils.http_client import set_offline


@pytest.fixture(autouse=True)
def _online():
    set_offline(False)
    yield
    set_offline(False)


class _FakeGremlin:
    def __init__(self):
        self.queries: list[str] = []
        self.answers: dict[str, list] = {}

    def handler(self, request: httpx.Request) -> httpx.Response:
        g = json.loads(request.content)["gremlin"]
        self.queries.append(g)
        data = []
        for marker, answer in self.answers.items():
            if marker in g:
                data = answer
        return httpx.Response(200, json={"result": {"data": {"@value": data}}})


@pytest.fixture
def fake():
    return _FakeGremlin()


@pytest.fixture
def store(fake):
    client = httpx.Client(transport=httpx.MockTransport(fake.handler))
    return NeptuneGraphStore("https://neptune.example:8182/gremlin",
                             client=client)


def _graph():
    g = UnifiedGraph()
    g.add_node(UnifiedNode(id="a1", entity_type=EntityType.AGENT, label="a1"))
    g.add_node(UnifiedNode(id="s1", entity_type=EntityType.SERVER, label="s1"))
    g.add_edge(UnifiedEdge(source="a1", target="s1",
                           relationship=RelationshipType.USES))
    return g


def test_requires_explicit_endpoint(monkeypatch):
    monkeypatch.delenv("AGENT_BOM_NEPTUNE_ENDPOINT", raising=False)
    with pytest.raises(RuntimeError, match="explicit opt-in"):
        NeptuneGraphStore()


def test_save_emits_vertices_edges_and_marker(store, fake):
    sid = store.save_snapshot(_graph(), scan_id="s-9", tenant_id="acme")
    addv = [q for q in fake.queries if q.startswith("g.addV(")]
    adde = [q for q in fake.queries if ".addE(" in q]
    assert len(addv) == 3  # 2 nodes + snapshot marker
    assert len(adde) == 1 and "'uses'" in adde[0]
    assert any("abom_snapshot" in q and "s-9" in q for q in addv)
    assert all(sid in q for q in adde)


def test_load_snapshot_rebuilds_nodes(store, fake):
    doc = json.dumps({"id": "a1", "entity_type": "agent", "label": "a1"})
    fake.answers["values('doc')"] = [doc]
    g = store.load_snapshot("snap-1")
    assert g is not None and "a1" in g.nodes
    fake.answers["values('doc')"] = []
    assert store.load_snapshot("snap-2") is None


def test_unsupported_ops_raise(store):
    with pytest.raises(NeptuneUnsupported):
        store.diff_snapshots("a", "b")
    with pytest.raises(NeptuneUnsupported):
        store.evidence_manifest()


def test_injection_escaped(store, fake):
    g = UnifiedGraph()
    g.add_node(UnifiedNode(id="x').drop().addV('pwn", entity_type=EntityType.AGENT,
                           label="evil"))
    store.save_snapshot(g)
    assert not any(".drop()" in q.replace("\\'", "") and "pwn" in q
                   and "\\" not in q for q in fake.queries)
    assert any("x\\').drop" in q for q in fake.queries)  # quote escaped
