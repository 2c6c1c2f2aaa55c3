"""Shared pytest configuration.
import pytest

Markers:
- ``gpu``: requires an MI355X (run via gpurun / driver round-end on a GPU box)
- ``slow``: long-running scale tests
"""

from __future__ import annotations

import os

import pytest


def pytest_configure(config: pytest.Config) -> None:
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU")
    config.addinivalue_line("markers", "slow: long-running scale tests")
    config.addinivalue_line("markers", "network: requires real network egress")


def _has_gpu() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


HAS_GPU = _has_gpu()


def pytest_collection_modifyitems(config: pytest.Config, items: list[pytest.Item]) -> None:
    import os

    skip_gpu = pytest.mark.skip(reason="no GPU in this environment")
    skip_net = pytest.mark.skip(reason="network tests disabled (AGENT_BOM_TEST_NETWORK=1)")
    net_on = os.environ.get("AGENT_BOM_TEST_NETWORK") == "1"
    for item in items:
        if not HAS_GPU and "gpu" in item.keywords:
            item.add_marker(skip_gpu)
        if not net_on and "network" in item.keywords:
            item.add_marker(skip_net)


@pytest.fixture(autouse=True)
def _reset_global_state():
    """Snapshot/restore process-global mutable state between tests."""
    from agentbom_amd.utils import version_utils

    sink = version_utils._scan_warning_sink
    yield
    version_utils._scan_warning_sink = sink


@pytest.fixture
def env(monkeypatch):
    """Helper to set AGENT_BOM_* env vars for one test."""

    def _set(**kwargs):
        for k, v in kwargs.items():
            monkeypatch.setenv(k, str(v))

    return _set


@pytest.fixture(autouse=True)
def _reset_scanner_registry():
    """Global-state reset: tests registering scanners never leak them
    (reference analog: tests/conftest.py reset_global_test_state)."""
    from agentbom_amd.scan import registry as _reg

    snapshot = dict(_reg._REGISTRY)
    yield
    _reg._REGISTRY.clear()
    _reg._REGISTRY.update(snapshot)
