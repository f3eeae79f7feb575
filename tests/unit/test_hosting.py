"""Hosting seam: the Worker resolves node -> processor through the
@processes registry and never names a kind (reference _hosting.py:1-2)."""

from dataclasses import dataclass

import pytest

from calfkit_amd._hosting import (
    PROCESSOR_REGISTRY,
    HostableNode,
    host,
    processes,
)


def test_builtin_kinds_are_registered():
    """Agent/tool/mcp processors self-register on import (the Worker
    composition root resolves every node through this table)."""
    import calfkit_amd._processors  # noqa: F401  (registration side effect)

    for kind in ("agent", "tool", "toolbox"):
        assert kind in PROCESSOR_REGISTRY, kind


def test_host_resolves_by_kind_not_by_type(monkeypatch):
    @dataclass
    class Custom:
        kind: str = "custom-kind"
        name: str = "n1"

    class Proc:
        def __init__(self, node, machinery):
            self.node_name = node.name

    monkeypatch.setitem(PROCESSOR_REGISTRY, "custom-kind",
                        lambda node, m: Proc(node, m))
    node = Custom()
    assert isinstance(node, HostableNode)
    proc = host(node, machinery=None)
    assert proc.node_name == "n1"


def test_host_rejects_unhostable_and_unknown():
    with pytest.raises(TypeError, match="not hostable"):
        host(object(), machinery=None)

    @dataclass
    class Odd:
        kind: str = "no-such-kind"
        name: str = "x"

    with pytest.raises(TypeError, match="no processor registered"):
        host(Odd(), machinery=None)


def test_processes_decorator_registers_and_returns(monkeypatch):
    monkeypatch.delitem(PROCESSOR_REGISTRY, "tmp-kind", raising=False)

    @processes("tmp-kind")
    def factory(node, machinery):
        return None

    assert PROCESSOR_REGISTRY["tmp-kind"] is factory
    PROCESSOR_REGISTRY.pop("tmp-kind", None)
