"""calfkit_amd — an MI355X-native event-driven multi-agent SDK.

Public modules are the user surface; ``_``-modules are machinery.  Exports
are lazy (PEP 562) so ``import calfkit_amd`` stays dependency-free — no
torch, no broker SDK, nothing heavy loads until used
(parity: /root/reference/src/calfkit/__init__.py:1-6 and the import-weight
guarantee /root/reference/tests/unit/test_import_weight.py:11-21).
"""

from __future__ import annotations

from typing import Any

__version__ = "0.1.0"

_EXPORTS: dict[str, tuple[str, str]] = {
    # name -> (module, attr)
    "Agent": ("calfkit_amd.definitions.agent", "Agent"),
    "ToolDef": ("calfkit_amd.definitions.tools", "ToolDef"),
    "tool": ("calfkit_amd.definitions.tools", "tool"),
    "agent_tool": ("calfkit_amd.definitions.tools", "agent_tool"),
    "MCPToolbox": ("calfkit_amd.definitions.toolbox", "MCPToolbox"),
    "Tools": ("calfkit_amd.definitions.selections", "Tools"),
    "Toolboxes": ("calfkit_amd.definitions.selections", "Toolboxes"),
    "PeerAgents": ("calfkit_amd.definitions.selections", "PeerAgents"),
    "Messaging": ("calfkit_amd.definitions.selections", "Messaging"),
    "Handoff": ("calfkit_amd.definitions.selections", "Handoff"),
    "Model": ("calfkit_amd.model", "Model"),
    "ModelSettings": ("calfkit_amd.schemas", "ModelSettings"),
    "Worker": ("calfkit_amd.worker", "Worker"),
    "Client": ("calfkit_amd.client", "Client"),
    "ToolError": ("calfkit_amd.errors", "ToolError"),
    "CalfError": ("calfkit_amd.errors", "CalfError"),
    "NodeFaultError": ("calfkit_amd.errors", "NodeFaultError"),
    "TurnContext": ("calfkit_amd.context", "TurnContext"),
    "ToolContext": ("calfkit_amd.context", "ToolContext"),
    "AgentHook": ("calfkit_amd.hooks", "AgentHook"),
    "resource": ("calfkit_amd.lifecycle", "resource"),
    "LocalModelClient": ("calfkit_amd.engine.local", "LocalModelClient"),
}

__all__ = sorted(_EXPORTS) + ["__version__"]


def __getattr__(name: str) -> Any:
    try:
        module_name, attr = _EXPORTS[name]
    except KeyError:
        raise AttributeError(f"module 'calfkit_amd' has no attribute {name!r}") from None
    import importlib

    return getattr(importlib.import_module(module_name), attr)


def __dir__() -> list[str]:
    return __all__
