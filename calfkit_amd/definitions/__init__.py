"""Pure definitions a client imports for free — no machinery.

Capability parity: reference ``definitions/``
(/root/reference/src/calfkit/definitions/__init__.py:1-3).
"""

from .agent import Agent
from .refs import AgentRef, EventRef, ThreadRef, ToolRef, ToolboxRef
from .selections import Handoff, Messaging, PeerAgents, ToolMode, Toolboxes, Tools
from .toolbox import MCPToolbox
from .tools import ToolDef, agent_tool, tool

__all__ = [
    "Agent",
    "AgentRef",
    "EventRef",
    "ThreadRef",
    "ToolRef",
    "ToolboxRef",
    "Handoff",
    "Messaging",
    "PeerAgents",
    "ToolMode",
    "Toolboxes",
    "Tools",
    "MCPToolbox",
    "ToolDef",
    "agent_tool",
    "tool",
]
