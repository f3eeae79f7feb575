"""Per-kind hop pipelines; session/bus factories injected by the root.

Capability parity: reference ``_processors/``
(/root/reference/src/calfkit/_processors/__init__.py:1-3).  Importing this
package registers the three processors with the ``@processes`` registry.
"""

from .agent import AgentProcessor
from .mcp import MCPToolboxProcessor
from .tool import ToolProcessor

__all__ = ["AgentProcessor", "ToolProcessor", "MCPToolboxProcessor"]
