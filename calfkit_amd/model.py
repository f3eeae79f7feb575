"""Model identity + settings — the decorator seam. User-constructed, I/O-free.

Capability parity: reference ``models.py``
(/root/reference/src/calfkit/models.py:1-2).  A ``Model`` pairs a name with a
provider factory and default settings; ``@model.instructions`` /
``temp_instructions`` per /root/reference/CHANGELOG.md:462.

(The reference calls this module ``models``; here the GPU model *families*
live in the ``models/`` package, so the identity class lives in ``model``.)
"""

from __future__ import annotations

import contextlib
from dataclasses import dataclass, field
from typing import Any, Callable

from .context import TurnContext
from .schemas import ModelSettings

__all__ = ["Model"]

InstructionsFn = Callable[[TurnContext], str]


@dataclass
class Model:
    """Identity + settings for a model a node will call.

    ``client`` is any object satisfying the ``Provider`` port, or a zero-arg
    factory returning one (so heavy engines construct lazily inside the
    Worker, never at definition import time).
    """

    name: str
    client: Any = None
    settings: ModelSettings = field(default_factory=ModelSettings)
    _instructions_fn: InstructionsFn | None = field(default=None, repr=False)
    _temp_instructions: list[str] = field(default_factory=list, repr=False)

    def resolve_client(self) -> Any:
        c = self.client
        if c is not None and callable(c) and not hasattr(c, "complete"):
            c = c()
            self.client = c
        return c

    # ------------------------------------------------------------ seam
    def instructions(self, fn: InstructionsFn) -> InstructionsFn:
        """Decorator: dynamic per-turn instructions appended to the system prompt."""
        self._instructions_fn = fn
        return fn

    @contextlib.contextmanager
    def temp_instructions(self, text: str):
        """Scoped extra instructions (testing / REPL overrides)."""
        self._temp_instructions.append(text)
        try:
            yield self
        finally:
            self._temp_instructions.remove(text)

    def render_instructions(self, ctx: TurnContext) -> str:
        out: list[str] = []
        if self._instructions_fn is not None:
            out.append(self._instructions_fn(ctx))
        out.extend(self._temp_instructions)
        return "\n".join(s for s in out if s)
