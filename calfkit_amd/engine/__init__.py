"""The MI355X inference engine package.

- ``engine``: the continuous-batching core (scheduler thread, paged KV,
  multi-step hipGraph decode runs, grammar-constrained sampling).
- ``local``: ``LocalModelClient`` — the Provider-port bridge the agent
  plane calls (prompt rendering, tool plans, structured output).
- ``kvcache``: paged KV allocator + chain-hashed prefix cache over the
  288 GB HBM3E rank.
- ``tokenizer``: the synthetic tokenizer (reserved control ids incl. the
  tool-call grammar tokens).

Lazy: importing this package must not load torch (the import-weight
guarantee) — reach submodules explicitly.
"""

from typing import TYPE_CHECKING

if TYPE_CHECKING:  # pragma: no cover
    from .engine import GenRequest, GenResult, InferenceEngine  # noqa: F401
    from .local import LocalModelClient  # noqa: F401

__all__ = ["GenRequest", "GenResult", "InferenceEngine", "LocalModelClient"]


def __getattr__(name: str):
    if name in ("GenRequest", "GenResult", "InferenceEngine"):
        from . import engine as _e

        return getattr(_e, name)
    if name == "LocalModelClient":
        from .local import LocalModelClient

        return LocalModelClient
    raise AttributeError(name)
