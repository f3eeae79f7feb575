"""Error vocabulary — the bottom tier of the layered core.

Everything above may import this; this imports nothing.

Capability parity: reference ``errors.py`` (/root/reference/src/calfkit/errors.py:1-3)
names ``ErrorReport``/``ErrorDetail``, ``ToolError`` (retryable, raised by tool
authors) and a user-catchable exception family; the typed fault rail
(``FaultMessage``/``NodeFaultError``) is from /root/reference/CHANGELOG.md:187-196.
Design here is original: plain dataclasses, structured exception harvest via
``ErrorReport.from_exception``, oversized-fault elision as a method on the report.
"""

from __future__ import annotations

import traceback
from dataclasses import dataclass, field
from typing import Any

__all__ = [
    "CalfError",
    "ToolError",
    "NodeFaultError",
    "DecodeError",
    "SelectionError",
    "TimeoutError_",
    "EngineError",
    "ErrorDetail",
    "ErrorReport",
    "FaultMessage",
    "MAX_FAULT_BYTES",
]

#: Faults whose serialized form exceeds this elide their run-state payload
#: instead of being dropped on the wire.
MAX_FAULT_BYTES = 256 * 1024


class CalfError(Exception):
    """Base of the user-catchable exception family."""


class ToolError(CalfError):
    """Raised by tool authors to signal a *retryable* tool failure.

    The message is surfaced to the calling agent as the tool result so the
    model can recover; it never crashes the node.
    """

    def __init__(self, message: str, *, retryable: bool = True):
        super().__init__(message)
        self.retryable = retryable


class NodeFaultError(CalfError):
    """A fault propagated from another node over the fault rail."""

    def __init__(self, report: "ErrorReport"):
        super().__init__(report.summary())
        self.report = report


class DecodeError(CalfError):
    """An inbound payload failed the decode floor (bad JSON/UTF-8/shape)."""


class SelectionError(CalfError):
    """A tool/peer selection could not be resolved against the directory."""


class TimeoutError_(CalfError):
    """A pending call outlived its deadline."""


class EngineError(CalfError):
    """The local model engine failed (kernel error, OOM, bad request)."""


@dataclass(frozen=True)
class ErrorDetail:
    """One frame of a structured exception harvest."""

    exc_type: str
    message: str
    frames: tuple[str, ...] = ()

    @classmethod
    def from_exception(cls, exc: BaseException, *, max_frames: int = 24) -> "ErrorDetail":
        frames = tuple(
            f"{f.filename}:{f.lineno} in {f.name}"
            for f in traceback.extract_tb(exc.__traceback__)[-max_frames:]
        )
        return cls(exc_type=type(exc).__name__, message=str(exc), frames=frames)


@dataclass
class ErrorReport:
    """The wire-model of a node fault.

    ``run_state`` carries arbitrary context (the thread id, the hop, the
    request) and is elided when the report grows past ``MAX_FAULT_BYTES``
    rather than silently dropping the whole fault.
    """

    node: str
    kind: str  # "tool" | "agent" | "mcp" | "engine" | "wire"
    detail: ErrorDetail
    causes: tuple[ErrorDetail, ...] = ()
    run_state: dict[str, Any] = field(default_factory=dict)
    elided: bool = False

    @classmethod
    def from_exception(cls, node: str, kind: str, exc: BaseException, **run_state: Any) -> "ErrorReport":
        causes: list[ErrorDetail] = []
        cause = exc.__cause__ or exc.__context__
        seen = {id(exc)}
        while cause is not None and id(cause) not in seen and len(causes) < 8:
            causes.append(ErrorDetail.from_exception(cause))
            seen.add(id(cause))
            cause = cause.__cause__ or cause.__context__
        return cls(node=node, kind=kind, detail=ErrorDetail.from_exception(exc),
                   causes=tuple(causes), run_state=dict(run_state))

    def summary(self) -> str:
        return f"[{self.kind}@{self.node}] {self.detail.exc_type}: {self.detail.message}"

    def to_wire(self) -> dict[str, Any]:
        d = {
            "node": self.node,
            "kind": self.kind,
            "detail": {"exc_type": self.detail.exc_type, "message": self.detail.message,
                        "frames": list(self.detail.frames)},
            "causes": [
                {"exc_type": c.exc_type, "message": c.message, "frames": list(c.frames)}
                for c in self.causes
            ],
            "run_state": self.run_state,
            "elided": self.elided,
        }
        return d

    @classmethod
    def from_wire(cls, d: dict[str, Any]) -> "ErrorReport":
        det = d.get("detail", {})
        return cls(
            node=d.get("node", "?"),
            kind=d.get("kind", "?"),
            detail=ErrorDetail(det.get("exc_type", "?"), det.get("message", ""),
                               tuple(det.get("frames", ()))),
            causes=tuple(
                ErrorDetail(c.get("exc_type", "?"), c.get("message", ""), tuple(c.get("frames", ())))
                for c in d.get("causes", ())
            ),
            run_state=dict(d.get("run_state", {})),
            elided=bool(d.get("elided", False)),
        )

    def elide_if_oversized(self, encoded_size: int, limit: int = MAX_FAULT_BYTES) -> "ErrorReport":
        """Return a copy with ``run_state`` dropped when too large for the wire."""
        if encoded_size <= limit:
            return self
        return ErrorReport(node=self.node, kind=self.kind, detail=self.detail,
                           causes=self.causes, run_state={}, elided=True)


@dataclass(frozen=True)
class FaultMessage:
    """A fault delivery as carried on the wire (the fault rail)."""

    report: ErrorReport
    #: echo of the call identity this fault answers, if any
    call_id: str | None = None
    frame_id: str | None = None
