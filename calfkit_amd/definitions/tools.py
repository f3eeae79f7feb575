"""ToolDef + the @tool / @agent_tool decorators: schema derivation, arg
parsing, invocation.

Capability parity: reference ``definitions/tools.py``
(/root/reference/src/calfkit/definitions/tools.py:1-2); function-tool
decorator with schema derivation per /root/reference/CHANGELOG.md:638;
ToolContext injection per /root/reference/CHANGELOG.md:563; retryable
``ToolError`` per /root/reference/src/calfkit/errors.py:2.

Schema derivation is hand-rolled from type hints (no pydantic import in the
core — keeps ``import calfkit_amd`` dependency-free like the reference's
import-weight guarantee, /root/reference/tests/unit/test_import_weight.py:11-21).
"""

from __future__ import annotations

import inspect
import typing
from dataclasses import MISSING, dataclass, field, fields as dc_fields, is_dataclass
from typing import Any, Callable, get_args, get_origin, get_type_hints

from ..context import ToolContext
from ..errors import ToolError
from ..schemas import ToolSchema
from .refs import validate_node_name

__all__ = ["ToolDef", "tool", "agent_tool", "schema_for_type", "parse_args_against"]


# ------------------------------------------------------- JSON schema from hints

_SCALARS: dict[Any, dict[str, Any]] = {
    str: {"type": "string"},
    int: {"type": "integer"},
    float: {"type": "number"},
    bool: {"type": "boolean"},
    type(None): {"type": "null"},
    Any: {},
}


def schema_for_type(tp: Any) -> dict[str, Any]:
    """Derive a JSON schema fragment from a Python type annotation."""
    if tp in _SCALARS:
        return dict(_SCALARS[tp])
    origin = get_origin(tp)
    if origin in (list, tuple, set, frozenset):
        args = get_args(tp)
        item = schema_for_type(args[0]) if args else {}
        return {"type": "array", "items": item}
    if origin is dict:
        args = get_args(tp)
        extra = schema_for_type(args[1]) if len(args) == 2 else {}
        return {"type": "object", "additionalProperties": extra or True}
    if origin is typing.Union:
        args = [a for a in get_args(tp)]
        if type(None) in args and len(args) == 2:
            inner = schema_for_type([a for a in args if a is not type(None)][0])
            return {"anyOf": [inner, {"type": "null"}]}
        return {"anyOf": [schema_for_type(a) for a in args]}
    if origin is typing.Literal:
        vals = list(get_args(tp))
        return {"enum": vals}
    if is_dataclass(tp):
        props: dict[str, Any] = {}
        required: list[str] = []
        hints = get_type_hints(tp)
        for f in dc_fields(tp):
            props[f.name] = schema_for_type(hints.get(f.name, Any))
            if f.default is MISSING and f.default_factory is MISSING:
                required.append(f.name)
        out: dict[str, Any] = {"type": "object", "properties": props}
        if required:
            out["required"] = required
        return out
    # unknown types ride as opaque strings
    return {"type": "string"}


def _coerce(value: Any, tp: Any) -> Any:
    """Best-effort coercion of a decoded JSON value to the annotated type."""
    if tp in (Any, inspect.Parameter.empty) or value is None:
        return value
    origin = get_origin(tp)
    if origin is typing.Union:
        for a in get_args(tp):
            if a is type(None):
                continue
            try:
                return _coerce(value, a)
            except (TypeError, ValueError):
                continue
        return value
    if origin is typing.Literal:
        return value
    if origin in (list, tuple, set, frozenset):
        args = get_args(tp)
        inner = args[0] if args else Any
        seq = [_coerce(v, inner) for v in value]
        if origin is list:
            return seq
        return origin(seq)
    if origin is dict:
        return dict(value)
    if is_dataclass(tp) and isinstance(value, dict):
        hints = get_type_hints(tp)
        kwargs = {f.name: _coerce(value[f.name], hints.get(f.name, Any))
                  for f in dc_fields(tp) if f.name in value}
        return tp(**kwargs)
    if tp is float and isinstance(value, int):
        return float(value)
    if tp in (str, int, float, bool):
        if not isinstance(value, tp):
            raise TypeError(f"expected {tp.__name__}, got {type(value).__name__}")
        return value
    return value


def parse_args_against(schema_fn_params: dict[str, Any], defaults: dict[str, Any],
                       required: set[str], arguments: dict[str, Any]) -> dict[str, Any]:
    """Validate + coerce decoded arguments against derived parameter info."""
    unknown = set(arguments) - set(schema_fn_params)
    if unknown:
        raise ToolError(f"unknown arguments: {sorted(unknown)}", retryable=True)
    missing = required - set(arguments)
    if missing:
        raise ToolError(f"missing required arguments: {sorted(missing)}", retryable=True)
    out: dict[str, Any] = {}
    for name, tp in schema_fn_params.items():
        if name in arguments:
            try:
                out[name] = _coerce(arguments[name], tp)
            except (TypeError, ValueError) as e:
                raise ToolError(f"bad argument {name!r}: {e}", retryable=True) from e
        elif name in defaults:
            out[name] = defaults[name]
    return out


# ------------------------------------------------------------------ ToolDef

@dataclass
class ToolDef:
    """A function tool: derived schema + parser + invoker. A mesh node."""

    name: str
    fn: Callable[..., Any]
    description: str = ""
    param_types: dict[str, Any] = field(default_factory=dict)
    defaults: dict[str, Any] = field(default_factory=dict)
    required: frozenset[str] = frozenset()
    schema: ToolSchema = field(default_factory=lambda: ToolSchema(name="_"))
    wants_context: str | None = None  # parameter name annotated ToolContext
    kind: str = "tool"

    @classmethod
    def from_function(cls, fn: Callable[..., Any], *, name: str | None = None,
                      description: str | None = None) -> "ToolDef":
        tool_name = validate_node_name(name or fn.__name__)
        doc = inspect.getdoc(fn) or ""
        desc = description if description is not None else doc.split("\n\n")[0]
        sig = inspect.signature(fn)
        hints = get_type_hints(fn)
        param_types: dict[str, Any] = {}
        defaults: dict[str, Any] = {}
        required: set[str] = set()
        wants_context: str | None = None
        props: dict[str, Any] = {}
        for pname, p in sig.parameters.items():
            if p.kind in (inspect.Parameter.VAR_POSITIONAL, inspect.Parameter.VAR_KEYWORD):
                raise TypeError(f"tool {tool_name}: *args/**kwargs not supported")
            tp = hints.get(pname, Any)
            if tp is ToolContext:
                wants_context = pname
                continue
            param_types[pname] = tp
            props[pname] = schema_for_type(tp)
            if p.default is inspect.Parameter.empty:
                required.add(pname)
            else:
                defaults[pname] = p.default
        parameters: dict[str, Any] = {"type": "object", "properties": props}
        if required:
            parameters["required"] = sorted(required)
        schema = ToolSchema(name=tool_name, description=desc, parameters=parameters)
        return cls(name=tool_name, fn=fn, description=desc, param_types=param_types,
                   defaults=defaults, required=frozenset(required), schema=schema,
                   wants_context=wants_context)

    async def invoke(self, arguments: dict[str, Any], ctx: ToolContext | None = None) -> Any:
        kwargs = parse_args_against(self.param_types, self.defaults, set(self.required),
                                    arguments)
        if self.wants_context:
            kwargs[self.wants_context] = ctx or ToolContext(tool_name=self.name, call_id="")
        result = self.fn(**kwargs)
        if inspect.isawaitable(result):
            result = await result
        return result

    # Agent ctor accepts ToolDef directly; calling the def calls the raw fn.
    def __call__(self, *args: Any, **kwargs: Any) -> Any:
        return self.fn(*args, **kwargs)


def tool(fn: Callable[..., Any] | None = None, *, name: str | None = None,
         description: str | None = None) -> Any:
    """``@tool`` — derive a ToolDef from a function."""

    def wrap(f: Callable[..., Any]) -> ToolDef:
        return ToolDef.from_function(f, name=name, description=description)

    return wrap(fn) if fn is not None else wrap


#: ``@agent_tool`` is the public alias used in the reference's README
#: (/root/reference/README.md:43-54): a tool intended to be hosted as its own
#: mesh node and called by agents.
agent_tool = tool
