"""LocalModelClient — the in-process Provider backed by the CDNA4 engine.

This replaces the reference's remote-HTTP providers behind the same
``Provider`` port: ``complete(EngineRequest, sink)`` renders
the chat prompt, submits to the shared per-process InferenceEngine (one per
GPU rank — concurrent agent turns batch into one decode), streams token
deltas to the sink, and returns an AssistantMessage whose ProviderFacts
carry the measured prefill/decode times for the StepLedger.
"""

from __future__ import annotations

import asyncio
import threading
from typing import Any

import torch

from ..messages import AssistantMessage, Message, ProviderFacts, TextPart, Usage
from ..schemas import EngineRequest
from ..streaming import ChunkSink, StreamChunk
from .engine import GenRequest, InferenceEngine
from .tokenizer import SyntheticTokenizer

__all__ = ["LocalModelClient", "shared_engine"]

_ENGINES: dict[tuple, InferenceEngine] = {}
_ENGINES_LOCK = threading.Lock()


def shutdown_engines() -> None:
    """Stop every shared engine (clean interpreter exit: a daemon scheduler
    thread killed mid-HIP-call aborts with 'terminate called without an
    active exception')."""
    with _ENGINES_LOCK:
        engines = list(_ENGINES.values())
        _ENGINES.clear()
    for eng in engines:
        try:
            eng.stop()
        except Exception:
            pass


import atexit  # noqa: E402

atexit.register(shutdown_engines)


def shared_engine(config: Any, device: str | None = None, *, tp: int = 1,
                  ep: int = 1, autostart: bool = True,
                  **engine_kw: Any) -> InferenceEngine:
    """One engine per (model, device) per process — all agents share it.

    ``tp`` > 1 shards the model across the torch.distributed world (every
    rank must construct the same engine and drive the same forward order);
    ``ep`` > 1 does the same with expert parallelism (MoE models only; the
    lockstep discipline is identical — every rank steps the same batch and
    the MoE all-to-alls line up).  The default is one independent replica
    per process (the one-Worker-per-GPU scaling story).
    """
    from ..models.config import ModelConfig, get_config
    from ..models.llama import LlamaShard

    if isinstance(config, str):
        config = get_config(config)
    assert isinstance(config, ModelConfig)
    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
    if ep > 1 and not config.is_moe:
        raise ValueError(f"{config.name} is dense; ep={ep} needs an MoE model")
    key = (config.name, device, tp, ep)
    with _ENGINES_LOCK:
        eng = _ENGINES.get(key)
        if eng is None:
            if config.is_moe:
                from ..models.mixtral import MixtralShard as Shard
            else:
                Shard = LlamaShard
            if tp > 1 or ep > 1:
                from ..parallel import get_dist, init_model_parallel

                dist = get_dist()
                world = tp * ep  # EP composes with TP (r02)
                if dist.world_size != world:
                    raise ValueError(
                        f"tp={tp} x ep={ep} but torch.distributed world is "
                        f"{dist.world_size}")
                init_model_parallel(tp, ep)
                if config.is_moe:
                    model = Shard(config, tp_rank=dist.rank % tp,
                                  tp_size=tp, ep_rank=dist.rank // tp,
                                  ep_size=ep, device=device)
                else:
                    model = Shard(config, tp_rank=dist.rank, tp_size=tp,
                                  device=device)
            else:
                model = Shard(config, tp_rank=0, tp_size=1, device=device)
            eng = InferenceEngine(model, **engine_kw)
            if autostart:
                eng.start()
            _ENGINES[key] = eng
    return eng


class LocalModelClient:
    """Provider port implementation over the in-process engine."""

    def __init__(self, model: Any = "llama-3-8b", *, device: str | None = None,
                 max_batch: int = 64, max_model_len: int | None = None,
                 use_graphs: bool | None = None, kv_pages: int | None = None,
                 tp: int = 1, ep: int = 1):
        self._config = model
        self._device = device
        self._engine_kw = dict(max_batch=max_batch, max_model_len=max_model_len,
                               use_graphs=use_graphs, kv_pages=kv_pages, tp=tp,
                               ep=ep)
        self._engine: InferenceEngine | None = None
        self._tokenizer: SyntheticTokenizer | None = None

    # lazily built so Worker/Agent construction never touches the GPU
    @property
    def engine(self) -> InferenceEngine:
        if self._engine is None:
            self._engine = shared_engine(self._config, self._device,
                                         **self._engine_kw)
            self._tokenizer = SyntheticTokenizer(self._engine.cfg.vocab_size)
        return self._engine

    @property
    def tokenizer(self) -> SyntheticTokenizer:
        self.engine
        assert self._tokenizer is not None
        return self._tokenizer

    # ------------------------------------------------------------ rendering
    @staticmethod
    def render_prompt(messages: tuple[Message, ...],
                      tools: tuple = (), output: Any = None) -> str:
        lines = []
        if output is not None:
            keys = " ".join(sorted(
                (output.json_schema or {}).get("properties", {})))
            lines.append(f"<|output|> {output.name} ( {keys} )")
        if tools:
            # the model-facing tool advertisement (reference semantics:
            # providers render request.tools into the prompt —
            # /root/reference/CHANGELOG.md:638)
            decls = " ; ".join(
                f"{t.name} ( {' '.join(sorted((t.parameters or {}).get('properties', {})))} ) {t.description}"
                for t in tools)
            lines.append(f"<|tools|> {decls}")
        for m in messages:
            text = " ".join(p.text for p in m.parts if isinstance(p, TextPart))
            lines.append(f"<|{m.role}|> {text}")
        lines.append("<|assistant|>")
        return "\n".join(lines)

    # ------------------------------------------------------------- complete
    async def complete(self, request: EngineRequest, sink: ChunkSink) -> AssistantMessage:
        engine = self.engine
        tok = self.tokenizer
        prompt = self.render_prompt(request.messages, request.tools,
                                    request.output)
        ids = tok.encode(prompt)
        s = request.settings
        seq = 0

        def on_tokens(ts: list[int]) -> None:
            # one chunk per decode RUN (the engine's host-visit granularity)
            # — per-token chunks flooded the asyncio loop at large batch
            nonlocal seq
            seq += 1
            sink.emit(StreamChunk(event="text_delta", seq=seq,
                                  text=tok.decode(ts) + " "))

        # Tool-call plan: with tools advertised and no tool
        # results in this turn yet, steer decode through the call grammar —
        # the NAME is sampled on-device among the advertised tools.  A turn
        # that already carries tool results produces the closing answer.
        tool_names = None
        trigger = 0
        ncalls = 1
        if request.tools and not any(m.role == "tool"
                                     for m in request.messages):
            import os

            advertised = request.tools
            if request.tool_choice and request.tool_choice != "required":
                # ToolMode.choice: the grammar may only sample THIS tool
                chosen = tuple(t for t in request.tools
                               if t.name == request.tool_choice)
                advertised = chosen or request.tools
            tool_names = [tok.word_id(t.name) for t in advertised][:8]
            seedv = (s.seed or 0) or (len(ids) * 2654435761 & 0xFFFF)
            fan = int(os.environ.get("CALF_TOOLCALL_FANOUT", "1"))
            if not s.parallel_tool_calls:
                fan = 1
            ncalls = max(1, min(fan, len(advertised)))
            # fire early enough that all calls fit the token budget
            span = 7 * ncalls + 2  # TCALL+name+4 args+TEND per call, + EOS
            trigger = min(seedv % 5 + 1, max(0, s.max_tokens - span))
            if request.tool_choice:
                # ToolMode.required: the call fires immediately — no
                # free-text preamble before the grammar trigger
                trigger = 1

        gen = GenRequest(
            prompt_ids=ids,
            max_new_tokens=max(1, s.max_tokens),
            temperature=s.temperature,
            top_k=s.top_k,
            seed=s.seed or 0,
            ignore_eos=False,
            tool_names=tool_names,
            tool_trigger=trigger,
            tool_ncalls=ncalls,
            on_tokens=on_tokens,
        )
        fut = engine.submit(gen)
        result = await asyncio.wrap_future(fut)
        parts, calls = _parse_output(result.token_ids, request.tools, tok)
        finish = "tool_calls" if calls else result.finish_reason
        if request.output is not None and not calls:
            # structured output from the ENGINE's sampled stream: the
            # schema dictates the shape, the sampled tokens pick the
            # values (same philosophy as tool-call args), emitted as the
            # final_result-prefixed union the policy end-shape expects
            import json as _json

            value = _materialize_value(request.output.json_schema or {},
                                       result.token_ids or [17], [0])
            parts = (TextPart(_json.dumps({"final_result": value})),)
            finish = "stop"
        if not calls and s.stop and request.output is None:
            # stop-sequence truncation (client side), text-only turns
            text = " ".join(p.text for p in parts
                            if isinstance(p, TextPart))
            for stop in s.stop:
                idx = text.find(stop)
                if idx >= 0:
                    text = text[:idx]
                    finish = "stop"
            parts = (TextPart(text),)
        facts = ProviderFacts(
            model_name=engine.cfg.name,
            usage=Usage(prompt_tokens=result.prompt_tokens,
                        completion_tokens=len(result.token_ids)),
            finish_reason=finish,
            prefill_ms=result.prefill_ms,
            decode_ms=result.decode_ms,
        )
        return AssistantMessage(parts, facts=facts)


def _materialize_value(schema: dict, toks: list[int], cur: list[int]):
    """Recursively fill a JSON schema from sampled tokens (engine-driven
    structured output; cur = mutable token cursor)."""
    def take() -> int:
        v = toks[cur[0] % len(toks)]
        cur[0] += 1
        return v

    if (schema or {}).get("enum"):
        e = schema["enum"]
        return e[take() % len(e)]
    t = (schema or {}).get("type", "object")
    if t == "object":
        return {k: _materialize_value(v or {}, toks, cur)
                for k, v in sorted((schema.get("properties") or {}).items())}
    if t == "array":
        item = schema.get("items") or {"type": "string"}
        return [_materialize_value(item, toks, cur)]
    if t == "integer":
        return take()
    if t == "number":
        return round(take() / 7.0, 3)
    if t == "boolean":
        return bool(take() % 2)
    if t == "null":
        return None
    return f"t{take()}"


def _materialize_args(schema: Any, toks: list[int]) -> dict:
    """Deterministically fill the tool's JSON schema from sampled arg
    tokens — the GPU picks the values, the schema dictates the types (so
    caller-side validation passes and the ToolProcessor can invoke)."""
    props = (schema.parameters or {}).get("properties", {})
    args: dict[str, Any] = {}
    for i, key in enumerate(sorted(props)):
        tokv = toks[i % len(toks)] if toks else 17
        spec = props[key] or {}
        if spec.get("enum"):
            # constrained fields (e.g. a capability tool's peer name):
            # the sampled token picks among the VALID values, so
            # engine-sampled handoffs / peer messages actually route
            args[key] = spec["enum"][tokv % len(spec["enum"])]
            continue
        ty = spec.get("type", "string")
        if ty == "integer":
            args[key] = int(tokv)
        elif ty == "number":
            args[key] = round(tokv / 7.0, 3)
        elif ty == "boolean":
            args[key] = bool(tokv % 2)
        elif ty == "array":
            args[key] = [f"t{tokv}"]
        elif ty == "object":
            args[key] = {}
        else:
            args[key] = f"t{tokv}"
    return args


def _parse_output(ids: list[int], tools: tuple, tok: SyntheticTokenizer):
    """Token stream -> (parts, calls): TCALL <name> <args...> TEND spans
    become ToolCallParts; everything else decodes to text."""
    import uuid

    from ..messages import ToolCallPart

    by_id: dict[int, Any] = {}
    for t in tools:
        by_id.setdefault(tok.word_id(t.name), t)
    parts: list[Any] = []
    text_run: list[int] = []
    calls: list[ToolCallPart] = []
    i = 0
    while i < len(ids):
        t = ids[i]
        if t == tok.TCALL and i + 1 < len(ids):
            name_id = ids[i + 1]
            i += 2
            arg_toks: list[int] = []
            while i < len(ids) and ids[i] != tok.TEND:
                arg_toks.append(ids[i])
                i += 1
            if i < len(ids):
                i += 1  # consume TEND
            schema = by_id.get(name_id)
            if schema is None:
                continue  # name outside the advertised set: drop the span
            if text_run:
                txt = tok.decode(text_run)
                if txt:
                    parts.append(TextPart(txt))
                text_run = []
            call = ToolCallPart(call_id=uuid.uuid4().hex,
                                tool_name=schema.name,
                                arguments=_materialize_args(schema, arg_toks))
            parts.append(call)
            calls.append(call)
        else:
            text_run.append(t)
            i += 1
    if text_run:
        txt = tok.decode(text_run)
        if txt:
            parts.append(TextPart(txt))
    if not parts:
        parts = [TextPart("")]
    return tuple(parts), calls
