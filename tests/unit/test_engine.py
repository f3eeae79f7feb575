"""Engine tests on CPU with the TINY config (reference ops path)."""

import math

import pytest
import torch

from calfkit_amd.engine.engine import GenRequest, InferenceEngine
from calfkit_amd.engine.kvcache import KVCache, PAGE_SIZE
from calfkit_amd.engine.tokenizer import SyntheticTokenizer
from calfkit_amd.models.config import TINY, get_config
from calfkit_amd.models.llama import LlamaShard


def make_engine(**kw) -> InferenceEngine:
    model = LlamaShard(TINY, device="cpu")
    kw.setdefault("max_batch", 4)
    kw.setdefault("use_graphs", False)
    kw.setdefault("max_model_len", 512)
    return InferenceEngine(model, **kw)


def test_kvcache_alloc_release():
    kv = KVCache(layers=1, kv_heads=1, head_dim=128, num_pages=8, device="cpu")
    slots = kv.slots_for(1, 100)
    assert len(slots) == 100
    assert kv.length(1) == 100
    assert kv.free_pages == 8 - 2
    kv.release(1)
    assert kv.free_pages == 8
    with pytest.raises(MemoryError):
        kv.slots_for(2, 8 * PAGE_SIZE + 1)


def test_tokenizer_deterministic():
    t = SyntheticTokenizer(512)
    a = t.encode("hello world hello")
    b = t.encode("hello world hello")
    assert a == b
    assert a[0] == t.BOS
    assert a[1] == a[3]  # same word, same id
    assert all(i < 512 for i in a)


def test_engine_generates_tokens():
    eng = make_engine()
    try:
        r = eng.generate(GenRequest(prompt_ids=[1, 5, 9, 200], max_new_tokens=8),
                         timeout=120)
        assert len(r.token_ids) == 8
        assert r.prompt_tokens == 4
        assert r.prefill_ms > 0 and r.decode_ms > 0
    finally:
        eng.stop()


def test_engine_greedy_is_deterministic_and_continuous_batching():
    eng = make_engine()
    try:
        reqs = [GenRequest(prompt_ids=[1, 5, 9], max_new_tokens=6)
                for _ in range(3)]
        futs = [eng.submit(r) for r in reqs]
        outs = [f.result(120).token_ids for f in futs]
        assert outs[0] == outs[1] == outs[2]
        # a decode step carried more than one sequence
        assert eng.stats["decode_tokens"] > eng.stats["decode_steps"]
    finally:
        eng.stop()


def test_engine_decode_matches_single_shot():
    """Batched decode == solo decode (continuous batching is transparent)."""
    eng = make_engine()
    try:
        solo = eng.generate(GenRequest(prompt_ids=[1, 7, 42], max_new_tokens=5),
                            timeout=120).token_ids
        futs = [eng.submit(GenRequest(prompt_ids=[1, 7, 42], max_new_tokens=5)),
                eng.submit(GenRequest(prompt_ids=[1, 9, 90, 3], max_new_tokens=5))]
        both = [f.result(120).token_ids for f in futs]
        assert both[0] == solo
    finally:
        eng.stop()


def test_model_forward_prefill_decode_consistency():
    """Decode of token t+1 after prefill of t tokens == prefill of t+1."""
    torch.manual_seed(0)
    model = LlamaShard(TINY, device="cpu")
    kv1 = KVCache(layers=TINY.layers, kv_heads=model.hkv, head_dim=model.dh,
                  num_pages=4, device="cpu")
    kv2 = KVCache(layers=TINY.layers, kv_heads=model.hkv, head_dim=model.dh,
                  num_pages=4, device="cpu")
    scratch = [(torch.zeros(256, model.hkv, model.dh, dtype=model.dtype),
                torch.zeros(256, model.hkv, model.dh, dtype=model.dtype))
               for _ in range(TINY.layers)]
    toks = [1, 17, 33, 99, 250]

    # path A: prefill all 5, logits of last row
    slots = torch.as_tensor(kv1.slots_for(1, 5), dtype=torch.int32)
    logits_a = model.forward(
        torch.as_tensor(toks), torch.arange(5, dtype=torch.int32),
        kv1.kcache, kv1.vcache, slots, prefill_scratch=scratch,
        logits_rows=slice(4, 5))

    # path B: prefill 4, then decode token 5 via the paged cache
    slots_b = torch.as_tensor(kv2.slots_for(2, 4), dtype=torch.int32)
    model.forward(torch.as_tensor(toks[:4]), torch.arange(4, dtype=torch.int32),
                  kv2.kcache, kv2.vcache, slots_b, prefill_scratch=scratch,
                  logits_rows=slice(3, 4))
    slot5 = torch.as_tensor(kv2.slots_for(2, 1), dtype=torch.int32)
    pt = torch.zeros(1, 4, dtype=torch.int32)
    pages = kv2._slots[2].pages
    pt[0, :len(pages)] = torch.as_tensor(pages, dtype=torch.int32)
    logits_b = model.forward(
        torch.as_tensor(toks[4:]), torch.as_tensor([4], dtype=torch.int32),
        kv2.kcache, kv2.vcache, slot5, page_table=pt,
        seq_lens=torch.as_tensor([5], dtype=torch.int32))

    assert torch.allclose(logits_a.float(), logits_b.float(), atol=0.2, rtol=0.05), \
        (logits_a.float() - logits_b.float()).abs().max()
    # and the argmax (greedy next token) agrees
    assert int(logits_a.argmax()) == int(logits_b.argmax())


def test_long_prompt_chunked_prefill_matches_short_path():
    """A prompt longer than PREFILL_CHUNK goes through the chunked solo
    path; its continuation must match an equivalent packed prefill."""
    import calfkit_amd.engine.engine as eng_mod

    old_chunk, old_budget = eng_mod.PREFILL_CHUNK, eng_mod.PREFILL_BUDGET
    eng_mod.PREFILL_CHUNK, eng_mod.PREFILL_BUDGET = 16, 48
    try:
        eng = make_engine(max_model_len=256)
        try:
            prompt = [1] + [(7 * i) % 200 + 10 for i in range(60)]  # 61 > 48
            r_long = eng.generate(GenRequest(prompt_ids=prompt,
                                             max_new_tokens=4), timeout=120)
        finally:
            eng.stop()
        # same prompt through the packed path (budget large enough)
        eng_mod.PREFILL_CHUNK, eng_mod.PREFILL_BUDGET = 512, 8192
        eng2 = make_engine(max_model_len=256)
        try:
            r_pack = eng2.generate(GenRequest(prompt_ids=prompt,
                                              max_new_tokens=4), timeout=120)
        finally:
            eng2.stop()
        assert r_long.token_ids == r_pack.token_ids
    finally:
        eng_mod.PREFILL_CHUNK, eng_mod.PREFILL_BUDGET = old_chunk, old_budget


@pytest.mark.gpu
def test_prefill_overlap_completes_gpu():
    """Opt-in side-stream prefill overlap (CALF_PREFILL_OVERLAP=1): a
    staggered request stream completes with the right token counts and no
    fault.  (Bit-equality with the serial path is NOT asserted: overlapped
    admission shifts decode batch shapes, and library-GEMM tilings round
    differently per shape, so greedy outputs legitimately depend on
    arrival timing — see profiles/r01_bench256_gaps.md.)"""
    import os
    import time as _time

    from calfkit_amd.models.llama import LlamaShard

    os.environ["CALF_PREFILL_OVERLAP"] = "1"
    try:
        model = LlamaShard(TINY, device="cuda")
        eng = InferenceEngine(model, max_batch=8, max_model_len=256)
        try:
            futs = [eng.submit(GenRequest(
                prompt_ids=[1, 3 + i, 7, 11 + i], max_new_tokens=24))
                for i in range(4)]
            _time.sleep(0.05)  # second wave arrives mid-decode -> overlap
            futs += [eng.submit(GenRequest(
                prompt_ids=[2, 5 + i, 9], max_new_tokens=16))
                for i in range(4)]
            outs = [f.result(120).token_ids for f in futs]
        finally:
            eng.stop()
    finally:
        os.environ.pop("CALF_PREFILL_OVERLAP", None)
    assert [len(o) for o in outs] == [24] * 4 + [16] * 4


def test_on_tokens_batched_delivery_matches_result():
    """on_tokens delivers every generated token exactly once, in order,
    batched per host visit (first chunk = the prefill's first token)."""
    eng = make_engine()
    try:
        chunks: list[list[int]] = []
        r = eng.generate(GenRequest(prompt_ids=[1, 5, 9], max_new_tokens=11,
                                    on_tokens=chunks.append), timeout=120)
        flat = [t for c in chunks for t in c]
        assert flat == r.token_ids
        assert len(r.token_ids) == 11
        assert chunks[0] == r.token_ids[:1]  # prefill's first token alone
        assert all(c for c in chunks)
    finally:
        eng.stop()


def test_on_tokens_suppresses_per_token_callback():
    eng = make_engine()
    try:
        per_token: list[int] = []
        chunks: list[list[int]] = []
        eng.generate(GenRequest(prompt_ids=[1, 2], max_new_tokens=4,
                                on_token=per_token.append,
                                on_tokens=chunks.append), timeout=120)
        assert per_token == []  # batched delivery wins
        assert sum(len(c) for c in chunks) == 4
    finally:
        eng.stop()


def test_seeded_sampling_reproducible_across_engines():
    """temperature>0 with a fixed seed: two engine instances produce the
    same tokens (counter-based Gumbel sampling, no hidden RNG state)."""
    def run() -> list[int]:
        eng = make_engine()
        try:
            return eng.generate(GenRequest(prompt_ids=[1, 4, 7],
                                           max_new_tokens=8, temperature=0.8,
                                           seed=42), timeout=120).token_ids
        finally:
            eng.stop()

    assert run() == run()


def test_large_batch_rounding_to_new_graph_sizes():
    """max_batch rounds up to the new 384/512/1024 graph sizes and the
    pad table never exceeds max_batch."""
    eng = make_engine(max_batch=300)
    try:
        assert eng.max_batch == 384
        assert eng._pad_batch(3) == 4
        assert eng._pad_batch(257) == 384
        assert eng._pad_batch(384) == 384
        r = eng.generate(GenRequest(prompt_ids=[1, 2, 3], max_new_tokens=3),
                         timeout=120)
        assert len(r.token_ids) == 3
    finally:
        eng.stop()


def test_context_cap_sequence_batched_with_longer_budget():
    """a request whose prompt+max_new reaches
    max_model_len, batched with a longer-budget request, blind-runs past the
    cap — the page table and RoPE table must carry the overshoot headroom
    instead of raising (or silently reading OOB RoPE rows on GPU)."""
    eng = make_engine(max_model_len=64)
    try:
        futs = [
            # prompt 8 + max_new 56 == max_model_len 64: at the cap
            eng.submit(GenRequest(prompt_ids=list(range(1, 9)),
                                  max_new_tokens=56)),
            # longer budget drives the run length past the first one's end
            eng.submit(GenRequest(prompt_ids=[1, 2], max_new_tokens=61)),
        ]
        r0 = futs[0].result(timeout=120)
        r1 = futs[1].result(timeout=120)
        assert len(r0.token_ids) == 56
        assert len(r1.token_ids) == 61
    finally:
        eng.stop()


def test_per_request_seed_independent_of_batch_composition():
    """same request seed => same tokens at temperature>0,
    whether the request runs alone or batched with others."""
    def run(extra: int) -> list[int]:
        eng = make_engine()
        try:
            futs = [eng.submit(GenRequest(prompt_ids=[1, 4, 7],
                                          max_new_tokens=8, temperature=0.8,
                                          seed=42))]
            for i in range(extra):
                futs.append(eng.submit(GenRequest(
                    prompt_ids=[9, 9, 9], max_new_tokens=8, temperature=0.9,
                    seed=1000 + i)))
            return futs[0].result(timeout=120).token_ids
        finally:
            eng.stop()

    alone = run(0)
    assert alone == run(2)


def test_eos_id_comes_from_model_config():
    from calfkit_amd.models.config import get_config

    cfg = get_config("tiny", eos_id=7)
    from calfkit_amd.models.llama import LlamaShard
    eng = InferenceEngine(LlamaShard(cfg, device="cpu"), max_batch=4,
                          use_graphs=False, max_model_len=128)
    try:
        assert eng.cfg.eos_id == 7
    finally:
        eng.stop()


def _run_tool_grammar(device="cpu", use_graphs=False):
    """Engine-level: a request with a tool plan emits a grammar-valid
    TCALL <name> <args x 4> TEND span with the name sampled among the
    advertised ids."""
    from calfkit_amd import ops
    from calfkit_amd.models.config import TINY

    model = LlamaShard(TINY, device=device)
    eng = InferenceEngine(model, max_batch=4, use_graphs=use_graphs,
                          max_model_len=256)
    try:
        names = [100, 200, 300]
        r = eng.generate(GenRequest(
            prompt_ids=[1, 20, 30, 40], max_new_tokens=24, temperature=0.7,
            seed=7, ignore_eos=False, tool_names=names, tool_trigger=2,
            tool_ncalls=2), timeout=300)
        toks = r.token_ids
        calls = []
        i = 0
        while i < len(toks):
            if toks[i] == ops.G_TCALL:
                name = toks[i + 1]
                j = i + 2
                args = []
                while j < len(toks) and toks[j] != ops.G_TEND:
                    args.append(toks[j])
                    j += 1
                calls.append((name, args))
                i = j + 1
            else:
                i += 1
        assert len(calls) == 2, toks
        for name, args in calls:
            assert name in names
            assert len(args) == ops.G_ARG_TOKS
            assert all(a >= 16 for a in args)  # no specials in args
        # the turn ends on EOS right after the final call
        assert toks[-1] == 2 and r.finish_reason == "stop"
    finally:
        eng.stop()


def test_tool_grammar_cpu():
    _run_tool_grammar()


@pytest.mark.gpu
def test_tool_grammar_gpu_graphs():
    """The same grammar inside the hipGraph-captured multi-step decode."""
    _run_tool_grammar(device="cuda", use_graphs=True)


def test_tool_call_parser_edges():
    """Parser robustness: truncated span (budget ran out mid-call), unknown
    name (dropped), interleaved text segments."""
    from calfkit_amd.engine.local import _parse_output
    from calfkit_amd.engine.tokenizer import SyntheticTokenizer
    from calfkit_amd.messages import TextPart, ToolCallPart
    from calfkit_amd.schemas import ToolSchema

    tok = SyntheticTokenizer(512)
    schema = ToolSchema(name="lookup", parameters={
        "type": "object", "properties": {"q": {"type": "string"},
                                         "n": {"type": "integer"}}})
    nid = tok.word_id("lookup")
    # text, full call, text, truncated call (no TEND)
    ids = [30, 31, tok.TCALL, nid, 40, 41, 42, 43, tok.TEND, 50,
           tok.TCALL, nid, 60, 61]
    parts, calls = _parse_output(ids, (schema,), tok)
    assert len(calls) == 2
    assert all(isinstance(c, ToolCallPart) for c in calls)
    assert set(calls[0].arguments) == {"q", "n"}
    assert isinstance(calls[0].arguments["n"], int)
    texts = [p for p in parts if isinstance(p, TextPart)]
    assert texts and texts[0].text  # the leading text survived
    # unknown name: span dropped entirely
    parts2, calls2 = _parse_output([tok.TCALL, 499, 1, 2, tok.TEND, 70],
                                   (schema,), tok)
    assert not calls2


def test_engine_fuzz_random_request_mix_is_deterministic():
    """Scheduler soak: a randomized mix of requests (lengths, budgets,
    temperatures, seeds, tool plans, eos behavior) all complete with the
    right token counts, and the same mix replays identically (the engine
    is deterministic given request seeds — no hidden RNG, no
    batch-composition leakage)."""
    import random

    def run_mix() -> list[list[int]]:
        rng = random.Random(1234)
        eng = make_engine(max_batch=8, max_model_len=128)
        futs = []
        try:
            for i in range(24):
                plan = rng.random() < 0.3
                req = GenRequest(
                    prompt_ids=[1] + [rng.randrange(16, 500)
                                      for _ in range(rng.randrange(2, 20))],
                    max_new_tokens=rng.randrange(1, 12),
                    temperature=rng.choice([0.0, 0.7, 1.2]),
                    seed=100 + i,
                    ignore_eos=not plan,
                    tool_names=[100, 200] if plan else None,
                    tool_trigger=rng.randrange(0, 3),
                    tool_ncalls=1)
                futs.append((req, eng.submit(req)))
            outs = []
            for req, f in futs:
                r = f.result(180)
                assert len(r.token_ids) <= req.max_new_tokens
                if req.ignore_eos:
                    assert len(r.token_ids) == req.max_new_tokens
                outs.append(r.token_ids)
            return outs
        finally:
            eng.stop()

    assert run_mix() == run_mix()
