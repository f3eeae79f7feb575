"""Unit: KV prefix caching — matching, sharing, eviction, equivalence."""

import torch

from calfkit_amd.engine.engine import GenRequest, InferenceEngine
from calfkit_amd.engine.kvcache import KVCache, PAGE_SIZE
from calfkit_amd.models.config import TINY
from calfkit_amd.models.llama import LlamaShard

LONG = [1] + [10 + i % 50 for i in range(130)]  # > 2 full pages


def make_engine(**kw):
    model = LlamaShard(TINY, device="cpu")
    kw.setdefault("max_batch", 4)
    kw.setdefault("use_graphs", False)
    kw.setdefault("max_model_len", 512)
    return InferenceEngine(model, **kw)


def test_match_and_register_round_trip():
    kv = KVCache(layers=1, kv_heads=1, head_dim=128, num_pages=16, device="cpu")
    prompt = list(range(200))
    assert kv.match_prefix(1, prompt) == 0  # cold
    kv.slots_for(1, len(prompt))
    kv.register_prefix(1, prompt)
    # a second sequence with the same prompt reuses the full pages
    cached = kv.match_prefix(2, prompt)
    assert cached == (len(prompt) - 1) // PAGE_SIZE * PAGE_SIZE == 192
    s1 = kv._slots[1]
    s2 = kv._slots[2]
    assert s2.pages == s1.pages[:3]  # shared, not copied
    # shared pages survive the owner's release
    kv.release(1)
    assert kv.match_prefix(3, prompt) == 192
    kv.release(2)
    kv.release(3)


def test_partial_prefix_match():
    kv = KVCache(layers=1, kv_heads=1, head_dim=128, num_pages=16, device="cpu")
    a = list(range(200))
    kv.slots_for(1, len(a))
    kv.register_prefix(1, a)
    # same first page, different second page
    b = a[:64] + [999] * 136
    assert kv.match_prefix(2, b) == 64
    kv.release(1)
    kv.release(2)


def test_eviction_reclaims_lru_pages():
    kv = KVCache(layers=1, kv_heads=1, head_dim=128, num_pages=4, device="cpu")
    kv.slots_for(1, 128)  # 2 pages
    kv.register_prefix(1, list(range(128)))
    kv.release(1)  # both pages now cached + evictable
    assert kv.free_pages == 4
    kv.slots_for(2, 256)  # needs all 4 pages -> evicts the cached ones
    assert kv.match_prefix(3, list(range(128)) + [1]) == 0  # index emptied
    kv.release(2)


def test_cached_prefill_matches_uncached_tokens():
    eng = make_engine()
    try:
        r1 = eng.generate(GenRequest(prompt_ids=LONG, max_new_tokens=5),
                          timeout=120)
        r2 = eng.generate(GenRequest(prompt_ids=LONG, max_new_tokens=5),
                          timeout=120)
        assert r1.token_ids == r2.token_ids
        # a diverging suffix still matches only the shared prefix
        forked = LONG[:100] + [333] * 40
        r3 = eng.generate(GenRequest(prompt_ids=forked, max_new_tokens=5),
                          timeout=120)
        assert len(r3.token_ids) == 5
    finally:
        eng.stop()


async def test_multi_turn_agent_thread_reuses_prefix():
    """The agent-serving payoff: turn 2 of a thread re-prefills only the
    delta of the rendered conversation."""
    from calfkit_amd import Agent, Client, Worker
    from calfkit_amd.adapters.memory import (
        InMemoryBus, InMemoryDirectory, InProcessThreadStore)
    from calfkit_amd.engine.local import LocalModelClient
    from calfkit_amd.schemas import ModelSettings

    filler = " ".join(f"w{i}" for i in range(140))  # >2 pages of prompt
    provider = LocalModelClient(TINY, device="cpu", max_batch=4,
                                max_model_len=512, use_graphs=False)
    agent = Agent(name="cached", model_client=provider,
                  system_prompt=filler,
                  model_settings=ModelSettings(max_tokens=4, temperature=0.0))
    bus, store, directory = InMemoryBus(), InProcessThreadStore(), InMemoryDirectory()
    async with Worker(agent, bus=bus, store=store, directory=directory):
        client = await Client.connect(bus, directory)
        await client.execute("cached", "turn one", thread_id="T", timeout=120)
        kv = provider.engine.kv
        index_size = len(kv._index)
        assert index_size >= 2  # the system prompt's pages got registered
        await client.execute("cached", "turn two", thread_id="T", timeout=120)
        # the second turn's prompt (same system prefix) hit the cache
        assert kv.prefix_hit_tokens >= 128


def test_identical_prompts_across_packed_batches():
    """Identical prompts (the synthetic tokenizer can legitimately produce
    them) share pages across packed batches; outputs stay identical.
    Regression for the r19 scratch-overflow crash."""
    eng = make_engine(max_model_len=256)
    try:
        p = [1] + [20 + i % 40 for i in range(140)]
        futs = [eng.submit(GenRequest(prompt_ids=p, max_new_tokens=3))
                for _ in range(3)]
        outs = [f.result(60).token_ids for f in futs]
        assert outs[0] == outs[1] == outs[2]
        futs = [eng.submit(GenRequest(prompt_ids=p, max_new_tokens=3)),
                eng.submit(GenRequest(prompt_ids=[1, 2, 3, 4],
                                      max_new_tokens=3))]
        outs2 = [f.result(60).token_ids for f in futs]
        assert outs2[0] == outs[0]
        assert eng.kv.prefix_hit_tokens >= 128
    finally:
        eng.stop()
