"""The CDNA4 kernel surface.

Dispatch contract:
- On a GPU box the in-tree HIP extension (``calfkit_amd/ops/_C*.so``) MUST be
  present — a missing extension raises immediately; there is no silent eager
  fallback on GPU (the native path is the one that runs).
- On CPU the pure-torch reference implementations below serve tests and
  development; every gpu-lane numerics test compares the HIP kernel against
  the fp32 torch reference of the same op.
"""

from __future__ import annotations

from typing import Any

import torch

__all__ = [
    "have_extension",
    "rmsnorm",
    "fused_add_rmsnorm",
    "silu_mul",
    "rope_qk",
    "rope_qkv_write",
    "kv_write",
    "attn_decode",
    "attn_prefill",
    "attn_prefill_varlen",
    "build_prefill_blocks",
    "sample",
    "skinny_gemm",
    "skinny_gemm_slabs",
    "tile_gemm",
    "tile_gemm_slabs",
    "tile_plan",
    "small_plan",
    "linear_indexed",
    "grammar_bias",
    "grammar_transition",
    "rope_qkv_write_slab",
    "fused_add_rmsnorm_slab",
    "skinny_gemm_silu",
    "moe_gateup_silu",
    "moe_down",
    "moe_buckets",
    "moe_route",
    "moe_combine",
    "choose_splitk",
    "build_rope_table",
]

_K: Any = None
_IMPORT_ERROR: Exception | None = None
try:
    import importlib

    _K = importlib.import_module("._C", __package__)
except ImportError as e:  # extension not built (CPU dev box is fine)
    _IMPORT_ERROR = e


def have_extension() -> bool:
    return _K is not None


def _native(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if _K is None:
        raise RuntimeError(
            "calfkit_amd.ops: tensor is on GPU but the gfx950 extension is not "
            "built — run `python setup.py build_ext --inplace` "
            f"(import error: {_IMPORT_ERROR})")
    return True


# ------------------------------------------------------------------ rmsnorm

def rmsnorm(out: torch.Tensor, inp: torch.Tensor, weight: torch.Tensor,
            eps: float = 1e-5) -> None:
    if _native(inp):
        _K.rmsnorm(out, inp, weight, eps)
        return
    x = inp.float()
    n = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + eps)
    out.copy_((n * weight.float()).to(out.dtype))


def fused_add_rmsnorm(hidden: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float = 1e-5) -> None:
    """residual += hidden; hidden = rmsnorm(residual). Both in place."""
    if _native(hidden):
        _K.fused_add_rmsnorm(hidden, residual, weight, eps)
        return
    s = (residual.float() + hidden.float())
    residual.copy_(s.to(residual.dtype))
    s = residual.float()  # norm over the bf16-rounded sum, like the kernel
    n = s * torch.rsqrt(s.pow(2).mean(-1, keepdim=True) + eps)
    hidden.copy_((n * weight.float()).to(hidden.dtype))


# ----------------------------------------------------------------- silu_mul

def silu_mul(out: torch.Tensor, gateup: torch.Tensor) -> None:
    if _native(gateup):
        _K.silu_mul(out, gateup)
        return
    f = out.shape[-1]
    gate = gateup[..., :f].float()
    up = gateup[..., f:].float()
    out.copy_((torch.nn.functional.silu(gate) * up).to(out.dtype))


# --------------------------------------------------------------------- rope

def build_rope_table(max_pos: int, dh: int, theta: float = 500000.0,
                     device: str | torch.device = "cpu") -> torch.Tensor:
    """[max_pos, dh] f32 table laid out [cos(dh/2) | sin(dh/2)] per position."""
    inv = 1.0 / (theta ** (torch.arange(0, dh, 2, dtype=torch.float64) / dh))
    pos = torch.arange(max_pos, dtype=torch.float64)
    ang = torch.outer(pos, inv)  # [max_pos, dh/2]
    return torch.cat([ang.cos(), ang.sin()], dim=-1).float().to(device)


def rope_qk(q: torch.Tensor, k: torch.Tensor, cos_sin: torch.Tensor,
            positions: torch.Tensor) -> None:
    """In-place neox rotate-half RoPE. q: [T,Hq,dh], k: [T,Hkv,dh]."""
    if _native(q):
        _K.rope_qk(q, k, cos_sin, positions)
        return
    dh = q.shape[-1]
    half = dh // 2
    cs = cos_sin[positions.long()]  # [T, dh]
    cos = cs[:, :half].unsqueeze(1)
    sin = cs[:, half:].unsqueeze(1)
    for t in (q, k):
        x0 = t[..., :half].float()
        x1 = t[..., half:].float()
        t[..., :half] = (x0 * cos - x1 * sin).to(t.dtype)
        t[..., half:] = (x1 * cos + x0 * sin).to(t.dtype)


def rope_qkv_write(qkv: torch.Tensor, kcache: torch.Tensor,
                   vcache: torch.Tensor, cos_sin: torch.Tensor,
                   positions: torch.Tensor, slots: torch.Tensor,
                   hq: int, hkv: int) -> None:
    """Fused QKV-projection epilogue: rotate Q/K heads in place inside the
    packed [T, (hq+2*hkv)*dh] GEMM output and scatter K/V into the paged
    cache (one launch instead of 3 copies + rope + kv_write)."""
    if _native(qkv):
        _K.rope_qkv_write(qkv, kcache, vcache, cos_sin, positions, slots,
                          hq, hkv)
        return
    dh = kcache.shape[-1]
    T = qkv.shape[0]
    q = qkv[:, :hq * dh].view(T, hq, dh)
    k = qkv[:, hq * dh:(hq + hkv) * dh].view(T, hkv, dh)
    v = qkv[:, (hq + hkv) * dh:].view(T, hkv, dh)
    rope_qk(q, k, cos_sin, positions)
    kv_write(kcache, vcache, k, v, slots)


# ----------------------------------------------------------------- kv cache

def kv_write(kcache: torch.Tensor, vcache: torch.Tensor, k: torch.Tensor,
             v: torch.Tensor, slots: torch.Tensor) -> None:
    """Scatter [T,Hkv,dh] rows into [P,Hkv,PAGE,dh] caches at flat slots."""
    if _native(k):
        _K.kv_write(kcache, vcache, k, v, slots)
        return
    page = kcache.shape[2]
    for t in range(k.shape[0]):
        s = int(slots[t])
        if s < 0:
            continue
        p, off = divmod(s, page)
        kcache[p, :, off] = k[t]
        vcache[p, :, off] = v[t]


# ---------------------------------------------------------------- attention

def attn_decode(out: torch.Tensor, q: torch.Tensor, kcache: torch.Tensor,
                vcache: torch.Tensor, page_table: torch.Tensor,
                seq_lens: torch.Tensor, scale: float) -> None:
    """Paged GQA decode attention. q/out: [B,Hq,dh]."""
    if _native(q):
        _K.attn_decode(out, q, kcache, vcache, page_table, seq_lens, scale)
        return
    B, hq, dh = q.shape
    hkv, page = kcache.shape[1], kcache.shape[2]
    G = hq // hkv
    for b in range(B):
        n = int(seq_lens[b])
        npages = (n + page - 1) // page
        pages = page_table[b, :npages].long()
        # gather [n, hkv, dh]
        kk = kcache[pages].permute(0, 2, 1, 3).reshape(-1, hkv, dh)[:n].float()
        vv = vcache[pages].permute(0, 2, 1, 3).reshape(-1, hkv, dh)[:n].float()
        for h in range(hq):
            kvh = h // G
            s = (kk[:, kvh] @ (q[b, h].float() * scale)) # [n]
            p = torch.softmax(s, dim=-1)
            out[b, h] = (p @ vv[:, kvh]).to(out.dtype)


def attn_prefill(out: torch.Tensor, q: torch.Tensor, k: torch.Tensor,
                 v: torch.Tensor, scale: float, past_len: int = 0) -> None:
    """Causal GQA prefill. q/out: [S,Hq,dh]; k/v: [past+S,Hkv,dh]."""
    if _native(q):
        _K.attn_prefill(out, q, k, v, scale, past_len)
        return
    S, hq, dh = q.shape
    hkv = k.shape[1]
    G = hq // hkv
    T = k.shape[0]
    mask = torch.arange(T, device=q.device)[None, :] > \
        (past_len + torch.arange(S, device=q.device))[:, None]
    for h in range(hq):
        kvh = h // G
        s = (q[:, h].float() @ k[:, kvh].float().T) * scale  # [S, T]
        s = s.masked_fill(mask, float("-inf"))
        p = torch.softmax(s, dim=-1)
        out[:, h] = (p @ v[:, kvh].float()).to(out.dtype)


def build_prefill_blocks(seq_lens: list[int], qblk: int = 64) -> torch.Tensor:
    """Host-side block map for varlen prefill: [(seq, qblock), ...] i32."""
    rows = []
    for s, n in enumerate(seq_lens):
        for qb in range((n + qblk - 1) // qblk):
            rows.append((s, qb))
    return torch.tensor(rows, dtype=torch.int32)


def attn_prefill_varlen(out: torch.Tensor, q: torch.Tensor, k: torch.Tensor,
                        v: torch.Tensor, block_map: torch.Tensor,
                        q_starts: torch.Tensor, kv_starts: torch.Tensor,
                        past_lens: torch.Tensor, scale: float) -> None:
    """Packed multi-sequence causal GQA prefill.

    q/out: [Tq, Hq, dh] packed; k/v: [Tkv, Hkv, dh] packed; sequence i's new
    rows are q[q_starts[i]:q_starts[i+1]] and its keys
    k[kv_starts[i]:kv_starts[i+1]] (= past_lens[i] context + new rows).
    """
    if _native(q):
        _K.attn_prefill_varlen(out, q, k, v, block_map, q_starts, kv_starts,
                               past_lens, scale)
        return
    B = past_lens.numel()
    for i in range(B):
        qs, qe = int(q_starts[i]), int(q_starts[i + 1])
        ks, ke = int(kv_starts[i]), int(kv_starts[i + 1])
        attn_prefill(out[qs:qe], q[qs:qe], k[ks:ke], v[ks:ke], scale,
                     int(past_lens[i]))


# ----------------------------------------------------------------- sampling

_SAMPLE_SCRATCH: dict[tuple, torch.Tensor] = {}


def sample(out: torch.Tensor, logits: torch.Tensor, temperature: torch.Tensor,
           seed_step: torch.Tensor, topk: torch.Tensor | None = None,
           cand: torch.Tensor | None = None,
           row_seeds: torch.Tensor | None = None) -> None:
    """Gumbel-max sampling; temperature 0 = greedy. out: int32 [B].
    Accepts fp32 or bf16 logits (bf16 avoids materializing [B,V] fp32).
    ``cand`` overrides the shared two-stage scratch — callers running on a
    side stream (overlapped prefill) MUST pass their own so a concurrent
    decode graph replaying the same batch size cannot race on it.
    ``row_seeds`` [B] int64 keys the RNG per row (request seed x token
    position, built by the engine) so a request's draws are independent of
    batch composition; without it the legacy (seed, step, row) key applies."""
    if _native(logits):
        if cand is None:
            key = (logits.device.index, int(out.shape[0]))
            cand = _SAMPLE_SCRATCH.get(key)
            if cand is None or cand.numel() < out.shape[0] * 64 * 2:
                cand = torch.empty(out.shape[0] * 64 * 2, dtype=torch.float32,
                                   device=logits.device)
                _SAMPLE_SCRATCH[key] = cand
        else:
            assert cand.numel() >= out.shape[0] * 64 * 2
        _K.sample(out, logits, temperature,
                  seed_step, topk if topk is not None else torch.Tensor(),
                  cand,
                  row_seeds if row_seeds is not None else torch.Tensor())
        return
    B, V = logits.shape
    lg = logits.float().clone()
    if topk is not None:
        for b in range(B):
            kk = int(topk[b])
            if 0 < kk < V:
                thr = torch.topk(lg[b], kk).values[-1]
                lg[b][lg[b] < thr] = float("-inf")
    for b in range(B):
        t = float(temperature[b])
        if t <= 0:
            out[b] = int(lg[b].argmax())
        else:
            # counter-based like the HIP kernel: the per-row key (or the
            # legacy (seed, step, row) triple) fully determines the draw
            if row_seeds is not None:
                key = int(row_seeds[b]) & 0x7FFFFFFF
            else:
                key = (int(seed_step[0]) * 1_000_003
                       + int(seed_step[1]) * 7919 + b) & 0x7FFFFFFF
            gen = torch.Generator().manual_seed(key)
            u = torch.rand(V, generator=gen).clamp_min(1e-12)
            g = -torch.log(-torch.log(u))
            out[b] = int((lg[b] / t + g).argmax())


# ----------------------------------------------------- tool-call grammar

#: grammar token ids (SyntheticTokenizer reserved range) + shape constants;
#: the HIP kernel (sampling.hip grammar_bias_kernel) hard-codes the same
G_TCALL, G_TEND, G_EOS = 5, 6, 2
G_ARG_TOKS = 4
G_MAX_TOOLS = 8


def grammar_transition(state: int, prev_token: int) -> int:
    """One host-side mirror step of the device grammar state machine."""
    s, calls = state & 0xFF, state >> 8
    if s == 1:
        s = 2
    elif s >= 2:
        if prev_token == G_TEND:
            s, calls = 0, calls + 1
        else:
            s += 1
    elif s == 0 and prev_token == G_TCALL:
        s = 1
    return s | (calls << 8)


def grammar_bias(logits: torch.Tensor, prev: torch.Tensor,
                 gstate: torch.Tensor, trigger: torch.Tensor,
                 names: torch.Tensor, pos: torch.Tensor,
                 ncalls: torch.Tensor) -> None:
    """Constrained-sampling bias steering rows with a tool plan through the
    tool-call grammar.  In-place on logits; transitions
    gstate on the previous token.  Rows with trigger < 0 are untouched."""
    if _native(logits):
        _K.grammar_bias(logits, prev, gstate, trigger, names, pos, ncalls)
        return
    BIG = 65536.0
    B = logits.shape[0]
    for b in range(B):
        if int(trigger[b]) < 0:
            continue
        st = grammar_transition(int(gstate[b]), int(prev[b]))
        gstate[b] = st
        s, calls = st & 0xFF, st >> 8
        row = logits[b]
        if s == 0:
            if calls < int(ncalls[b]) and int(pos[b]) >= int(trigger[b]):
                row[G_TCALL] += BIG
            elif 0 < calls and calls >= int(ncalls[b]):
                row[G_EOS] += BIG
            else:
                row[G_TCALL] -= BIG
                row[G_EOS] -= BIG
        elif s == 1:
            for t in names[b].tolist():
                if t >= 0:
                    row[t] += BIG
        elif s - 2 >= G_ARG_TOKS:
            row[G_TEND] += BIG
        else:
            row[:16] -= BIG


# ----------------------------------------------------- searched-algo GEMM

_ALGO_TABLE: "dict | None" = None
_ALGO_WS: dict = {}


def _algo_table() -> dict:
    global _ALGO_TABLE
    if _ALGO_TABLE is None:
        import json
        import os

        path = os.path.join(os.path.dirname(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__)))), "profiles", "gemm_algo_table.json")
        table = {}
        try:
            with open(path) as f:
                for v in json.load(f).values():
                    if v.get("numerics_ok") and v.get("speedup", 0) > 1.0:
                        table[(v["M"], v["N"], v["K"])] = v["algo_index"]
        except Exception:
            pass
        if os.environ.get("CALF_GEMM_INDEXED", "1") != "1":
            table = {}
        _ALGO_TABLE = table
    return _ALGO_TABLE


def linear_indexed(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """hipBLASLt with the SEARCHED algorithm index for known serving
    shapes (profiles/gemm_algo_table.json: best-vs-heuristic 1.03-1.11x);
    the heuristic path otherwise.  Wired on the DECODE path only — the
    per-call setup host cost is paid once at hipGraph capture and never
    at replay."""
    if _K is None or not x.is_cuda:
        return torch.nn.functional.linear(x, w)
    key = (x.shape[0], w.shape[0], w.shape[1])
    idx = _algo_table().get(key)
    if idx is None:
        return torch.nn.functional.linear(x, w)
    out = torch.empty(x.shape[0], w.shape[0], dtype=x.dtype, device=x.device)
    ws = _ALGO_WS.get(x.device.index)
    if ws is None:
        ws = torch.empty(64 << 20, dtype=torch.uint8, device=x.device)
        _ALGO_WS[x.device.index] = ws
    try:
        _K.hipblaslt_gemm_indexed(out, x, w, idx, ws)
    except RuntimeError:
        # stale index (library version drift): drop it and fall back
        _algo_table().pop(key, None)
        return torch.nn.functional.linear(x, w)
    return out


# -------------------------------------------------------------- skinny GEMM

def choose_splitk(N: int, K: int, m: int = 32) -> int:
    """Split-K so (N/64)*splitk lands near 1-2x the 256 CUs, K dividing.

    The M>64 path doesn't wave-split K, so its divisibility needs only
    K % (64*sk); its block target is ~256 (the guide's 1/2-1x CU rule)."""
    target = 384 if m <= 64 else 256
    best = 1
    for sk in (1, 2, 4, 8):
        if m <= 64:
            if K % (64 * sk * 4) != 0 and K % (32 * sk * 4) != 0:
                continue
        else:
            if K % (64 * sk) != 0:
                continue
        if (N // 64) * sk <= target:
            best = sk
    return best


def small_plan(M: int, N: int, K: int) -> bool:
    """True when the 64x128 small-tile kernel is the measured winner.

    r02 tile_micro: shallow-K wide-N shapes (the TP-sharded o/down
    projections, K = d/tp <= 1792, i.e. <= 28 K-iterations) land 256+
    natural 64x128 tiles, where the kernel beat hipBLASLt on every
    measured shape — o_8b_tp8 2.77x (7.0 vs 19.4 us), o_70b_tp8 1.51x,
    down_8b_tp8 1.32x (3-buf ring).  The win holds down to M=192 / 96
    blocks (o_8b_tp8 M=192: 6.3 vs 18.0 us — the library has an ~18 us
    floor on this class at every M), which covers the TP-overlap path's
    half-chunks of the 384/512 graph batches.  From K=2048 up the
    library edges ahead (o_70b_tp4 0.97x, down70b_tp8 0.85x, qkv_8b
    0.90x even with the ring) and those shapes stay on the big
    kernel / library.
    """
    if N % 128 or K % 64 or K > 1792 or N < 4096:
        return False
    return (N // 128) * ((M + 63) // 64) >= 96


def tile_plan(M: int, N: int, K: int) -> "tuple[int, int] | None":
    """(BN, splitk) for the tiled serving GEMM, or None if unroutable.

    Rule distilled from the r02 variant sweeps (profiles/r02_tile_gemm.md):
    wide-N shapes (gateup, lm_head) take one full wave of 256-wide tiles
    with maximum W reuse; narrow shapes split K on 128-wide tiles until
    ~3/4 of the 256 CUs fill, but never below 1024 K per block (the
    pipeline needs depth and the slab reduce isn't free).
    """
    import os

    env_bn = int(os.environ.get("CALF_TILE_BN", "0"))
    env_sk = int(os.environ.get("CALF_TILE_SK", "0"))
    if env_bn and env_sk:
        return env_bn, env_sk
    m_tiles = (M + 255) // 256
    if N % 256 == 0 and K % 64 == 0 and (N // 256) * m_tiles >= 192:
        return 256, 1
    best = None
    for bn in (128, 256):
        if N % bn:
            continue
        for sk in (1, 2, 4, 8):
            if K % sk or (K // sk) % 64 or K // sk < 1024:
                continue
            best = (bn, sk)
            if (N // bn) * m_tiles * sk >= 192:
                return best
        if best is not None:
            return best
    return best


def tile_gemm(out: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
              work: torch.Tensor, bn: int, splitk: int) -> None:
    """out[M,N] = x[M,K] @ w[N,K]^T for serving batches (M = 65..512+)."""
    if _native(x):
        _K.tile_gemm(out, x, w, work, bn, splitk)
        return
    out.copy_((x.float() @ w.float().T).to(out.dtype))


def tile_gemm_small(out: torch.Tensor, x: torch.Tensor,
                    w: torch.Tensor) -> None:
    """out[M,N] = x[M,K] @ w[N,K]^T via 64x128 tiles (natural fill, no
    split-K) — the small-tile variant for the M=384/512 projections."""
    if _native(x):
        _K.tile_gemm_small(out, x, w)
        return
    out.copy_((x.float() @ w.float().T).to(out.dtype))


def tile_gemm_slabs(x: torch.Tensor, w: torch.Tensor, work: torch.Tensor,
                    bn: int, splitk: int) -> int:
    """Slab-only tiled GEMM; returns Mp. A fused consumer reduces."""
    Mp = ((x.shape[0] + 255) // 256) * 256
    if _native(x):
        _K.tile_gemm_slabs(x, w, work, bn, splitk)
        return Mp
    N = w.shape[0]
    y = x.float() @ w.float().T
    wk = work[: splitk * Mp * N].view(splitk, Mp, N)
    wk.zero_()
    wk[0, : x.shape[0]] = y
    return Mp


def skinny_gemm(out: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                work: torch.Tensor, splitk: int) -> None:
    """out[M,N] = x[M,K] @ w[N,K]^T for decode batches (M <= 64)."""
    if _native(x):
        _K.skinny_gemm(out, x, w, work, splitk)
        return
    out.copy_((x.float() @ w.float().T).to(out.dtype))


def skinny_gemm_silu(out: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                     work: torch.Tensor, splitk: int) -> None:
    """out[M,F] = silu(x@Wg^T) * (x@Wu^T); w = [gate rows | up rows] [2F,K]."""
    if _native(x):
        _K.skinny_gemm_silu(out, x, w, work, splitk)
        return
    F = out.shape[-1]
    y = x.float() @ w.float().T
    g, u = y[:, :F], y[:, F:]
    out.copy_((torch.nn.functional.silu(g) * u).to(out.dtype))


# ------------------------------------------------------------- grouped MoE

def moe_buckets(experts: torch.Tensor, weights: torch.Tensor,
                num_experts: int) -> tuple[torch.Tensor, torch.Tensor,
                                           torch.Tensor]:
    """Device-side expert bucketing for the grouped MoE kernels.

    experts/weights: [T, topk] router picks.  Returns (tok, off, scale):
    tok[r] i32 source token for bucket row r (rows sorted by expert),
    off[E+1] i32 bucket offsets, scale[r] f32 router weight.  Built from
    sort/cumsum/gather only — no host sync, hipGraph-capturable (unlike
    bincount().tolist() in the host-grouped fallback).
    """
    T, k = experts.shape
    flat_expert = experts.reshape(-1)
    order = torch.argsort(flat_expert, stable=True)
    flat_token = torch.arange(T, device=experts.device,
                              dtype=torch.int32).repeat_interleave(k)
    counts = torch.zeros(num_experts, device=experts.device,
                         dtype=torch.int32)
    counts.scatter_add_(0, flat_expert.long(),
                        torch.ones_like(flat_expert, dtype=torch.int32))
    off = torch.zeros(num_experts + 1, device=experts.device,
                      dtype=torch.int32)
    torch.cumsum(counts, 0, out=off[1:])
    tok = flat_token[order].contiguous()
    scale = weights.reshape(-1)[order].float().contiguous()
    return tok, off, scale


def moe_route(logits: torch.Tensor, num_experts: int,
              topk: int) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor,
                                  torch.Tensor]:
    """Fused router: softmax -> top-K -> renormalize -> stable expert
    buckets in one launch (T <= 256).  Returns (tok, off, scale, inv) —
    the ``moe_buckets`` contract plus the inverse map inv[t*K+j] = bucket
    row, which ``moe_combine`` uses to fold the k expert outputs per
    token deterministically (profiles/r02_moe_kernelstats.md)."""
    T = logits.shape[0]
    if _native(logits):
        dev = logits.device
        tok = torch.empty(T * topk, dtype=torch.int32, device=dev)
        off = torch.empty(num_experts + 1, dtype=torch.int32, device=dev)
        scale = torch.empty(T * topk, dtype=torch.float32, device=dev)
        inv = torch.empty(T * topk, dtype=torch.int32, device=dev)
        _K.moe_route(tok, off, scale, inv, logits.contiguous(), topk)
        return tok, off, scale, inv
    weights, experts = torch.topk(torch.softmax(logits.float(), -1), topk,
                                  dim=-1)
    weights = weights / weights.sum(-1, keepdim=True)
    tok, off, scale = moe_buckets(experts, weights, num_experts)
    flat_expert = experts.reshape(-1)
    order = torch.argsort(flat_expert, stable=True)
    inv = torch.empty_like(order, dtype=torch.int32)
    inv[order] = torch.arange(order.numel(), dtype=torch.int32)
    return tok, off, scale, inv


def moe_combine(out: torch.Tensor, y: torch.Tensor, inv: torch.Tensor,
                topk: int) -> None:
    """out[t] = sum_j y[inv[t*K+j]] with fp32 accumulation — replaces
    zeros + index_add_ + two dtype converts (4 launches -> 1)."""
    if _native(y):
        _K.moe_combine(out, y, inv, topk)
        return
    T, d = out.shape
    idx = inv.long().view(T, topk)
    out.copy_(y.float()[idx].sum(1).to(out.dtype))


def moe_gateup_silu(act: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                    tok: torch.Tensor, off: torch.Tensor,
                    work: torch.Tensor) -> None:
    """Grouped gate-up: act[r,:F] = silu(g)*u of x[tok[r]] @ w[e(r)]^T.

    w: [E, 2F, K] stacked [gate rows | up rows]; rows r in [off[e], off[e+1])
    belong to expert e; work: fp32 scratch >= R*2F.
    """
    if _native(x):
        _K.moe_gateup_silu(act, x, w, tok, off, work)
        return
    F = act.shape[-1]
    offs = off.tolist()
    for e in range(w.shape[0]):
        lo, hi = offs[e], offs[e + 1]
        if lo == hi:
            continue
        xs = x[tok[lo:hi].long()].float()
        y = xs @ w[e].float().T
        g, u = y[:, :F], y[:, F:]
        act[lo:hi] = (torch.nn.functional.silu(g) * u).to(act.dtype)


def moe_down(y: torch.Tensor, act: torch.Tensor, w: torch.Tensor,
             off: torch.Tensor, scale: torch.Tensor) -> None:
    """Grouped down: y[r] = scale[r] * (act[r] @ w[e(r)]^T); w: [E, d, F]."""
    if _native(act):
        _K.moe_down(y, act, w, off, scale)
        return
    offs = off.tolist()
    for e in range(w.shape[0]):
        lo, hi = offs[e], offs[e + 1]
        if lo == hi:
            continue
        out = act[lo:hi].float() @ w[e].float().T
        y[lo:hi] = (out * scale[lo:hi, None].float()).to(y.dtype)


def pad_m(M: int) -> int:
    for p in (16, 32, 64, 128, 256):
        if M <= p:
            return p
    raise ValueError(f"decode batch {M} > 256")


def skinny_gemm_slabs(x: torch.Tensor, w: torch.Tensor, work: torch.Tensor,
                      splitk: int) -> int:
    """Run only the slab stage; returns Mp. Consumer fuses the reduce."""
    Mp = pad_m(x.shape[0])
    if _native(x):
        _K.skinny_gemm_slabs(x, w, work, splitk)
        return Mp
    # CPU reference: one fp32 slab with the full product, rest zero
    N = w.shape[0]
    y = x.float() @ w.float().T
    wk = work[: splitk * Mp * N].view(splitk, Mp, N)
    wk.zero_()
    wk[0, : x.shape[0]] = y
    return Mp


def rope_qkv_write_slab(qkv: torch.Tensor, work: torch.Tensor,
                        kcache: torch.Tensor, vcache: torch.Tensor,
                        cos_sin: torch.Tensor, positions: torch.Tensor,
                        slots: torch.Tensor, hq: int, hkv: int,
                        Mp: int, splitk: int) -> None:
    """qkv = sum(slabs); then RoPE + paged cache write (one launch)."""
    if _native(qkv):
        _K.rope_qkv_write_slab(qkv, work, kcache, vcache, cos_sin, positions,
                               slots, hq, hkv, Mp, splitk)
        return
    T, W = qkv.shape
    summed = work[: splitk * Mp * W].view(splitk, Mp, W)[:, :T].sum(0)
    qkv.copy_(summed.to(qkv.dtype))
    rope_qkv_write(qkv, kcache, vcache, cos_sin, positions, slots, hq, hkv)


def fused_add_rmsnorm_slab(hidden: torch.Tensor, residual: torch.Tensor,
                           work: torch.Tensor, weight: torch.Tensor,
                           eps: float, Mp: int, splitk: int) -> None:
    """residual += sum(slabs); hidden = rmsnorm(residual)."""
    if _native(hidden):
        _K.fused_add_rmsnorm_slab(hidden, residual, work, weight, eps, Mp,
                                  splitk)
        return
    T, d = hidden.shape
    summed = work[: splitk * Mp * d].view(splitk, Mp, d)[:, :T].sum(0)
    hidden.copy_(summed.to(hidden.dtype))
    fused_add_rmsnorm(hidden, residual, weight, eps)
