"""Llama-family model (3-8B / 3-70B shapes) on the CDNA4 op set.

Design (MI355X-first, not a port):
- projections run on hipBLASLt through torch.matmul (plain library GEMMs);
  every fused op — residual+RMSNorm, RoPE, paged attention, SiLU·mul,
  sampling — is a hand-written HIP kernel (calfkit_amd.ops);
- tensor parallel: column-parallel QKV/gate-up, row-parallel O/down with one
  RCCL all-reduce per block pair; KV heads shard with the TP group
  (Hkv=8 → TP≤8, one KV head per rank at TP=8);
- weights are random-init bf16 shards created directly on the device —
  there is no full-model materialization and no checkpoint download
  (BASELINE: synthetic data / random-init weights);
- decode path is shape-static per batch size so the whole step can be
  captured in a hipGraph (engine/runner.py).
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch

from .. import ops
from ..parallel import all_reduce_sum
from .config import ModelConfig

__all__ = ["LlamaShard", "LayerWeights", "PrefillPlan"]


@dataclass
class PrefillPlan:
    """Packed varlen prefill description (built once per prefill batch)."""

    scratch: list  # [(k_buf, v_buf)] per layer, packed kv write targets
    block_map: torch.Tensor   # [nblocks, 2] i32 (seq, qblock)
    q_starts: torch.Tensor    # [B+1] i32 prefix of packed q rows
    kv_starts: torch.Tensor   # [B+1] i32 prefix of packed kv rows
    past_lens: torch.Tensor   # [B] i32
    kv_write_start: int       # where this forward's new kv rows land
    kv_total: int             # total kv rows the attention reads
    #: scattered landing spots for the new kv rows (packed multi-seq prefill
    #: with per-seq cached pasts); None = contiguous at kv_write_start
    kv_write_index: "torch.Tensor | None" = None
    #: True when every past_len is 0 and kv rows == q rows: attention reads
    #: K/V straight from the packed qkv GEMM output (no scratch copy at all)
    kv_in_qkv: bool = False


@dataclass
class LayerWeights:
    # nn.Linear layout [out, in]: rows are output features, so the skinny
    # decode GEMM's B-fragments are contiguous 16 B loads (see
    # ops/csrc/skinny_gemm.hip) and hipBLASLt gets its preferred TN form.
    wqkv: torch.Tensor      # [(hq_s + 2*hkv_s) * dh, hidden]
    wo: torch.Tensor        # [hidden, hq_s * dh]
    wgateup: torch.Tensor   # [2 * ffn_s, hidden]
    wdown: torch.Tensor     # [hidden, ffn_s]
    norm_attn: torch.Tensor
    norm_mlp: torch.Tensor


class LlamaShard:
    """One TP rank's shard of a Llama model, random-init, inference-only."""

    def __init__(self, cfg: ModelConfig, *, tp_rank: int = 0, tp_size: int = 1,
                 device: torch.device | str = "cpu",
                 dtype: torch.dtype = torch.bfloat16, seed: int = 1234):
        if cfg.heads % tp_size or cfg.kv_heads % tp_size or cfg.ffn % tp_size:
            raise ValueError(f"{cfg.name}: heads/kv/ffn must divide tp={tp_size}")
        self.cfg = cfg
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.device = torch.device(device)
        self.dtype = dtype
        self.hq = cfg.heads // tp_size
        self.hkv = cfg.kv_heads // tp_size
        self.ffn_s = cfg.ffn // tp_size
        self.dh = cfg.head_dim
        self.scale = 1.0 / math.sqrt(self.dh)

        # random-init directly on the device: CPU-side randn of 8-70B params
        # costs minutes; on-GPU generation is seconds (seeded per TP rank)
        gen_device = "cpu" if self.device.type == "cpu" else self.device
        gen = torch.Generator(device=gen_device).manual_seed(seed + tp_rank)
        d = cfg.hidden

        def w(rows: int, cols: int, fan_in: int) -> torch.Tensor:
            std = 1.0 / math.sqrt(fan_in)
            t = torch.randn(rows, cols, generator=gen, dtype=torch.float32,
                            device=gen_device) * std
            return t.to(device=self.device, dtype=dtype)

        self.embed = w(cfg.vocab_size, d, d)
        self.layers: list[LayerWeights] = []
        qkv_out = (self.hq + 2 * self.hkv) * self.dh
        for _ in range(cfg.layers):
            self.layers.append(LayerWeights(
                wqkv=w(qkv_out, d, d),
                wo=w(d, self.hq * self.dh, cfg.q_dim),
                wgateup=w(2 * self.ffn_s, d, d),
                wdown=w(d, self.ffn_s, cfg.ffn),
                norm_attn=torch.ones(d, device=self.device, dtype=dtype),
                norm_mlp=torch.ones(d, device=self.device, dtype=dtype),
            ))
        # fp32 workspaces for the skinny/tiled decode GEMMs' split-K slabs;
        # the second one carries down-proj slabs across the layer boundary
        # while the next layer's qkv GEMM already writes the first.  Sized
        # for the worst routed case: skinny at M<=64 x sk8 over the vocab,
        # tiled at M<=512 x sk8 over the widest projection.
        work_elems = max(8 * 64 * max(qkv_out, 2 * self.ffn_s, d,
                                      cfg.vocab_size),
                         8 * 512 * max(qkv_out, 2 * self.ffn_s, d))
        if self.device.type == "cuda":
            self._gemm_work = torch.empty(work_elems, dtype=torch.float32,
                                          device=self.device)
            self._gemm_work2 = torch.empty(8 * 512 * d, dtype=torch.float32,
                                           device=self.device)
        else:
            self._gemm_work = torch.empty(0)
            self._gemm_work2 = torch.empty(0)
        self.norm_final = torch.ones(d, device=self.device, dtype=dtype)
        # lm_head replicated (vocab-parallel sharding is a later optimization;
        # 128256 x 8192 bf16 is ~2 GB of the 288 GB rank)
        self.lm_head = self.embed if cfg.tie_embeddings else w(cfg.vocab_size, d, d)
        # + MAX_RUN_OVERSHOOT: multi-step decode runs blind-run finishing
        # sequences a few positions past max_seq before truncation; those
        # positions must still index a real RoPE row.
        from .config import MAX_RUN_OVERSHOOT

        self.rope = ops.build_rope_table(cfg.max_seq + MAX_RUN_OVERSHOOT,
                                         self.dh, cfg.rope_theta,
                                         device=self.device)
        self._splitk_cache: dict[int, int] = {}
        #: comm stream for the TP-overlap decode pipeline (north star:
        #: "RCCL all-reduce over xGMI overlapped with the GEMMs")
        self._comm_stream = (torch.cuda.Stream(device=self.device)
                             if tp_size > 1 and self.device.type == "cuda"
                             else None)

    # ------------------------------------------------------- GEMM dispatch
    def _use_skinny(self, T: int, n_out: int, k_in: int) -> bool:
        """Decode projections route to the hand-written kernel in its
        measured win region (profiles/r01_gemm_sweep.md): N*K <= ~40M weight
        elements (qkv/o and the TP-sharded 70B shapes).  On the giant tiles
        hipBLASLt streams at ~6.5 TB/s (roofline) and keeps them; the
        all-skinny experiment (r6) measured a net regression 70.4 -> 66.6
        turns/s and was reverted."""
        import os

        big = os.environ.get("CALF_DECODE_BIG_SKINNY", "0") == "1"
        limit = 10 ** 18 if big else 42_000_000
        # the M>64 path exists (skinny_gemm_m_kernel) but measured 0.2-0.4x
        # of hipBLASLt (r14: the 8 waves re-read the shared W tile; it needs
        # the guide's LDS W-staging before it can route) — library for now
        return (self.device.type == "cuda" and T <= 64 and n_out % 64 == 0
                and k_in % 128 == 0 and n_out * k_in <= limit)

    def _tile_plan(self, T: int, n_out: int, k_in: int):
        """(BN, splitk) when the serving-batch tiled GEMM should take this
        projection (M=65..512 ran hipBLASLt fill-bound at
        0.65-0.84 PF/s; the 256-wide MFMA tile + split-K beats it — see
        profiles/r02_tile_gemm.md), else None."""
        import os

        mode = os.environ.get("CALF_TILED", "1")
        if self.device.type != "cuda" or not (64 < T <= 512) or mode == "0":
            return None
        plan = ops.tile_plan(T, n_out, k_in)
        if plan is None or mode == "all":
            return plan
        # measured-policy dispatch (profiles/r02_tile_gemm.md): shape
        # classes route when the hand-written kernel matches or beats
        # hipBLASLt in the serving path.  Default: deep-K down
        # projections and the vocab-wide lm_head (clear wins); qkv is at
        # fused-parity and opt-in.  CALF_TILED takes "all", "0", or a
        # comma list of classes (lmhead,down,qkv,o,gateup).
        if n_out >= 32000 and n_out % 256 == 0:
            cls = "lmhead"
        elif k_in >= 3 * n_out and n_out >= 2048:
            cls = "down"
        elif 1.2 * k_in <= n_out <= 2 * k_in:
            cls = "qkv"
        elif n_out == k_in:
            cls = "o"
        else:
            cls = "gateup"
        routed = (mode.split(",") if "," in mode or mode.isalpha()
                  else ["lmhead", "down"])
        return plan if cls in routed else None

    def _small_ok(self, T: int, n_out: int, k_in: int) -> bool:
        """Route the 64x128 small-tile kernel: the shallow-K wide-N class
        (TP-sharded o/down projections, K = d/tp <= 1792) where it measured
        1.3-2.8x vs hipBLASLt (ops.small_plan / profiles/r02_tile_gemm.md)."""
        import os

        if self.device.type != "cuda" or not (64 < T <= 512):
            return False
        if os.environ.get("CALF_TILED", "1") == "0":
            return False
        return ops.small_plan(T, n_out, k_in)

    def _splitk_for(self, weight: torch.Tensor, m: int = 32) -> int:
        key = (weight.shape[0], m > 64)
        sk = self._splitk_cache.get(key)
        if sk is None:
            sk = ops.choose_splitk(weight.shape[0], weight.shape[1], m)
            self._splitk_cache[key] = sk
        return sk

    def _proj(self, x: torch.Tensor, weight: torch.Tensor,
              decode: bool) -> torch.Tensor:
        n_out = weight.shape[0]
        if decode and self._use_skinny(x.shape[0], n_out, weight.shape[1]):
            sk = self._splitk_for(weight, x.shape[0])
            out = torch.empty(x.shape[0], n_out, dtype=x.dtype, device=x.device)
            ops.skinny_gemm(out, x, weight, self._gemm_work, sk)
            return out
        if decode:
            if self._small_ok(x.shape[0], n_out, weight.shape[1]):
                out = torch.empty(x.shape[0], n_out, dtype=x.dtype,
                                  device=x.device)
                ops.tile_gemm_small(out, x, weight)
                return out
            plan = self._tile_plan(x.shape[0], n_out, weight.shape[1])
            if plan is not None:
                out = torch.empty(x.shape[0], n_out, dtype=x.dtype,
                                  device=x.device)
                ops.tile_gemm(out, x, weight, self._gemm_work, *plan)
                return out
            # library path with the SEARCHED algo index where one exists
            return ops.linear_indexed(x, weight)
        return torch.nn.functional.linear(x, weight)

    # ------------------------------------------------------------------ util
    def param_bytes(self) -> int:
        n = self.embed.numel() + self.lm_head.numel() + self.norm_final.numel()
        for lw in self.layers:
            n += sum(t.numel() for t in
                     (lw.wqkv, lw.wo, lw.wgateup, lw.wdown, lw.norm_attn,
                      lw.norm_mlp))
        return n * self.embed.element_size()

    # ---------------------------------------------------------------- layers
    def _attn(self, lw: LayerWeights, hidden: torch.Tensor,
              positions: torch.Tensor, kcache: torch.Tensor,
              vcache: torch.Tensor, slots: torch.Tensor, *,
              prefill_kv: tuple[torch.Tensor, torch.Tensor] | None,
              plan: "PrefillPlan | None",
              page_table: torch.Tensor | None,
              seq_lens: torch.Tensor | None,
              do_reduce: bool = True) -> torch.Tensor:
        T = hidden.shape[0]
        decode = plan is None
        qkv_w = lw.wqkv
        tplan = (self._tile_plan(T, qkv_w.shape[0], qkv_w.shape[1])
                 if decode else None)
        if decode and self._use_skinny(T, qkv_w.shape[0], qkv_w.shape[1]):
            # slab-fused path: split-K slabs -> {reduce + RoPE + cache write}
            # in the consumer kernel (no reduce_slabs pass, no re-read)
            sk = self._splitk_for(qkv_w, T)
            Mp = ops.skinny_gemm_slabs(hidden, qkv_w, self._gemm_work, sk)
            qkv = torch.empty(T, qkv_w.shape[0], dtype=hidden.dtype,
                              device=hidden.device)
            ops.rope_qkv_write_slab(qkv, self._gemm_work, kcache, vcache,
                                    self.rope, positions, slots, self.hq,
                                    self.hkv, Mp, sk)
        elif tplan is not None and tplan[1] > 1:
            # tiled serving-batch path, same slab contract as skinny
            bn, sk = tplan
            Mp = ops.tile_gemm_slabs(hidden, qkv_w, self._gemm_work, bn, sk)
            qkv = torch.empty(T, qkv_w.shape[0], dtype=hidden.dtype,
                              device=hidden.device)
            ops.rope_qkv_write_slab(qkv, self._gemm_work, kcache, vcache,
                                    self.rope, positions, slots, self.hq,
                                    self.hkv, Mp, sk)
        else:
            qkv = self._proj(hidden, qkv_w, decode=decode)
            # fused epilogue: in-place RoPE on the packed q/k heads +
            # paged-cache scatter — q/k/v stay strided views (no copies)
            ops.rope_qkv_write(qkv, kcache, vcache, self.rope, positions,
                               slots, self.hq, self.hkv)
        q = qkv[:, :self.hq * self.dh].view(T, self.hq, self.dh)
        o = torch.empty(T, self.hq, self.dh, dtype=qkv.dtype, device=qkv.device)
        if plan is not None:
            if plan.kv_in_qkv:
                # fresh packed prefill: the new rows ARE the whole kv set
                kk = qkv[:, self.hq * self.dh:(self.hq + self.hkv) * self.dh]
                vv = qkv[:, (self.hq + self.hkv) * self.dh:]
                kk = kk.view(T, self.hkv, self.dh)
                vv = vv.view(T, self.hkv, self.dh)
            else:
                kfull, vfull = prefill_kv
                knew = qkv[:, self.hq * self.dh:
                           (self.hq + self.hkv) * self.dh].view(
                               T, self.hkv, self.dh)
                vnew = qkv[:, (self.hq + self.hkv) * self.dh:].view(
                    T, self.hkv, self.dh)
                if plan.kv_write_index is not None:
                    kfull.index_copy_(0, plan.kv_write_index, knew)
                    vfull.index_copy_(0, plan.kv_write_index, vnew)
                else:
                    w0 = plan.kv_write_start
                    kfull[w0:w0 + T] = knew
                    vfull[w0:w0 + T] = vnew
                kk, vv = kfull[:plan.kv_total], vfull[:plan.kv_total]
            ops.attn_prefill_varlen(
                o, q, kk, vv, plan.block_map, plan.q_starts, plan.kv_starts,
                plan.past_lens, self.scale)
        else:
            ops.attn_decode(o, q, kcache, vcache, page_table, seq_lens, self.scale)
        o2 = o.view(T, self.hq * self.dh)
        if (decode and self.tp_size == 1
                and self._use_skinny(T, lw.wo.shape[0], lw.wo.shape[1])):
            # leave the o-projection as slabs; the following residual norm
            # fuses the split-K reduce
            sk = self._splitk_for(lw.wo, T)
            Mp = ops.skinny_gemm_slabs(o2, lw.wo, self._gemm_work, sk)
            return ("slabs", Mp, sk)
        if decode and self.tp_size == 1:
            oplan = self._tile_plan(T, lw.wo.shape[0], lw.wo.shape[1])
            if oplan is not None and oplan[1] > 1:
                bn, sk = oplan
                Mp = ops.tile_gemm_slabs(o2, lw.wo, self._gemm_work, bn, sk)
                return ("slabs", Mp, sk)
        out = self._proj(o2, lw.wo, decode=decode)
        return all_reduce_sum(out) if (self.tp_size > 1 and do_reduce) else out

    def _mlp(self, lw: LayerWeights, hidden: torch.Tensor, *,
             decode: bool = False, do_reduce: bool = True):
        T = hidden.shape[0]
        act = torch.empty(T, self.ffn_s, dtype=hidden.dtype,
                          device=hidden.device)
        if decode and self._use_skinny(T, 2 * self.ffn_s, self.cfg.hidden):
            sk = ops.choose_splitk(2 * self.ffn_s, self.cfg.hidden, T)
            ops.skinny_gemm_silu(act, hidden, lw.wgateup, self._gemm_work, sk)
        else:
            # serving batches route gate-up through the tiled GEMM (via
            # _proj); prefill stays on the library
            gateup = self._proj(hidden, lw.wgateup, decode=decode)
            ops.silu_mul(act, gateup)
        if (decode and self.tp_size == 1
                and self._use_skinny(T, lw.wdown.shape[0], lw.wdown.shape[1])):
            # down stays as slabs; the NEXT residual norm fuses the reduce
            sk = self._splitk_for(lw.wdown, T)
            Mp = ops.skinny_gemm_slabs(act, lw.wdown, self._gemm_work2, sk)
            return ("slabs2", Mp, sk)
        if decode and self.tp_size == 1:
            dplan = self._tile_plan(T, lw.wdown.shape[0], lw.wdown.shape[1])
            if dplan is not None and dplan[1] > 1:
                bn, sk = dplan
                Mp = ops.tile_gemm_slabs(act, lw.wdown, self._gemm_work2,
                                         bn, sk)
                return ("slabs2", Mp, sk)
        down = self._proj(act, lw.wdown, decode=decode)
        return all_reduce_sum(down) if (self.tp_size > 1 and do_reduce) \
            else down

    # ------------------------------------------------- TP-overlap decode
    def _forward_tp_overlap(self, tokens, positions, kcaches, vcaches,
                            slots, *, page_table, seq_lens, logits_rows):
        """Decode forward as a two-chunk software pipeline: chunk c's
        row-parallel all-reduce runs on the comm stream WHILE chunk 1-c
        computes, with event ordering both ways (north star: RCCL
        all-reduce over xGMI overlapped with the GEMMs).

        Every rank issues the same chunk/collective order, so TP lockstep
        is preserved; on CPU (gloo test lane) the same code runs with the
        collectives inline — numerics are identical because every op in
        the chunk pipeline is row-independent."""
        T = tokens.shape[0]
        h = T // 2
        cs = (slice(0, h), slice(h, T))
        on_gpu = self.device.type == "cuda" and self._comm_stream is not None
        main = torch.cuda.current_stream() if on_gpu else None
        comm = self._comm_stream

        emb = self.embed[tokens.long()]
        residual = torch.zeros_like(emb)
        hid = [emb[cs[0]], emb[cs[1]]]
        ev: list = [None, None]

        def reduce_async(c: int, t: torch.Tensor) -> None:
            if not on_gpu:
                all_reduce_sum(t)
                return
            e1 = torch.cuda.Event()
            e1.record(main)
            comm.wait_event(e1)
            with torch.cuda.stream(comm):
                t.record_stream(comm)
                all_reduce_sum(t)
            e2 = torch.cuda.Event()
            e2.record(comm)
            ev[c] = e2

        def wait_reduce(c: int) -> None:
            if on_gpu and ev[c] is not None:
                main.wait_event(ev[c])
                ev[c] = None

        eps = self.cfg.norm_eps
        for i, lw in enumerate(self.layers):
            nxt: list = [None, None]
            for c in (0, 1):
                wait_reduce(c)
                ops.fused_add_rmsnorm(hid[c], residual[cs[c]], lw.norm_attn,
                                      eps)
                o = self._attn(lw, hid[c], positions[cs[c]], kcaches[i],
                               vcaches[i], slots[cs[c]], prefill_kv=None,
                               plan=None, page_table=page_table[cs[c]],
                               seq_lens=seq_lens[cs[c]], do_reduce=False)
                reduce_async(c, o)
                nxt[c] = o
            for c in (0, 1):
                wait_reduce(c)
                hid[c] = nxt[c]
                ops.fused_add_rmsnorm(hid[c], residual[cs[c]], lw.norm_mlp,
                                      eps)
                d = self._mlp(lw, hid[c], decode=True, do_reduce=False)
                reduce_async(c, d)
                nxt[c] = d
            hid = nxt
        logits = [None, None]
        for c in (0, 1):
            wait_reduce(c)
            ops.fused_add_rmsnorm(hid[c], residual[cs[c]], self.norm_final,
                                  eps)
            hc = hid[c] if logits_rows is None else hid[c]
            logits[c] = self._proj(hc, self.lm_head, decode=True)
        out = torch.cat(logits)
        return out if logits_rows is None else out[logits_rows]

    # --------------------------------------------------------------- forward
    def make_single_plan(self, scratch, T: int, past_len: int) -> "PrefillPlan":
        """Plan for one sequence of T new tokens over past_len context."""
        dev = self.device
        nb = (T + 63) // 64
        bm = torch.stack([torch.zeros(nb, dtype=torch.int32),
                          torch.arange(nb, dtype=torch.int32)], 1).to(dev)
        return PrefillPlan(
            scratch=scratch,
            block_map=bm.contiguous(),
            q_starts=torch.tensor([0, T], dtype=torch.int32, device=dev),
            kv_starts=torch.tensor([0, past_len + T], dtype=torch.int32,
                                   device=dev),
            past_lens=torch.tensor([past_len], dtype=torch.int32, device=dev),
            kv_write_start=past_len,
            kv_total=past_len + T,
            # chunked prefill MUST land every chunk in the scratch buffers:
            # a past==0 first chunk that skipped the write would leave stale
            # K/V for the later chunks (caught by
            # test_long_prompt_chunked_prefill_matches_short_path)
            kv_in_qkv=False)

    def forward(self, tokens: torch.Tensor, positions: torch.Tensor,
                kcaches: list[torch.Tensor], vcaches: list[torch.Tensor],
                slots: torch.Tensor, *,
                prefill_scratch: list[tuple[torch.Tensor, torch.Tensor]] | None = None,
                prefill_plan: "PrefillPlan | None" = None,
                page_table: torch.Tensor | None = None,
                seq_lens: torch.Tensor | None = None,
                past_len: int = 0,
                logits_rows: "slice | torch.Tensor | None" = None) -> torch.Tensor:
        """One forward pass.  Prefill mode when a plan (or the single-seq
        ``prefill_scratch``+``past_len`` convenience form) is given; decode
        mode otherwise (B sequences, one token each, via the paged tables).

        Returns logits [rows, vocab] (rows = T or B or the logits_rows
        slice/index).
        """
        plan = prefill_plan
        if plan is None and prefill_scratch is not None:
            plan = self.make_single_plan(prefill_scratch, tokens.shape[0],
                                         past_len)
        elif plan is not None:
            prefill_scratch = plan.scratch
        if plan is None and prefill_scratch is None and self.tp_size > 1:
            import os

            if (tokens.shape[0] >= 2
                    and os.environ.get("CALF_TP_OVERLAP", "1") == "1"):
                return self._forward_tp_overlap(
                    tokens, positions, kcaches, vcaches, slots,
                    page_table=page_table, seq_lens=seq_lens,
                    logits_rows=logits_rows)
        hidden = self.embed[tokens.long()]  # [T, d]
        residual = torch.zeros_like(hidden)
        for i, lw in enumerate(self.layers):
            if isinstance(hidden, tuple):  # down-proj slabs from layer i-1
                _, Mp, sk = hidden
                hidden = torch.empty_like(residual)
                ops.fused_add_rmsnorm_slab(hidden, residual, self._gemm_work2,
                                           lw.norm_attn, self.cfg.norm_eps,
                                           Mp, sk)
            else:
                ops.fused_add_rmsnorm(hidden, residual, lw.norm_attn,
                                      self.cfg.norm_eps)
            attn_out = self._attn(
                lw, hidden, positions, kcaches[i], vcaches[i], slots,
                prefill_kv=prefill_scratch[i] if prefill_scratch else None,
                plan=plan, page_table=page_table, seq_lens=seq_lens)
            if isinstance(attn_out, tuple):  # ("slabs", Mp, sk)
                _, Mp, sk = attn_out
                hidden = torch.empty_like(residual)
                ops.fused_add_rmsnorm_slab(hidden, residual, self._gemm_work,
                                           lw.norm_mlp, self.cfg.norm_eps,
                                           Mp, sk)
            else:
                hidden = attn_out
                ops.fused_add_rmsnorm(hidden, residual, lw.norm_mlp,
                                      self.cfg.norm_eps)
            hidden = self._mlp(lw, hidden, decode=plan is None)
        # final residual add + norm
        if isinstance(hidden, tuple):
            _, Mp, sk = hidden
            hidden = torch.empty_like(residual)
            ops.fused_add_rmsnorm_slab(hidden, residual, self._gemm_work2,
                                       self.norm_final, self.cfg.norm_eps,
                                       Mp, sk)
        else:
            ops.fused_add_rmsnorm(hidden, residual, self.norm_final,
                                  self.cfg.norm_eps)
        if logits_rows is not None:
            hidden = hidden[logits_rows]
        # decode lm_head routes through the tiled GEMM at serving batches
        # (501x2 natural tiles at llama vocab: well-filled, no split-K)
        return self._proj(hidden, self.lm_head, decode=plan is None)
