"""Mixtral-8x7B-family MoE model on the CDNA4 op set.

Attention is identical to Llama (GQA, dh=128, paged KV) and is inherited;
the MLP becomes a top-2 router + 8 SwiGLU experts.

Expert compute:
- DECODE (T <= 256, GPU): the grouped CDNA4 kernel pair
  (ops/csrc/moe_gemm.hip) — buckets built device-side (argsort + cumsum,
  no host sync), token gather fused into the gate-up GEMM's A-fragment
  loads, router scale fused into the down GEMM's epilogue.  The whole MoE
  layer is hipGraph-capturable, so Mixtral decode runs under the engine's
  multi-step graphs exactly like the dense models (SURVEY §2.4 row
  "Grouped GEMM (MoE)").
- PREFILL / CPU: tokens bucketed with a host-side count, each expert runs
  one dense gate-up/down GEMM pair over its bucket (hipBLASLt) — the
  compute-bound regime where library tiles win and the one-time host sync
  amortizes over the whole chunk.

Parallel layouts (router always replicated):
- TP: every expert's ffn shards over the TP group exactly like the dense
  MLP (row/column parallel + one all-reduce).
- EP: experts partition over the EP group; tokens travel to their experts
  over RCCL all-to-all on xGMI and combine back (``_mlp_ep``).  Per-expert
  weight seeds make any layout produce identical experts, so EP_n == EP_1
  numerically (tests/unit/test_expert_parallel.py, gloo world 2).
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch

from .. import ops
from ..parallel import all_reduce_sum, all_to_all_rows
from .config import ModelConfig
from .llama import LayerWeights, LlamaShard

__all__ = ["MixtralShard", "MoeLayerWeights"]


@dataclass
class MoeLayerWeights(LayerWeights):
    # wgateup/wdown become stacked per-expert (nn.Linear [out, in] layout):
    # wgateup [E_loc, 2*ffn_s, hidden], wdown [E_loc, hidden, ffn_s]
    router: torch.Tensor = None  # [E, hidden]


class MixtralShard(LlamaShard):
    def __init__(self, cfg: ModelConfig, *, ep_rank: int = 0, ep_size: int = 1,
                 **kw):
        assert cfg.is_moe, "MixtralShard needs an MoE config"
        super().__init__(cfg, **kw)
        # Expert parallelism: experts partition over the EP group (attention
        # and router stay replicated; tokens travel to their experts over
        # RCCL all-to-all on xGMI).  EP composes with TP (r02): world =
        # tp x ep, each expert's ffn TP-shards within the owner's TP group
        # (parallel.init_model_parallel builds the groups) and the combine
        # all-reduces over TP after the EP combine.
        if cfg.num_experts % ep_size:
            raise ValueError(f"{cfg.num_experts} experts must divide ep={ep_size}")
        self.ep_rank = ep_rank
        self.ep_size = ep_size
        self.e_loc = cfg.num_experts // ep_size

        # replace the dense MLP weights with stacked expert weights + router.
        # Per-expert generators (seeded by layer*E+expert) make the weights
        # layout-independent: any EP sharding reproduces the same experts,
        # so EP2 == EP1 holds by construction (tests/unit/test_expert_parallel).
        gen_device = "cpu" if self.device.type == "cpu" else self.device
        d, e = cfg.hidden, cfg.num_experts

        def w(*shape: int, fan_in: int, seed: int) -> torch.Tensor:
            gen = torch.Generator(device=gen_device).manual_seed(seed)
            std = 1.0 / math.sqrt(fan_in)
            t = torch.randn(*shape, generator=gen, dtype=torch.float32,
                            device=gen_device) * std
            return t.to(device=self.device, dtype=self.dtype)

        f0 = self.tp_rank * self.ffn_s  # TP slice of each expert's ffn

        def expert_gateup(li: int, ei: int) -> torch.Tensor:
            full = w(2 * cfg.ffn, d, fan_in=d, seed=4321 + li * e + ei)
            gate, up = full[:cfg.ffn], full[cfg.ffn:]
            return torch.cat([gate[f0:f0 + self.ffn_s],
                              up[f0:f0 + self.ffn_s]]).contiguous()

        def expert_down(li: int, ei: int) -> torch.Tensor:
            full = w(d, cfg.ffn, fan_in=cfg.ffn, seed=9876 + li * e + ei)
            return full[:, f0:f0 + self.ffn_s].contiguous()

        e0 = ep_rank * self.e_loc
        layers: list[MoeLayerWeights] = []
        for li, lw in enumerate(self.layers):
            layers.append(MoeLayerWeights(
                wqkv=lw.wqkv, wo=lw.wo,
                wgateup=torch.stack([expert_gateup(li, ei)
                                     for ei in range(e0, e0 + self.e_loc)]),
                wdown=torch.stack([expert_down(li, ei)
                                   for ei in range(e0, e0 + self.e_loc)]),
                norm_attn=lw.norm_attn, norm_mlp=lw.norm_mlp,
                router=w(e, d, fan_in=d, seed=5555 + li),
            ))
        self.layers = layers
        self.topk = cfg.experts_per_token
        # fp32 gate-up slab for the grouped decode kernel: R = 256 (max
        # decode batch) * topk rows of 2F columns (~59 MB for 8x7B)
        if self.device.type == "cuda":
            self._moe_work = torch.empty(
                256 * self.topk * 2 * self.ffn_s, dtype=torch.float32,
                device=self.device)
        else:
            self._moe_work = torch.empty(0)

    def _mlp(self, lw: MoeLayerWeights, hidden: torch.Tensor, *,
             decode: bool = False, do_reduce: bool = True) -> torch.Tensor:
        T, d = hidden.shape
        e = self.cfg.num_experts
        k = self.topk
        logits = torch.nn.functional.linear(hidden, lw.router).float()  # [T,E]

        if self.ep_size > 1:
            weights, experts = torch.topk(torch.softmax(logits, -1), k, dim=-1)
            weights = weights / weights.sum(-1, keepdim=True)
            out = self._mlp_ep(lw, hidden, weights, experts,
                               do_reduce=do_reduce)
            return out

        if decode and self.device.type == "cuda" and ops.have_extension() \
                and T <= 256:
            import os

            # grouped CDNA4 path: no host sync, hipGraph-capturable; the
            # FUSED router + combine replace softmax/topk/argsort/cumsum/
            # gather and zeros/index_add/converts (~12-14 launches -> 2,
            # profiles/r02_moe_kernelstats.md); CALF_MOE_ROUTE=0 A/Bs the
            # torch pipeline
            fused = os.environ.get("CALF_MOE_ROUTE", "1") != "0"
            if fused:
                tok, off, scale, inv = ops.moe_route(logits, e, k)
            else:
                weights, experts = torch.topk(torch.softmax(logits, -1), k,
                                              dim=-1)
                weights = weights / weights.sum(-1, keepdim=True)
                tok, off, scale = ops.moe_buckets(experts, weights, e)
            R = T * k
            act = torch.empty(R, self.ffn_s, dtype=hidden.dtype,
                              device=hidden.device)
            ops.moe_gateup_silu(act, hidden, lw.wgateup, tok, off,
                                self._moe_work)
            y = torch.empty(R, d, dtype=hidden.dtype, device=hidden.device)
            ops.moe_down(y, act, lw.wdown, off, scale)
            if fused:
                out = torch.empty_like(hidden)
                ops.moe_combine(out, y, inv, k)  # fp32-accum fold of k rows
            else:
                out32 = torch.zeros(T, d, dtype=torch.float32,
                                    device=hidden.device)
                out32.index_add_(0, tok.long(), y.float())
                out = out32.to(hidden.dtype)
            return all_reduce_sum(out) \
                if (self.tp_size > 1 and do_reduce) else out

        weights, experts = torch.topk(torch.softmax(logits, -1), k, dim=-1)
        weights = weights / weights.sum(-1, keepdim=True)   # renormalize top-k
        flat_expert = experts.reshape(-1)                   # [T*k]
        flat_token = (torch.arange(T, device=hidden.device)
                      .repeat_interleave(k))                # [T*k]
        order = torch.argsort(flat_expert, stable=True)
        counts = torch.bincount(flat_expert, minlength=e)

        out = torch.zeros_like(hidden)
        gathered = hidden[flat_token[order]]                # [T*k, d] bucketed
        w_sorted = weights.reshape(-1)[order]
        start = 0
        counts_l = counts.tolist()  # host sync; grouped kernel removes this
        for ei in range(e):
            n = counts_l[ei]
            if n == 0:
                continue
            xs = gathered[start:start + n]
            gu = torch.nn.functional.linear(xs, lw.wgateup[ei])
            act = torch.empty(n, self.ffn_s, dtype=xs.dtype, device=xs.device)
            ops.silu_mul(act, gu)
            ys = torch.nn.functional.linear(act, lw.wdown[ei])
            out.index_add_(0, flat_token[order[start:start + n]],
                           (ys.float() * w_sorted[start:start + n, None])
                           .to(out.dtype))
            start += n
        return all_reduce_sum(out) if (self.tp_size > 1 and do_reduce) else out

    def _mlp_ep(self, lw: MoeLayerWeights, hidden: torch.Tensor,
                weights: torch.Tensor, experts: torch.Tensor,
                do_reduce: bool = True) -> torch.Tensor:
        """Expert-parallel MLP: dispatch tokens to their experts' owner ranks
        over all-to-all, compute the local expert shard, combine back.

        The exchange is the reference's EP dispatch/combine pattern
         on RCCL point-to-point xGMI.  Variable splits need a
        counts exchange (host sync), so EP runs eager — decode stays on the
        TP/grouped layout for graph capture; EP is the capacity layout for
        models whose experts exceed one rank's HBM.
        """
        T, d = hidden.shape
        e, e_loc = self.cfg.num_experts, self.e_loc
        tok, off, scale = ops.moe_buckets(experts, weights, e)
        counts = off.diff()
        # rows for rank r = buckets of its expert range, already contiguous
        send_splits = [int(counts[r * e_loc:(r + 1) * e_loc].sum())
                       for r in range(self.ep_size)]
        eids = torch.repeat_interleave(
            torch.arange(e, device=hidden.device, dtype=torch.int32), counts)
        rows = hidden[tok.long()]
        recv_rows, recv_splits = all_to_all_rows(rows, send_splits)
        recv_eids, _ = all_to_all_rows(eids, send_splits, recv_splits)

        # regroup across sources by local expert (each source segment is
        # expert-sorted; the concatenation is not)
        local_eid = recv_eids - self.ep_rank * e_loc
        order2 = torch.argsort(local_eid, stable=True).int()
        counts2 = torch.zeros(e_loc, device=hidden.device, dtype=torch.int32)
        counts2.scatter_add_(0, local_eid.long(),
                             torch.ones_like(local_eid, dtype=torch.int32))
        off2 = torch.zeros(e_loc + 1, device=hidden.device, dtype=torch.int32)
        torch.cumsum(counts2, 0, out=off2[1:])
        R2 = recv_rows.shape[0]

        if R2 > 0:
            act = torch.empty(R2, self.ffn_s, dtype=hidden.dtype,
                              device=hidden.device)
            work = self._moe_work if self._moe_work.numel() >= R2 * 2 * self.ffn_s \
                else torch.empty(R2 * 2 * self.ffn_s, dtype=torch.float32,
                                 device=hidden.device)
            ops.moe_gateup_silu(act, recv_rows, lw.wgateup, order2, off2, work)
            y2 = torch.empty(R2, d, dtype=hidden.dtype, device=hidden.device)
            ones = torch.ones(R2, dtype=torch.float32, device=hidden.device)
            ops.moe_down(y2, act, lw.wdown, off2, ones)
            # un-permute to the received order so sources can reassemble
            y_src = torch.empty_like(y2)
            y_src[order2.long()] = y2
        else:
            y_src = torch.empty(0, d, dtype=hidden.dtype, device=hidden.device)
        y_back, _ = all_to_all_rows(y_src, recv_splits, send_splits)

        out32 = torch.zeros(T, d, dtype=torch.float32, device=hidden.device)
        out32.index_add_(0, tok.long(), y_back.float() * scale[:, None])
        out = out32.to(hidden.dtype)
        # each expert's ffn is TP-sharded: partial sums reduce over the TP
        # group after the EP combine (one [T,d] all-reduce)
        return all_reduce_sum(out) if (self.tp_size > 1 and do_reduce) \
            else out
