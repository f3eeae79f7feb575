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

TP: every expert's ffn shards over the TP group exactly like the dense MLP
(row/column parallel + one all-reduce); the router is replicated.  Expert
parallelism with RCCL all-to-all is a post-baseline extension for EP>1.
"""

from __future__ import annotations

import math
from dataclasses import dataclass

import torch

from .. import ops
from ..parallel import all_reduce_sum
from .config import ModelConfig
from .llama import LayerWeights, LlamaShard

__all__ = ["MixtralShard", "MoeLayerWeights"]


@dataclass
class MoeLayerWeights(LayerWeights):
    # wgateup/wdown become stacked per-expert: [E, hidden, 2*ffn_s], [E, ffn_s, hidden]
    router: torch.Tensor = None  # [hidden, E]


class MixtralShard(LlamaShard):
    def __init__(self, cfg: ModelConfig, **kw):
        assert cfg.is_moe, "MixtralShard needs an MoE config"
        super().__init__(cfg, **kw)
        # replace the dense MLP weights with stacked expert weights + router
        gen_device = "cpu" if self.device.type == "cpu" else self.device
        gen = torch.Generator(device=gen_device).manual_seed(4321 + self.tp_rank)
        d, e = cfg.hidden, cfg.num_experts

        def w(*shape: int, fan_in: int) -> torch.Tensor:
            std = 1.0 / math.sqrt(fan_in)
            t = torch.randn(*shape, generator=gen, dtype=torch.float32,
                            device=gen_device) * std
            return t.to(device=self.device, dtype=self.dtype)

        layers: list[MoeLayerWeights] = []
        for lw in self.layers:
            layers.append(MoeLayerWeights(
                wqkv=lw.wqkv, wo=lw.wo,
                wgateup=w(e, 2 * self.ffn_s, d, fan_in=d),
                wdown=w(e, d, self.ffn_s, fan_in=cfg.ffn),
                norm_attn=lw.norm_attn, norm_mlp=lw.norm_mlp,
                router=w(e, d, fan_in=d),
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
             decode: bool = False) -> torch.Tensor:
        T, d = hidden.shape
        e = self.cfg.num_experts
        k = self.topk
        logits = torch.nn.functional.linear(hidden, lw.router).float()  # [T,E]
        weights, experts = torch.topk(torch.softmax(logits, -1), k, dim=-1)
        weights = weights / weights.sum(-1, keepdim=True)   # renormalize top-k

        if decode and self.device.type == "cuda" and ops.have_extension() \
                and T <= 256:
            # grouped CDNA4 path: no host sync, hipGraph-capturable
            tok, off, scale = ops.moe_buckets(experts, weights, e)
            R = T * k
            act = torch.empty(R, self.ffn_s, dtype=hidden.dtype,
                              device=hidden.device)
            ops.moe_gateup_silu(act, hidden, lw.wgateup, tok, off,
                                self._moe_work)
            y = torch.empty(R, d, dtype=hidden.dtype, device=hidden.device)
            ops.moe_down(y, act, lw.wdown, off, scale)
            out32 = torch.zeros(T, d, dtype=torch.float32,
                                device=hidden.device)
            out32.index_add_(0, tok.long(), y.float())  # k rows per token
            out = out32.to(hidden.dtype)
            return all_reduce_sum(out) if self.tp_size > 1 else out

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
        return all_reduce_sum(out) if self.tp_size > 1 else out
