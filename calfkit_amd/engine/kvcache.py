"""Paged KV cache manager for 288 GB HBM3E ranks.

Layout per layer: [num_pages, Hkv_shard, PAGE=64, head_dim] bf16 — one
(page, head) panel is a contiguous 16 KB run, matching the decode kernel's
streaming pattern (ops/csrc/attn_decode.hip).

The allocator is a free-list over pages shared by all layers (a sequence
holds the SAME page ids in every layer, so the page table is per-sequence,
not per-layer).  Python-side allocation is not on the decode hot path (it
runs once per generated token per sequence, off-graph).
"""

from __future__ import annotations

from dataclasses import dataclass, field

import torch

__all__ = ["PAGE_SIZE", "KVCache", "SequenceSlot"]

PAGE_SIZE = 64


@dataclass
class SequenceSlot:
    """A live sequence's cache state."""

    seq_id: int
    pages: list[int] = field(default_factory=list)
    length: int = 0

    def capacity(self) -> int:
        return len(self.pages) * PAGE_SIZE


class KVCache:
    def __init__(self, *, layers: int, kv_heads: int, head_dim: int,
                 num_pages: int, device: torch.device | str,
                 dtype: torch.dtype = torch.bfloat16):
        self.layers = layers
        self.kv_heads = kv_heads
        self.head_dim = head_dim
        self.num_pages = num_pages
        self.device = torch.device(device)
        shape = (num_pages, kv_heads, PAGE_SIZE, head_dim)
        self.kcache = [torch.zeros(shape, dtype=dtype, device=device)
                       for _ in range(layers)]
        self.vcache = [torch.zeros(shape, dtype=dtype, device=device)
                       for _ in range(layers)]
        self._free: list[int] = list(range(num_pages - 1, -1, -1))
        self._slots: dict[int, SequenceSlot] = {}

    # ------------------------------------------------------------ accounting
    @property
    def free_pages(self) -> int:
        return len(self._free)

    def bytes_per_page(self) -> int:
        return self.layers * 2 * self.kv_heads * PAGE_SIZE * self.head_dim * 2

    # ------------------------------------------------------------ allocation
    def open(self, seq_id: int) -> SequenceSlot:
        if seq_id in self._slots:
            return self._slots[seq_id]
        slot = SequenceSlot(seq_id=seq_id)
        self._slots[seq_id] = slot
        return slot

    def reserve(self, seq_id: int, new_tokens: int) -> SequenceSlot:
        """Grow the sequence's page list to hold ``new_tokens`` more tokens."""
        slot = self.open(seq_id)
        need = slot.length + new_tokens
        while slot.capacity() < need:
            if not self._free:
                raise MemoryError(
                    f"KV cache out of pages ({self.num_pages} total)")
            slot.pages.append(self._free.pop())
        return slot

    def slots_for(self, seq_id: int, n: int) -> list[int]:
        """Flat cache slot ids for the next ``n`` tokens (and advance)."""
        slot = self.reserve(seq_id, n)
        out = []
        for i in range(n):
            pos = slot.length + i
            out.append(slot.pages[pos // PAGE_SIZE] * PAGE_SIZE + pos % PAGE_SIZE)
        slot.length += n
        return out

    def length(self, seq_id: int) -> int:
        s = self._slots.get(seq_id)
        return s.length if s else 0

    def truncate(self, seq_id: int, length: int) -> None:
        """Drop cache back to ``length`` tokens (resume/rewind)."""
        slot = self._slots.get(seq_id)
        if slot is None:
            return
        keep_pages = (length + PAGE_SIZE - 1) // PAGE_SIZE
        while len(slot.pages) > keep_pages:
            self._free.append(slot.pages.pop())
        slot.length = length

    def release(self, seq_id: int) -> None:
        slot = self._slots.pop(seq_id, None)
        if slot is not None:
            self._free.extend(slot.pages)

    # ------------------------------------------------------------ page table
    def page_table(self, seq_ids: list[int], max_pages: int,
                   out: torch.Tensor | None = None) -> torch.Tensor:
        """Build (or fill) an i32 [B, max_pages] page-table tensor."""
        if out is None:
            out = torch.zeros(len(seq_ids), max_pages, dtype=torch.int32)
        for i, sid in enumerate(seq_ids):
            pages = self._slots[sid].pages
            n = min(len(pages), max_pages)
            out[i, :n] = torch.as_tensor(pages[:n], dtype=torch.int32)
        return out
