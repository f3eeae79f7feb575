"""Paged KV cache manager for 288 GB HBM3E ranks.

Layout per layer: [num_pages, Hkv_shard, PAGE=64, head_dim] bf16 — one
(page, head) panel is a contiguous 16 KB run, matching the decode kernel's
streaming pattern (ops/csrc/attn_decode.hip).

The allocator is a refcounted free-list over pages shared by all layers (a
sequence holds the SAME page ids in every layer, so the page table is
per-sequence, not per-layer).

Prefix caching: full 64-token blocks are indexed by a chain hash of their
token ids; a new prompt reuses the longest cached page run (pages are
read-shared — new tokens always land in fresh pages, so no copy-on-write
is ever needed).  Pages whose refcount drops to zero but that remain in
the index go to an LRU; they are reclaimed only when the free list runs
dry.  Python-side allocation is off the decode hot path.
"""

from __future__ import annotations

import hashlib
from collections import OrderedDict
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


def _chain_hashes(token_ids: list[int]) -> list[bytes]:
    """Chain hash per full PAGE_SIZE block (hash of the whole prefix)."""
    out: list[bytes] = []
    h = b"calfkit-prefix"
    for i in range(len(token_ids) // PAGE_SIZE):
        block = token_ids[i * PAGE_SIZE:(i + 1) * PAGE_SIZE]
        h = hashlib.blake2s(h + b"|" + ",".join(map(str, block)).encode(),
                            digest_size=16).digest()
        out.append(h)
    return out


class KVCache:
    def __init__(self, *, layers: int, kv_heads: int, head_dim: int,
                 num_pages: int, device: torch.device | str,
                 dtype: torch.dtype = torch.bfloat16,
                 prefix_cache: bool = True):
        self.layers = layers
        self.kv_heads = kv_heads
        self.head_dim = head_dim
        self.num_pages = num_pages
        self.device = torch.device(device)
        self.prefix_cache = prefix_cache
        shape = (num_pages, kv_heads, PAGE_SIZE, head_dim)
        self.kcache = [torch.zeros(shape, dtype=dtype, device=device)
                       for _ in range(layers)]
        self.vcache = [torch.zeros(shape, dtype=dtype, device=device)
                       for _ in range(layers)]
        self._free: list[int] = list(range(num_pages - 1, -1, -1))
        self._slots: dict[int, SequenceSlot] = {}
        self._ref: dict[int, int] = {}
        self._index: dict[bytes, int] = {}          # chain hash -> page
        self._page_hash: dict[int, bytes] = {}      # page -> chain hash
        self._lru: "OrderedDict[int, None]" = OrderedDict()  # evictable pages
        #: telemetry: tokens served from the prefix cache instead of prefill
        self.prefix_hit_tokens = 0

    # ------------------------------------------------------------ accounting
    @property
    def free_pages(self) -> int:
        return len(self._free) + len(self._lru)

    def bytes_per_page(self) -> int:
        return self.layers * 2 * self.kv_heads * PAGE_SIZE * self.head_dim * 2

    # ------------------------------------------------------------ allocation
    def _alloc_page(self) -> int:
        if self._free:
            return self._free.pop()
        if self._lru:  # evict the least-recently-used cached page
            page, _ = self._lru.popitem(last=False)
            h = self._page_hash.pop(page, None)
            if h is not None:
                self._index.pop(h, None)
            return page
        raise MemoryError(f"KV cache out of pages ({self.num_pages} total)")

    def _release_page(self, page: int) -> None:
        n = self._ref.get(page, 1) - 1
        if n > 0:
            self._ref[page] = n
            return
        self._ref.pop(page, None)
        if page in self._page_hash:  # stays cached, becomes evictable
            self._lru[page] = None
            self._lru.move_to_end(page)
        else:
            self._free.append(page)

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
            page = self._alloc_page()
            self._ref[page] = 1
            slot.pages.append(page)
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
            self._release_page(slot.pages.pop())
        slot.length = length

    def release(self, seq_id: int) -> None:
        slot = self._slots.pop(seq_id, None)
        if slot is not None:
            for page in slot.pages:
                self._release_page(page)

    # ---------------------------------------------------------- prefix cache
    def match_prefix(self, seq_id: int, token_ids: list[int]) -> int:
        """Open ``seq_id`` seeded with the longest cached page run of the
        prompt; returns the cached token count (a multiple of PAGE_SIZE).

        At least one prompt token is always left to prefill (the caller
        needs last-token logits).
        """
        slot = self.open(seq_id)
        if not self.prefix_cache or slot.length or len(token_ids) < 2:
            return 0
        matched: list[int] = []
        for h in _chain_hashes(token_ids[:len(token_ids) - 1]):
            page = self._index.get(h)
            if page is None:
                break
            matched.append(page)
        for page in matched:
            self._ref[page] = self._ref.get(page, 0) + 1
            self._lru.pop(page, None)
        slot.pages.extend(matched)
        slot.length = len(matched) * PAGE_SIZE
        self.prefix_hit_tokens += slot.length
        return slot.length

    def register_prefix(self, seq_id: int, token_ids: list[int]) -> None:
        """Index this sequence's full prompt blocks for future reuse."""
        if not self.prefix_cache:
            return
        slot = self._slots.get(seq_id)
        if slot is None:
            return
        for i, h in enumerate(_chain_hashes(token_ids)):
            if i >= len(slot.pages):
                break
            if h not in self._index:
                page = slot.pages[i]
                # a page can carry only one identity; skip if already named
                if page not in self._page_hash:
                    self._index[h] = page
                    self._page_hash[page] = h

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
