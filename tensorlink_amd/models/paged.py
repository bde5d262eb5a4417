"""Paged KV cache (vLLM-style block tables) for long-context serving.

SURVEY.md §2.4/§5 names the paged-KV decode kernel as the long-context
story: instead of one contiguous [B, Hkv, Smax, D] slab per sequence, K/V
live in a shared pool of 128-position pages and a per-sequence block table
maps position/128 -> page id. The decode and rope+append kernels take the
table (decode_attn_mfma.hip / rope_append.hip, PAGED variants), so
sequences only consume memory for pages they actually touch and a long
context never needs a contiguous reservation.

Enabled per-runner with kv_mode="paged" (contiguous remains the default
serving path). Pages are allocated from a free list in shuffled order so
tests genuinely exercise the indirection.
"""

from __future__ import annotations

import random
from collections import OrderedDict, defaultdict
from typing import Dict, List, Optional, Tuple

import torch

PAGE = 128


class PagedKVCache:
    """Same duck-type surface as KVCache (k/v per layer, seq_lens, reset,
    advance) plus a block table; pool tensors are shared by reference when
    the cache is sliced for micro-batches."""

    def __init__(self, n_layers: int, batch: int, max_seq: int, config,
                 device, dtype=torch.bfloat16, shuffle_pages: bool = True):
        pages_per_seq = (max_seq + PAGE - 1) // PAGE
        n_pages = batch * pages_per_seq
        self.k = [torch.zeros(n_pages, config.num_key_value_heads, PAGE,
                              config.head_dim, device=device, dtype=dtype)
                  for _ in range(n_layers)]
        self.v = [torch.zeros_like(self.k[0]) for _ in range(n_layers)]
        self.seq_lens = torch.zeros(batch, device=device, dtype=torch.int32)
        self.max_seq = pages_per_seq * PAGE
        self.batch = batch
        self.pages_per_seq = pages_per_seq
        ids = list(range(n_pages))
        if shuffle_pages:
            random.Random(1234).shuffle(ids)
        self.table = torch.tensor(ids, device=device,
                                  dtype=torch.int32).view(batch,
                                                          pages_per_seq)

    def reset(self):
        self.seq_lens.zero_()

    def advance(self, n: int):
        self.seq_lens += n

    def append(self, layer: int, k_new: torch.Tensor,
               v_new: torch.Tensor, positions: torch.Tensor):
        """Scatter new K/V into pages at `positions` (the contiguous
        KVCache.append API — the GPT-2 family writes through it instead
        of the fused rope_append kernel). k_new/v_new [B,S,Hkv,D],
        positions [B,S]."""
        B, S, H, D = k_new.shape
        pos = positions.to(self.table.device, torch.long).reshape(B, S)
        page = self.table.long().gather(1, pos // PAGE).reshape(-1)
        off = (pos % PAGE).reshape(-1)
        self.k[layer][page, :, off] = k_new.reshape(B * S, H, D).to(
            self.k[layer].dtype)
        self.v[layer][page, :, off] = v_new.reshape(B * S, H, D).to(
            self.v[layer].dtype)

    def gather_contiguous(self, layer: int, S: int):
        """Materialize the first S positions as [B, S, Hkv, D] (prefill
        attention readback; once per request)."""
        np = (S + PAGE - 1) // PAGE
        idx = self.table[:, :np].long()                        # [B, np]
        k = self.k[layer][idx]                                 # [B,np,H,PAGE,D]
        v = self.v[layer][idx]
        B, _, H, _, D = k.shape
        k = k.permute(0, 1, 3, 2, 4).reshape(B, np * PAGE, H, D)
        v = v.permute(0, 1, 3, 2, 4).reshape(B, np * PAGE, H, D)
        return k[:, :S].contiguous(), v[:, :S].contiguous()


class PageAllocator:
    """Free-list allocator over a shared page pool: slots lease pages on
    demand and release them on retirement, so long-lived short sequences
    do not reserve worst-case context (the memory-elasticity half of
    paged attention; per-slot static reservation is the fallback)."""

    def __init__(self, n_pages: int):
        self.free = list(range(n_pages - 1, -1, -1))
        self.n_pages = n_pages

    def alloc(self, n: int):
        if n > len(self.free):
            raise RuntimeError(
                f"KV page pool exhausted (need {n}, free {len(self.free)})")
        return [self.free.pop() for _ in range(n)]

    def release(self, pages):
        self.free.extend(int(p) for p in pages)

    @property
    def n_free(self):
        return len(self.free)


class DynamicPagedKVCache(PagedKVCache):
    """Paged cache whose slot tables start EMPTY; `ensure(slot, length)`
    leases pages as the sequence grows and `release_slot` returns them."""

    def __init__(self, n_layers: int, slots: int, pool_pages: int,
                 max_seq: int, config, device, dtype=torch.bfloat16):
        self.k = [torch.zeros(pool_pages, config.num_key_value_heads, PAGE,
                              config.head_dim, device=device, dtype=dtype)
                  for _ in range(n_layers)]
        self.v = [torch.zeros_like(self.k[0]) for _ in range(n_layers)]
        self.seq_lens = torch.zeros(slots, device=device, dtype=torch.int32)
        self.pages_per_seq = (max_seq + PAGE - 1) // PAGE
        self.max_seq = self.pages_per_seq * PAGE
        self.batch = slots
        self.allocator = PageAllocator(pool_pages)
        self.table = torch.zeros(slots, self.pages_per_seq, device=device,
                                 dtype=torch.int32)
        self._slot_pages = [[] for _ in range(slots)]

    def ensure(self, slot: int, length: int):
        """Lease enough pages for `length` positions in `slot`."""
        need = (length + PAGE - 1) // PAGE
        have = len(self._slot_pages[slot])
        if need > have:
            new = self.allocator.alloc(need - have)
            self._slot_pages[slot].extend(new)
            self.table[slot, have:need] = torch.tensor(
                new, dtype=torch.int32, device=self.table.device)

    def release_slot(self, slot: int):
        self.allocator.release(self._slot_pages[slot])
        self._slot_pages[slot] = []
        self.seq_lens[slot] = 0

    def available_pages(self) -> int:
        return self.allocator.n_free

    def reset(self):
        for s in range(self.batch):
            self.release_slot(s)


def kv_slice_paged(cache: PagedKVCache, s: int, e: int) -> PagedKVCache:
    view = object.__new__(PagedKVCache)
    view.k = cache.k                       # pools shared
    view.v = cache.v
    view.seq_lens = cache.seq_lens[s:e]
    view.table = cache.table[s:e]
    view.max_seq = cache.max_seq
    view.batch = e - s
    view.pages_per_seq = cache.pages_per_seq
    return view


class _SwapBlob:
    """Host-side copy of one slot's KV pages (preemption)."""

    def __init__(self, k, v, n_pages):
        self.k, self.v, self.n_pages = k, v, n_pages


def swap_out(cache: DynamicPagedKVCache, slot: int) -> _SwapBlob:
    """Copy `slot`'s leased pages to host memory and release them back
    to the pool (vLLM-style preemption: the scheduler parks a running
    sequence when the pool starves and resumes it later). Returns the
    blob `swap_in` restores from."""
    pages = list(cache._slot_pages[slot])
    idx = torch.tensor(pages, dtype=torch.long, device=cache.k[0].device)
    k = [layer[idx].to("cpu", copy=True) for layer in cache.k]
    v = [layer[idx].to("cpu", copy=True) for layer in cache.v]
    blob = _SwapBlob(k, v, len(pages))
    cache.release_slot(slot)
    return blob


def swap_in(cache: DynamicPagedKVCache, slot: int, blob: _SwapBlob,
            length: int) -> None:
    """Lease fresh pages for `slot` and restore the swapped KV."""
    assert not cache._slot_pages[slot], "slot must be empty to swap in"
    cache.ensure(slot, blob.n_pages * PAGE)
    pages = cache._slot_pages[slot][:blob.n_pages]
    idx = torch.tensor(pages, dtype=torch.long, device=cache.k[0].device)
    for li in range(len(cache.k)):
        cache.k[li][idx] = blob.k[li].to(cache.k[li].device)
        cache.v[li][idx] = blob.v[li].to(cache.v[li].device)


class PrefixCachingKVCache(DynamicPagedKVCache):
    """Dynamic paged cache with automatic prefix reuse (vLLM-style).

    Every FULL page of a finished prompt is published under a chained
    content hash (page i's key folds in the hash of pages 0..i-1, so a
    page is only reused when the entire prefix matches). A new request
    whose prompt shares a published prefix adopts those pages by
    reference (refcount) and starts its chunked prefill at the first
    uncached token — prompt compute and KV writes for the shared prefix
    are skipped entirely. Published pages with no live references sit in
    an LRU from which the allocator evicts when the free list runs dry,
    so the prefix cache consumes only otherwise-idle pool memory.

    At least the last prompt token is always re-prefilled (its logits
    seed decoding), so ``match_prefix`` never covers the whole prompt.
    """

    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        self.refcnt: Dict[int, int] = defaultdict(int)
        self.hash_to_page: Dict[int, int] = {}
        self.page_hash: Dict[int, int] = {}
        self.lru: "OrderedDict[int, None]" = OrderedDict()
        self.hits = 0                  # tokens of prefill skipped (stats)

    # ---------------- hashing ----------------
    @staticmethod
    def _chain(tokens) -> List[int]:
        hashes, h = [], 0
        for i in range(len(tokens) // PAGE):
            h = hash((h, tuple(int(t) for t in
                               tokens[i * PAGE:(i + 1) * PAGE])))
            hashes.append(h)
        return hashes

    def _usable(self, n_tokens: int) -> int:
        """Cacheable prefix length: full pages, minus one page if that
        would cover the whole prompt."""
        n = (n_tokens // PAGE) * PAGE
        if n >= n_tokens:
            n -= PAGE
        return max(0, n)

    # ---------------- allocation with LRU eviction ----------------
    def _take_pages(self, n: int) -> List[int]:
        out = []
        while len(out) < n:
            if self.allocator.n_free:
                out.extend(self.allocator.alloc(1))
            elif self.lru:
                page, _ = self.lru.popitem(last=False)
                h = self.page_hash.pop(page)
                self.hash_to_page.pop(h, None)
                out.append(page)
            else:
                raise RuntimeError(
                    "KV page pool exhausted (all pages referenced)")
        return out

    def available_pages(self) -> int:
        return self.allocator.n_free + len(self.lru)

    def ensure(self, slot: int, length: int):
        need = (length + PAGE - 1) // PAGE
        have = len(self._slot_pages[slot])
        if need > have:
            new = self._take_pages(need - have)
            for p in new:
                self.refcnt[p] += 1
            self._slot_pages[slot].extend(new)
            self.table[slot, have:need] = torch.tensor(
                new, dtype=torch.int32, device=self.table.device)

    def release_slot(self, slot: int):
        for p in self._slot_pages[slot]:
            self.refcnt[p] -= 1
            if self.refcnt[p] == 0:
                if p in self.page_hash:
                    self.lru[p] = None          # evictable, still published
                else:
                    self.allocator.release([p])
        self._slot_pages[slot] = []
        self.seq_lens[slot] = 0

    # ---------------- prefix API (used by the batcher) ----------------
    def match_prefix(self, tokens) -> int:
        """Longest published prefix (in tokens) WITHOUT adopting it."""
        n = 0
        for h in self._chain(tokens[:self._usable(len(tokens))]):
            if h not in self.hash_to_page:
                break
            n += PAGE
        return n

    def adopt_prefix(self, slot: int, tokens) -> int:
        """Point `slot`'s first pages at the published prefix; returns
        the number of tokens covered (0 if none). Call on an empty
        slot."""
        assert not self._slot_pages[slot]
        pages = []
        for h in self._chain(tokens[:self._usable(len(tokens))]):
            p = self.hash_to_page.get(h)
            if p is None:
                break
            pages.append(p)
        if not pages:
            return 0
        for p in pages:
            self.refcnt[p] += 1
            self.lru.pop(p, None)
        self._slot_pages[slot].extend(pages)
        self.table[slot, :len(pages)] = torch.tensor(
            pages, dtype=torch.int32, device=self.table.device)
        self.hits += len(pages) * PAGE
        return len(pages) * PAGE

    def register_prefix(self, slot: int, tokens):
        """Publish `slot`'s full prompt pages for future reuse (call
        after its prefill completes)."""
        for i, h in enumerate(self._chain(
                tokens[:self._usable(len(tokens))])):
            if h in self.hash_to_page:
                continue              # identical page already published
            page = self._slot_pages[slot][i]
            if page in self.page_hash:
                continue              # page already carries another hash
            self.hash_to_page[h] = page
            self.page_hash[page] = h
