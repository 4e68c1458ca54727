"""Paged KV cache sized for 288 GB of HBM3E per GPU.

Layout [num_pages, n_kv_heads, page_size, head_dim] per layer (K and V
separate) — chosen so the decode kernel streams each page's per-head K as
one long contiguous run (page_size*head_dim elements, vectorizable at
16 B/lane).  Page size defaults to 256 tokens: with 288 GB there is no
reason to pay fine-grained page indirection (fewer, larger pages = fewer
block-table lookups and longer contiguous streams).

Replaces nothing in the reference (it rents context windows; SURVEY.md
§5.7) — this is the net-new engine memory manager.
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import torch


class PagedKVCache:
    def __init__(
        self,
        n_layers: int,
        n_kv_heads: int,
        head_dim: int,
        page_size: int = 256,
        num_pages: int = 64,
        device: str = "cpu",
        dtype: torch.dtype = torch.bfloat16,
    ):
        self.n_layers = n_layers
        self.n_kv_heads = n_kv_heads
        self.head_dim = head_dim
        self.page_size = page_size
        self.num_pages = num_pages
        self.device = device
        self.dtype = dtype
        shape = (num_pages, n_kv_heads, page_size, head_dim)
        self.k_caches: List[torch.Tensor] = []
        self.v_caches: List[torch.Tensor] = []
        for _ in range(n_layers):
            self.k_caches.append(torch.zeros(shape, device=device, dtype=dtype))
            self.v_caches.append(torch.zeros(shape, device=device, dtype=dtype))
        self._free: List[int] = list(range(num_pages - 1, -1, -1))
        # page refcounts: cross-request prefix sharing gives one page many
        # owners (each seq/session holding it counts one ref)
        self._rc: Dict[int, int] = {}
        # rc==0 pages whose content is retained for future prefix reuse
        # (LRU; reclaimed by alloc() on demand — they still count as free)
        from collections import OrderedDict as _OD
        self._cached: "_OD[int, None]" = _OD()
        #: called when a page's LAST ref drops; return True to RETAIN the
        #: page content in the cached tier (the engine keeps prefix-table
        #: entries alive for retained pages)
        self.on_final_free = None
        #: called when a retained page is reclaimed by alloc() — the engine
        #: must drop its prefix-table entry before the page is rewritten
        self.on_cache_evict = None

    @staticmethod
    def pages_for_memory(
        mem_bytes: int, n_layers: int, n_kv_heads: int, head_dim: int,
        page_size: int = 256, dtype_bytes: int = 2,
    ) -> int:
        per_page = 2 * n_layers * n_kv_heads * page_size * head_dim * dtype_bytes
        return max(int(mem_bytes // per_page), 1)

    def layer_cache(self, i: int) -> Tuple[torch.Tensor, torch.Tensor]:
        return self.k_caches[i], self.v_caches[i]

    @property
    def caches(self) -> List[Tuple[torch.Tensor, torch.Tensor]]:
        return list(zip(self.k_caches, self.v_caches))

    # -- page allocation ---------------------------------------------------

    def num_free(self) -> int:
        return len(self._free) + len(self._cached)  # cached is reclaimable

    def alloc(self, n: int) -> List[int]:
        if n > self.num_free():
            raise MemoryError(f"KV cache exhausted: want {n} pages, have {self.num_free()}")
        pages: List[int] = []
        for _ in range(n):
            if self._free:
                pages.append(self._free.pop())
            else:  # reclaim the least-recently-retained prefix page
                p, _ = self._cached.popitem(last=False)
                if self.on_cache_evict is not None:
                    self.on_cache_evict(p)
                pages.append(p)
        for p in pages:
            self._rc[p] = 1
        return pages

    def incref(self, pages: List[int]) -> None:
        """Add one reference per page; revives cached (rc==0) pages."""
        for p in pages:
            if p in self._cached:
                del self._cached[p]
                self._rc[p] = 1
            else:
                self._rc[p] += 1

    def free(self, pages: List[int]) -> None:
        """Drop one reference per page; at refcount zero the page is
        recycled, or retained in the cached tier when the owner asks."""
        for p in pages:
            rc = self._rc[p] - 1
            if rc:
                self._rc[p] = rc
                continue
            del self._rc[p]
            if self.on_final_free is not None and self.on_final_free(p):
                self._cached[p] = None  # retained, LRU tail
            else:
                self._free.append(p)

    def pages_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.page_size - 1) // self.page_size
