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

from typing import List, Tuple

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
        return len(self._free)

    def alloc(self, n: int) -> List[int]:
        if n > len(self._free):
            raise MemoryError(f"KV cache exhausted: want {n} pages, have {len(self._free)}")
        return [self._free.pop() for _ in range(n)]

    def free(self, pages: List[int]) -> None:
        self._free.extend(pages)

    def pages_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.page_size - 1) // self.page_size
