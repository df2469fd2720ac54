"""Paged KV cache — the HBM3E pool behind prefill/decode attention.

One contiguous allocation per engine holds every layer's K and V pages
(layout [layers, 2, pages, kv_heads, page_size, head_dim] bf16) so the pool
is sized ONCE against the GPU's 288 GB and the per-layer kernel views are
zero-copy slices.  Pages are the transfer unit for P->D KV migration
(parallel/kv_transfer.py): a sequence's pages can be copied peer-to-peer
with one hipMemcpyPeerAsync per (layer, page) run, batched into large
contiguous spans when pages are adjacent.

Capability analog: the KV pool of the engines the reference orchestrates,
plus the Mooncake-style capacity role (SURVEY §5 long-context).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from .config import EngineConfig, ModelConfig


class OutOfPages(RuntimeError):
    pass


class PagedKVCache:
    def __init__(self, cfg: EngineConfig, device: torch.device,
                 num_pages: Optional[int] = None):
        self.cfg = cfg
        m = cfg.model
        self.page_size = cfg.page_size
        if num_pages is None:
            num_pages = self._size_pool(cfg, device)
        self.num_pages = num_pages
        kvh = m.num_kv_heads // max(1, cfg.tp_size)   # TP shards the heads
        self.kv = torch.zeros(
            (m.num_layers, 2, num_pages, kvh, cfg.page_size, m.head_dim),
            dtype=torch.bfloat16, device=device)
        # page 0 is reserved scratch: hipGraph decode padding rows write
        # their (dead) KV slot there (model_runner._decode_graph)
        self._free: List[int] = list(range(num_pages - 1, 0, -1))

    @staticmethod
    def _size_pool(cfg: EngineConfig, device: torch.device) -> int:
        m = cfg.model
        kvh = m.num_kv_heads // max(1, cfg.tp_size)
        page_bytes = (m.num_layers * 2 * kvh * cfg.page_size *
                      m.head_dim * 2)
        if device.type == "cuda" and torch.cuda.is_available():
            free, _total = torch.cuda.mem_get_info(device)
            budget = int(free * cfg.gpu_memory_utilization)
        else:
            budget = 64 << 20     # CPU tests: 64 MB
        if cfg.kv_pool_tokens:
            return max(1, cfg.kv_pool_tokens // cfg.page_size)
        return max(8, budget // page_bytes)

    # -- page accounting ----------------------------------------------------

    @property
    def free_pages(self) -> int:
        return len(self._free)

    def alloc(self, n: int) -> List[int]:
        if n > len(self._free):
            raise OutOfPages(f"need {n} pages, {len(self._free)} free")
        out = [self._free.pop() for _ in range(n)]
        return out

    def free(self, pages: List[int]) -> None:
        self._free.extend(pages)

    # -- kernel views --------------------------------------------------------

    def key_cache(self, layer: int) -> torch.Tensor:
        return self.kv[layer, 0]

    def value_cache(self, layer: int) -> torch.Tensor:
        return self.kv[layer, 1]

    def tokens_capacity(self) -> int:
        return self.num_pages * self.page_size


class BlockTable:
    """Per-sequence page list + slot mapping helpers."""

    def __init__(self, cache: PagedKVCache):
        self.cache = cache
        self.pages: List[int] = []
        self.num_tokens = 0

    def ensure(self, num_tokens: int) -> None:
        need = (num_tokens + self.cache.page_size - 1) // self.cache.page_size
        if need > len(self.pages):
            self.pages.extend(self.cache.alloc(need - len(self.pages)))

    def slots_for(self, start: int, count: int) -> List[int]:
        """Global slot ids (page*page_size + offset) for token positions
        [start, start+count)."""
        self.ensure(start + count)
        ps = self.cache.page_size
        return [self.pages[(start + i) // ps] * ps + (start + i) % ps
                for i in range(count)]

    def release(self) -> None:
        self.cache.free(self.pages)
        self.pages = []
        self.num_tokens = 0
