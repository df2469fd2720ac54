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

from .config import EngineConfig


class OutOfPages(RuntimeError):
    pass


class PagedKVCache:
    def __init__(self, cfg: EngineConfig, device: torch.device,
                 num_pages: Optional[int] = None,
                 num_layers: Optional[int] = None):
        self.cfg = cfg
        m = cfg.model
        self.page_size = cfg.page_size
        layers = num_layers or m.num_layers      # PP: only local layers
        self.num_layers = layers
        if num_pages is None:
            num_pages = self._size_pool(cfg, device, layers)
        self.num_pages = num_pages
        kvh = m.num_kv_heads // max(1, cfg.tp_size)   # TP shards the heads
        shape = (layers, 2, num_pages, kvh, cfg.page_size, m.head_dim)
        self.ipc_exportable = False
        if device.type == "cuda":
            # hipMalloc base allocation (not the caching allocator) so the
            # pool is hipIpc-exportable: prefill engines map it and push
            # migrated pages straight over xGMI (parallel/kv_peer.py)
            from .. import ops
            if ops.HAVE_HIP:
                self.kv = ops._require_hip().ipc_alloc_bf16(list(shape))
                self.kv.zero_()
                self.ipc_exportable = True
                # incarnation token: a RECREATED engine's pool can receive
                # identical hipIpc handle bytes (deterministic allocator),
                # so peers must key their mappings by (uid, handle), not
                # handle alone — a stale mapping into a dead process's
                # freed pool is an uncatchable GPU fault (GPU run r2/13)
                import uuid as _uuid
                self.ipc_uid = _uuid.uuid4().hex
            else:
                self.kv = torch.zeros(shape, dtype=torch.bfloat16,
                                      device=device)
        else:
            self.kv = torch.zeros(shape, dtype=torch.bfloat16, device=device)
        # page 0 is reserved scratch: hipGraph decode padding rows write
        # their (dead) KV slot there (model_runner._decode_graph)
        self._free: List[int] = list(range(num_pages - 1, 0, -1))
        import threading
        self._lock = threading.Lock()   # alloc/free may race import threads
        self.prefix: Optional["PrefixCache"] = (
            PrefixCache(self) if cfg.enable_prefix_cache else None)

    @staticmethod
    def _size_pool(cfg: EngineConfig, device: torch.device,
                   num_layers: Optional[int] = None) -> int:
        m = cfg.model
        kvh = m.num_kv_heads // max(1, cfg.tp_size)
        page_bytes = ((num_layers or m.num_layers) * 2 * kvh *
                      cfg.page_size * m.head_dim * 2)
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

    @property
    def free_pages_evictable(self) -> int:
        extra = len(self.prefix.lru) if self.prefix is not None else 0
        return len(self._free) + extra

    def alloc(self, n: int) -> List[int]:
        with self._lock:
            if n > len(self._free) and self.prefix is not None:
                # reclaim idle cached-prefix pages (LRU) under pressure
                self._free.extend(self.prefix.evict(n - len(self._free)))
            if n > len(self._free):
                raise OutOfPages(f"need {n} pages, {len(self._free)} free")
            return [self._free.pop() for _ in range(n)]

    def free(self, pages: List[int]) -> None:
        with self._lock:
            self._free.extend(pages)

    # -- kernel views --------------------------------------------------------

    def key_cache(self, layer: int) -> torch.Tensor:
        return self.kv[layer, 0]

    def value_cache(self, layer: int) -> torch.Tensor:
        return self.kv[layer, 1]

    def tokens_capacity(self) -> int:
        return self.num_pages * self.page_size


class BlockTable:
    """Per-sequence page list + slot mapping helpers.  The first
    ``num_shared`` pages are owned by the prefix cache (full, immutable,
    possibly shared across sequences) and are released back to it rather
    than to the free list."""

    def __init__(self, cache: PagedKVCache):
        self.cache = cache
        self.pages: List[int] = []
        self.num_shared = 0
        self.num_tokens = 0

    def ensure(self, num_tokens: int) -> None:
        need = (num_tokens + self.cache.page_size - 1) // self.cache.page_size
        if need > len(self.pages):
            self.pages.extend(self.cache.alloc(need - len(self.pages)))

    def adopt_shared(self, pages: List[int]) -> None:
        assert not self.pages, "adopt before any allocation"
        self.pages = list(pages)
        self.num_shared = len(pages)

    def slots_for(self, start: int, count: int) -> List[int]:
        """Global slot ids (page*page_size + offset) for token positions
        [start, start+count)."""
        self.ensure(start + count)
        ps = self.cache.page_size
        return [self.pages[(start + i) // ps] * ps + (start + i) % ps
                for i in range(count)]

    def release(self) -> None:
        if self.num_shared and self.cache.prefix is not None:
            for pg in self.pages[:self.num_shared]:
                self.cache.prefix.release_page(pg)
            self.cache.free(self.pages[self.num_shared:])
        else:
            self.cache.free(self.pages)
        self.pages = []
        self.num_shared = 0
        self.num_tokens = 0


class PrefixCache:
    """Hash-based full-page prefix reuse — the in-engine realization of the
    reference's Mooncake KV-reuse role (reference keps/74-mooncake-integration;
    the multi-turn TTFT win).  Each FULL page of a prompt is keyed by the
    hash chain of every token up to its end; matched pages are adopted
    instead of re-prefilled.  Pages with refcount 0 park in an LRU pool the
    allocator evicts from under memory pressure, so a hot 288 GB HBM pool
    doubles as the reuse store."""

    def __init__(self, cache: "PagedKVCache"):
        self.cache = cache
        self.by_hash: Dict[int, int] = {}           # chain hash -> page id
        self.page_info: Dict[int, Tuple[int, int]] = {}  # page -> (hash, refs)
        from collections import OrderedDict
        self.lru: "OrderedDict[int, None]" = OrderedDict()
        self.hits = 0
        self.misses = 0
        import threading
        # refcounts are touched from the engine thread AND prefill RPC
        # threads (_release_parked); compound read-modify-writes need a lock
        self._lock = threading.Lock()

    def _chain(self, tokens: List[int]):
        ps = self.cache.page_size
        h = 0x9e3779b9
        for i in range(0, (len(tokens) // ps) * ps, ps):
            h = hash((h, tuple(tokens[i:i + ps])))
            yield h

    def match(self, prompt: List[int]) -> List[int]:
        """Longest cached page chain for prompt[:-1] (at least one token is
        always left to prefill so logits exist); acquires the pages."""
        with self._lock:
            got: List[int] = []
            for h in self._chain(prompt[:-1]):
                pg = self.by_hash.get(h)
                if pg is None:
                    break
                got.append(pg)
            for pg in got:
                h, refs = self.page_info[pg]
                self.page_info[pg] = (h, refs + 1)
                self.lru.pop(pg, None)
            self.hits += len(got)
            self.misses += max(0, len(prompt[:-1]) // self.cache.page_size
                               - len(got))
            return got

    def register(self, prompt: List[int], pages: List[int],
                 already_shared: int) -> int:
        """After prefill: publish the prompt's full pages.  Returns the new
        shared-page count (callers update BlockTable.num_shared)."""
        with self._lock:
            shared = already_shared
            for idx, h in enumerate(self._chain(prompt)):
                if idx < already_shared:
                    continue
                if h in self.by_hash:
                    break   # someone registered concurrently; keep ours
                pg = pages[idx]
                self.by_hash[h] = pg
                self.page_info[pg] = (h, 1)
                shared = idx + 1
            return shared

    def release_page(self, pg: int) -> None:
        with self._lock:
            h, refs = self.page_info[pg]
            if refs <= 1:
                self.page_info[pg] = (h, 0)
                self.lru[pg] = None
            else:
                self.page_info[pg] = (h, refs - 1)

    def evict(self, n: int) -> List[int]:
        with self._lock:
            out = []
            while self.lru and len(out) < n:
                pg, _ = self.lru.popitem(last=False)
                h, _refs = self.page_info.pop(pg)
                self.by_hash.pop(h, None)
                out.append(pg)
            return out

    def stats(self) -> Dict[str, int]:
        return {"hits": self.hits, "misses": self.misses,
                "cached_pages": len(self.page_info),
                "evictable": len(self.lru)}
