"""Continuous-batching scheduler.

Each engine step is either one prefill batch (as many waiting prompts as fit
the token budget) or one decode step over every running sequence — the
standard continuous-batching loop of the engines the reference orchestrates
(SURVEY §2.3 decode engine row).  Admission reserves KV pages for prompt +
max_new_tokens up front, so a running sequence can never hit OutOfPages
mid-decode (no preemption needed at node scale; revisit for oversubscribed
pools).
"""
from __future__ import annotations

from collections import deque
from typing import Deque, List, Tuple

from .config import EngineConfig
from .kv_cache import BlockTable, PagedKVCache
from .sequence import FINISHED, RUNNING, Sequence


class Scheduler:
    def __init__(self, cfg: EngineConfig, cache: PagedKVCache):
        self.cfg = cfg
        self.cache = cache
        self.waiting: Deque[Sequence] = deque()
        self.running: List[Sequence] = []

    def add(self, seq: Sequence) -> None:
        self.waiting.append(seq)

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def _pages_needed(self, seq: Sequence) -> int:
        total = seq.num_prompt_tokens + seq.sampling.max_new_tokens
        return (total + self.cfg.page_size - 1) // self.cfg.page_size

    def schedule(self) -> Tuple[str, List[Sequence]]:
        """Returns ("prefill"|"decode"|"idle", sequences)."""
        # prefill first: new prompts keep the decode batch full.  A prompt
        # whose remaining suffix exceeds the step budget is CHUNKED: it stays
        # at the head of the queue and prefills budget-sized pieces per step
        # (the attend-over-[past; new] path of the prefill kernel).
        batch: List[Sequence] = []
        budget = self.cfg.max_prefill_tokens
        while self.waiting and len(self.running) + len(batch) < \
                self.cfg.max_batch_size and budget > 0:
            seq = self.waiting[0]
            if seq.block_table is None:
                if self._pages_needed(seq) > self.cache.free_pages_evictable:
                    break
                seq.block_table = BlockTable(self.cache)
                if self.cache.prefix is not None and not seq.imported_kv:
                    cached = self.cache.prefix.match(seq.prompt_tokens)
                    if cached:
                        seq.block_table.adopt_shared(cached)
                        seq.cached_prefix_len = \
                            len(cached) * self.cfg.page_size
                seq.num_prefilled = seq.cached_prefix_len
                seq.block_table.ensure(min(
                    seq.num_prompt_tokens + seq.sampling.max_new_tokens,
                    self.cfg.max_seq_len))
            remaining = seq.num_prompt_tokens - seq.num_prefilled
            seq.chunk_len = min(remaining, budget)
            budget -= seq.chunk_len
            batch.append(seq)
            if seq.num_prefilled + seq.chunk_len < seq.num_prompt_tokens:
                break     # partial chunk: seq stays queued, batch is full
            self.waiting.popleft()
            seq.status = RUNNING
        if batch:
            return "prefill", batch
        if self.running:
            return "decode", list(self.running)
        return "idle", []

    def finish_prefill(self, seqs: List[Sequence]) -> None:
        self.running.extend(seqs)
        self._retire_finished()

    def finish_decode(self) -> None:
        self._retire_finished()

    def _retire_finished(self) -> None:
        still = []
        for seq in self.running:
            if seq.should_stop():
                seq.status = FINISHED
                import time
                seq.finish_time = time.monotonic()
                if seq.block_table is not None:
                    seq.block_table.release()
                    seq.block_table = None
            else:
                still.append(seq)
        self.running = still
