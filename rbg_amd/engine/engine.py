"""LLM serving engine — scheduler + model runner + metrics.

The GPU-resident realization of one "role instance" engine process
(the reference externalizes this to SGLang — examples/
inference/pd-disagg-standalone.yaml:96-135 runs sglang.launch_server per
role): add_request() enqueues, step() runs one prefill or decode
iteration, and the stats feed the Prometheus-style gauges the controller
scrapes (tok/s, TTFT, pool occupancy).
"""
from __future__ import annotations

import collections
import logging
import time
from dataclasses import dataclass, field
from typing import Deque, Dict, List, Optional

from ..models.llama import TPContext
from .config import EngineConfig
from .model_runner import ModelRunner
from .scheduler import Scheduler
from .sequence import FINISHED, SamplingParams, Sequence

log = logging.getLogger(__name__)


@dataclass
class EngineStats:
    prefill_tokens: int = 0
    decode_tokens: int = 0
    prefill_steps: int = 0
    decode_steps: int = 0
    prefill_wall_s: float = 0.0
    decode_wall_s: float = 0.0
    idle_wall_s: float = 0.0
    sched_wall_s: float = 0.0
    finished: int = 0
    # recent-window TTFTs: bounded so a long-running server neither leaks
    # nor pays an ever-growing sort on every stats scrape
    ttfts: Deque[float] = field(
        default_factory=lambda: collections.deque(maxlen=65536))
    started: float = field(default_factory=time.monotonic)

    def snapshot(self) -> Dict[str, float]:
        wall = max(1e-9, time.monotonic() - self.started)
        ttfts = sorted(self.ttfts)
        p50 = ttfts[len(ttfts) // 2] if ttfts else 0.0
        return {
            "prefill_tokens": self.prefill_tokens,
            "decode_tokens": self.decode_tokens,
            "prefill_steps": self.prefill_steps,
            "decode_steps": self.decode_steps,
            "prefill_wall_s": round(self.prefill_wall_s, 2),
            "decode_wall_s": round(self.decode_wall_s, 2),
            "idle_wall_s": round(self.idle_wall_s, 2),
            "sched_wall_s": round(self.sched_wall_s, 2),
            "output_tok_per_s": self.decode_tokens / wall,
            "finished": self.finished,
            "p50_ttft_s": p50,
        }


class LLMEngine:
    def __init__(self, cfg: EngineConfig, tp: Optional[TPContext] = None,
                 pp=None):
        self.cfg = cfg
        self.runner = ModelRunner(cfg, tp, pp)
        self.scheduler = Scheduler(cfg, self.runner.cache)
        self.stats = EngineStats()

    # ------------------------------------------------------------------

    def add_request(self, prompt_tokens: List[int],
                    sampling: Optional[SamplingParams] = None) -> Sequence:
        seq = Sequence(prompt_tokens, sampling)
        limit = min(self.cfg.max_seq_len, self.cfg.model.max_position)
        total = seq.num_prompt_tokens + seq.sampling.max_new_tokens
        if total > limit:
            raise ValueError(
                f"prompt+max_new_tokens = {total} exceeds the engine limit "
                f"{limit} (max_seq_len/model.max_position)")
        self.scheduler.add(seq)
        return seq

    def step(self) -> str:
        """Run one engine iteration; returns the mode executed."""
        t0 = time.monotonic()
        mode, seqs = self.scheduler.schedule()
        self.stats.sched_wall_s += time.monotonic() - t0
        t0 = time.monotonic()
        if mode == "prefill":
            self.runner.prefill(seqs)
            self.stats.prefill_tokens += sum(s.chunk_len for s in seqs)
            self.stats.prefill_steps += 1
            prefix = self.runner.cache.prefix
            completed = [s for s in seqs
                         if s.num_prefilled == s.num_prompt_tokens]
            for s in completed:
                t = s.ttft()
                if t is not None:
                    self.stats.ttfts.append(t)
                if prefix is not None and s.block_table is not None:
                    # publish this prompt's full pages for reuse
                    s.block_table.num_shared = prefix.register(
                        s.prompt_tokens, s.block_table.pages,
                        s.block_table.num_shared)
            self.scheduler.finish_prefill(completed)
            self.stats.prefill_wall_s += time.monotonic() - t0
        elif mode == "decode":
            self.runner.decode(seqs)
            self.stats.decode_tokens += len(seqs)
            self.stats.decode_steps += 1
            self.scheduler.finish_decode()
            self.stats.decode_wall_s += time.monotonic() - t0
        else:
            self.stats.idle_wall_s += time.monotonic() - t0
        return mode

    def generate(self, prompts: List[List[int]],
                 sampling: Optional[SamplingParams] = None,
                 max_steps: int = 1_000_000) -> List[Sequence]:
        seqs = [self.add_request(p, sampling) for p in prompts]
        steps = 0
        while self.scheduler.has_work() and steps < max_steps:
            self.step()
            steps += 1
        self.stats.finished += sum(1 for s in seqs if s.status == FINISHED)
        return seqs

    # -- live update (weight reload keeping the KV pool) ------

    def reload_weights(self, seed: int) -> None:
        self.runner.model.reload_weights(seed)
