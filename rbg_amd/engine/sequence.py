"""Request/sequence state for continuous batching.

The in-engine analog of one routed request in the reference's P/D flow
(reference examples/inference/pd-disagg-standalone.yaml router role:
a request is prefills on one engine, decodes on another; here Sequence
carries the KV page list that migrates between them).
"""
from __future__ import annotations

import itertools
import time
from dataclasses import dataclass
from typing import List, Optional

from .kv_cache import BlockTable

_ids = itertools.count()

WAITING = "waiting"
RUNNING = "running"
FINISHED = "finished"


@dataclass
class SamplingParams:
    max_new_tokens: int = 128
    temperature: float = 0.0       # 0 = greedy
    top_p: float = 1.0
    stop_token: Optional[int] = None
    ignore_eos: bool = True        # random-init weights: run to max tokens


class Sequence:
    def __init__(self, prompt_tokens: List[int],
                 sampling: Optional[SamplingParams] = None,
                 seq_id: Optional[int] = None):
        self.seq_id = seq_id if seq_id is not None else next(_ids)
        self.prompt_tokens = list(prompt_tokens)
        self.output_tokens: List[int] = []
        self.sampling = sampling or SamplingParams()
        self.status = WAITING
        self.block_table: Optional[BlockTable] = None
        self.arrival_time = time.monotonic()
        self.first_token_time: Optional[float] = None
        self.finish_time: Optional[float] = None
        # P/D disaggregation: sequences migrated from a prefill engine carry
        # their KV pages + first generated token instead of re-prefilling
        self.imported_kv = False
        # prefix cache: leading tokens whose KV pages were adopted from the
        # cache (prefill runs only on the suffix)
        self.cached_prefix_len = 0
        # chunked prefill: prompt tokens whose KV is already computed; the
        # scheduler feeds [num_prefilled, num_prefilled+chunk) per step
        self.num_prefilled = 0
        self.chunk_len = 0

    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_tokens)

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_tokens) + len(self.output_tokens)

    @property
    def last_token(self) -> int:
        return (self.output_tokens[-1] if self.output_tokens
                else self.prompt_tokens[-1])

    def append_token(self, tok: int) -> None:
        if self.first_token_time is None:
            self.first_token_time = time.monotonic()
        self.output_tokens.append(tok)

    def should_stop(self) -> bool:
        if len(self.output_tokens) >= self.sampling.max_new_tokens:
            return True
        if (not self.sampling.ignore_eos and self.sampling.stop_token is not None
                and self.output_tokens
                and self.output_tokens[-1] == self.sampling.stop_token):
            return True
        return False

    def ttft(self) -> Optional[float]:
        if self.first_token_time is None:
            return None
        return self.first_token_time - self.arrival_time
