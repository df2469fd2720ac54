"""Prefix-cache reuse (the Mooncake multi-turn capability):
exactness of cached-prefix generation, page accounting, LRU eviction."""
import pytest
import torch

from rbg_amd.engine.config import EngineConfig, ModelConfig
from rbg_amd.engine.engine import LLMEngine
from rbg_amd.engine.kv_cache import PagedKVCache
from rbg_amd.engine.sequence import SamplingParams


def engine(prefix=True, pool=4096, **kw):
    cfg = EngineConfig(model=ModelConfig.preset("tiny"), device="cpu",
                       kv_pool_tokens=pool, enforce_eager=True,
                       enable_prefix_cache=prefix, **kw)
    return LLMEngine(cfg)


def test_multi_turn_reuses_prefix_and_matches_uncached():
    torch.manual_seed(0)
    system = torch.randint(0, 500, (40,)).tolist()   # 2.5 pages shared
    turn1 = system + torch.randint(0, 500, (9,)).tolist()
    turn2 = system + torch.randint(0, 500, (13,)).tolist()

    eng = engine(prefix=True)
    (a1,) = eng.generate([turn1], SamplingParams(max_new_tokens=4))
    prefilled_first = eng.stats.prefill_tokens
    (a2,) = eng.generate([turn2], SamplingParams(max_new_tokens=4))
    prefilled_second = eng.stats.prefill_tokens - prefilled_first
    # the shared 2 full pages (32 tokens) were NOT re-prefilled
    assert a2.cached_prefix_len == 32
    assert prefilled_second == len(turn2) - 32
    assert eng.runner.cache.prefix.hits >= 2

    # outputs identical to a prefix-cache-off engine (exact greedy match)
    ref = engine(prefix=False)
    (b1,) = ref.generate([turn1], SamplingParams(max_new_tokens=4))
    (b2,) = ref.generate([turn2], SamplingParams(max_new_tokens=4))
    assert a1.output_tokens == b1.output_tokens
    assert a2.output_tokens == b2.output_tokens


def test_identical_prompt_full_reuse():
    torch.manual_seed(1)
    prompt = torch.randint(0, 500, (64,)).tolist()   # 4 full pages
    eng = engine(prefix=True)
    (a,) = eng.generate([prompt], SamplingParams(max_new_tokens=3))
    (b,) = eng.generate([prompt], SamplingParams(max_new_tokens=3))
    # 3 pages reusable (last page excluded so >=1 token prefills)
    assert b.cached_prefix_len == 48
    assert a.output_tokens == b.output_tokens


def test_pages_return_to_lru_and_evict():
    torch.manual_seed(2)
    eng = engine(prefix=True, pool=512)   # 32 pages (1 reserved)
    cache = eng.runner.cache
    prompt = torch.randint(0, 500, (48,)).tolist()
    eng.generate([prompt], SamplingParams(max_new_tokens=2))
    # finished: full prompt pages parked in LRU, rest freed
    assert cache.prefix.stats()["evictable"] == 3
    total = cache.free_pages + 3
    # a big allocation forces eviction of the cached pages
    pages = cache.alloc(total)
    assert cache.prefix.stats()["evictable"] == 0
    cache.free(pages)


def test_prefix_cache_disabled_path():
    torch.manual_seed(3)
    eng = engine(prefix=False)
    prompt = torch.randint(0, 500, (40,)).tolist()
    (a,) = eng.generate([prompt], SamplingParams(max_new_tokens=2))
    (b,) = eng.generate([prompt], SamplingParams(max_new_tokens=2))
    assert a.output_tokens == b.output_tokens
    assert b.cached_prefix_len == 0


def test_shared_pages_not_freed_while_running():
    torch.manual_seed(4)
    eng = engine(prefix=True)
    prompt = torch.randint(0, 500, (64,)).tolist()
    # two concurrent requests with the same prompt, second admitted after
    # the first finished (registration happens at prefill completion)
    (a,) = eng.generate([prompt], SamplingParams(max_new_tokens=2))
    seq_b = eng.add_request(prompt, SamplingParams(max_new_tokens=6))
    seq_c = eng.add_request(prompt, SamplingParams(max_new_tokens=2))
    while eng.scheduler.has_work():
        eng.step()
    assert seq_b.cached_prefix_len == 48
    assert seq_c.cached_prefix_len == 48
    assert a.output_tokens[:2] == seq_c.output_tokens
    # refcounts drained back to LRU after both finished
    assert eng.runner.cache.prefix.stats()["evictable"] >= 3


def test_chunked_prefill_exact():
    """A prompt longer than max_prefill_tokens prefills in chunks across
    steps (attend-over-past path); greedy output must exactly match an
    engine with an unbounded budget."""
    torch.manual_seed(5)
    prompt = torch.randint(0, 500, (150,)).tolist()
    small = engine(prefix=False, max_prefill_tokens=48)
    (a,) = small.generate([prompt], SamplingParams(max_new_tokens=5))
    assert small.stats.prefill_steps >= 4     # 150 tokens / 48 budget
    big = engine(prefix=False, max_prefill_tokens=4096)
    (b,) = big.generate([prompt], SamplingParams(max_new_tokens=5))
    assert big.stats.prefill_steps == 1
    assert a.output_tokens == b.output_tokens


def test_chunked_prefill_with_prefix_cache():
    torch.manual_seed(6)
    shared = torch.randint(0, 500, (64,)).tolist()
    p1 = shared + torch.randint(0, 500, (90,)).tolist()
    p2 = shared + torch.randint(0, 500, (70,)).tolist()
    eng = engine(prefix=True, max_prefill_tokens=40)
    (a1,) = eng.generate([p1], SamplingParams(max_new_tokens=3))
    (a2,) = eng.generate([p2], SamplingParams(max_new_tokens=3))
    assert a2.cached_prefix_len == 64        # 4 pages reused
    ref = engine(prefix=False, max_prefill_tokens=4096)
    (b1,) = ref.generate([p1], SamplingParams(max_new_tokens=3))
    (b2,) = ref.generate([p2], SamplingParams(max_new_tokens=3))
    assert a1.output_tokens == b1.output_tokens
    assert a2.output_tokens == b2.output_tokens


def test_mixed_batch_chunked_and_short():
    torch.manual_seed(7)
    long_p = torch.randint(0, 500, (120,)).tolist()
    short_p = torch.randint(0, 500, (10,)).tolist()
    eng = engine(prefix=False, max_prefill_tokens=64)
    seq_l = eng.add_request(long_p, SamplingParams(max_new_tokens=3))
    seq_s = eng.add_request(short_p, SamplingParams(max_new_tokens=3))
    while eng.scheduler.has_work():
        eng.step()
    ref = engine(prefix=False, max_prefill_tokens=4096)
    (bl,) = ref.generate([long_p], SamplingParams(max_new_tokens=3))
    (bs,) = ref.generate([short_p], SamplingParams(max_new_tokens=3))
    assert seq_l.output_tokens == bl.output_tokens
    assert seq_s.output_tokens == bs.output_tokens
