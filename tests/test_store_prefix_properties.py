"""Randomized invariants: store optimistic concurrency under thread race,
prefix-cache refcount accounting under random workloads."""
import random
import threading

from hypothesis import given, settings, strategies as st

from rbg_amd.api import constants as C
from rbg_amd.api.types import ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec
from rbg_amd.store.store import Store


def test_store_apply_is_lost_update_free():
    """N threads x M apply-increments on one object: optimistic
    concurrency must retry internally so no increment is lost."""
    store = Store()
    rbg = RoleBasedGroup(metadata=ObjectMeta(name="ctr"),
                         spec=RoleBasedGroupSpec())
    rbg.metadata.annotations["count"] = "0"
    store.create(rbg)
    N, M = 6, 50

    def worker():
        for _ in range(M):
            def bump(cur):
                cur.metadata.annotations["count"] = str(
                    int(cur.metadata.annotations["count"]) + 1)
                return cur
            store.apply(C.KIND_RBG, "ctr", bump)
    ts = [threading.Thread(target=worker) for _ in range(N)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert int(store.get(C.KIND_RBG, "ctr").metadata.annotations["count"]) \
        == N * M


@settings(max_examples=30, deadline=None)
@given(seed=st.integers(min_value=0, max_value=10**6))
def test_prefix_cache_refcount_invariants(seed):
    """Random admit/finish/evict workload over shared-prefix prompts:
    every page is freed exactly once, refcounts never go negative, and
    the pool's free count is fully restored at the end."""
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.kv_cache import PagedKVCache
    rng = random.Random(seed)
    cfg = EngineConfig(model=ModelConfig.preset("tiny"), device="cpu",
                       kv_pool_tokens=16 * 64, enable_prefix_cache=True)
    cache = PagedKVCache(cfg, __import__("torch").device("cpu"))
    total_free0 = cache.free_pages
    ps = cache.page_size
    prefixes = [[rng.randrange(100) for _ in range(ps * rng.randint(1, 3))]
                for _ in range(3)]
    live = []       # (pages, prompt, shared)
    for step in range(60):
        op = rng.random()
        if op < 0.6 and cache.free_pages > 6:
            prompt = rng.choice(prefixes) + \
                [rng.randrange(100) for _ in range(rng.randint(1, 2 * ps))]
            got = cache.prefix.match(prompt)
            need = (len(prompt) + ps - 1) // ps - len(got)
            pages = got + cache.alloc(need)
            shared = cache.prefix.register(prompt, pages, len(got))
            live.append((pages, prompt, shared))
        elif live:
            pages, prompt, shared = live.pop(rng.randrange(len(live)))
            # release: shared pages via the cache's refcounts, private free
            for pg in pages[:shared]:
                cache.prefix.release_page(pg)
            cache.free(pages[shared:])
        if rng.random() < 0.2:
            # evict() hands back the pages; the engine immediately
            # repurposes them — here they return to the pool
            cache.free(cache.prefix.evict(2))
    # drain everything
    while live:
        pages, prompt, shared = live.pop()
        for pg in pages[:shared]:
            cache.prefix.release_page(pg)
        cache.free(pages[shared:])
    evicted = cache.prefix.evict(10**6)
    cache.free(evicted)
    assert cache.free_pages == total_free0, \
        (cache.free_pages, total_free0)


def test_gang_allocator_concurrent_invariants():
    """N threads reserve/release random gangs: exclusive GPUs are never
    double-granted, HBM is never oversubscribed, and full capacity
    returns once everything is released."""
    import random
    import threading
    from rbg_amd.scheduler.gang import (GangAllocator, GangUnschedulable,
                                        GpuClaim)
    from rbg_amd.scheduler.topology import fully_connected
    topo = fully_connected(8)
    alloc = GangAllocator(topo)
    hbm = topo.hbm_bytes
    stop = threading.Event()
    violations = []

    def worker(wid):
        rng = random.Random(wid)
        held = []
        for i in range(120):
            if held and rng.random() < 0.5:
                alloc.release(held.pop(rng.randrange(len(held))))
                continue
            gid = f"g{wid}-{i}"
            if rng.random() < 0.5:
                claims = [GpuClaim(gpus=rng.randint(1, 3))]
            else:
                claims = [GpuClaim(gpus=1, hbm_bytes=hbm // 4)
                          for _ in range(rng.randint(1, 3))]
            try:
                res = alloc.reserve(gid, claims, timeout=0.0)
            except GangUnschedulable:
                continue
            grants = [g for gs in res.assignments for g in gs]
            if any(g < 0 or g >= 8 for g in grants):
                violations.append(("bad gpu id", grants))
            held.append(gid)
        for gid in held:
            alloc.release(gid)

    ts = [threading.Thread(target=worker, args=(w,)) for w in range(6)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not violations, violations[:3]
    assert sorted(alloc.free_gpus()) == list(range(8))


@settings(max_examples=40, deadline=None)
@given(seed=st.integers(min_value=0, max_value=10**6))
def test_serde_roundtrip_random_rbg(seed):
    """asdict/fromdict round-trip over randomized RoleBasedGroups: the
    serialized form (camelCase YAML shape) must rebuild identically."""
    import random
    from rbg_amd.api.serde import asdict, fromdict
    from rbg_amd.api.types import (EngineResources, EngineSpec,
                                   EngineTemplate, EnvVar,
                                   LeaderWorkerPattern, RoleSpec)
    rng = random.Random(seed)
    roles = []
    for i in range(rng.randint(1, 4)):
        eng = EngineSpec(
            name=f"e{i}", runner=rng.choice(["echo", "llm-engine"]),
            args={"k": rng.randint(0, 9), "mode": "colocated"},
            env=[EnvVar(name="A", value=str(rng.random()))],
            resources=EngineResources(gpus=rng.randint(0, 4),
                                      hbm_bytes=rng.randint(0, 1 << 30)))
        role = RoleSpec(name=f"r{i}", replicas=rng.randint(1, 5),
                        template=EngineTemplate(engines=[eng]),
                        min_ready_seconds=rng.randint(0, 9))
        if rng.random() < 0.4:
            role.pattern = C.PATTERN_LEADER_WORKER
            role.leader_worker_pattern = LeaderWorkerPattern(
                size=rng.randint(1, 4))
        roles.append(role)
    rbg = RoleBasedGroup(metadata=ObjectMeta(name=f"x{seed % 97}"),
                         spec=RoleBasedGroupSpec(roles=roles))
    doc = asdict(rbg)
    back = fromdict(RoleBasedGroup, doc)
    assert asdict(back) == doc
