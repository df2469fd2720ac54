"""Pure-function pins for the decode launch policies: hipGraph batch
bucketing and the XCD-filling split-KV heuristic.  Both are graph-safety
contracts — they must depend only on (padded batch, config), never on
per-step tensor values — and the split policy encodes a measured optimum
(~1024 workgroups fills 256 CUs / 8 XCDs; profiles/decode_breakdown.md
r59 sweep), so a regression here silently costs decode bandwidth."""
from rbg_amd.engine.model_runner import GRAPH_BATCH_SIZES
from rbg_amd.ops import pick_decode_splits


def test_graph_buckets_cover_and_round_up():
    # monotone ascending, 1..256: every batch <= 256 has a bucket >= it
    assert list(GRAPH_BATCH_SIZES) == sorted(set(GRAPH_BATCH_SIZES))
    assert GRAPH_BATCH_SIZES[0] == 1 and GRAPH_BATCH_SIZES[-1] == 256

    def bucket(n):
        for b in GRAPH_BATCH_SIZES:
            if n <= b:
                return b
        return GRAPH_BATCH_SIZES[-1]

    for n in range(1, 257):
        assert bucket(n) >= n
        assert bucket(n) in GRAPH_BATCH_SIZES
    assert bucket(300) == 256          # oversize clamps to the last bucket
    assert bucket(128) == 128          # exact sizes map to themselves


def test_split_policy_targets_chip_fill():
    kv_heads, ctx = 8, 2048
    # measured optima from the r59 sweep (v4 packs 2 seqs/WG):
    # B128 -> 1024 base WGs at splits=2; B64 -> splits=4
    assert pick_decode_splits(128, kv_heads, ctx, variant=4) == 2
    assert pick_decode_splits(64, kv_heads, ctx, variant=4) == 4
    # a huge batch already fills the chip: no split-KV overhead
    assert pick_decode_splits(512, kv_heads, ctx, variant=4) == 1
    # splits never exceed what the context can feed (>= one page pair
    # per split) nor the hard cap of 16
    assert pick_decode_splits(1, kv_heads, 64, variant=4) <= 2
    for bs in (1, 2, 7, 64, 128, 256):
        s = pick_decode_splits(bs, kv_heads, ctx, variant=4)
        assert 1 <= s <= 16


def test_split_policy_is_batch_deterministic():
    """Graph safety: same (batch, config) -> same splits, always."""
    a = [pick_decode_splits(b, 8, 4096, variant=4) for b in range(1, 129)]
    b = [pick_decode_splits(b, 8, 4096, variant=4) for b in range(1, 129)]
    assert a == b
