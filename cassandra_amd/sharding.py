"""Token-range sharding for multi-GPU compaction (host-side).

Mirrors the reference's own decomposition: disjoint Murmur3 token ranges per
worker (CompactionManager.forceCompactionForTokenRange, ShardManager.boundaries).
No collectives on the data path — ranges are independent.
"""
TOKEN_MIN = -(2**63)
TOKEN_MAX = 2**63 - 1


def split_token_range(world_size, rank):
    """Inclusive (lo, hi) token bounds of `rank`'s shard of the full ring.

    The 2^64-sized signed token space is split into world_size near-equal
    contiguous ranges; shards are disjoint and their union is the full space.
    """
    if not (0 <= rank < world_size):
        raise ValueError("rank out of range")
    span = 2**64
    lo_u = (span * rank) // world_size
    hi_u = (span * (rank + 1)) // world_size - 1
    return (lo_u + TOKEN_MIN, hi_u + TOKEN_MIN)


def shards_cover_ring(world_size):
    """Validation helper: shards are disjoint, ordered and cover the ring."""
    prev_hi = None
    for r in range(world_size):
        lo, hi = split_token_range(world_size, r)
        if r == 0 and lo != TOKEN_MIN:
            return False
        if prev_hi is not None and lo != prev_hi + 1:
            return False
        if hi < lo:
            return False
        prev_hi = hi
    return prev_hi == TOKEN_MAX
