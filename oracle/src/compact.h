// ORACLE — test infrastructure only (see util.h header note).
#pragma once
#include "sstable.h"

namespace oracle {

// Purge predicate info (CompactionController.getPurgeEvaluator,
// CompactionController.java:247-286): host-precomputed token-interval ->
// min-timestamp table of overlapping NON-compacting sources that may contain
// the key. Empty table == no overlaps == purge-everything-allowed.
struct PurgeRange {
    int64_t tok_lo, tok_hi;        // inclusive token bounds
    int64_t min_ts;
    // optional bloom-filter bits of the overlapping sstable (Filter.db
    // payload, little-endian words): enables the per-key purge evaluator of
    // CompactionController.getPurgeEvaluator (CompactionController.java:
    // 247-286,308-329) — the entry gates purge only for keys it might
    // contain. Empty = the conservative interval-only behavior.
    std::vector<uint8_t> bloom;    // byte-addressed bits (bit i -> bloom[i>>3] & 1<<(i&7))
    int32_t bloom_k = 0;
};

struct CompactionJob {
    std::vector<SSTable> inputs;  // in task order (reconcile tie order)
    int64_t now_sec = 0;          // nowInSec
    int64_t gc_before = 0;        // gcBefore (seconds); LONG_MIN == never gc
    bool never_purge = false;     // -Dcassandra.never_purge_tombstones / getNeverPurgeTombstones
    bool enforce_strict_liveness = false;
    std::vector<PurgeRange> overlaps;
    bool has_shard = false;
    int64_t shard_lo = INT64_MIN, shard_hi = INT64_MAX;  // inclusive token range filter
    // nodetool garbagecollect: tombstone sources (other sstables) whose
    // deletions/cells remove shadowed data WITHOUT being written out
    // (CompactionIterator.GarbageSkipper); cell_level_gc == TombstoneOption.CELL
    std::vector<SSTable> tomb_sources;
    bool cell_level_gc = false;
    // anticompaction split: keep tokens inside (or, inverted, outside) ranges
    std::vector<PurgeRange> keep_ranges;  // min_ts unused
    bool invert_ranges = false;
};

struct CompactionResult {
    SSTable out;                 // merged+purged partitions with SerializationHeader.make header
    uint64_t partitions_in = 0, partitions_out = 0;
    uint64_t rows_in = 0, rows_out = 0;
    std::vector<uint64_t> merged_partition_counts;  // histogram by merge arity (index k-1)
};

CompactionResult compact(const CompactionJob& job);

// Exposed pieces for law tests:
// merge ≤k versions of one partition (UnfilteredRowIterators.merge semantics)
Partition merge_partition_versions(const std::vector<const Partition*>& versions, const Header& h);
// purge one partition in place; returns false if empty post-purge (drop it)
// repair-validation digest (Validator.rowHash): 32 bytes per partition
void validator_digest(const Partition& p, const Header& h, uint8_t out[32]);

bool purge_partition(Partition& p, int64_t now_sec, int64_t gc_before, bool never_purge,
                     const std::vector<PurgeRange>& overlaps, bool enforce_strict_liveness);
int compare_clustering_prefix(const Header& h, BoundKind ka, const Clustering& a,
                              BoundKind kb, const Clustering& b);
void garbage_filter(Partition& data, const Partition& tomb, const Header& h, bool cell_level);

}  // namespace oracle
