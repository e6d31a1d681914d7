/*
 * libcassandra_gpucompact — MI355X-native SSTable compaction, C ABI.
 *
 * Drop-in boundary: replaces the hot loop of CompactionTask.runMayThrow
 * (reference: src/java/org/apache/cassandra/db/compaction/CompactionTask.java:184-236)
 * — scanners + CompactionIterator + CompactionAwareWriter — for eligible
 * tables (big-format `oa`, Murmur3Partitioner, any partition key, up to 32
 * clustering columns of fixed (bigint/int) or variable (text/ascii/blob)
 * width including prefix range-tombstone bounds, 1..63 regular and 1..63
 * static columns with cell subsets, LZ4 chunk compression; row, cell,
 * partition and range tombstones all supported). The Java host above the seam
 * (strategies, CompactionManager, LifecycleTransaction, metrics) is
 * unchanged and binds these entry points via JNI/Panama.
 *
 * Inputs mirror what the loop consumes (CompactionTask.java:184-205):
 *   - input sstable component paths (format version `oa`)
 *   - merge params nowInSec/gcBefore (CompactionIterator.java:343-347)
 *   - purge info: token-interval -> min-timestamp table for overlapping
 *     non-compacting sources (CompactionController.getPurgeEvaluator,
 *     CompactionController.java:247-286); empty table == no overlaps
 *   - optional token-range restriction (CompactionManager.forceCompactionForTokenRange)
 * Outputs mirror what the loop produces (CompactionTask.java:206-282):
 *   - output sstable component files written under output_base (tmp names;
 *     the host renames/commits under its LifecycleTransaction)
 *   - the stats the task reports (bytes, partition/row counts, merge histogram)
 *
 * Threading: one call == one compaction task, synchronous; the host may run
 * several concurrently on different devices (concurrent_compactors).
 * All compute (decompress, decode, merge, reconcile, purge, serialize,
 * compress, CRC, bloom) runs on the GPU; the library fails with
 * GPUC_ERR_NO_GPU if no HIP device is available — there is no CPU fallback.
 */
#ifndef CASSANDRA_GPUCOMPACT_H
#define CASSANDRA_GPUCOMPACT_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

#define GPUC_OK 0
#define GPUC_ERR_IO 1
#define GPUC_ERR_FORMAT 2        /* unparseable/unsupported component bytes */
#define GPUC_ERR_UNSUPPORTED 3   /* schema/feature outside current GPU scope */
#define GPUC_ERR_NO_GPU 4
#define GPUC_ERR_HIP 5
#define GPUC_ERR_INTERNAL 6
#define GPUC_ERR_CANCELLED 7     /* cancel_flag observed set (CompactionInterruptedException) */

typedef struct gpuc_purge_range {
    int64_t token_lo;   /* inclusive Murmur3 token bounds */
    int64_t token_hi;
    int64_t min_timestamp;
    /* optional bloom filter of the overlapping sstable (the raw Filter.db
       bit payload after its 8-byte header; little-endian words). When set,
       purge gating applies only to keys the bloom might contain — exactly
       CompactionController.getPurgeEvaluator's overlapIterator +
       BF.isPresent chain (CompactionController.java:247-286,308-329). NULL
       keeps the conservative interval-only behavior (never purges more than
       the JVM, may retain tombstones it would drop). */
    const uint32_t* bloom_bits;
    uint64_t bloom_bit_len;   /* number of bits (words*64 from Filter.db) */
    int32_t bloom_hash_count; /* Filter.db hash count */
} gpuc_purge_range;

typedef struct gpuc_job {
    /* input sstable path prefixes, e.g. "/data/ks/t-uuid/oa-12-big"
       (components resolved as "<base>-Data.db" etc.) */
    const char* const* input_bases;
    int32_t n_inputs;

    /* output path prefix; all component files are created as
       "<output_base>-<Component>" (host renames under its txn log) */
    const char* output_base;

    int64_t now_sec;            /* nowInSec */
    int64_t gc_before;          /* gcBefore seconds; INT64_MIN == gc nothing */
    int32_t never_purge;        /* NEVER_PURGE_TOMBSTONES / table option */
    int32_t enforce_strict_liveness;

    const gpuc_purge_range* overlaps;  /* may be NULL when n_overlaps == 0 */
    int32_t n_overlaps;

    int32_t has_token_range;    /* shard restriction (multi-GPU token sharding) */
    int64_t token_lo;           /* inclusive */
    int64_t token_hi;           /* inclusive */

    /* anticompaction split (CompactionManager.antiCompactGroup / RepairFinishedCompactionTask):
       keep only partitions whose token falls in one of these inclusive ranges
       (invert_ranges=1 keeps the complement). One anticompaction = two calls:
       ranges -> the repaired output, ranges+invert -> the unrepaired one. */
    const gpuc_purge_range* keep_ranges;  /* min_timestamp field unused */
    int32_t n_keep_ranges;
    int32_t invert_ranges;

    int32_t device;             /* HIP device ordinal */

    /* nodetool garbagecollect (CompactionIterator.GarbageSkipper): sstables
       whose tombstones/cells REMOVE shadowed data from the compacted inputs
       without being written out. cell_level_gc == TombstoneOption.CELL
       (overwritten cells also removed); 0 == TombstoneOption.ROW. */
    const char* const* tombstone_source_bases;
    int32_t n_tomb_sources;
    int32_t cell_level_gc;

    /* 0/1 = one output sstable. N>1: the job is split into N equal Murmur3
       token ranges; each produces its own complete output sstable
       (generation, generation+1, ...) — the UCS shard model / a splitting
       compaction writer. Two ranges are processed concurrently so front-phase
       kernels overlap the LDS-bound compressor. */
    int32_t n_output_shards;

    /* optional cooperative cancellation (CompactionIterator.isStopRequested,
       CompactionIterator.java:721-740): when non-NULL, polled between
       pipeline phases; a non-zero value aborts the task with
       GPUC_ERR_CANCELLED. Output files may be partially written — the host
       discards them exactly as it discards tmp files of an interrupted Java
       compaction (LifecycleTransaction abort). */
    const volatile int32_t* cancel_flag;
} gpuc_job;

typedef struct gpuc_result {
    uint64_t input_uncompressed_bytes;  /* ISSTableScanner.getLengthInBytes basis */
    uint64_t output_uncompressed_bytes;
    uint64_t output_compressed_bytes;   /* final Data.db size */
    uint64_t partitions_in;             /* summed over inputs (versions) */
    uint64_t partitions_out;
    uint64_t rows_in;
    uint64_t rows_out;
    uint64_t merged_counts[64];         /* merge-arity histogram (CompactionTask.java:258-273) */
    /* phase timings, milliseconds (GPU phases are HIP-event timed) */
    double ms_read_io, ms_h2d, ms_decompress, ms_parse, ms_merge, ms_reconcile,
           ms_serialize, ms_compress, ms_d2h, ms_write_io, ms_total;
    /* dominant kernel bookkeeping for the roofline report */
    char dominant_kernel[64];
    double dominant_kernel_ms;
    uint64_t dominant_kernel_launches;
    char error[256];
} gpuc_result;

/* One compaction task. Returns GPUC_OK or an error code (message in result->error). */
int gpuc_compact(const gpuc_job* job, gpuc_result* result);

/* Synthetic sstable generation ON THE GPU using the product write path
 * (serialize+compress+index+bloom kernels — the flush-path seed).
 * Writes n_sstables sstables under dir as "oa-<first_generation+i>-big-*".
 * Spec fields mirror oracle/src/gen.h (the shared generator contract). */
typedef struct gpuc_gen_spec {
    uint64_t seed;
    uint32_t n_sstables;
    uint64_t rows_per_sstable;
    uint32_t overlap_pct;
    uint32_t value_len;
    uint32_t value_repeat_pct;
    uint32_t tombstone_pct;
    uint32_t partition_del_pct;
    uint32_t clustering_rows;   /* >0: wide partitions (LongType ck), this many rows each */
    uint32_t range_tomb_pct;    /* % of wide partitions with one range tombstone */
    uint32_t key_len;           /* partition key bytes, 8..255 (0 == 8); >8 appends salt bytes */
    uint32_t ck_text;           /* clustering values as UTF8 strings (variable width) */
    uint32_t ck_cols;           /* 0/1 = one clustering column; 2 = composite (bigint,bigint)
                                   with ck0-prefix range-tombstone bounds */
    uint32_t static_pct;        /* percent of wide partitions with a static row */
    uint32_t n_value_cols;      /* regular columns val0..valN-1 (0 == 1 column "val") */
    uint32_t col_missing_pct;   /* P(cell absent) per live row and column */
    int64_t base_ts;
    int64_t base_ldt;
    uint64_t first_generation;
    int32_t device;
    int32_t snappy;             /* 1: SnappyCompressor chunks (C3 shape) */
    uint32_t ttl_pct;           /* percent of live rows written with expiring
                                   liveness/cells (LivenessInfo.java:67,
                                   AbstractCell.java:53-76) */
    uint32_t complex_pct;       /* percent of live rows with cells in one complex
                                   column 'zm' map<blob,blob> (last regular column;
                                   ComplexColumnData.java:47) */
    uint32_t complex_del_pct;   /* percent of those rows also carrying a
                                   complexDeletion */
    int32_t bti;                /* 1: write the `da` (trie-indexed) component
                                   set (Partitions.db/Rows.db, BtiFormat.md) */
    int32_t counter;            /* 1: one CounterColumnType column with
                                   synthetic CounterContexts (CounterContext.java) */
} gpuc_gen_spec;

int gpuc_generate(const gpuc_gen_spec* spec, const char* dir, char* error, size_t error_len);

/* VALIDATION compaction (CompactionManager.doValidationCompaction +
 * repair/Validator.rowHash): merge + purge the inputs exactly as a
 * compaction would, then write per-partition repair digests — records of
 * (Murmur3 token, int64 BE) + 32-byte concat(murmur3_128(1000),
 * murmur3_128(2000)) hash (db/Digest.java:53-59) — to out_path instead of
 * an output sstable. The host feeds these to its MerkleTree
 * (Validator.add). Uses the same gpuc_job fields (now/gcBefore/overlaps);
 * sharding/tombstone-source/keep-range modes are rejected. */
int gpuc_validate(const gpuc_job* job, const char* out_path, uint64_t* n_partitions,
                  char* error, size_t error_len);

/* Memtable flush (the write-path seed): n unsorted UNIQUE-key rows ->
 * token-sorted on device -> one complete `oa` sstable under output_base.
 * Schema: `pk blob PRIMARY KEY, val blob`. values[i] == NULL makes row i a
 * row tombstone with local deletion time del_ldts[i] (else del_ldts ignored;
 * pass UINT32_MAX). Replaces the flush serialize/compress path of
 * Memtable.FlushablePartitionSet -> BigTableWriter. */
typedef struct gpuc_flush_rows {
    uint64_t n_rows;
    const uint8_t* const* keys;
    const uint16_t* key_lens;
    const int64_t* timestamps;
    const uint8_t* const* values;       /* NULL entry = row tombstone */
    const uint32_t* value_lens;
    const uint32_t* del_ldts;           /* UINT32_MAX = live */
} gpuc_flush_rows;

int gpuc_flush(const gpuc_flush_rows* rows, const char* output_base, int32_t device,
               char* error, size_t error_len);

/* ---- full-schema memtable flush ----------------------------------------
 * The real Memtable -> BigTableWriter path (Memtable.FlushablePartitionSet
 * -> SortedTablePartitionWriter): the caller hands over partitions in any
 * order (the engine token-sorts them on device, like gpuc_flush), each with
 * its partition deletion, optional static row, and its unfiltereds (rows +
 * range-tombstone markers) ALREADY in clustering order — exactly what a
 * memtable holds (reference memtables are sorted maps; flush iterates them
 * in order, Memtable.java getFlushSet). The engine validates the
 * within-partition order and key uniqueness and fails loudly on violations.
 *
 * Cell/row field conventions match the serialization they produce:
 * ldt is the u32 localDeletionTime encoding (GPUC_LDT_NONE = none), ttl 0 =
 * none, expiring cells carry ttl>0 + ldt = localExpirationTime. A cell with
 * flags bit HAS_VALUE cleared and ldt set is a cell tombstone. */
#define GPUC_LDT_NONE 0xFFFFFFFFu
#define GPUC_CELLF_PRESENT 1u   /* column set on this row */
#define GPUC_CELLF_HAS_VALUE 2u
#define GPUC_CELLF_EXPIRING 4u
#define GPUC_ROWF_HAS_ROW 1u    /* always set for rows */
#define GPUC_ROWF_LIVE_TS 2u    /* primaryKeyLivenessInfo present */
#define GPUC_ROWF_DELETED 4u    /* row deletion present */

typedef struct gpuc_cell {
    uint8_t flags;              /* GPUC_CELLF_*; 0 = column absent */
    int64_t ts;
    uint32_t ldt;               /* GPUC_LDT_NONE = live */
    int32_t ttl;                /* 0 = none */
    const uint8_t* value;
    uint32_t value_len;
} gpuc_cell;

typedef struct gpuc_cpx_cell {  /* one complex (collection) cell */
    gpuc_cell cell;
    const uint8_t* path;        /* CellPath: the map key bytes */
    uint32_t path_len;
} gpuc_cpx_cell;

typedef struct gpuc_unfiltered {
    uint8_t kind;               /* ClusteringPrefix.Kind ordinal
                                 * (ClusteringPrefix.java:65-85): 4 = row,
                                 * 0/1/6/7 bounds, 2/5 boundaries */
    uint8_t ck_count;           /* clustering components present (rows: n_ck;
                                 * marker bounds may be shorter prefixes) */
    uint8_t row_flags;          /* GPUC_ROWF_* (rows only) */
    const uint8_t* const* ck;   /* per component: serialized value bytes */
    const uint32_t* ck_lens;
    int64_t live_ts;            /* with GPUC_ROWF_LIVE_TS */
    int32_t live_ttl;           /* >0 = ExpiringLivenessInfo */
    int64_t live_let;           /* localExpirationTime (long semantics) */
    int64_t del_mfda;           /* row deletion / marker close(or single) */
    uint32_t del_ldt;
    int64_t open_mfda;          /* boundary markers: open deletion */
    uint32_t open_ldt;
    const gpuc_cell* cells;     /* rows: schema->n_cols entries (the complex
                                 * column's entry is ignored; see cpx_*) */
    uint8_t has_cpx;            /* complex column present on this row */
    int64_t cpx_del_mfda;       /* complexDeletion */
    uint32_t cpx_del_ldt;
    uint32_t n_cpx_cells;
    const gpuc_cpx_cell* cpx_cells;  /* in CellPath order */
} gpuc_unfiltered;

typedef struct gpuc_flush_part {
    const uint8_t* key;
    uint16_t key_len;
    int64_t pdel_mfda;          /* partition deletion (INT64_MIN = live) */
    uint32_t pdel_ldt;
    uint8_t static_flags;       /* GPUC_ROWF_*; 0 = no/empty static row */
    int64_t static_live_ts;
    int32_t static_live_ttl;
    int64_t static_live_let;
    int64_t static_del_mfda;
    uint32_t static_del_ldt;
    const gpuc_cell* static_cells;  /* schema->n_static entries when
                                     * static_flags != 0 */
    uint64_t n_unf;
    const gpuc_unfiltered* unf; /* clustering order */
} gpuc_flush_part;

typedef struct gpuc_flush_schema {
    /* AbstractType class names as the reference serializes them into the
     * Statistics.db HEADER (e.g. "org.apache.cassandra.db.marshal.LongType");
     * fixed widths are derived from the type (LongType 8, Int32Type 4,
     * otherwise variable). NULL key_type = BytesType. */
    const char* key_type;
    uint32_t n_ck;
    const char* const* ck_types;
    uint32_t n_cols;            /* regular columns, header order; a complex
                                 * (MapType) column must be LAST */
    const uint8_t* const* col_names;
    const uint32_t* col_name_lens;
    const char* const* col_types;
    uint32_t n_static;
    const uint8_t* const* static_names;
    const uint32_t* static_name_lens;
    const char* const* static_types;
    uint32_t n_cpx;             /* 0/1: last regular column is complex */
    uint32_t column_index_size; /* 0 = 64 KiB */
    uint32_t snappy;            /* chunk codec for the output */
    uint32_t bti;               /* write the `da` (trie-indexed) components */
    /* header EncodingStats, as the reference's memtable hands its collected
     * stats to SerializationHeader.make; has_stats == 0 derives them from
     * the rows instead */
    uint8_t has_stats;
    int64_t stats_min_ts;
    int64_t stats_min_ldt;      /* long-seconds semantics */
    int32_t stats_min_ttl;
} gpuc_flush_schema;

int gpuc_flush_table(const gpuc_flush_schema* schema, const gpuc_flush_part* parts,
                     uint64_t n_partitions, const char* output_base, int32_t device,
                     char* error, size_t error_len);

/* Scrub one sstable (SortedTableScrubber): salvage every partition whose
 * byte range touches only CRC/decode-clean 16 KiB chunks and rewrite them as
 * a clean sstable under output_base (recovery granularity documented in
 * DESIGN.md; identical to the oracle's). kept/dropped may be NULL. */
int gpuc_scrub(const char* input_base, const char* output_base, int32_t device,
               uint64_t* kept, uint64_t* dropped, char* error, size_t error_len);

/* Verify one sstable (Verifier.java / sstableverify --extended semantics):
 * CompressionInfo frame walk, per-chunk CRC32, full row-format walk,
 * strict DecoratedKey order, Digest.crc32 recomputation, bloom-filter
 * rebuild compared against Filter.db. GPUC_OK or GPUC_ERR_FORMAT with a
 * message. */
int gpuc_verify(const char* input_base, int32_t device, char* error, size_t error_len);

/* library/build identification */
const char* gpuc_version(void);
/* returns number of visible HIP devices (0 => gpuc_compact will fail) */
int gpuc_device_count(void);

#ifdef __cplusplus
}
#endif
#endif /* CASSANDRA_GPUCOMPACT_H */
