// ORACLE — BTI (Big Trie-Indexed, version `da`) index READER: trie node
// decoding per io/tries/TrieNode.java (ordinals TrieNode.java:945-962, node
// layouts per BtiFormat.md "Trie nodes") and the Partitions.db / Rows.db
// container layouts (BtiFormat.md "Partition index"/"Row index",
// PartitionIndex.java, RowIndexReader.java). Round-2 scaffolding for
// SURVEY §8(f) rank 2; the writer restatement comes later (docs/bti_notes.md).
#pragma once
#include "sstable.h"

namespace oracle {

struct BtiEntry {
    bytes prefix;        // unique byte-comparable prefix stored in the trie
    uint8_t hash = 0;    // present iff pb >= 8 (Cassandra 5 files)
    bool has_hash = false;
    int64_t idxpos = 0;  // >= 0: Rows.db position; < 0: ~pos in (uncompressed) Data.db
};

struct BtiPartitionsFile {
    bytes first_key, last_key;  // short-length-prefixed keys from the footer
    uint64_t key_count = 0;
    uint64_t root_pos = 0;
    std::vector<BtiEntry> entries;  // DFS order == byte order
};

// parse + fully enumerate a -Partitions.db file
BtiPartitionsFile read_bti_partitions(const bytes& file);

// one row-index block payload (Rows.db trie leaf)
struct BtiRowIndexEntry {
    bytes prefix;
    uint64_t offset = 0;        // offset within the partition
    bool has_open = false;      // pb >= 8: deletion active at block start
    DeletionTime open_dt;
    int pb = 0;                 // raw payload bits + bytes (for regeneration)
    bytes raw_payload;
};

struct BtiRowIndexBlock {
    bytes partition_key;
    uint64_t data_pos = 0;      // partition position in (uncompressed) Data.db
    uint64_t root_pos = 0;
    uint64_t row_count = 0;
    DeletionTime partition_del;
    std::vector<BtiRowIndexEntry> entries;
};

// parse the row index whose FOOTER ends the structure rooted at `index_pos`
// (the position a Partitions.db payload with idxpos >= 0 points at)
BtiRowIndexBlock read_bti_row_index(const bytes& file, uint64_t index_pos);

// ---------------------------------------------------------------------------
// BTI Partitions.db WRITER restatement: IncrementalTrieWriterPageAware
// (io/tries/IncrementalTrieWriterPageAware.java) + TrieNode type selection
// (TrieNode.java:157-180) + PartitionIndexBuilder key-cutting and footer
// (PartitionIndexBuilder.java:130-183) + the PartitionIndex payload
// serializer (PartitionIndex.java:111-141). Byte-exactness pinned against
// the reference's legacy_da fixtures and round-tripped through the reader.
// ---------------------------------------------------------------------------
struct BtiKeyEntry {
    bytes byte_comparable;  // full byte-ordered representation of the key
    bytes raw_key;          // the key bytes for the footer (first/last)
    uint8_t hash_bits;      // DecoratedKey.filterHashLowerBits
    int64_t idxpos;         // >=0 row index pos; <0 = ~data_pos
};

// builds a complete -Partitions.db image (trie + keys + footer) from entries
// in key order
bytes write_bti_partitions(const std::vector<BtiKeyEntry>& entries);

// Rows.db writer: one row-index structure (trie + TrieIndexEntry footer)
// appended to `file` (page arithmetic is file-absolute). Entries carry the
// pre-cut separator prefixes and raw payloads (pb = offset-bytes |
// 8-if-open-deletion; payload = SizedInts offset [+ compact DeletionTime]).
// Returns the footer position (what a Partitions.db idxpos >= 0 points at).
struct BtiRowIndexBlockSpec {
    struct Entry {
        bytes prefix;
        int pb;
        bytes payload;
    };
    std::vector<Entry> entries;
    bytes partition_key;
    uint64_t data_pos = 0;
    uint64_t block_count = 0;
    DeletionTime partition_del;
};
uint64_t append_bti_row_index(bytes& file, const BtiRowIndexBlockSpec& spec);

// OSS50 byte-comparable encoding of a ClusteringPrefix
// (ClusteringComparator.ByteComparableClustering: per component
// NEXT_COMPONENT 0x40 + the type's OSS50 encoding; kind terminator per
// Kind.asByteComparableValue at Version.OSS50 — CLUSTERING_K 0x38, GT-side
// bounds/boundary 0x60, LT-side 0x20, STATIC 0x18).
// Types supported: UTF8/ASCII/BYTES (escaped) and LONG/INT
// (ByteSource.variableLengthInteger).
bytes bti_byte_comparable_clustering(const Clustering& c,
                                     const std::vector<CqlType>& types, BoundKind kind);

// ByteComparable.separatorGt(prev, cur): shortest byte string s with
// prev < s <= cur (common prefix + cur's first differing byte)
bytes bti_separator_gt(const bytes& prev, const bytes& cur);

// RowIndexWriter.nudge(value, nudgeAt): value's bytes through index nudgeAt
// with the byte at nudgeAt incremented (0xFF spills to the next position)
bytes bti_nudge(const bytes& value, size_t nudge_at);

// OSS50 byte-comparable encoding of a Murmur3 DecoratedKey
// (DecoratedKey.asComparableBytes: Multi(NEXT_COMPONENT-prefixed components,
// TERMINATOR) over [ByteSource.of(token) = 8 BE bytes sign-flipped,
// escaped key bytes]; AbstractEscaper: 0x00 runs -> 0x00 FE* [FF byte |
// FE-at-end], non-zero tail -> trailing 0x00)
bytes bti_byte_comparable_m3(int64_t token, const bytes& key);

// ByteOrderedPartitioner form: single escaped key component (legacy_da fixtures)
bytes bti_byte_comparable_bop(const bytes& key);

}  // namespace oracle
