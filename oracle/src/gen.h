// ORACLE + shared generator CONTRACT.
// Deterministic synthetic sstable content derived from (seed, sstable index).
// The GPU product implements the SAME derivation; both sides must produce
// identical logical rows so parity tests can compare oracle-written and
// GPU-written/compacted sstables byte for byte.
//
// Key scheme: a bijective Feistel permutation over [0, universe) gives each
// sstable a distinct, dedup-free key set; sstable s covers permuted indices
// j_global = (s*stride + j) mod universe for j in [0, rows). stride < rows
// makes consecutive sstables overlap by exactly (rows-stride) keys.
#pragma once
#include "sstable.h"

namespace oracle {

struct GenSpec {
    uint64_t seed = 42;
    uint32_t n_sstables = 4;
    uint64_t rows_per_sstable = 1000;
    uint32_t overlap_pct = 10;       // % of each sstable's keys shared with the previous one
    uint32_t value_len = 1024;
    uint32_t value_repeat_pct = 55;  // P(8-byte word repeats previous) -> LZ4 ratio knob
    uint32_t tombstone_pct = 0;      // percent of rows that are row-deletions
    uint32_t partition_del_pct = 0;  // percent of partitions with partition deletion
    int64_t base_ts = 1700000000000000LL;   // µs
    int64_t base_ldt = 1700000000LL;        // seconds
    uint64_t first_generation = 1;
    uint32_t snappy = 0;             // 1: SnappyCompressor chunks (C3 shape)
    uint32_t bti = 0;                // 1: write the `da` (trie-indexed) component set
    // wide-partition (C4) shape: clustering_rows > 0 makes each partition hold
    // clustering_rows rows keyed by a LongType clustering column; tombstone_pct
    // then applies per ROW, and range_tomb_pct partitions get one range
    // tombstone [a, b] over the clustering space.
    uint32_t clustering_rows = 0;
    uint32_t range_tomb_pct = 0;
    // ck_text: clustering values become UTF8 strings instead of bigint —
    // "%08u" decimal of the numeric position (so byte order == numeric
    // order) plus, for ROW values only, (j %% 3) 'x' suffix bytes to
    // exercise variable width. GPU mirror: gen2_ck_text.
    uint32_t ck_text = 0;
    // ck_cols=2: composite clustering (bigint ck0, bigint ck1); rows are
    // (16*(j/4), j%4); range tombstones become ck0-PREFIX bounds covering
    // whole ck0 groups (GPU mirror: GenParams2.ck_cols)
    uint32_t ck_cols = 0;
    // static_pct: percent of partitions (wide mode) carrying a static row in
    // one static column "s0" blob (GPU mirror: GenParams2.static_pct)
    uint32_t static_pct = 0;
    // partition key width in bytes, 8..255: first 8 = big-endian key id,
    // bytes 8.. = splitmix64(id ^ (0xC0FFEE5EED + j)) (GPU: gen2_key_salt).
    // key_len > 8 switches the declared key type LongType -> BytesType.
    uint32_t key_len = 8;
    // multi-column: n_value_cols regular blob columns named val0..valN-1
    // (1 keeps the single column "val"); col_missing_pct drops cells per
    // (row, column). GPU mirror: GenParams2.n_value_cols/col_missing_pct.
    uint32_t n_value_cols = 1;
    uint32_t col_missing_pct = 0;
    // complex_pct: percent of LIVE rows carrying cells in ONE complex column
    // "zm" map<blob,blob> (MapType(BytesType,BytesType)) appended after the
    // simple value columns ('z' keeps it last in name order). Paths are drawn
    // from a small space so versions of a key collide and exercise per-path
    // reconcile; complex_del_pct of those rows also carry a complexDeletion
    // (shadowing ts-older cells on merge). GPU mirror: GenParams2.complex_pct.
    uint32_t complex_pct = 0;
    uint32_t complex_del_pct = 0;
    // counter: 1 = the schema becomes ONE counter column "cnt"
    // (CounterColumnType; counter tables hold only counter columns). Cell
    // values are synthetic CounterContexts drawn from an 8-id pool with
    // global/local/remote roles chosen to exercise every compare() branch
    // (global clock ties with differing counts, local+local sums, remote
    // clock rules incl. negative legacy clocks). GPU mirror: GenParams2.
    uint32_t counter = 0;
    // ttl_pct: percent of LIVE rows written with EXPIRING liveness/cells
    // (LivenessInfo.java:67 ExpiringLivenessInfo; cells carry ttl +
    // localDeletionTime == localExpirationTime, AbstractCell.java:53-76).
    // GPU mirror: GenParams2.ttl_pct.
    uint32_t ttl_pct = 0;

    uint64_t stride() const { return rows_per_sstable * (100 - overlap_pct) / 100; }
    uint64_t universe() const {
        uint64_t u = stride() * n_sstables;
        return u < rows_per_sstable ? rows_per_sstable : u;
    }
};

// 4-round balanced Feistel on 2*hb bits with cycle-walking down to [0, universe)
inline uint64_t feistel_perm(uint64_t seed, uint64_t universe, uint64_t x) {
    int hb = 1;
    while ((1ULL << (2 * hb)) < universe) hb++;
    uint64_t mask = (1ULL << hb) - 1;
    do {
        uint64_t l = (x >> hb) & mask, r = x & mask;
        for (int round = 0; round < 4; round++) {
            uint64_t f = splitmix64(seed ^ r ^ ((uint64_t)(round + 1) << 56)) & mask;
            uint64_t nl = r;
            r = l ^ f;
            l = nl;
        }
        x = (l << hb) | r;
    } while (x >= universe);
    return x;
}

inline uint64_t gen_key_id(const GenSpec& g, uint32_t sst, uint64_t j) {
    uint64_t u = g.universe();
    return feistel_perm(g.seed, u, (sst * g.stride() + j) % u);
}
inline bytes gen_key_bytes(const GenSpec& g, uint64_t id) {
    bytes key(g.key_len ? g.key_len : 8);
    for (int b = 0; b < 8; b++) key[b] = (uint8_t)(id >> (8 * (7 - b)));
    for (size_t b = 8; b < key.size(); b++)
        key[b] = (uint8_t)splitmix64(id ^ (0xC0FFEE5EEDULL + (uint64_t)b));
    return key;
}
inline int64_t gen_ts(const GenSpec& g, uint32_t sst, uint64_t key_id) {
    return g.base_ts + (int64_t)(splitmix64(g.seed ^ key_id * 31 ^ ((uint64_t)sst << 48)) % 1000000000ULL);
}
inline bool gen_col_missing(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t j,
                            uint32_t c) {
    if (g.col_missing_pct == 0) return false;
    return splitmix64(g.seed ^ 0xC011C011ULL ^ key_id ^ ((uint64_t)(j + 1) << 40) ^
                      ((uint64_t)(c + 1) << 56) ^ ((uint64_t)sst << 32)) % 100 < g.col_missing_pct;
}
inline uint64_t gen_col_value_id(uint64_t seed_id, uint32_t c) {
    return seed_id + (uint64_t)c * 0xA5A5A5A5A5A5A5ULL;
}
inline bool gen_is_tombstone(const GenSpec& g, uint32_t sst, uint64_t key_id) {
    if (g.tombstone_pct == 0) return false;
    return splitmix64(g.seed ^ 0xDEADULL ^ key_id ^ ((uint64_t)sst << 32)) % 100 < g.tombstone_pct;
}
// wide-partition derivations (rows keyed by (key_id, row j))
inline bytes gen_ck_bytes(const GenSpec& g, int64_t ckval, bool is_row) {
    if (!g.ck_text) {
        bytes b(8);
        for (int i = 0; i < 8; i++) b[i] = (uint8_t)((uint64_t)ckval >> (8 * (7 - i)));
        return b;
    }
    char buf[12];
    snprintf(buf, sizeof(buf), "%08llu", (unsigned long long)ckval);
    bytes out(buf, buf + 8);
    if (is_row)
        for (int i = 0; i < (int)((ckval / 16) % 3); i++) out.push_back('x');
    return out;
}
inline int64_t gen_ck(const GenSpec& g, uint64_t key_id, uint32_t j) {
    (void)g; (void)key_id;
    return (int64_t)j * 16;  // ascending, gaps so range bounds can fall between rows
}
inline int64_t gen_row_ts(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t j) {
    return g.base_ts + (int64_t)(splitmix64(g.seed ^ key_id * 31 ^ ((uint64_t)sst << 48) ^ (uint64_t)(j + 1) * 0x9E37ULL) % 1000000000ULL);
}
inline bool gen_row_is_tombstone(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t j) {
    if (g.tombstone_pct == 0) return false;
    return splitmix64(g.seed ^ 0xDEADULL ^ key_id ^ ((uint64_t)sst << 32) ^ (uint64_t)(j + 7) * 131) % 100 < g.tombstone_pct;
}
// expiring-row derivations (shared contract; GPU mirror in k_gen_fill2)
inline bool gen_row_expiring(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t j) {
    if (g.ttl_pct == 0) return false;
    return splitmix64(g.seed ^ 0x771E771EULL ^ key_id ^ ((uint64_t)sst << 32) ^
                      (uint64_t)(j + 3) * 101) % 100 < g.ttl_pct;
}
inline int32_t gen_ttl(const GenSpec& g, uint64_t key_id, uint32_t j) {
    (void)g;
    return (int32_t)(60 + splitmix64(key_id ^ 0x77AA11ULL ^ (uint64_t)(j + 1) * 131) % 86400);
}
// localExpirationTime: seconds near base_ldt so tests can place `now` on
// either side of expiry
inline int64_t gen_let(const GenSpec& g, uint64_t key_id, uint32_t j) {
    return g.base_ldt + (int64_t)(splitmix64(key_id ^ 0x1E7E1E7EULL ^ (uint64_t)(j + 1) * 17) % 2000);
}
// complex-column derivations (GPU mirror: k_gen_fill2 / k_gen_values2)
inline bool gen_has_complex(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t j) {
    if (g.complex_pct == 0) return false;
    return splitmix64(g.seed ^ 0xC0113C71ULL ^ key_id ^ ((uint64_t)sst << 32) ^
                      (uint64_t)(j + 5) * 157) % 100 < g.complex_pct;
}
inline bool gen_has_cpx_del(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t j) {
    if (g.complex_del_pct == 0) return false;
    return splitmix64(g.seed ^ 0xCDE1CDE1ULL ^ key_id ^ ((uint64_t)sst << 32) ^
                      (uint64_t)(j + 2) * 211) % 100 < g.complex_del_pct;
}
// candidate cell count 1..4 BEFORE path dedup (paths from a 40-value space,
// so intra-row duplicates collapse and cross-sstable versions collide)
inline uint32_t gen_cpx_count(const GenSpec& g, uint64_t key_id, uint32_t j) {
    (void)g;
    return 1 + (uint32_t)(splitmix64(key_id ^ 0xCE11C07ULL ^ (uint64_t)(j + 1) * 19) % 4);
}
inline uint32_t gen_cpx_path_val(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t j, uint32_t e) {
    (void)g;
    return (uint32_t)(splitmix64(key_id ^ 0x9A7B9A7BULL ^ ((uint64_t)sst << 24) ^
                                 (uint64_t)(j + 1) * 23 ^ (uint64_t)(e + 1) * 71) % 40);
}
inline uint64_t gen_cpx_value_word(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t j, uint32_t pathv) {
    return splitmix64(g.seed ^ key_id * 131 ^ ((uint64_t)sst << 40) ^ (uint64_t)(j + 1) * 29 ^
                      (uint64_t)(pathv + 1) * 389);
}
// counter-context derivations (roles fixed per id; presence/clock/count per
// (sstable, key, id) so versions of a key collide on every role pairing)
inline void gen_counter_id(uint32_t idx, uint8_t out[16]) {
    uint64_t a = splitmix64(0xC0C0C0C0ULL + idx), b = splitmix64(0xF00DF00DULL + idx);
    for (int i = 0; i < 8; i++) out[i] = (uint8_t)(a >> (8 * (7 - i)));
    for (int i = 0; i < 8; i++) out[8 + i] = (uint8_t)(b >> (8 * (7 - i)));
}
inline bool gen_ctr_present(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t idx) {
    return splitmix64(g.seed ^ 0xC717C717ULL ^ key_id ^ ((uint64_t)sst << 32) ^
                      (uint64_t)(idx + 1) * 37) % 100 < 55;
}
// role per id: 0=global, 1=local, 2=remote
inline int gen_ctr_role(uint32_t idx) { return idx % 3; }
inline int64_t gen_ctr_clock(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t idx) {
    int role = gen_ctr_role(idx);
    uint64_t r = splitmix64(g.seed ^ 0xC10CULL ^ key_id ^ (uint64_t)(idx + 1) * 131 ^
                            ((uint64_t)sst << 40));
    if (role == 0) {
        // globals: half the keys share the clock across sstables (the
        // equal-clock, differing-count self-heal branch), half differ
        uint64_t base = splitmix64(g.seed ^ 0x610BULL ^ key_id ^ idx);
        return (int64_t)(1000 + (key_id % 2 ? base % 50 : (base + sst) % 50));
    }
    if (role == 1) return (int64_t)(1 + r % 5);  // locals sum
    if (idx == 8 - 1 && (key_id % 5) == 0) return -(int64_t)(1 + r % 90);  // legacy negative
    // remote clocks are STRICTLY positive (clock 0 hits the asymmetric
    // no-branch case of CounterContext.compare and is never produced by
    // live Cassandra: clocks start at 1)
    return (int64_t)(1 + r % 99);
}
inline int64_t gen_ctr_count(const GenSpec& g, uint32_t sst, uint64_t key_id, uint32_t idx) {
    return (int64_t)(splitmix64(g.seed ^ 0xC0117ULL ^ key_id ^ (uint64_t)(idx + 1) * 17 ^
                                ((uint64_t)sst << 24)) % 1000) - 100;
}
inline bool gen_has_static(const GenSpec& g, uint32_t sst, uint64_t key_id) {
    if (g.static_pct == 0) return false;
    return splitmix64(g.seed ^ 0x57A71CULL ^ key_id ^ ((uint64_t)sst << 32)) % 100 < g.static_pct;
}
inline int64_t gen_static_ts(const GenSpec& g, uint32_t sst, uint64_t key_id) {
    return g.base_ts + (int64_t)(splitmix64(g.seed ^ key_id * 977 ^ ((uint64_t)sst << 48)) % 1000000000ULL);
}
inline bool gen_has_range_tomb(const GenSpec& g, uint32_t sst, uint64_t key_id) {
    if (g.range_tomb_pct == 0) return false;
    return splitmix64(g.seed ^ 0xBEEFULL ^ key_id ^ ((uint64_t)sst << 32)) % 100 < g.range_tomb_pct;
}
// range [lo, hi] clustering values (inclusive bounds), lo < hi
inline void gen_range_bounds(const GenSpec& g, uint32_t sst, uint64_t key_id, int64_t* lo, int64_t* hi) {
    uint64_t r = splitmix64(g.seed ^ 0xB00BULL ^ key_id ^ ((uint64_t)sst << 32));
    uint32_t a = (uint32_t)(r % g.clustering_rows);
    uint32_t b = a + 1 + (uint32_t)((r >> 32) % (g.clustering_rows - a));
    *lo = gen_ck(g, key_id, a) + 8;  // between rows a and a+1
    *hi = gen_ck(g, key_id, b) + 8;
    // NOTE: with +8 offsets the bounds never collide with row clusterings
}
inline int64_t gen_range_ts(const GenSpec& g, uint32_t sst, uint64_t key_id) {
    return g.base_ts + (int64_t)(splitmix64(g.seed ^ 0xAB1EULL ^ key_id ^ ((uint64_t)sst << 48)) % 1000000000ULL);
}
inline bool gen_has_partition_del(const GenSpec& g, uint32_t sst, uint64_t key_id) {
    if (g.partition_del_pct == 0) return false;
    return splitmix64(g.seed ^ 0xFEEDULL ^ key_id ^ ((uint64_t)sst << 32)) % 100 < g.partition_del_pct;
}
inline uint32_t gen_ldt(const GenSpec& g, uint64_t key_id, uint64_t salt) {
    return (uint32_t)(g.base_ldt + (int64_t)(splitmix64(key_id ^ salt) % 1000));
}
// value: value_len bytes in 8-byte words; word w repeats word w-1 with prob repeat_pct
inline void gen_value(const GenSpec& g, uint32_t sst, uint64_t key_id, bytes& out) {
    out.resize(g.value_len);
    uint64_t state = g.seed ^ key_id * 0x100000001B3ULL ^ ((uint64_t)sst << 40);
    uint64_t prev = splitmix64(state);
    size_t nw = (g.value_len + 7) / 8;
    for (size_t w = 0; w < nw; w++) {
        uint64_t r = splitmix64(state + 1 + w);
        uint64_t word = (r % 100 < g.value_repeat_pct && w > 0) ? prev : splitmix64(r);
        prev = word;
        size_t off = w * 8;
        for (size_t b = 0; b < 8 && off + b < g.value_len; b++)
            out[off + b] = (uint8_t)(word >> (8 * b));
    }
}

// build one full synthetic sstable (sorted keys)
SSTable generate_sstable(const GenSpec& g, uint32_t sst_index);

}  // namespace oracle
