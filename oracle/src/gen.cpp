// ORACLE — test infrastructure + synthetic input generator (shared contract, see gen.h).
#include "gen.h"
#include <algorithm>
#include <unordered_set>

namespace oracle {

SSTable generate_sstable(const GenSpec& g_in, uint32_t sst) {
    GenSpec g = g_in;
    if (g.key_universe == 0) g.key_universe = 10 * g.rows_per_sstable * g.n_sstables;
    SSTable t;
    t.generation = g.first_generation + sst;
    t.header.key_type = CqlType::LONG;  // pk bigint
    t.header.regular_cols = {{bytes{'v', 'a', 'l'}, CqlType::BYTES}};  // val blob

    // distinct key ids: walk j until rows_per_sstable unique
    std::unordered_set<uint64_t> seen;
    seen.reserve(g.rows_per_sstable * 2);
    std::vector<uint64_t> ids;
    ids.reserve(g.rows_per_sstable);
    for (uint64_t j = 0; ids.size() < g.rows_per_sstable; j++) {
        uint64_t id = gen_key_id(g, sst, j);
        if (seen.insert(id).second) ids.push_back(id);
    }

    struct Ent { int64_t token; bytes key; uint64_t id; };
    std::vector<Ent> ents;
    ents.reserve(ids.size());
    for (uint64_t id : ids) {
        bytes key(8);
        for (int b = 0; b < 8; b++) key[b] = (uint8_t)(id >> (8 * (7 - b)));  // LongType BE
        int64_t tok = murmur3_token(key.data(), key.size());
        ents.push_back({tok, std::move(key), id});
    }
    std::sort(ents.begin(), ents.end(), [](const Ent& a, const Ent& b) {
        return compare_decorated_key(a.token, a.key.data(), a.key.size(),
                                     b.token, b.key.data(), b.key.size()) < 0;
    });

    EncodingStats hs;  // header stats collected from this sstable's own data
    int64_t min_ts = INT64_MAX, min_ldt_l = INT64_MAX;
    for (const Ent& e : ents) {
        Partition p;
        p.key = e.key;
        p.token = e.token;
        int64_t ts = gen_ts(g, sst, e.id);
        min_ts = std::min(min_ts, ts);
        Unfiltered u;
        u.kind = Unfiltered::ROW;
        Row& r = u.row;
        r.cells.resize(1);
        if (gen_has_partition_del(g, sst, e.id)) {
            p.del.mfda = ts;
            p.del.ldt = (uint32_t)(g.base_ldt + (int64_t)(splitmix64(e.id ^ 0xDD) % 1000));
            min_ldt_l = std::min<int64_t>(min_ldt_l, p.del.ldt);
            t.parts.push_back(std::move(p));
            continue;  // partition deletion only, no row
        }
        if (gen_is_tombstone(g, sst, e.id)) {
            r.del.mfda = ts;
            r.del.ldt = (uint32_t)(g.base_ldt + (int64_t)(splitmix64(e.id ^ 0xEE) % 1000));
            min_ldt_l = std::min<int64_t>(min_ldt_l, r.del.ldt);
        } else {
            r.live.ts = ts;
            Cell c;
            c.ts = ts;
            gen_value(g, sst, e.id, c.value);
            r.cells[0] = std::move(c);
        }
        p.items.push_back(std::move(u));
        t.parts.push_back(std::move(p));
    }
    // EncodingStats for the header: collected from data (EncodingStats.Collector
    // during flush); NO_* map to epochs (EncodingStats.java:78-89)
    hs.min_ts = min_ts == INT64_MAX ? TIMESTAMP_EPOCH : min_ts;
    hs.min_ldt = min_ldt_l == INT64_MAX ? DELETION_TIME_EPOCH : min_ldt_l;
    hs.min_ttl = 0;
    t.header.stats = hs;
    return t;
}

}  // namespace oracle
