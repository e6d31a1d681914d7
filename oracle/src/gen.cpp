// ORACLE — test infrastructure + synthetic input generator (shared contract, see gen.h).
#include "gen.h"
#include <algorithm>

namespace oracle {

SSTable generate_sstable(const GenSpec& g, uint32_t sst) {
    SSTable t;
    t.generation = g.first_generation + sst;
    t.header.key_type = CqlType::LONG;  // pk bigint
    t.header.regular_cols = {{bytes{'v', 'a', 'l'}, CqlType::BYTES}};  // val blob

    struct Ent { int64_t token; bytes key; uint64_t id; };
    std::vector<Ent> ents;
    ents.reserve(g.rows_per_sstable);
    for (uint64_t j = 0; j < g.rows_per_sstable; j++) {
        uint64_t id = gen_key_id(g, sst, j);
        bytes key(8);
        for (int b = 0; b < 8; b++) key[b] = (uint8_t)(id >> (8 * (7 - b)));  // LongType BE
        int64_t tok = murmur3_token(key.data(), key.size());
        ents.push_back({tok, std::move(key), id});
    }
    std::sort(ents.begin(), ents.end(), [](const Ent& a, const Ent& b) {
        return compare_decorated_key(a.token, a.key.data(), a.key.size(),
                                     b.token, b.key.data(), b.key.size()) < 0;
    });

    int64_t min_ts = INT64_MAX, min_ldt_l = INT64_MAX;
    for (const Ent& e : ents) {
        Partition p;
        p.key = e.key;
        p.token = e.token;
        int64_t ts = gen_ts(g, sst, e.id);
        min_ts = std::min(min_ts, ts);
        if (gen_has_partition_del(g, sst, e.id)) {
            p.del.mfda = ts;
            p.del.ldt = gen_ldt(g, e.id, 0xDD);
            min_ldt_l = std::min<int64_t>(min_ldt_l, p.del.ldt);
            t.parts.push_back(std::move(p));
            continue;  // partition deletion only, no row
        }
        Unfiltered u;
        u.kind = Unfiltered::ROW;
        Row& r = u.row;
        r.cells.resize(1);
        if (gen_is_tombstone(g, sst, e.id)) {
            r.del.mfda = ts;
            r.del.ldt = gen_ldt(g, e.id, 0xEE);
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
    // EncodingStats for the header (collected at flush; NO_* map to epochs,
    // EncodingStats.java:78-89)
    t.header.stats.min_ts = min_ts == INT64_MAX ? TIMESTAMP_EPOCH : min_ts;
    t.header.stats.min_ldt = min_ldt_l == INT64_MAX ? DELETION_TIME_EPOCH : min_ldt_l;
    t.header.stats.min_ttl = 0;
    return t;
}

}  // namespace oracle
