// ORACLE — test infrastructure + synthetic input generator (shared contract, see gen.h).
#include "gen.h"
#include <algorithm>

namespace oracle {

SSTable generate_sstable(const GenSpec& g, uint32_t sst) {
    SSTable t;
    if (g.snappy) t.comp.algo = Compressor::SNAPPY;
    t.generation = g.first_generation + sst;
    t.header.key_type = g.key_len > 8 ? CqlType::BYTES : CqlType::LONG;  // pk bigint / blob
    if (g.clustering_rows > 0) {
        uint32_t nck = g.ck_cols ? g.ck_cols : 1;
        t.header.clustering_types.assign(nck, g.ck_text ? CqlType::UTF8 : CqlType::LONG);
    }
    if (g.static_pct > 0)
        t.header.static_cols = {{bytes{'s', '0'}, CqlType::BYTES}};
    if (g.n_value_cols <= 1) {
        t.header.regular_cols = {{bytes{'v', 'a', 'l'}, CqlType::BYTES}};  // val blob
    } else {
        for (uint32_t cc = 0; cc < g.n_value_cols; cc++) {
            std::string nm = "val" + std::to_string(cc);
            t.header.regular_cols.push_back({bytes(nm.begin(), nm.end()), CqlType::BYTES});
        }
    }
    if (g.complex_pct > 0)  // one complex column, named to sort last
        t.header.regular_cols.push_back({bytes{'z', 'm'}, CqlType::MAP_BB});
    if (g.counter)  // counter tables: the single counter column replaces the blobs
        t.header.regular_cols = {{bytes{'c', 'n', 't'}, CqlType::COUNTER}};

    struct Ent { int64_t token; bytes key; uint64_t id; };
    std::vector<Ent> ents;
    ents.reserve(g.rows_per_sstable);
    for (uint64_t j = 0; j < g.rows_per_sstable; j++) {
        uint64_t id = gen_key_id(g, sst, j);
        bytes key = gen_key_bytes(g, id);
        int64_t tok = murmur3_token(key.data(), key.size());
        ents.push_back({tok, std::move(key), id});
    }
    std::sort(ents.begin(), ents.end(), [](const Ent& a, const Ent& b) {
        return compare_decorated_key(a.token, a.key.data(), a.key.size(),
                                     b.token, b.key.data(), b.key.size()) < 0;
    });

    int64_t min_ts = INT64_MAX, min_ldt_l = INT64_MAX;
    int32_t min_ttl = INT32_MAX;
    // counter context for (sst, key): shards from the sorted 8-id pool
    auto make_ctx = [&](uint32_t sst2, uint64_t id) {
        struct P { uint8_t id[16]; uint32_t idx; };
        static std::vector<P> pool = [] {
            std::vector<P> v(8);
            for (uint32_t i = 0; i < 8; i++) { gen_counter_id(i, v[i].id); v[i].idx = i; }
            std::sort(v.begin(), v.end(), [](const P& a, const P& b) {
                return memcmp(a.id, b.id, 16) < 0;
            });
            return v;
        }();
        bytes body;
        std::vector<int16_t> elts;
        int shard_no = 0;
        for (auto& pp2 : pool) {
            if (!gen_ctr_present(g, sst2, id, pp2.idx)) continue;
            int role = gen_ctr_role(pp2.idx);
            if (role == 0) elts.push_back((int16_t)(shard_no + INT16_MIN));
            else if (role == 1) elts.push_back((int16_t)shard_no);
            body.insert(body.end(), pp2.id, pp2.id + 16);
            int64_t ck2 = gen_ctr_clock(g, sst2, id, pp2.idx);
            int64_t cn = gen_ctr_count(g, sst2, id, pp2.idx);
            for (int b = 7; b >= 0; b--) body.push_back((uint8_t)((uint64_t)ck2 >> (8 * b)));
            for (int b = 7; b >= 0; b--) body.push_back((uint8_t)((uint64_t)cn >> (8 * b)));
            shard_no++;
        }
        bytes out;
        out.push_back((uint8_t)(elts.size() >> 8));
        out.push_back((uint8_t)elts.size());
        for (int16_t e2 : elts) {
            out.push_back((uint8_t)((uint16_t)e2 >> 8));
            out.push_back((uint8_t)e2);
        }
        out.insert(out.end(), body.begin(), body.end());
        return out;
    };
    // complex column "zm": dedup+sorted map cells, optional complexDeletion
    auto put_complex = [&](Row& r, uint32_t sst2, uint64_t id, uint32_t j, int64_t ts) {
        if (!gen_has_complex(g, sst2, id, j)) return;
        size_t ci = t.header.regular_cols.size() - 1;
        r.cells.resize(t.header.regular_cols.size());
        r.complex.resize(t.header.regular_cols.size());
        ComplexData cd;
        if (gen_has_cpx_del(g, sst2, id, j)) {
            cd.del.mfda = ts - 1;
            cd.del.ldt = gen_ldt(g, id * 5 + j, 0xCD);
            min_ts = std::min(min_ts, cd.del.mfda);
            min_ldt_l = std::min<int64_t>(min_ldt_l, cd.del.ldt);
        }
        uint32_t nc = gen_cpx_count(g, id, j);
        std::vector<uint32_t> pv;
        for (uint32_t e = 0; e < nc; e++) pv.push_back(gen_cpx_path_val(g, sst2, id, j, e));
        std::sort(pv.begin(), pv.end());
        pv.erase(std::unique(pv.begin(), pv.end()), pv.end());
        for (uint32_t p4 : pv) {
            Cell c;
            c.ts = ts - (int64_t)(p4 % 3);
            min_ts = std::min(min_ts, c.ts);
            c.path.resize(4);
            for (int b = 0; b < 4; b++) c.path[b] = (uint8_t)(p4 >> (8 * (3 - b)));
            uint64_t w = gen_cpx_value_word(g, sst2, id, j, p4);
            c.value.resize(8);
            for (int b = 0; b < 8; b++) c.value[b] = (uint8_t)(w >> (8 * b));
            cd.cells.push_back(std::move(c));
        }
        r.complex[ci] = std::move(cd);
    };
    for (const Ent& e : ents) {
        if (g.clustering_rows > 0) {
            // ---- wide partition: clustering rows + optional range tombstone ----
            Partition p;
            p.key = e.key;
            p.token = e.token;
            if (g.static_pct > 0 && gen_has_static(g, sst, e.id)) {
                Row& sr = p.static_row;
                sr.static_flag = true;
                sr.cells.resize(1);
                int64_t sts = gen_static_ts(g, sst, e.id);
                min_ts = std::min(min_ts, sts);
                sr.live.ts = sts;
                Cell sc;
                sc.ts = sts;
                gen_value(g, sst, e.id ^ 0xABCDEF57ULL, sc.value);
                sr.cells[0] = std::move(sc);
            }
            bool has_rt = gen_has_range_tomb(g, sst, e.id);
            int64_t rlo = 0, rhi = 0, rts = 0;
            uint32_t rldt = 0;
            if (has_rt) {
                gen_range_bounds(g, sst, e.id, &rlo, &rhi);
                rts = gen_range_ts(g, sst, e.id);
                rldt = gen_ldt(g, e.id, 0xCC);
                min_ts = std::min(min_ts, rts);
                min_ldt_l = std::min<int64_t>(min_ldt_l, rldt);
            }
            bool ck2 = g.ck_cols == 2;
            if (ck2 && has_rt) {
                // ck0-group PREFIX bounds (GPU mirror: k_gen_fill2 ck_cols==2)
                uint64_t r2 = splitmix64(g.seed ^ 0xB00BULL ^ e.id ^ ((uint64_t)sst << 32));
                uint32_t a2 = (uint32_t)(r2 % g.clustering_rows);
                uint32_t b2 = a2 + 1 + (uint32_t)((r2 >> 32) % (g.clustering_rows - a2));
                rlo = (int64_t)(a2 / 4) * 64;
                rhi = (int64_t)(b2 / 4) * 64 + 63;
            }
            auto ck_vals = [&](int64_t ckpos, bool is_row) {
                Clustering cv;
                if (!ck2) {
                    cv = {ClusterVal{ClusterVal::VALUE, gen_ck_bytes(g, ckpos, is_row)}};
                } else if (is_row) {
                    cv = {ClusterVal{ClusterVal::VALUE, gen_ck_bytes(g, (ckpos / 64) * 16, true)},
                          ClusterVal{ClusterVal::VALUE, gen_ck_bytes(g, ckpos % 64, true)}};
                } else {
                    cv = {ClusterVal{ClusterVal::VALUE, gen_ck_bytes(g, (ckpos / 64) * 16, false)}};
                }
                return cv;
            };
            bool rt_open = false, rt_done = false;
            for (uint32_t j = 0; j < g.clustering_rows; j++) {
                int64_t ck = ck2 ? (int64_t)(j / 4) * 64 + (int64_t)(j % 4) : gen_ck(g, e.id, j);
                if (has_rt && !rt_open && !rt_done && (ck2 ? ck >= rlo : ck > rlo)) {
                    Unfiltered u;
                    u.kind = Unfiltered::MARKER;
                    u.marker.kind = INCL_START;
                    u.marker.values = ck_vals(rlo, false);
                    u.marker.end_dt = DeletionTime{rts, rldt};
                    // marker sits at bound position rlo (before this row)
                    p.items.push_back(std::move(u));
                    rt_open = true;
                }
                Unfiltered u;
                u.kind = Unfiltered::ROW;
                Row& r = u.row;
                r.clustering = ck_vals(ck, true);
                uint32_t ncols = g.n_value_cols ? g.n_value_cols : 1;
                r.cells.resize(t.header.regular_cols.size());
                int64_t ts = gen_row_ts(g, sst, e.id, j);
                min_ts = std::min(min_ts, ts);
                if (gen_row_is_tombstone(g, sst, e.id, j)) {
                    r.del.mfda = ts;
                    r.del.ldt = gen_ldt(g, e.id * 1000 + j, 0xEE);
                    min_ldt_l = std::min<int64_t>(min_ldt_l, r.del.ldt);
                } else {
                    r.live.ts = ts;
                    bool expg = gen_row_expiring(g, sst, e.id, j);
                    if (expg) {
                        r.live.ttl = gen_ttl(g, e.id, j);
                        r.live.let = gen_let(g, e.id, j);
                        min_ldt_l = std::min(min_ldt_l, r.live.let);
                        min_ttl = std::min(min_ttl, r.live.ttl);
                    }
                    uint64_t seed_id = e.id ^ ((uint64_t)j << 52) ^ j;
                    for (uint32_t cc = 0; cc < ncols; cc++) {
                        if (gen_col_missing(g, sst, e.id, j, cc)) continue;
                        Cell cell;
                        cell.ts = ts;
                        if (expg) {
                            cell.ttl = r.live.ttl;
                            cell.ldt = ldt_to_u32(r.live.let);
                        }
                        if (g.counter) cell.value = make_ctx(sst, e.id ^ ((uint64_t)j << 20));
                        else gen_value(g, sst, gen_col_value_id(seed_id, cc), cell.value);
                        r.cells[cc] = std::move(cell);
                    }
                    put_complex(r, sst, e.id, j, ts);
                }
                p.items.push_back(std::move(u));
                int64_t ck_next = ck2 ? (int64_t)((j + 1) / 4) * 64 + (int64_t)((j + 1) % 4)
                                      : gen_ck(g, e.id, j + 1);
                if (has_rt && rt_open && j + 1 < g.clustering_rows && ck_next > rhi) {
                    Unfiltered m;
                    m.kind = Unfiltered::MARKER;
                    m.marker.kind = INCL_END;
                    m.marker.values = ck_vals(rhi, false);
                    m.marker.end_dt = DeletionTime{rts, rldt};
                    p.items.push_back(std::move(m));
                    rt_open = false;
                    rt_done = true;
                }
            }
            if (has_rt && rt_open) {
                Unfiltered m;
                m.kind = Unfiltered::MARKER;
                m.marker.kind = INCL_END;
                m.marker.values = ck_vals(rhi, false);
                m.marker.end_dt = DeletionTime{rts, rldt};
                p.items.push_back(std::move(m));
            }
            t.parts.push_back(std::move(p));
            continue;
        }
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
        uint32_t ncols = g.n_value_cols ? g.n_value_cols : 1;
        r.cells.resize(t.header.regular_cols.size());
        if (gen_is_tombstone(g, sst, e.id)) {
            r.del.mfda = ts;
            r.del.ldt = gen_ldt(g, e.id, 0xEE);
            min_ldt_l = std::min<int64_t>(min_ldt_l, r.del.ldt);
        } else {
            r.live.ts = ts;
            bool expg = gen_row_expiring(g, sst, e.id, 0);
            if (expg) {
                r.live.ttl = gen_ttl(g, e.id, 0);
                r.live.let = gen_let(g, e.id, 0);
                min_ldt_l = std::min(min_ldt_l, r.live.let);
                min_ttl = std::min(min_ttl, r.live.ttl);
            }
            for (uint32_t cc = 0; cc < ncols; cc++) {
                if (gen_col_missing(g, sst, e.id, 0, cc)) continue;
                Cell c;
                c.ts = ts;
                if (expg) {
                    c.ttl = r.live.ttl;
                    c.ldt = ldt_to_u32(r.live.let);
                }
                if (g.counter) c.value = make_ctx(sst, e.id);
                else gen_value(g, sst, gen_col_value_id(e.id, cc), c.value);
                r.cells[cc] = std::move(c);
            }
            put_complex(r, sst, e.id, 0, ts);
        }
        p.items.push_back(std::move(u));
        t.parts.push_back(std::move(p));
    }
    // EncodingStats for the header (collected at flush; NO_* map to epochs,
    // EncodingStats.java:78-89)
    t.header.stats.min_ts = min_ts == INT64_MAX ? TIMESTAMP_EPOCH : min_ts;
    t.header.stats.min_ldt = min_ldt_l == INT64_MAX ? DELETION_TIME_EPOCH : min_ldt_l;
    t.header.stats.min_ttl = min_ttl == INT32_MAX ? 0 : min_ttl;
    return t;
}

}  // namespace oracle
