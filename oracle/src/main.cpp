// ORACLE CLI — test infrastructure only (see util.h header note).
// Subcommands: roundtrip, dump, gen, compact, selftest, bench-compact.
#include "sstable.h"
#include "compact.h"
#include "gen.h"
#include "bti.h"
#include <chrono>
#include <cstring>
#include <algorithm>
#include <map>
#include <string>

using namespace oracle;

static std::map<std::string, std::string> parse_kv(int argc, char** argv, int start,
                                                   std::vector<std::string>& positional) {
    std::map<std::string, std::string> kv;
    for (int i = start; i < argc; i++) {
        std::string a = argv[i];
        auto eq = a.find('=');
        if (eq == std::string::npos) positional.push_back(a);
        else kv[a.substr(0, eq)] = a.substr(eq + 1);
    }
    return kv;
}

static int cmd_dump(const std::string& base) {
    SSTable t = read_sstable(base, true);
    printf("sstable %s\n  partitions=%zu uncompressed=%zu\n", base.c_str(), t.parts.size(),
           t.raw_data_uncompressed.size());
    printf("  header: key=%s clustering=%zu regulars=%zu stats(minTs=%lld minLdt=%lld minTTL=%d)\n",
           cql_type_name(t.header.key_type), t.header.clustering_types.size(),
           t.header.regular_cols.size(), (long long)t.header.stats.min_ts,
           (long long)t.header.stats.min_ldt, t.header.stats.min_ttl);
    size_t show = std::min<size_t>(5, t.parts.size());
    for (size_t i = 0; i < show; i++) {
        auto& p = t.parts[i];
        printf("  part[%zu] token=%lld keylen=%zu del=%s items=%zu\n", i, (long long)p.token,
               p.key.size(), p.del.live() ? "live" : "set", p.items.size());
    }
    return 0;
}

static int cmd_roundtrip(const std::string& base) {
    SSTable t = read_sstable(base, true);
    // column_index_size is config, not recorded in the sstable: our writer uses
    // the 64 KiB default; the reference test fixtures were written with 4 KiB
    // (test/conf/cassandra.yaml:24). Accept either.
    bytes data, index;
    for (auto& p : t.parts) serialize_partition(p, t.header, data, index, COLUMN_INDEX_SIZE);
    if (!t.bti) {
        bytes want_idx = read_file(base + "-Index.db");
        if (index != want_idx) {
            bytes d2, i2;
            for (auto& p : t.parts) serialize_partition(p, t.header, d2, i2, 4096);
            if (i2 == want_idx) { index = std::move(i2); t.column_index_size = 4096; }
        }
    } else {
        // column_index_size is config, not recorded: try the 64 KiB default,
        // fall back to the fixtures' 4 KiB test setting on mismatch
        bytes want_rows = read_file(base + "-Rows.db");
        SSTable probe;
        probe.header = t.header;
        probe.comp = t.comp;
        probe.parts = t.parts;
        probe.partitioner = t.partitioner;
        probe.column_index_size = COLUMN_INDEX_SIZE;
        if (write_sstable(probe, true).rows_db != want_rows) t.column_index_size = 4096;
    }
    int rc = 0;
    auto check = [&](const char* what, const bytes& got, const bytes& want) {
        if (got == want) { printf("OK  %-18s %zu bytes\n", what, want.size()); return; }
        rc = 1;
        size_t n = std::min(got.size(), want.size()), i = 0;
        while (i < n && got[i] == want[i]) i++;
        printf("FAIL %-18s got=%zu want=%zu first_diff=%zu (got=%02x want=%02x)\n", what,
               got.size(), want.size(), i, i < got.size() ? got[i] : 0, i < want.size() ? want[i] : 0);
    };
    check("Data(uncompressed)", data, t.raw_data_uncompressed);
    ChunkedOut co = chunk_compress(data, t.comp);
    check("Data.db", co.file, read_file(base + "-Data.db"));
    check("CompressionInfo.db", make_compression_info(t.comp, data.size(), co.offsets),
          read_file(base + "-CompressionInfo.db"));
    if (!t.bti) check("Index.db", index, read_file(base + "-Index.db"));
    if (getenv("ORACLE_DUMP_INDEX")) write_file(std::string(getenv("ORACLE_DUMP_INDEX")), index);
    {
        std::string d = std::to_string(crc32(co.file.data(), co.file.size()));
        check("Digest.crc32", bytes(d.begin(), d.end()), read_file(base + "-Digest.crc32"));
    }
    {
        // Filter.db: recompute bloom from keys; Statistics COMPACTION blob:
        // the HLL is real now (sparse fixture-pinned), compare its slice
        SSTable tmp;
        tmp.header = t.header;
        tmp.comp = t.comp;
        tmp.parts = t.parts;
        tmp.partitioner = t.partitioner;
        tmp.column_index_size = t.column_index_size;
        WriterOut w = write_sstable(tmp, t.bti);
        check("Filter.db", w.filter, read_file(base + "-Filter.db"));
        if (t.bti) {
            check("Partitions.db", w.partitions_db, read_file(base + "-Partitions.db"));
            check("Rows.db", w.rows_db, read_file(base + "-Rows.db"));
        }
        auto comp_slice = [](const bytes& st) -> bytes {
            if (st.size() < 4) return {};
            Reader r(st);
            uint32_t n = r.be32();
            r.be32();  // toc CRC32 (checksummed metadata v2)
            uint32_t start = 0, end2 = 0;
            for (uint32_t i = 0; i < n; i++) {
                uint32_t typ = r.be32();
                uint32_t off = r.be32();
                if (typ == 1) start = off;
                if (typ == 2) end2 = off;
            }
            if (!start || end2 <= start + 4) return {};
            return bytes(st.begin() + start, st.begin() + end2 - 4);  // strip component CRC
        };
        check("Stats[COMPACTION]", comp_slice(w.statistics), comp_slice(t.raw_statistics));
        if (!t.bti) check("Summary.db", w.summary, read_file(base + "-Summary.db"));
    }
    return rc;
}

// rewrite: read an sstable and write all components to a new base (HLL /
// writer-path inspection without compaction semantics)
static int cmd_rewrite(const std::string& inbase, const std::string& outbase) {
    SSTable t = read_sstable(inbase, true);
    if (const char* e = getenv("ORACLE_CIS")) t.column_index_size = (uint32_t)atoi(e);
    WriterOut w = write_sstable(t, t.bti);
    write_components(w, outbase);
    printf("rewrote %zu partitions\n", t.parts.size());
    return 0;
}

static GenSpec spec_from_kv(std::map<std::string, std::string>& kv) {
    GenSpec g;
    auto geti = [&](const char* k, auto def) -> int64_t {
        return kv.count(k) ? (int64_t)strtoll(kv[k].c_str(), nullptr, 10) : (int64_t)def;
    };
    g.seed = geti("seed", g.seed);
    g.n_sstables = (uint32_t)geti("n", g.n_sstables);
    g.rows_per_sstable = geti("rows", g.rows_per_sstable);
    g.overlap_pct = (uint32_t)geti("overlap", g.overlap_pct);
    g.value_len = (uint32_t)geti("vlen", g.value_len);
    g.value_repeat_pct = (uint32_t)geti("vrep", g.value_repeat_pct);
    g.tombstone_pct = (uint32_t)geti("tomb", g.tombstone_pct);
    g.partition_del_pct = (uint32_t)geti("pdel", g.partition_del_pct);
    g.clustering_rows = (uint32_t)geti("crows", g.clustering_rows);
    g.range_tomb_pct = (uint32_t)geti("rtomb", g.range_tomb_pct);
    g.key_len = (uint32_t)geti("keylen", g.key_len);
    g.ck_text = (uint32_t)geti("cktext", g.ck_text);
    g.ck_cols = (uint32_t)geti("ckcols", g.ck_cols);
    g.static_pct = (uint32_t)geti("statics", g.static_pct);
    g.base_ts = geti("ts0", g.base_ts);
    g.base_ldt = geti("ldt0", g.base_ldt);
    g.n_value_cols = (uint32_t)geti("ncols", g.n_value_cols);
    g.col_missing_pct = (uint32_t)geti("colmiss", g.col_missing_pct);
    g.first_generation = geti("gen0", g.first_generation);
    g.snappy = (uint32_t)geti("snappy", g.snappy);
    g.ttl_pct = (uint32_t)geti("ttl", g.ttl_pct);
    g.bti = (uint32_t)geti("bti", g.bti);
    g.counter = (uint32_t)geti("counter", g.counter);
    g.complex_pct = (uint32_t)geti("cpx", g.complex_pct);
    g.complex_del_pct = (uint32_t)geti("cpxdel", g.complex_del_pct);
    return g;
}

static int cmd_gen(const std::string& outdir, std::map<std::string, std::string>& kv) {
    GenSpec g = spec_from_kv(kv);
    uint64_t total_unc = 0;
    for (uint32_t s = 0; s < g.n_sstables; s++) {
        SSTable t = generate_sstable(g, s);
        WriterOut w = write_sstable(t, g.bti != 0);
        std::string base = g.bti ? outdir + "/da-" + std::to_string(t.generation) + "-bti"
                                 : outdir + "/oa-" + std::to_string(t.generation) + "-big";
        write_components(w, base);
        if (kv.count("dump")) {
            t.bti = g.bti != 0;  // write_sstable takes the flag separately
            write_memdump(t, base + ".memdump");
        }
        total_unc += w.uncompressed_data_len;
        printf("wrote %s: parts=%llu uncompressed=%llu compressed=%zu\n", base.c_str(),
               (unsigned long long)w.partition_count, (unsigned long long)w.uncompressed_data_len,
               w.data_db.size());
    }
    printf("total_uncompressed=%llu\n", (unsigned long long)total_unc);
    return 0;
}

// flush rows from a text file: "keyhex ts valhex" | "keyhex ts T ldt"
static int cmd_flush(const std::string& outbase, const std::string& rowfile) {
    auto unhex = [](const std::string& s) {
        bytes b;
        for (size_t i = 0; i + 1 < s.size(); i += 2)
            b.push_back((uint8_t)strtoul(s.substr(i, 2).c_str(), nullptr, 16));
        return b;
    };
    SSTable t;
    t.generation = 1;
    t.header.key_type = CqlType::BYTES;
    t.header.regular_cols = {{bytes{'v', 'a', 'l'}, CqlType::BYTES}};
    FILE* f = fopen(rowfile.c_str(), "r");
    if (!f) { fprintf(stderr, "cannot open %s\n", rowfile.c_str()); return 1; }
    char k[4096], v[65536];
    long long ts;
    int64_t min_ts = INT64_MAX, min_ldt_l = INT64_MAX;
    while (fscanf(f, "%4095s %lld %65535s", k, &ts, v) == 3) {
        Partition p;
        p.key = unhex(k);
        p.set_token();
        Unfiltered u;
        u.kind = Unfiltered::ROW;
        Row& r = u.row;
        r.cells.resize(1);
        min_ts = std::min(min_ts, (int64_t)ts);
        if (v[0] == 'T') {
            long long ldt;
            if (fscanf(f, "%lld", &ldt) != 1) break;
            r.del.mfda = ts;
            r.del.ldt = (uint32_t)ldt;
            min_ldt_l = std::min(min_ldt_l, (int64_t)ldt);
        } else {
            r.live.ts = ts;
            Cell c;
            c.ts = ts;
            c.value = unhex(v);
            r.cells[0] = std::move(c);
        }
        p.items.push_back(std::move(u));
        t.parts.push_back(std::move(p));
    }
    fclose(f);
    std::sort(t.parts.begin(), t.parts.end(), [](const Partition& a, const Partition& b) {
        return compare_decorated_key(a.token, a.key.data(), a.key.size(), b.token,
                                     b.key.data(), b.key.size()) < 0;
    });
    t.header.stats.min_ts = min_ts == INT64_MAX ? TIMESTAMP_EPOCH : min_ts;
    t.header.stats.min_ldt = min_ldt_l == INT64_MAX ? DELETION_TIME_EPOCH : min_ldt_l;
    t.header.stats.min_ttl = 0;
    WriterOut w = write_sstable(t);
    write_components(w, outbase);
    printf("flushed %zu partitions\n", t.parts.size());
    return 0;
}

static int cmd_scrub(const std::string& outbase, const std::string& inbase) {
    ScrubResult sr = scrub_sstable(inbase, outbase);
    printf("{\"partitions_kept\": %llu, \"partitions_dropped\": %llu}\n",
           (unsigned long long)sr.kept, (unsigned long long)sr.dropped);
    return 0;
}

static int cmd_compact(const std::string& outbase, std::vector<std::string>& inputs,
                       std::map<std::string, std::string>& kv) {
    using clk = std::chrono::steady_clock;
    CompactionJob job;
    auto t0 = clk::now();
    uint64_t input_unc = 0;
    for (auto& in : inputs) {
        job.inputs.push_back(read_sstable(in, true));
        input_unc += job.inputs.back().raw_data_uncompressed.size();
        job.inputs.back().raw_data_uncompressed.clear();
        job.inputs.back().raw_statistics.clear();
    }
    job.now_sec = kv.count("now") ? strtoll(kv["now"].c_str(), nullptr, 10) : 1800000000LL;
    job.gc_before = kv.count("gcbefore") ? strtoll(kv["gcbefore"].c_str(), nullptr, 10) : INT64_MIN;
    job.never_purge = kv.count("nevergc") && kv["nevergc"] == "1";
    job.cell_level_gc = kv.count("cellgc") && kv["cellgc"] == "1";
    if (kv.count("tombsrc")) {
        // comma-separated sstable bases whose tombstones shadow the data
        std::string s = kv["tombsrc"];
        size_t p0 = 0;
        while (p0 < s.size()) {
            size_t c = s.find(',', p0);
            if (c == std::string::npos) c = s.size();
            job.tomb_sources.push_back(read_sstable(s.substr(p0, c - p0), true));
            p0 = c + 1;
        }
    }
    if (kv.count("ov")) {
        // overlap table "lo:hi:ts[:FilterPath],..." — FilterPath loads the
        // sstable's Filter.db for the per-key bloom-checked evaluator
        std::string sv = kv["ov"];
        size_t p0 = 0;
        while (p0 < sv.size()) {
            size_t cm = sv.find(',', p0);
            if (cm == std::string::npos) cm = sv.size();
            std::string one = sv.substr(p0, cm - p0);
            PurgeRange r{};
            size_t a = one.find(':'), b = one.find(':', a + 1), c = one.find(':', b + 1);
            r.tok_lo = strtoll(one.substr(0, a).c_str(), nullptr, 10);
            r.tok_hi = strtoll(one.substr(a + 1, b - a - 1).c_str(), nullptr, 10);
            r.min_ts = strtoll(one.substr(b + 1, (c == std::string::npos ? one.size() : c) - b - 1).c_str(), nullptr, 10);
            if (c != std::string::npos) {
                bytes f = read_file(one.substr(c + 1));
                Reader rd(f);
                r.bloom_k = (int32_t)rd.be32();
                uint32_t words = rd.be32();
                bytes bits = rd.take((size_t)words * 8);
                r.bloom.assign(bits.begin(), bits.end());
            }
            job.overlaps.push_back(r);
            p0 = cm + 1;
        }
    }
    if (kv.count("ranges")) {
        // "lo:hi,lo:hi" inclusive keep-ranges; invertranges=1 keeps the complement
        std::string s = kv["ranges"];
        size_t p0 = 0;
        while (p0 < s.size()) {
            size_t cm = s.find(',', p0);
            if (cm == std::string::npos) cm = s.size();
            std::string one = s.substr(p0, cm - p0);
            size_t c2 = one.find(':');
            PurgeRange r{};
            r.tok_lo = strtoll(one.substr(0, c2).c_str(), nullptr, 10);
            r.tok_hi = strtoll(one.substr(c2 + 1).c_str(), nullptr, 10);
            job.keep_ranges.push_back(r);
            p0 = cm + 1;
        }
        job.invert_ranges = kv.count("invertranges") && kv["invertranges"] == "1";
    }
    if (kv.count("shard")) {
        auto s = kv["shard"];
        auto c = s.find(':');
        job.has_shard = true;
        job.shard_lo = strtoll(s.substr(0, c).c_str(), nullptr, 10);
        job.shard_hi = strtoll(s.substr(c + 1).c_str(), nullptr, 10);
    }
    auto t1 = clk::now();
    CompactionResult res = compact(job);
    auto t2 = clk::now();
    // output format follows the inputs (a da compaction emits da)
    bool out_bti = !job.inputs.empty() && job.inputs[0].bti;
    if (const char* e = getenv("ORACLE_CIS"))
        res.out.column_index_size = (uint32_t)atoi(e);
    WriterOut w = write_sstable(res.out, out_bti);
    write_components(w, outbase);
    auto t3 = clk::now();
    double read_s = std::chrono::duration<double>(t1 - t0).count();
    double merge_s = std::chrono::duration<double>(t2 - t1).count();
    double write_s = std::chrono::duration<double>(t3 - t2).count();
    printf("{\"input_uncompressed_bytes\": %llu, \"partitions_in\": %llu, \"partitions_out\": %llu, "
           "\"rows_in\": %llu, \"rows_out\": %llu, \"read_s\": %.3f, \"merge_s\": %.3f, "
           "\"write_s\": %.3f, \"total_s\": %.3f, \"mb_per_s\": %.2f}\n",
           (unsigned long long)input_unc, (unsigned long long)res.partitions_in,
           (unsigned long long)res.partitions_out, (unsigned long long)res.rows_in,
           (unsigned long long)res.rows_out, read_s, merge_s, write_s, read_s + merge_s + write_s,
           input_unc / 1e6 / (read_s + merge_s + write_s));
    return 0;
}

static int cmd_selftest() {
    // vint round trips incl. boundaries
    for (uint64_t v : std::initializer_list<uint64_t>{0, 1, 127, 128, 16383, 16384,
                       (uint64_t)INT64_MAX, 0xFFFFFFFFFFFFFFFFULL, 0x8000000000000000ULL}) {
        bytes b;
        put_unsigned_vint(b, v);
        Reader r(b);
        if (read_unsigned_vint(r) != v || r.pos != b.size()) { printf("vint FAIL %llx\n", (unsigned long long)v); return 1; }
    }
    // murmur3: empty key handling + known stability (self-consistency)
    if (murmur3_token(nullptr, 0) != INT64_MIN) { printf("token(empty) FAIL\n"); return 1; }
    printf("selftest OK\n");
    return 0;
}

int main(int argc, char** argv) {
    if (argc < 2) { fprintf(stderr, "usage: oracle_tool <dump|roundtrip|gen|compact|selftest> ...\n"); return 2; }
    std::string cmd = argv[1];
    try {
        std::vector<std::string> pos;
        auto kv = parse_kv(argc, argv, 2, pos);
        if (cmd == "selftest") return cmd_selftest();
        if (cmd == "dump") return cmd_dump(pos.at(0));
        if (cmd == "roundtrip") return cmd_roundtrip(pos.at(0));
        if (cmd == "rewrite") return cmd_rewrite(pos.at(0), pos.at(1));
        if (cmd == "dumpsst") {
            // memdump of ANY existing sstable (gen writes dumps only for its
            // own tables; this lets compact outputs — pdel-only partitions,
            // counter tables, purged shapes — round-trip through the
            // gpuc_flush_table parity harness)
            SSTable t = read_sstable(pos.at(0), false);
            write_memdump(t, pos.at(1));
            return 0;
        }
        if (cmd == "validate") {
            // validate <outfile> <inputs...> [now= gcbefore= ...]: the
            // VALIDATION compaction epilogue (CompactionManager.doValidation
            // + Validator.rowHash): merge+purge, then per-partition repair
            // digests written as (token i64 BE, 32-byte hash) records
            CompactionJob job;
            for (size_t i2 = 1; i2 < pos.size(); i2++)
                job.inputs.push_back(read_sstable(pos[i2], true));
            job.now_sec = kv.count("now") ? strtoll(kv["now"].c_str(), nullptr, 10) : 1800000000LL;
            job.gc_before = kv.count("gcbefore") ? strtoll(kv["gcbefore"].c_str(), nullptr, 10) : INT64_MIN;
            CompactionResult res = compact(job);
            bytes outb;
            for (const Partition& p2 : res.out.parts) {
                for (int b = 7; b >= 0; b--) outb.push_back((uint8_t)((uint64_t)p2.token >> (8 * b)));
                uint8_t hsh[32];
                validator_digest(p2, res.out.header, hsh);
                outb.insert(outb.end(), hsh, hsh + 32);
            }
            write_file(pos.at(0), outb);
            printf("validated %zu partitions\n", res.out.parts.size());
            return 0;
        }
        if (cmd == "btidump") {
            // BTI (da) Partitions.db [+ Rows.db] dump — reader scaffolding
            bytes pf = read_file(pos.at(0) + "-Partitions.db");
            BtiPartitionsFile bp = read_bti_partitions(pf);
            printf("{\"key_count\": %llu, \"root\": %llu, \"entries\": [",
                   (unsigned long long)bp.key_count, (unsigned long long)bp.root_pos);
            bytes rf;
            try { rf = read_file(pos.at(0) + "-Rows.db"); } catch (...) {}
            for (size_t i = 0; i < bp.entries.size(); i++) {
                auto& e = bp.entries[i];
                std::string pfx;
                for (uint8_t b : e.prefix) { char t[4]; snprintf(t, 4, "%02x", b); pfx += t; }
                printf("%s{\"prefix\": \"%s\", \"hash\": %d, ", i ? ", " : "",
                       pfx.c_str(), e.has_hash ? e.hash : -1);
                if (e.idxpos < 0) {
                    printf("\"data_pos\": %lld}", (long long)~e.idxpos);
                } else {
                    printf("\"rowindex_pos\": %lld", (long long)e.idxpos);
                    if (!rf.empty()) {
                        BtiRowIndexBlock rb = read_bti_row_index(rf, (uint64_t)e.idxpos);
                        printf(", \"data_pos\": %llu, \"index_blocks\": %llu, \"trie_payloads\": %zu",
                               (unsigned long long)rb.data_pos,
                               (unsigned long long)rb.row_count, rb.entries.size());
                    }
                    printf("}");
                }
            }
            printf("], \"first_key_len\": %zu, \"last_key_len\": %zu}\n",
                   bp.first_key.size(), bp.last_key.size());
            return 0;
        }
        if (cmd == "gen") return cmd_gen(pos.at(0), kv);
        if (cmd == "flush") return cmd_flush(pos.at(0), pos.at(1));
        if (cmd == "scrub") return cmd_scrub(pos.at(0), pos.at(1));
        if (cmd == "compact") {
            std::string outbase = pos.at(0);
            std::vector<std::string> ins(pos.begin() + 1, pos.end());
            return cmd_compact(outbase, ins, kv);
        }
    } catch (const std::exception& e) {
        fprintf(stderr, "error: %s\n", e.what());
        return 1;
    }
    fprintf(stderr, "unknown command %s\n", cmd.c_str());
    return 2;
}
