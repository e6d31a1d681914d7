#include "bti.h"
#include <functional>
#include <stdexcept>

namespace oracle {

// ---------------------------------------------------------------------------
// trie node decoding (TrieNode.java; layouts per BtiFormat.md "Trie nodes").
// Pointers are backward distances from the node position.
// ---------------------------------------------------------------------------
struct Node {
    int type = 0, pb = 0;
    uint64_t ppos = 0;                      // payload position (pb != 0)
    std::vector<std::pair<uint8_t, uint64_t>> children;  // transition -> pos
};

static uint64_t be(const bytes& f, uint64_t p, int n) {
    uint64_t v = 0;
    for (int i = 0; i < n; i++) v = (v << 8) | f.at(p + i);
    return v;
}

static Node decode_node(const bytes& f, uint64_t pos) {
    Node n;
    uint8_t b0 = f.at(pos);
    n.type = b0 >> 4;
    n.pb = b0 & 0x0F;
    auto add = [&](uint8_t t, uint64_t dist) {
        if (dist) n.children.push_back({t, pos - dist});
    };
    switch (n.type) {
        case 0:  // PAYLOAD_ONLY
            n.ppos = pos + 1;
            break;
        case 1:  // SINGLE_NOPAYLOAD_4: 4 ptr bits + transition byte
            add(f.at(pos + 1), n.pb);
            n.pb = 0;
            break;
        case 2:  // SINGLE_8: pb, transition, 8-bit ptr
            add(f.at(pos + 1), f.at(pos + 2));
            n.ppos = pos + 3;
            break;
        case 3:  // SINGLE_NOPAYLOAD_12: 4+8 ptr bits, transition
            add(f.at(pos + 2), ((uint64_t)n.pb << 8) | f.at(pos + 1));
            n.pb = 0;
            break;
        case 4:  // SINGLE_16
            add(f.at(pos + 1), be(f, pos + 2, 2));
            n.ppos = pos + 4;
            break;
        case 5: case 7: case 8: case 9: {  // SPARSE_8/16/24/40
            int w = n.type == 5 ? 1 : n.type == 7 ? 2 : n.type == 8 ? 3 : 5;
            int cc = f.at(pos + 1);
            for (int i = 0; i < cc; i++)
                add(f.at(pos + 2 + i), be(f, pos + 2 + cc + (uint64_t)i * w, w));
            n.ppos = pos + 2 + cc + (uint64_t)cc * w;
            break;
        }
        case 6: {  // SPARSE_12: packed 12-bit pointers
            int cc = f.at(pos + 1);
            uint64_t pbase = pos + 2 + cc;
            for (int i = 0; i < cc; i++) {
                uint64_t byteoff = pbase + (uint64_t)(i * 3) / 2;
                uint64_t v = (i % 2 == 0) ? (be(f, byteoff, 2) >> 4)
                                          : (be(f, byteoff, 2) & 0xFFF);
                add(f.at(pos + 2 + i), v);
            }
            n.ppos = pos + 2 + cc + ((uint64_t)cc * 3 + 1) / 2;
            break;
        }
        case 10: {  // DENSE_12
            int start = f.at(pos + 1), len = f.at(pos + 2) + 1;
            uint64_t pbase = pos + 3;
            for (int i = 0; i < len; i++) {
                uint64_t byteoff = pbase + (uint64_t)(i * 3) / 2;
                uint64_t v = (i % 2 == 0) ? (be(f, byteoff, 2) >> 4)
                                          : (be(f, byteoff, 2) & 0xFFF);
                add((uint8_t)(start + i), v);
            }
            n.ppos = pos + 3 + ((uint64_t)len * 3 + 1) / 2;
            break;
        }
        case 11: case 12: case 13: case 14: case 15: {  // DENSE_16..40, LONG
            int w = n.type == 11 ? 2 : n.type == 12 ? 3 : n.type == 13 ? 4
                                                        : n.type == 14 ? 5 : 8;
            int start = f.at(pos + 1), len = f.at(pos + 2) + 1;
            for (int i = 0; i < len; i++)
                add((uint8_t)(start + i), be(f, pos + 3 + (uint64_t)i * w, w));
            n.ppos = pos + 3 + (uint64_t)len * w;
            break;
        }
        default:
            throw std::runtime_error("bad trie node type");
    }
    return n;
}

static int64_t sign_extend(uint64_t v, int nbytes) {
    if (nbytes == 0) return 0;
    uint64_t sign = 1ull << (nbytes * 8 - 1);
    return (int64_t)((v ^ sign) - sign);
}

// DFS in byte order == key order
static void walk(const bytes& f, uint64_t pos, bytes& prefix,
                 const std::function<void(const bytes&, int, uint64_t)>& on_payload) {
    Node n = decode_node(f, pos);
    if (n.pb) on_payload(prefix, n.pb, n.ppos);
    for (auto& [t, cpos] : n.children) {
        prefix.push_back(t);
        walk(f, cpos, prefix, on_payload);
        prefix.pop_back();
    }
}

BtiPartitionsFile read_bti_partitions(const bytes& file) {
    BtiPartitionsFile out;
    if (file.size() < 24) throw std::runtime_error("Partitions.db too short");
    uint64_t p = file.size();
    out.root_pos = be(file, p - 8, 8);
    out.key_count = be(file, p - 16, 8);
    uint64_t keys_pos = be(file, p - 24, 8);
    {   // smallest + largest key, each with a 16-bit length prefix
        uint64_t kp = keys_pos;
        uint16_t l1 = (uint16_t)be(file, kp, 2);
        out.first_key = bytes(file.begin() + kp + 2, file.begin() + kp + 2 + l1);
        kp += 2 + l1;
        uint16_t l2 = (uint16_t)be(file, kp, 2);
        out.last_key = bytes(file.begin() + kp + 2, file.begin() + kp + 2 + l2);
    }
    bytes prefix;
    walk(file, out.root_pos, prefix, [&](const bytes& pf, int pb, uint64_t ppos) {
        BtiEntry e;
        e.prefix = pf;
        if (pb >= 8) {
            e.has_hash = true;
            e.hash = file.at(ppos);
            e.idxpos = sign_extend(be(file, ppos + 1, pb - 7), pb - 7);
        } else {
            e.idxpos = sign_extend(be(file, ppos, pb), pb);
        }
        out.entries.push_back(std::move(e));
    });
    return out;
}

BtiRowIndexBlock read_bti_row_index(const bytes& file, uint64_t index_pos) {
    BtiRowIndexBlock out;
    Reader r(file);
    r.skip(index_pos);
    uint16_t klen = r.be16();
    out.partition_key = r.take(klen);
    // TrieIndexEntry.serialize layout; the root delta is relative to the
    // entry start = the position right after the short-length key
    // (BtiTableReader.java:193 captures getFilePointer() there)
    uint64_t base = index_pos + 2 + klen;
    out.data_pos = read_unsigned_vint(r);
    int64_t delta = read_vint(r);
    out.root_pos = (uint64_t)((int64_t)base + delta);
    out.row_count = read_unsigned_vint(r);
    // DeletionTime.Serializer for hasUIntDeletionTime versions
    // (DeletionTime.java:205-245): 0x80 byte == LIVE, else 8-byte mfda
    // (sign bit clear) + 4-byte unsigned ldt
    {
        uint8_t flags = r.u8();
        if (flags & 0x80) {
            out.partition_del = DeletionTime{};  // LIVE
        } else {
            uint64_t rest = 0;
            for (int i = 0; i < 7; i++) rest = (rest << 8) | r.u8();
            out.partition_del.mfda = (int64_t)(((uint64_t)flags << 56) | rest);
            out.partition_del.ldt = r.be32();
        }
    }
    bytes prefix;
    walk(file, out.root_pos, prefix, [&](const bytes& pf, int pb, uint64_t ppos) {
        BtiRowIndexEntry e;
        e.prefix = pf;
        int obytes = pb & 7;
        e.offset = be(file, ppos, obytes);
        if (pb >= 8) {
            e.has_open = true;
            uint8_t flags = file.at(ppos + obytes);
            if (flags & 0x80) {
                e.open_dt = DeletionTime{};  // LIVE
            } else {
                e.open_dt.mfda = (int64_t)be(file, ppos + obytes, 8);
                e.open_dt.ldt = (uint32_t)be(file, ppos + obytes + 8, 4);
            }
        }
        out.entries.push_back(std::move(e));
    });
    return out;
}

}  // namespace oracle
