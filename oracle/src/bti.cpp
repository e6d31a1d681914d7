#include "bti.h"
#include <algorithm>
#include <functional>
#include <memory>
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
        e.pb = pb;
        int obytes = pb & 7;
        e.offset = be(file, ppos, obytes);
        size_t plen = obytes;
        if (pb >= 8) {
            e.has_open = true;
            uint8_t flags = file.at(ppos + obytes);
            if (flags & 0x80) {
                e.open_dt = DeletionTime{};  // LIVE
                plen += 1;
            } else {
                e.open_dt.mfda = (int64_t)be(file, ppos + obytes, 8);
                e.open_dt.ldt = (uint32_t)be(file, ppos + obytes + 8, 4);
                plen += 12;
            }
        }
        e.raw_payload = bytes(file.begin() + ppos, file.begin() + ppos + plen);
        out.entries.push_back(std::move(e));
    });
    return out;
}

// ---------------------------------------------------------------------------
// WRITER (IncrementalTrieWriterPageAware restatement; 4096-byte pages)
// ---------------------------------------------------------------------------
namespace btiw {

constexpr int PAGE = 4096;

struct WNode {
    int transition = 0;
    std::vector<std::unique_ptr<WNode>> children;  // ascending transition
    int pb = 0;           // 4 payload bits (0 == no payload)
    bytes payload;        // serialized payload bytes following the node
    int branch_size = -1;
    int node_size = -1;
    bool oop_children = true;   // BaseNode default (forced true pre-complete)
    bool oop_in_branch = false;
    int64_t file_pos = -1;
};

struct Dest {
    bytes buf;
    int64_t position() const { return (int64_t)buf.size(); }
    int bytes_left_in_page() const { return PAGE - (int)(buf.size() % PAGE); }
    void pad_to_page() { while (buf.size() % PAGE) buf.push_back(0); }
    void u8(uint8_t b) { buf.push_back(b); }
    void be(uint64_t v, int n) {
        for (int i = n - 1; i >= 0; i--) buf.push_back((uint8_t)(v >> (8 * i)));
    }
};

// SizedInts.nonZeroSize: significant bits + sign, rounded up to bytes
static int sized_int_size(int64_t v) {
    uint64_t u = v < 0 ? ~(uint64_t)v : (uint64_t)v;
    int bits = 0;
    while (u >> bits) bits++;
    return (bits + 1 + 7) / 8;
}

static int payload_size(const WNode& n) { return (int)n.payload.size(); }

// max (most negative) position delta per Node.maxPositionDelta
static int64_t max_position_delta(const WNode& n, int64_t node_pos) {
    if (!n.oop_children)
        return -(int64_t)(n.branch_size - n.children[0]->branch_size);
    int64_t min_placed = 0, min_unplaced = 1;
    for (auto& c : n.children) {
        if (c->file_pos != -1) min_placed = std::min(min_placed, c->file_pos - node_pos);
        else if (min_unplaced > 0) min_unplaced = -(int64_t)(n.branch_size - c->branch_size);
    }
    return std::min(min_placed, min_unplaced);
}

// node-type machinery (TrieNode.java). type ids = ordinals.
struct TypeInfo {
    int ordinal;
    int bytes_per_pointer;  // 0 == fractional (SNP4: 4-bit, SNP12/SPARSE_12/DENSE_12: 12-bit)
};

static int fits_bits(int ordinal) {
    switch (ordinal) {
        case 1: return 4;    // SINGLE_NOPAYLOAD_4
        case 2: return 8;    // SINGLE_8
        case 3: return 12;   // SINGLE_NOPAYLOAD_12
        case 4: return 16;   // SINGLE_16
        case 5: return 8;    // SPARSE_8
        case 6: return 12;   // SPARSE_12
        case 7: return 16;   // SPARSE_16
        case 8: return 24;   // SPARSE_24
        case 9: return 40;   // SPARSE_40
        case 10: return 12;  // DENSE_12
        case 11: return 16;
        case 12: return 24;
        case 13: return 32;
        case 14: return 40;
        case 15: return 64;  // LONG_DENSE
    }
    return 64;
}
static bool type_fits(int ordinal, uint64_t dist) {
    int b = fits_bits(ordinal);
    return b >= 64 || dist < (1ull << b);
}

static int child_span(const WNode& n) {
    return n.children.back()->transition - n.children.front()->transition + 1;
}

// TrieNode.sizeofNode per type (excl. payload)
static int type_sizeof(int ordinal, const WNode& n) {
    int cc = (int)n.children.size();
    switch (ordinal) {
        case 0: return 1;
        case 1: return 2;
        case 2: return 3;
        case 3: return 3;
        case 4: return 4;
        case 5: return 2 + cc * 2;
        case 6: return 2 + cc + (cc * 3 + 1) / 2;
        case 7: return 2 + cc * 3;
        case 8: return 2 + cc * 4;
        case 9: return 2 + cc * 6;
        case 10: return 3 + (child_span(n) * 3 + 1) / 2;
        case 11: return 3 + child_span(n) * 2;
        case 12: return 3 + child_span(n) * 3;
        case 13: return 3 + child_span(n) * 4;
        case 14: return 3 + child_span(n) * 5;
        default: return 3 + child_span(n) * 8;
    }
}

// TrieNode.typeFor (TrieNode.java:157-180)
static const int SINGLES[8] = {1, 2, 3, 4, 12, 13, 14, 15};
static const int SPARSES[8] = {5, 5, 6, 7, 8, 9, 9, 15};
static const int DENSES[8] = {10, 10, 10, 11, 12, 13, 14, 15};

static int type_for(const WNode& n, int64_t node_pos) {
    int cc = (int)n.children.size();
    if (cc == 0) return 0;
    int64_t delta = max_position_delta(n, node_pos);
    int idx = 0;
    while (!type_fits(SINGLES[idx], (uint64_t)(-delta))) idx++;
    if (cc == 1) {
        // fractional singles cannot carry a payload
        if (n.pb != 0 && (SINGLES[idx] == 1 || SINGLES[idx] == 3)) idx++;
        return SINGLES[idx];
    }
    int sparse = SPARSES[idx], dense = DENSES[idx];
    return type_sizeof(sparse, n) < type_sizeof(dense, n) ? sparse : dense;
}

// serializer.sizeofNode (PartitionIndex.PartitionIndexSerializer)
static int sizeof_node(const WNode& n, int64_t node_pos) {
    return type_sizeof(type_for(n, node_pos), n) + payload_size(n);
}

// TrieNode.serialize per type + payload (PartitionIndexSerializer.write)
static void write_node_bytes(Dest& d, const WNode& n, int64_t node_pos) {
    int t = type_for(n, node_pos);
    int pb = n.pb;
    auto dist = [&](const WNode& c) { return (uint64_t)(node_pos - c.file_pos); };
    switch (t) {
        case 0:
            d.u8((uint8_t)(0 << 4 | pb));
            break;
        case 1:
            d.u8((uint8_t)(1 << 4 | (int)dist(*n.children[0])));
            d.u8((uint8_t)n.children[0]->transition);
            break;
        case 2:
            d.u8((uint8_t)(2 << 4 | pb));
            d.u8((uint8_t)n.children[0]->transition);
            d.be(dist(*n.children[0]), 1);
            break;
        case 3: {
            uint64_t v = dist(*n.children[0]);
            d.u8((uint8_t)(3 << 4 | (int)(v >> 8)));
            d.u8((uint8_t)(v & 0xFF));
            d.u8((uint8_t)n.children[0]->transition);
            break;
        }
        case 4:
            d.u8((uint8_t)(4 << 4 | pb));
            d.u8((uint8_t)n.children[0]->transition);
            d.be(dist(*n.children[0]), 2);
            break;
        case 5: case 7: case 8: case 9: {
            int w = t == 5 ? 1 : t == 7 ? 2 : t == 8 ? 3 : 5;
            d.u8((uint8_t)(t << 4 | pb));
            d.u8((uint8_t)n.children.size());
            for (auto& c : n.children) d.u8((uint8_t)c->transition);
            for (auto& c : n.children) d.be(dist(*c), w);
            break;
        }
        case 6: {  // SPARSE_12: packed 12-bit
            d.u8((uint8_t)(6 << 4 | pb));
            d.u8((uint8_t)n.children.size());
            for (auto& c : n.children) d.u8((uint8_t)c->transition);
            uint32_t carry = 0;
            bool half = false;
            for (auto& c : n.children) {
                uint32_t v = (uint32_t)dist(*c);
                if (!half) { d.u8((uint8_t)(v >> 4)); carry = v & 0xF; half = true; }
                else { d.u8((uint8_t)(carry << 4 | (v >> 8))); d.u8((uint8_t)(v & 0xFF)); half = false; }
            }
            if (half) d.u8((uint8_t)(carry << 4));
            break;
        }
        case 10: {  // DENSE_12
            d.u8((uint8_t)(10 << 4 | pb));
            int start = n.children.front()->transition, span = child_span(n);
            d.u8((uint8_t)start);
            d.u8((uint8_t)(span - 1));
            size_t ci = 0;
            uint32_t carry = 0;
            bool half = false;
            for (int i = 0; i < span; i++) {
                uint32_t v = 0;
                if (ci < n.children.size() && n.children[ci]->transition == start + i)
                    v = (uint32_t)dist(*n.children[ci++]);
                if (!half) { d.u8((uint8_t)(v >> 4)); carry = v & 0xF; half = true; }
                else { d.u8((uint8_t)(carry << 4 | (v >> 8))); d.u8((uint8_t)(v & 0xFF)); half = false; }
            }
            if (half) d.u8((uint8_t)(carry << 4));
            break;
        }
        case 11: case 12: case 13: case 14: case 15: {
            int w = t == 11 ? 2 : t == 12 ? 3 : t == 13 ? 4 : t == 14 ? 5 : 8;
            d.u8((uint8_t)(t << 4 | pb));
            int start = n.children.front()->transition, span = child_span(n);
            d.u8((uint8_t)start);
            d.u8((uint8_t)(span - 1));
            size_t ci = 0;
            for (int i = 0; i < span; i++) {
                uint64_t v = 0;
                if (ci < n.children.size() && n.children[ci]->transition == start + i)
                    v = dist(*n.children[ci++]);
                d.be(v, w);
            }
            break;
        }
    }
    for (uint8_t b : n.payload) d.u8(b);
}

static int recalc_total_size(WNode& n, int64_t node_pos);
static void layout_children(Dest& d, WNode& n);

// recursive in-page write (IncrementalTrieWriterPageAware.write)
static int64_t write_rec(Dest& d, WNode& n) {
    int64_t node_pos = d.position();
    for (auto& c : n.children)
        if (c->file_pos == -1) c->file_pos = write_rec(d, *c);
    node_pos += n.branch_size;
    write_node_bytes(d, n, node_pos);
    return node_pos;
}

static int recalc_total_size(WNode& n, int64_t node_pos) {
    if (n.oop_in_branch) {
        int sz = 0;
        for (auto& c : n.children) sz += recalc_total_size(*c, node_pos + sz);
        n.branch_size = sz;
    }
    if (n.oop_children || n.oop_in_branch)
        n.node_size = sizeof_node(n, node_pos + n.branch_size);
    return n.branch_size + n.node_size;
}

static void layout_children(Dest& d, WNode& n) {
    // NavigableSet ordered by (branch+node size, transition); pick the
    // largest that fits the current page, else pad and take the largest
    std::vector<WNode*> pending;
    for (auto& c : n.children)
        if (c->file_pos == -1) pending.push_back(c.get());
    auto cmp = [](WNode* a, WNode* b) {
        int sa = a->branch_size + a->node_size, sb = b->branch_size + b->node_size;
        if (sa != sb) return sa < sb;
        return a->transition < b->transition;
    };
    std::sort(pending.begin(), pending.end(), cmp);
    int bytes_left = d.bytes_left_in_page();
    while (!pending.empty()) {
        // largest with branch+node <= bytes_left
        int i = (int)pending.size() - 1;
        while (i >= 0 && pending[i]->branch_size + pending[i]->node_size > bytes_left) i--;
        WNode* child;
        if (i < 0) {
            d.pad_to_page();
            bytes_left = PAGE;
            child = pending.back();
            pending.pop_back();
        } else {
            child = pending[i];
            pending.erase(pending.begin() + i);
        }
        if (child->oop_children || child->oop_in_branch) {
            int actual = recalc_total_size(*child, d.position());
            if (actual > bytes_left) {
                if (bytes_left == PAGE) {
                    layout_children(d, *child);
                    bytes_left = d.bytes_left_in_page();
                }
                // put back with the new size
                pending.push_back(child);
                std::sort(pending.begin(), pending.end(), cmp);
                continue;
            }
        }
        child->file_pos = write_rec(d, *child);
        bytes_left = d.bytes_left_in_page();
    }
    n.branch_size = 0;
    n.oop_children = true;
    n.oop_in_branch = false;
    n.node_size = sizeof_node(n, d.position());
}

static int64_t complete_and_write(Dest& d, WNode& root);

// bottom-up completion (IncrementalTrieWriterPageAware.complete), post-order
static void complete_rec(Dest& d, WNode& n) {
    for (auto& c : n.children) complete_rec(d, *c);
    int branch = 0;
    for (auto& c : n.children) branch += c->branch_size + c->node_size;
    n.branch_size = branch;
    int node_size = sizeof_node(n, d.position());
    if (node_size + branch < PAGE) {
        n.node_size = node_size;
        n.oop_children = false;
        n.oop_in_branch = false;
        for (auto& c : n.children) {
            if (c->file_pos != -1) n.oop_children = true;
            else if (c->oop_children || c->oop_in_branch) n.oop_in_branch = true;
        }
        return;
    }
    layout_children(d, n);
}

// performCompletion + final root write (IncrementalTrieWriterPageAware)
static int64_t complete_and_write(Dest& d, WNode& root) {
    complete_rec(d, root);
    int actual = recalc_total_size(root, d.position());
    int bytes_left = d.bytes_left_in_page();
    if (actual > bytes_left) {
        if (actual <= PAGE) {
            d.pad_to_page();
            bytes_left = PAGE;
            actual = recalc_total_size(root, d.position());
        }
        if (actual > bytes_left) {
            layout_children(d, root);
            if (root.node_size > d.bytes_left_in_page()) {
                d.pad_to_page();
                recalc_total_size(root, d.position());
            }
        }
    }
    return write_rec(d, root);
}

}  // namespace btiw

// escaped component emission (ByteSource.AbstractEscaper)
static void bc_escape(bytes& out, const bytes& data) {
    size_t i = 0;
    while (i < data.size()) {
        if (data[i] != 0) {
            out.push_back(data[i++]);
            continue;
        }
        out.push_back(0x00);  // ESCAPE
        i++;
        while (i < data.size() && data[i] == 0) {
            out.push_back(0xFE);  // ESCAPED_0_CONT
            i++;
        }
        if (i < data.size()) {
            out.push_back(0xFF);  // ESCAPED_0_DONE, then the non-zero byte
        } else {
            out.push_back(0xFE);  // zeros at end: CONT and stop
            return;
        }
    }
    out.push_back(0x00);  // trailing ESCAPE after non-zero-ending data
}

// ByteSource.variableLengthInteger (ByteSource.java:219-…)
static void bc_varint(bytes& out, int64_t value64) {
    uint64_t v = (uint64_t)value64;
    uint64_t neg = (uint64_t)(value64 >> 63);  // all-ones for negative
    v ^= neg;
    int bits = 64;
    while (bits > 1 && !((v | 1) >> (bits - 1))) bits--;  // 64 - clz(v|1)
    int nbytes = bits / 7 + 1;
    if (nbytes >= 9) {
        out.push_back((uint8_t)(neg ? 0x00 : 0xFF));
        uint64_t val = (v | 0x8000000000000000ull) ^ neg;
        for (int i = 7; i >= 0; i--) out.push_back((uint8_t)(val >> (8 * i)));
    } else {
        uint64_t mask = ((uint64_t)(int64_t)-0x100 >> nbytes) & 0xFF;
        int pos = nbytes * 8;
        uint64_t val = (v | (mask << (pos - 8))) ^ neg;
        for (int i = nbytes - 1; i >= 0; i--) out.push_back((uint8_t)(val >> (8 * i)));
    }
}

bytes bti_byte_comparable_clustering(const Clustering& c,
                                     const std::vector<CqlType>& types, BoundKind kind) {
    bytes out;
    for (size_t i = 0; i < c.size(); i++) {
        out.push_back(0x40);  // NEXT_COMPONENT
        switch (types.at(i)) {
            case CqlType::LONG: {
                if (c[i].v.size() != 8) throw std::runtime_error("bad LONG clustering");
                int64_t v = 0;
                for (int b = 0; b < 8; b++) v = (v << 8) | c[i].v[b];
                bc_varint(out, v);
                break;
            }
            case CqlType::INT32: {
                if (c[i].v.size() != 4) throw std::runtime_error("bad INT clustering");
                int32_t v = 0;
                for (int b = 0; b < 4; b++) v = (v << 8) | c[i].v[b];
                bc_varint(out, (int64_t)v);
                break;
            }
            default:
                bc_escape(out, c[i].v);
        }
    }
    // Kind.asByteComparableValue at ByteComparable.Version.OSS50 — the version
    // every trie writer/reader uses (io/tries/Walker.java:64;
    // ClusteringPrefix.java:70-79): CLUSTERING -> TERMINATOR 0x38 (the LEGACY
    // value is 0x40), end/start bounds and boundaries by their LT/GT side,
    // STATIC_CLUSTERING -> EXCLUDED 0x18.
    switch (kind) {
        case CLUSTERING_K: out.push_back(0x38); break;
        case INCL_END: case EXCL_START: case INCL_END_EXCL_START: out.push_back(0x60); break;
        case EXCL_END: case INCL_START: case EXCL_END_INCL_START: out.push_back(0x20); break;
        case STATIC_K: out.push_back(0x18); break;
        default: throw std::runtime_error("unsupported bound kind for byte-comparable");
    }
    return out;
}

bytes bti_separator_gt(const bytes& prev, const bytes& cur) {
    bytes out;
    size_t i = 0;
    while (i < prev.size() && i < cur.size() && prev[i] == cur[i]) {
        out.push_back(cur[i]);
        i++;
    }
    // prev must be strictly less; emit cur's differing byte (or cur ended ==
    // prev prefix case cannot happen for valid prefix-free inputs)
    if (i < cur.size()) out.push_back(cur[i]);
    return out;
}

bytes bti_nudge(const bytes& value, size_t nudge_at) {
    bytes out;
    size_t i = 0;
    for (; i <= nudge_at && i < value.size(); i++) out.push_back(value[i]);
    // increment at nudge_at; 0xFF spills rightward (emit and nudge next)
    while (!out.empty() && out.size() - 1 >= nudge_at && out.back() == 0xFF) {
        nudge_at++;
        if (out.size() - 1 < nudge_at) {
            if (i < value.size()) out.push_back(value[i++]);
            else break;
        } else break;
    }
    if (!out.empty() && out.back() != 0xFF) out.back()++;
    return out;
}

uint64_t append_bti_row_index(bytes& file, const BtiRowIndexBlockSpec& spec) {
    using namespace btiw;
    WNode root;
    for (auto& e : spec.entries) {
        WNode* n = &root;
        for (uint8_t b : e.prefix) {
            if (n->children.empty() || n->children.back()->transition != b) {
                n->children.push_back(std::make_unique<WNode>());
                n->children.back()->transition = b;
            }
            n = n->children.back().get();
        }
        n->pb = e.pb;
        n->payload = e.payload;
    }
    Dest d;
    d.buf = std::move(file);
    int64_t root_pos = complete_and_write(d, root);
    // TrieIndexEntry footer (TrieIndexEntry.serialize; root delta is
    // relative to the entry start = position after the short-length key)
    uint64_t index_pos = (uint64_t)d.position();
    d.be(spec.partition_key.size(), 2);
    for (uint8_t b : spec.partition_key) d.u8(b);
    int64_t base = d.position();
    put_unsigned_vint(d.buf, spec.data_pos);
    put_vint(d.buf, root_pos - base);
    put_unsigned_vint(d.buf, spec.block_count);
    if (spec.partition_del.live()) {
        d.u8(0x80);  // compact DeletionTime: LIVE flag
    } else {
        d.be((uint64_t)spec.partition_del.mfda, 8);
        d.be(spec.partition_del.ldt, 4);
    }
    file = std::move(d.buf);
    return index_pos;
}

// ByteOrderedPartitioner keys: single escaped component (the fixture's
// legacy_da tables; prefixes `40 <key bytes>` pin this form)
bytes bti_byte_comparable_bop(const bytes& key) {
    bytes out;
    out.push_back(0x40);  // NEXT_COMPONENT
    bc_escape(out, key);
    out.push_back(0x38);  // TERMINATOR
    return out;
}

bytes bti_byte_comparable_m3(int64_t token, const bytes& key) {
    bytes out;
    out.push_back(0x40);  // NEXT_COMPONENT
    uint64_t t = (uint64_t)token ^ (1ull << 63);
    for (int i = 7; i >= 0; i--) out.push_back((uint8_t)(t >> (8 * i)));
    out.push_back(0x40);  // NEXT_COMPONENT
    bc_escape(out, key);
    out.push_back(0x38);  // TERMINATOR
    return out;
}

bytes write_bti_partitions(const std::vector<BtiKeyEntry>& entries) {
    using namespace btiw;
    // PartitionIndexBuilder.addEntry: store each key cut to
    // max(diffPoint(prev,cur), diffPoint(cur,next)) bytes
    auto diff_point = [](const bytes& a, const bytes& b) {
        size_t i = 0;
        while (i < a.size() && i < b.size() && a[i] == b[i]) i++;
        return (int)i + 1;
    };
    WNode root;
    auto insert = [&](const bytes& bc, int cut, const BtiKeyEntry& e) {
        WNode* n = &root;
        for (int i = 0; i < cut && i < (int)bc.size(); i++) {
            if (n->children.empty() || n->children.back()->transition != bc[i]) {
                n->children.push_back(std::make_unique<WNode>());
                n->children.back()->transition = bc[i];
            }
            n = n->children.back().get();
        }
        int sz = sized_int_size(e.idxpos);
        n->pb = 8 + (sz - 1);  // PartitionIndexSerializer: always with hash
        n->payload.clear();
        n->payload.push_back(e.hash_bits);
        for (int b = sz - 1; b >= 0; b--)
            n->payload.push_back((uint8_t)((uint64_t)e.idxpos >> (8 * b)));
    };
    for (size_t i = 0; i < entries.size(); i++) {
        int dp_prev = i ? diff_point(entries[i - 1].byte_comparable, entries[i].byte_comparable) : 0;
        int dp_next = i + 1 < entries.size()
                          ? diff_point(entries[i].byte_comparable, entries[i + 1].byte_comparable)
                          : 0;
        insert(entries[i].byte_comparable, std::max(dp_prev, dp_next), entries[i]);
    }

    Dest d;
    int64_t root_pos = btiw::complete_and_write(d, root);

    // PartitionIndexBuilder.complete footer
    int64_t first_key_pos = d.position();
    if (!entries.empty()) {
        d.be(entries.front().raw_key.size(), 2);
        for (uint8_t b : entries.front().raw_key) d.u8(b);
        d.be(entries.back().raw_key.size(), 2);
        for (uint8_t b : entries.back().raw_key) d.u8(b);
    } else {
        d.be(0, 2);
        d.be(0, 2);
    }
    d.be((uint64_t)first_key_pos, 8);
    d.be(entries.size(), 8);
    d.be((uint64_t)root_pos, 8);
    return d.buf;
}

}  // namespace oracle
