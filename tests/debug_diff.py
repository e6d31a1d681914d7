#!/usr/bin/env python3
"""Debug helper (GPU box): reproduce a failing parity case, decompress both
Data.db files, and print the first differing partition at byte level."""
import ctypes
import os
import struct
import subprocess
import sys
import tempfile

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
ORACLE = os.path.join(REPO, "oracle", "bin", "oracle_tool")

lz4 = ctypes.CDLL("liblz4.so.1")
lz4.LZ4_decompress_safe.restype = ctypes.c_int


def read_ci(base):
    b = open(base + "-CompressionInfo.db", "rb").read()
    i = 0
    n = struct.unpack_from(">H", b, i)[0]; i += 2 + n
    opts = struct.unpack_from(">I", b, i)[0]; i += 4
    for _ in range(opts):
        l1 = struct.unpack_from(">H", b, i)[0]; i += 2 + l1
        l2 = struct.unpack_from(">H", b, i)[0]; i += 2 + l2
    chunk_len, maxc = struct.unpack_from(">II", b, i); i += 8
    data_len = struct.unpack_from(">Q", b, i)[0]; i += 8
    cnt = struct.unpack_from(">I", b, i)[0]; i += 4
    offs = [struct.unpack_from(">Q", b, i + 8 * k)[0] for k in range(cnt)]
    return chunk_len, data_len, offs


def decompress(base):
    chunk_len, data_len, offs = read_ci(base)
    f = open(base + "-Data.db", "rb").read()
    out = bytearray()
    for c, off in enumerate(offs):
        end = offs[c + 1] if c + 1 < len(offs) else len(f)
        comp = f[off + 4:end - 4]
        ulen = struct.unpack_from("<I", f, off)[0]
        dst = ctypes.create_string_buffer(ulen)
        r = lz4.LZ4_decompress_safe(bytes(comp), dst, len(comp), ulen)
        assert r == ulen, (c, r, ulen)
        out += dst.raw
    assert len(out) == data_len
    return bytes(out)


def rvint(b, i):
    f = b[i]
    if f < 0x80:
        return f, i + 1
    n = 0
    while f & (0x80 >> n):
        n += 1
    v = f & (0xFF >> n)
    for k in range(n):
        v = (v << 8) | b[i + 1 + k]
    return v, i + 1 + n


def walk(b, ck_width=0):
    i = 0
    parts = []
    while i < len(b):
        start = i
        klen = struct.unpack_from(">H", b, i)[0]; i += 2
        key = b[i:i + klen]; i += klen
        i += 1 if (b[i] & 0x80) else 12
        nrows = 0
        while True:
            flags = b[i]; i += 1
            if flags & 0x01:
                break
            if flags & 0x02:
                i += 1  # kind
                nv = struct.unpack_from(">H", b, i)[0]; i += 2
                for _ in range(nv):
                    i += 1 + ck_width  # header vint + fixed value
            elif ck_width:
                i += 1 + ck_width
            size, i = rvint(b, i)
            prev, j = rvint(b, i)
            i += size
            nrows += 1
        parts.append((start, key.hex(), nrows))
    return parts


def main():
    import cassandra_amd as ca
    td = tempfile.mkdtemp()
    case = sys.argv[1] if len(sys.argv) > 1 else "all_overlap"
    ckw = 0
    if case == "genwide":
        kw = dict(seed=23, n=2, rows=30, crows=200, vlen=600, overlap=30, tomb=10, rtomb=40)
        gdir, odir = td + "/g", td + "/o"
        os.makedirs(gdir), os.makedirs(odir)
        ca.generate(gdir, seed=23, n_sstables=2, rows_per_sstable=30, clustering_rows=200,
                    value_len=600, overlap_pct=30, tombstone_pct=10, range_tomb_pct=40)
        subprocess.run([ORACLE, "gen", odir] + [f"{k}={v}" for k, v in kw.items()],
                       check=True, capture_output=True)
        a = decompress(os.path.join(odir, "oa-1-big"))
        b = decompress(os.path.join(gdir, "oa-1-big"))
        ckw = 8
    else:
        cases = {
            "all_overlap": (["seed=29", "n=4", "rows=1500", "vlen=128", "overlap=100"], {}),
            "tomb": (["seed=17", "n=4", "rows=2500", "vlen=256", "overlap=30", "tomb=20", "pdel=5"], {}),
        }
        gen, jkw = cases[case]
        n = int(gen[1].split("=")[1])
        subprocess.run([ORACLE, "gen", td] + gen, check=True, capture_output=True)
        ins = [os.path.join(td, f"oa-{g}-big") for g in range(1, n + 1)]
        subprocess.run([ORACLE, "compact", os.path.join(td, "oa-90-big")] + ins,
                       check=True, capture_output=True)
        ca.compact(ins, os.path.join(td, "oa-91-big"), **jkw)
        a = decompress(os.path.join(td, "oa-90-big"))
        b = decompress(os.path.join(td, "oa-91-big"))
    print("uncompressed lens:", len(a), len(b))
    n = min(len(a), len(b))
    d = next((i for i in range(n) if a[i] != b[i]), n)
    print("first uncompressed diff at", d)
    pa = walk(a, ckw)
    pb = walk(b, ckw)
    print("partition counts:", len(pa), len(pb))
    ia = max(j for j, t2 in enumerate(pa) if t2[0] <= d)
    ib = max(j for j, t2 in enumerate(pb) if t2[0] <= d)
    print("diff in oracle partition", ia, pa[ia], "gpu partition", ib, pb[ib])
    print("row counts around:", [(x[2]) for x in pa[max(0,ia-2):ia+3]], "vs", [(x[2]) for x in pb[max(0,ib-2):ib+3]])
    sa = pa[ia][0]
    ea = pa[ia + 1][0] if ia + 1 < len(pa) else len(a)
    sb = pb[ib][0]
    eb = pb[ib + 1][0] if ib + 1 < len(pb) else len(b)
    print("oracle decoded:", decode_partition(a, sa, ckw if ckw else 0))
    print("gpu    decoded:", decode_partition(b, sb, ckw if ckw else 0))
    # also next partition keys
    for lbl, ps, buf in (("oracle", pa, a), ("gpu", pb, b)):
        print(lbl, "around:", [t2[1] for t2 in ps[max(0, ia - 2):ia + 3]])




def decode_partition(b, start, ck_width=8, limit=12):
    """decode (kind, ck, size, prev) per unfiltered"""
    i = start
    klen = struct.unpack_from(">H", b, i)[0]; i += 2
    key = b[i:i + klen]; i += klen
    i += 1 if (b[i] & 0x80) else 12
    out = []
    while len(out) < limit:
        flags = b[i]; i += 1
        if flags & 0x01:
            out.append(("END",)); break
        if flags & 0x02:
            kind = b[i]; i += 1
            nv = struct.unpack_from(">H", b, i)[0]; i += 2
            ck = None
            for _ in range(nv):
                i += 1
                ck = int.from_bytes(b[i:i + ck_width], "big", signed=True)
                i += ck_width
            size, i = rvint(b, i)
            prev, i2 = rvint(b, i)
            i += size
            out.append(("M", kind, ck, size, prev))
        else:
            ck = None
            if ck_width:
                i += 1
                ck = int.from_bytes(b[i:i + ck_width], "big", signed=True)
                i += ck_width
            size, i = rvint(b, i)
            prev, i2 = rvint(b, i)
            i += size
            out.append(("R", flags, ck, size, prev))
    return out


if __name__ == "__main__":
    main()
