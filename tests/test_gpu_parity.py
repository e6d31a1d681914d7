"""GPU parity tests: the product's GPU compaction output must be
BYTE-IDENTICAL to the CPU oracle on the same inputs, across every component
file. Inputs are synthetic (shared generator contract) — /root/reference is
never touched at run time (it does not exist on the GPU box).
"""
import ctypes
import filecmp
import os
import subprocess

import pytest

from conftest import ORACLE

pytestmark = pytest.mark.gpu

COMPONENTS = ["Data.db", "Index.db", "CompressionInfo.db", "Filter.db",
              "Digest.crc32", "Statistics.db", "Summary.db", "TOC.txt"]


@pytest.fixture(scope="module")
def ca(product_lib):
    import cassandra_amd as mod
    if mod.device_count() < 1:
        pytest.skip("no GPU")
    return mod


def _oracle_gen(d, **kw):
    args = [f"{k}={v}" for k, v in kw.items()]
    subprocess.run([ORACLE, "gen", d, *args], check=True, capture_output=True)


def _oracle_compact(outbase, inputs, **kw):
    args = [f"{k}={v}" for k, v in kw.items()]
    subprocess.run([ORACLE, "compact", outbase, *inputs, *args], check=True,
                   capture_output=True)


def _assert_dirs_equal(base_a, base_b, components=COMPONENTS):
    for c in components:
        fa, fb = f"{base_a}-{c}", f"{base_b}-{c}"
        assert os.path.exists(fa), f"missing {fa}"
        assert os.path.exists(fb), f"missing {fb}"
        if not filecmp.cmp(fa, fb, shallow=False):
            a = open(fa, "rb").read()
            b = open(fb, "rb").read()
            i = next((j for j in range(min(len(a), len(b))) if a[j] != b[j]), min(len(a), len(b)))
            raise AssertionError(
                f"{c}: len {len(a)} vs {len(b)}, first diff at {i} "
                f"({a[i:i+8].hex() if i < len(a) else '-'} vs {b[i:i+8].hex() if i < len(b) else '-'})")


def test_generator_parity(ca, oracle_bin, tmp_path):
    """GPU write path (serialize+compress+index+bloom kernels) == oracle writer."""
    dg, do = str(tmp_path / "gpu"), str(tmp_path / "cpu")
    os.makedirs(dg), os.makedirs(do)
    kw = dict(seed=13, n=3, rows=2000, vlen=700, overlap=25, tomb=10, pdel=2)
    ca.generate(dg, seed=13, n_sstables=3, rows_per_sstable=2000, value_len=700,
                overlap_pct=25, tombstone_pct=10, partition_del_pct=2)
    _oracle_gen(do, **kw)
    for g in (1, 2, 3):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")


def test_generator_parity_wide(ca, oracle_bin, tmp_path):
    """Wide partitions: clustering rows + range tombstones + promoted index."""
    dg, do = str(tmp_path / "gpu"), str(tmp_path / "cpu")
    os.makedirs(dg), os.makedirs(do)
    kw = dict(seed=23, n=2, rows=30, crows=200, vlen=600, overlap=30, tomb=10, rtomb=40)
    ca.generate(dg, seed=23, n_sstables=2, rows_per_sstable=30, clustering_rows=200,
                value_len=600, overlap_pct=30, tombstone_pct=10, range_tomb_pct=40)
    _oracle_gen(do, **kw)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")


def test_generator_parity_static_rows(ca, oracle_bin, tmp_path):
    """Static rows (one blob static column, SortedTablePartitionWriter slot
    after the partition deletion): write-path parity."""
    dg, do = str(tmp_path / "gpu"), str(tmp_path / "cpu")
    os.makedirs(dg), os.makedirs(do)
    kw = dict(seed=113, n=2, rows=40, crows=30, vlen=250, overlap=30, tomb=10, statics=60)
    ca.generate(dg, seed=113, n_sstables=2, rows_per_sstable=40, clustering_rows=30,
                value_len=250, overlap_pct=30, tombstone_pct=10, static_pct=60)
    _oracle_gen(do, **kw)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")


def test_generator_parity_composite_ck(ca, oracle_bin, tmp_path):
    """Composite (2-column) clustering keys with ck0-prefix range-tombstone
    bounds: write-path parity."""
    dg, do = str(tmp_path / "gpu"), str(tmp_path / "cpu")
    os.makedirs(dg), os.makedirs(do)
    kw = dict(seed=104, n=2, rows=40, crows=48, vlen=300, overlap=30, tomb=10, rtomb=40,
              ckcols=2)
    ca.generate(dg, seed=104, n_sstables=2, rows_per_sstable=40, clustering_rows=48,
                value_len=300, overlap_pct=30, tombstone_pct=10, range_tomb_pct=40,
                ck_cols=2)
    _oracle_gen(do, **kw)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")


def test_generator_parity_multi_column(ca, oracle_bin, tmp_path):
    """N regular columns + per-cell subset bitmaps: write-path parity."""
    dg, do = str(tmp_path / "gpu"), str(tmp_path / "cpu")
    os.makedirs(dg), os.makedirs(do)
    kw = dict(seed=86, n=2, rows=1000, vlen=180, overlap=25, tomb=10, ncols=4, colmiss=25)
    ca.generate(dg, seed=86, n_sstables=2, rows_per_sstable=1000, value_len=180,
                overlap_pct=25, tombstone_pct=10, n_value_cols=4, col_missing_pct=25)
    _oracle_gen(do, **kw)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")


def test_generator_parity_text_ck(ca, oracle_bin, tmp_path):
    """Variable-width (UTF8) clustering values: write-path parity."""
    dg, do = str(tmp_path / "gpu"), str(tmp_path / "cpu")
    os.makedirs(dg), os.makedirs(do)
    kw = dict(seed=75, n=2, rows=40, crows=120, vlen=400, overlap=30, tomb=10, rtomb=40,
              cktext=1)
    ca.generate(dg, seed=75, n_sstables=2, rows_per_sstable=40, clustering_rows=120,
                value_len=400, overlap_pct=30, tombstone_pct=10, range_tomb_pct=40,
                ck_text=True)
    _oracle_gen(do, **kw)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")


def test_generator_parity_long_keys(ca, oracle_bin, tmp_path):
    """Arbitrary-length partition keys (BytesType, 24 B): write-path parity."""
    dg, do = str(tmp_path / "gpu"), str(tmp_path / "cpu")
    os.makedirs(dg), os.makedirs(do)
    kw = dict(seed=61, n=3, rows=1500, vlen=400, overlap=25, tomb=10, keylen=24)
    ca.generate(dg, seed=61, n_sstables=3, rows_per_sstable=1500, value_len=400,
                overlap_pct=25, tombstone_pct=10, key_len=24)
    _oracle_gen(do, **kw)
    for g in (1, 2, 3):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")


@pytest.mark.parametrize("case", [
    dict(name="plain", gen=dict(seed=42, n=4, rows=3000, vlen=512, overlap=10), job={}),
    dict(name="tombstones_nogc", gen=dict(seed=17, n=4, rows=2500, vlen=256, overlap=30,
                                          tomb=20, pdel=5), job={}),
    dict(name="tombstones_gc", gen=dict(seed=17, n=4, rows=2500, vlen=256, overlap=30,
                                        tomb=20, pdel=5),
         job=dict(gc_before=2000000000)),
    dict(name="never_purge", gen=dict(seed=17, n=4, rows=2500, vlen=256, overlap=30, tomb=20),
         job=dict(gc_before=2000000000, never_purge=True)),
    dict(name="single_input", gen=dict(seed=8, n=1, rows=4000, vlen=300, overlap=0, tomb=10),
         job={}),
    dict(name="all_overlap", gen=dict(seed=29, n=4, rows=1500, vlen=128, overlap=100), job={}),
    dict(name="incompressible", gen=dict(seed=31, n=2, rows=2000, vlen=900, overlap=10,
                                         vrep=0), job={}),
    dict(name="highly_compressible", gen=dict(seed=33, n=2, rows=2000, vlen=900, overlap=10,
                                              vrep=97), job={}),
    dict(name="constant_values", gen=dict(seed=35, n=2, rows=1500, vlen=1000, overlap=0,
                                          vrep=100), job={}),
    dict(name="wide_plain", gen=dict(seed=51, n=3, rows=80, crows=50, vlen=300, overlap=20),
         job={}),
    dict(name="wide_tombstones", gen=dict(seed=52, n=3, rows=60, crows=40, vlen=200,
                                          overlap=30, tomb=15, rtomb=30), job={}),
    dict(name="wide_gc", gen=dict(seed=52, n=3, rows=60, crows=40, vlen=200, overlap=30,
                                  tomb=15, rtomb=30), job=dict(gc_before=2000000000)),
    dict(name="wide_promoted_index", gen=dict(seed=53, n=2, rows=12, crows=300, vlen=800,
                                              overlap=50, rtomb=40), job={}),
    dict(name="wide_single_input", gen=dict(seed=54, n=1, rows=40, crows=60, vlen=250,
                                            tomb=10, rtomb=25, overlap=0), job={}),
    dict(name="long_keys", gen=dict(seed=62, n=4, rows=2000, vlen=300, overlap=40,
                                    tomb=10, keylen=24), job={}),
    dict(name="long_keys_max", gen=dict(seed=63, n=2, rows=800, vlen=200, overlap=50,
                                        keylen=255), job={}),
    dict(name="long_keys_wide", gen=dict(seed=64, n=3, rows=40, crows=60, vlen=250,
                                         overlap=30, tomb=10, rtomb=30, keylen=32), job={}),
    dict(name="long_keys_gc", gen=dict(seed=65, n=3, rows=1500, vlen=200, overlap=40,
                                       tomb=25, keylen=24), job=dict(gc_before=2000000000)),
    dict(name="text_ck_plain", gen=dict(seed=71, n=3, rows=60, crows=50, vlen=300,
                                        overlap=25, cktext=1), job={}),
    dict(name="text_ck_tombstones", gen=dict(seed=72, n=3, rows=50, crows=40, vlen=200,
                                             overlap=30, tomb=15, rtomb=35, cktext=1), job={}),
    dict(name="text_ck_gc", gen=dict(seed=72, n=3, rows=50, crows=40, vlen=200, overlap=30,
                                     tomb=15, rtomb=35, cktext=1),
         job=dict(gc_before=2000000000)),
    dict(name="text_ck_promoted_index", gen=dict(seed=73, n=2, rows=10, crows=400, vlen=700,
                                                 overlap=50, rtomb=40, cktext=1), job={}),
    dict(name="text_ck_long_keys", gen=dict(seed=74, n=3, rows=40, crows=50, vlen=250,
                                            overlap=30, tomb=10, rtomb=25, cktext=1,
                                            keylen=32), job={}),
    dict(name="multi_column", gen=dict(seed=82, n=4, rows=1500, vlen=200, overlap=30,
                                       tomb=10, ncols=4, colmiss=25), job={}),
    dict(name="multi_column_wide", gen=dict(seed=83, n=3, rows=40, crows=50, vlen=150,
                                            overlap=30, tomb=10, rtomb=30, ncols=3,
                                            colmiss=20), job={}),
    dict(name="multi_column_gc", gen=dict(seed=84, n=3, rows=1200, vlen=120, overlap=40,
                                          tomb=25, ncols=5, colmiss=30),
         job=dict(gc_before=2000000000)),
    dict(name="multi_column_everything", gen=dict(seed=85, n=3, rows=30, crows=40, vlen=180,
                                                  overlap=30, tomb=12, rtomb=25, ncols=4,
                                                  colmiss=20, cktext=1, keylen=24), job={}),
    dict(name="composite_ck", gen=dict(seed=101, n=3, rows=40, crows=48, vlen=250,
                                       overlap=30, tomb=10, ckcols=2), job={}),
    dict(name="composite_ck_range_tombs", gen=dict(seed=102, n=3, rows=40, crows=48,
                                                   vlen=200, overlap=30, tomb=10, rtomb=40,
                                                   ckcols=2), job={}),
    dict(name="composite_ck_gc", gen=dict(seed=102, n=3, rows=40, crows=48, vlen=200,
                                          overlap=30, tomb=10, rtomb=40, ckcols=2),
         job=dict(gc_before=2000000000)),
    dict(name="composite_ck_kitchen_sink", gen=dict(seed=103, n=3, rows=24, crows=40,
                                                    vlen=180, overlap=30, tomb=12, rtomb=30,
                                                    ckcols=2, ncols=3, colmiss=20, keylen=24),
         job={}),
    dict(name="static_rows", gen=dict(seed=111, n=3, rows=40, crows=30, vlen=200,
                                      overlap=30, tomb=10, statics=50), job={}),
    dict(name="static_rows_gc", gen=dict(seed=111, n=3, rows=40, crows=30, vlen=200,
                                         overlap=30, tomb=10, statics=50),
         job=dict(gc_before=2000000000)),
    dict(name="static_rows_promoted_index", gen=dict(seed=98535134, n=6, rows=50, vlen=900,
                                                      overlap=100, tomb=40, vrep=97, keylen=120,
                                                      crows=200, rtomb=70, statics=80),
         job=dict(gc_before=2000000000, never_purge=True)),
    dict(name="static_rows_kitchen_sink", gen=dict(seed=112, n=3, rows=24, crows=36,
                                                   vlen=150, overlap=30, tomb=12, rtomb=25,
                                                   statics=40, ckcols=2, ncols=3, colmiss=20,
                                                   keylen=24), job={}),
])
def test_compaction_parity(ca, oracle_bin, tmp_path, case):
    d = str(tmp_path)
    gen = case["gen"]
    _oracle_gen(d, **gen)
    ins = [f"{d}/oa-{g}-big" for g in range(1, gen["n"] + 1)]
    job = dict(case["job"])
    okw = {}
    if "gc_before" in job:
        okw["gcbefore"] = job["gc_before"]
    if job.pop("never_purge", False):
        okw["nevergc"] = 1
        job["never_purge"] = True
    _oracle_compact(f"{d}/oa-90-big", ins, **okw)
    ca.compact(ins, f"{d}/oa-91-big", **job)
    _assert_dirs_equal(f"{d}/oa-90-big", f"{d}/oa-91-big")


def test_sharded_output_parity(ca, oracle_bin, tmp_path):
    """n_output_shards=S: one call emits S sstables over equal token ranges,
    each byte-identical to an oracle compaction restricted to that range
    (shards run two at a time inside the library)."""
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from cassandra_amd.sharding import split_token_range
    d = str(tmp_path)
    S = 3
    _oracle_gen(d, seed=91, n=4, rows=2500, vlen=300, overlap=30, tomb=10)
    ins = [f"{d}/oa-{g}-big" for g in range(1, 5)]
    ca.compact(ins, f"{d}/oa-200-big", n_output_shards=S)
    for i in range(S):
        lo, hi = split_token_range(S, i)
        _oracle_compact(f"{d}/oa-{300+i}-big", ins, shard=f"{lo}:{hi}")
        _assert_dirs_equal(f"{d}/oa-{300+i}-big", f"{d}/oa-{200+i}-big")


def test_sharded_output_parity_wide(ca, oracle_bin, tmp_path):
    """Sharded outputs with wide partitions + long keys + multi-column."""
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from cassandra_amd.sharding import split_token_range
    d = str(tmp_path)
    S = 2
    _oracle_gen(d, seed=92, n=3, rows=40, crows=50, vlen=200, overlap=30, tomb=10,
                rtomb=30, ncols=3, keylen=24)
    ins = [f"{d}/oa-{g}-big" for g in range(1, 4)]
    ca.compact(ins, f"{d}/oa-200-big", n_output_shards=S)
    for i in range(S):
        lo, hi = split_token_range(S, i)
        _oracle_compact(f"{d}/oa-{300+i}-big", ins, shard=f"{lo}:{hi}")
        _assert_dirs_equal(f"{d}/oa-{300+i}-big", f"{d}/oa-{200+i}-big")


@pytest.mark.parametrize("mode", ["row", "cell"])
def test_garbage_collect_parity(ca, oracle_bin, tmp_path, mode):
    """nodetool garbagecollect (GarbageSkipper): tombstone SOURCES remove
    shadowed data without being written; GPU output == oracle, byte for byte.
    Sources are a later generation of the same key space (newer timestamps +
    tombstones), so both partition-, row- and cell-level shadowing occur."""
    d = str(tmp_path)
    # data: seed A; sources: same key universe, later base_ts, heavy tombstones
    _oracle_gen(d, seed=121, n=3, rows=2000, vlen=200, overlap=40, tomb=5)
    os.makedirs(d + "/src")
    _oracle_gen(d + "/src", seed=121, n=2, rows=1500, vlen=150, overlap=40, tomb=40,
                pdel=10, ts0=1700000500000000)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    srcs = [f"{d}/src/oa-{g}-big" for g in (1, 2)]
    cell = 1 if mode == "cell" else 0
    _oracle_compact(f"{d}/oa-90-big", ins, tombsrc=",".join(srcs), cellgc=cell)
    ca.compact(ins, f"{d}/oa-91-big", tombstone_sources=srcs, cell_level_gc=bool(cell))
    _assert_dirs_equal(f"{d}/oa-90-big", f"{d}/oa-91-big")


def test_garbage_collect_parity_wide(ca, oracle_bin, tmp_path):
    """Garbage collect with clustering + range tombstones in the sources."""
    d = str(tmp_path)
    _oracle_gen(d, seed=122, n=2, rows=60, crows=50, vlen=200, overlap=40, tomb=5)
    os.makedirs(d + "/src")
    _oracle_gen(d + "/src", seed=122, n=2, rows=50, crows=50, vlen=150, overlap=40,
                tomb=30, rtomb=50, ts0=1700000500000000)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2)]
    srcs = [f"{d}/src/oa-{g}-big" for g in (1, 2)]
    _oracle_compact(f"{d}/oa-90-big", ins, tombsrc=",".join(srcs))
    ca.compact(ins, f"{d}/oa-91-big", tombstone_sources=srcs)
    _assert_dirs_equal(f"{d}/oa-90-big", f"{d}/oa-91-big")


def test_flush_parity(ca, oracle_bin, tmp_path):
    """gpuc_flush (memtable-flush analog): unsorted unique-key host rows ->
    GPU token sort + writer == oracle flush of the same rows, byte for byte."""
    import random
    d = str(tmp_path)
    rng = random.Random(117)
    rows, lines = [], []
    seen = set()
    for i in range(3000):
        klen = rng.choice([4, 8, 12, 24])
        key = bytes(rng.randrange(256) for _ in range(klen))
        if key in seen:
            continue
        seen.add(key)
        ts = 1700000000000000 + rng.randrange(10**9)
        if rng.random() < 0.15:
            ldt = 1700000000 + rng.randrange(1000)
            rows.append((key, ts, None, ldt))
            lines.append(f"{key.hex()} {ts} T {ldt}")
        else:
            val = bytes(rng.randrange(256) for _ in range(rng.randrange(1, 400)))
            rows.append((key, ts, val))
            lines.append(f"{key.hex()} {ts} {val.hex()}")
    with open(f"{d}/rows.txt", "w") as f:
        f.write("\n".join(lines) + "\n")
    subprocess.run([ORACLE, "flush", f"{d}/oa-1-big", f"{d}/rows.txt"],
                   check=True, capture_output=True)
    ca.flush(rows, f"{d}/oa-2-big")
    _assert_dirs_equal(f"{d}/oa-1-big", f"{d}/oa-2-big")
    ca.verify(f"{d}/oa-2-big")
    # duplicate keys must be rejected loudly
    try:
        ca.flush([(b"same", 1, b"x"), (b"same", 2, b"y")], f"{d}/oa-9-big")
        raise AssertionError("duplicate keys accepted")
    except Exception as e:
        assert "duplicate" in str(e), e


def test_verify_epilogue(ca, oracle_bin, tmp_path):
    """gpuc_verify passes on valid sstables (oracle- and GPU-written, all
    schema shapes) and fails on corruption of data, digest or filter."""
    import shutil
    d = str(tmp_path)
    os.makedirs(d + "/w"), os.makedirs(d + "/g")
    _oracle_gen(d, seed=95, n=1, rows=2000, vlen=300, overlap=0, tomb=10)
    _oracle_gen(d + "/w", seed=96, n=1, rows=30, crows=60, vlen=200, rtomb=30,
                cktext=1, keylen=24, ncols=3, overlap=0, statics=50)
    ca.generate(d + "/g", seed=97, n_sstables=1, rows_per_sstable=2000, value_len=256)
    for b in [f"{d}/oa-1-big", f"{d}/w/oa-1-big", f"{d}/g/oa-1-big"]:
        ca.verify(b)
    # corrupt one data byte mid-file -> chunk CRC must fail
    for comp, what in [("Data.db", "crc"), ("Digest.crc32", "digest"), ("Filter.db", "bloom")]:
        cd = f"{d}/corrupt_{what}"
        os.makedirs(cd)
        for c in COMPONENTS:
            shutil.copy(f"{d}/oa-1-big-{c}", f"{cd}/oa-1-big-{c}")
        with open(f"{cd}/oa-1-big-{comp}", "r+b") as f:
            f.seek(os.path.getsize(f"{cd}/oa-1-big-{comp}") // 2)
            b0 = f.read(1)
            f.seek(-1, 1)
            f.write(bytes([b0[0] ^ 0x01]))
        try:
            ca.verify(f"{cd}/oa-1-big")
            raise AssertionError(f"verify accepted corrupted {comp}")
        except Exception as e:
            assert "gpuc_verify" in str(e), e


def test_anticompaction_split_parity(ca, oracle_bin, tmp_path):
    """Anticompaction (antiCompactGroup): one pass keeps tokens inside the
    repaired ranges, the inverted pass keeps the complement; together they
    partition the data. Both outputs byte-identical to the oracle."""
    d = str(tmp_path)
    _oracle_gen(d, seed=141, n=3, rows=2000, vlen=250, overlap=30, tomb=10)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    ranges = [(-2**62, 0), (2**61, 2**62)]
    rs = ",".join(f"{lo}:{hi}" for lo, hi in ranges)
    _oracle_compact(f"{d}/oa-90-big", ins, ranges=rs)
    _oracle_compact(f"{d}/oa-92-big", ins, ranges=rs, invertranges=1)
    ca.compact(ins, f"{d}/oa-91-big", keep_ranges=ranges)
    ca.compact(ins, f"{d}/oa-93-big", keep_ranges=ranges, invert_ranges=True)
    _assert_dirs_equal(f"{d}/oa-90-big", f"{d}/oa-91-big")
    _assert_dirs_equal(f"{d}/oa-92-big", f"{d}/oa-93-big")
    # the two halves partition the rows
    import json
    a = json.loads(subprocess.run([ORACLE, "compact", f"{d}/oa-94-big", *ins],
                                  capture_output=True, text=True, check=True).stdout.splitlines()[-1])
    pr = 0
    for base in (f"{d}/oa-90-big", f"{d}/oa-92-big"):
        out = subprocess.run([ORACLE, "dump", base], capture_output=True, text=True, check=True)
        pr += int(out.stdout.splitlines()[1].split("partitions=")[1].split()[0])
    assert pr == a["partitions_out"], (pr, a["partitions_out"])


def test_scrub_parity(ca, oracle_bin, tmp_path):
    """gpuc_scrub (SortedTableScrubber): flip a byte mid-Data.db, scrub on
    GPU and in the oracle -> identical kept/dropped counts and byte-identical
    scrubbed sstables; the scrubbed output passes gpuc_verify. Also: scrub of
    a clean sstable keeps every partition; statics/composite schema covered."""
    import json
    d = str(tmp_path)
    _oracle_gen(d, seed=171, n=1, rows=3000, vlen=300, overlap=0, tomb=10)
    os.makedirs(d + "/w")
    _oracle_gen(d + "/w", seed=172, n=1, rows=60, crows=50, vlen=200, rtomb=30,
                cktext=1, keylen=20, ncols=3, overlap=0, statics=40)
    for sub, base in [("", f"{d}/oa-1-big"), ("/w", f"{d}/w/oa-1-big")]:
        with open(base + "-Data.db", "r+b") as f:
            f.seek(os.path.getsize(base + "-Data.db") // 2)
            b0 = f.read(1)
            f.seek(-1, 1)
            f.write(bytes([b0[0] ^ 0x10]))
        out = subprocess.run([ORACLE, "scrub", f"{d}{sub}/oa-80-big", base],
                             capture_output=True, text=True, check=True)
        ores = json.loads(out.stdout.splitlines()[-1])
        kept, dropped = ca.scrub(base, f"{d}{sub}/oa-81-big")
        assert dropped > 0, "corruption did not drop any partition"
        assert (kept, dropped) == (ores["partitions_kept"], ores["partitions_dropped"])
        _assert_dirs_equal(f"{d}{sub}/oa-80-big", f"{d}{sub}/oa-81-big")
        ca.verify(f"{d}{sub}/oa-81-big")
    # clean input -> everything kept, output verifies
    os.makedirs(d + "/c")
    _oracle_gen(d + "/c", seed=173, n=1, rows=800, vlen=150, overlap=0)
    kept, dropped = ca.scrub(f"{d}/c/oa-1-big", f"{d}/c/oa-81-big")
    assert dropped == 0 and kept > 0
    ca.verify(f"{d}/c/oa-81-big")
    # round-2 schemas through scrub: complex, counter and TTL sstables
    for tag, genkw in [("x", dict(seed=174, n=1, rows=900, vlen=80, overlap=0,
                                  tomb=10, cpx=45, cpxdel=25)),
                       ("k", dict(seed=175, n=1, rows=900, overlap=0, tomb=10,
                                  counter=1)),
                       ("t", dict(seed=176, n=1, rows=900, vlen=80, overlap=0,
                                  ttl=40))]:
        dd = f"{d}/{tag}"
        os.makedirs(dd)
        _oracle_gen(dd, **genkw)
        base = f"{dd}/oa-1-big"
        with open(base + "-Data.db", "r+b") as f:
            f.seek(os.path.getsize(base + "-Data.db") // 3)
            b0 = f.read(1)
            f.seek(-1, 1)
            f.write(bytes([b0[0] ^ 0x10]))
        out = subprocess.run([ORACLE, "scrub", f"{dd}/oa-80-big", base],
                             capture_output=True, text=True, check=True)
        ores = json.loads(out.stdout.splitlines()[-1])
        kept, dropped = ca.scrub(base, f"{dd}/oa-81-big")
        assert dropped > 0, tag
        assert (kept, dropped) == (ores["partitions_kept"], ores["partitions_dropped"]), tag
        _assert_dirs_equal(f"{dd}/oa-80-big", f"{dd}/oa-81-big")
        ca.verify(f"{dd}/oa-81-big")


def test_compaction_associativity_gpu(ca, oracle_bin, tmp_path):
    """Multi-level compaction on GPU: (1) with key overlap, re-compacting a
    compaction OUTPUT (a parity case the generator never produces: its
    StatsMetadata mins can exceed its header mins) stays byte-identical to
    the oracle doing the same chain; (2) with disjoint keys the chain is
    also byte-associative: compact(compact(A,B),C) == compact(A,B,C)."""
    d = str(tmp_path)
    _oracle_gen(d, seed=311, n=3, rows=1500, vlen=200, overlap=30, tomb=15)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    ca.compact(ins[:2], f"{d}/oa-50-big")
    _oracle_compact(f"{d}/oa-80-big", ins[:2])
    _assert_dirs_equal(f"{d}/oa-50-big", f"{d}/oa-80-big")
    ca.compact([f"{d}/oa-80-big", ins[2]], f"{d}/oa-60-big")
    _oracle_compact(f"{d}/oa-81-big", [f"{d}/oa-80-big", ins[2]])
    _assert_dirs_equal(f"{d}/oa-60-big", f"{d}/oa-81-big")
    os.makedirs(d + "/dj")
    _oracle_gen(d + "/dj", seed=313, n=3, rows=1200, vlen=150, overlap=0, tomb=15)
    dins = [f"{d}/dj/oa-{g}-big" for g in (1, 2, 3)]
    ca.compact(dins[:2], f"{d}/dj/oa-50-big")
    ca.compact([f"{d}/dj/oa-50-big", dins[2]], f"{d}/dj/oa-60-big")
    ca.compact(dins, f"{d}/dj/oa-70-big")
    _assert_dirs_equal(f"{d}/dj/oa-60-big", f"{d}/dj/oa-70-big")


def test_snappy_pipeline(ca, oracle_bin, tmp_path):
    """C3 codec end-to-end on GPU: SnappyCompressor sstables (chunk payload =
    one raw snappy block, SnappyCompressor.java:82-86; wave kernels proven
    bit-exact vs the 1.1.8 restatement in profiles/r01_snappy_kernels.txt)
    compact, verify and scrub byte-identically to the oracle."""
    import json
    d = str(tmp_path)
    _oracle_gen(d, seed=7, n=3, rows=1500, vlen=400, overlap=25, tomb=10, snappy=1)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    _oracle_compact(f"{d}/oa-90-big", ins)
    ca.compact(ins, f"{d}/oa-91-big")
    _assert_dirs_equal(f"{d}/oa-90-big", f"{d}/oa-91-big")
    ca.verify(f"{d}/oa-91-big")
    # wide partitions + statics through the same codec
    os.makedirs(d + "/w")
    _oracle_gen(d + "/w", seed=8, n=2, rows=40, crows=50, vlen=200, rtomb=30,
                cktext=1, statics=40, ncols=3, overlap=20, snappy=1)
    wins = [f"{d}/w/oa-{g}-big" for g in (1, 2)]
    _oracle_compact(f"{d}/w/oa-90-big", wins)
    ca.compact(wins, f"{d}/w/oa-91-big")
    _assert_dirs_equal(f"{d}/w/oa-90-big", f"{d}/w/oa-91-big")
    # scrub parity on a corrupt snappy chunk
    with open(f"{d}/oa-1-big-Data.db", "r+b") as f:
        f.seek(os.path.getsize(f"{d}/oa-1-big-Data.db") // 2)
        b0 = f.read(1)
        f.seek(-1, 1)
        f.write(bytes([b0[0] ^ 0x11]))
    out = subprocess.run([ORACLE, "scrub", f"{d}/oa-80-big", f"{d}/oa-1-big"],
                         capture_output=True, text=True, check=True)
    ores = json.loads(out.stdout.splitlines()[-1])
    kept, dropped = ca.scrub(f"{d}/oa-1-big", f"{d}/oa-81-big")
    assert dropped > 0 and (kept, dropped) == (ores["partitions_kept"],
                                               ores["partitions_dropped"])
    _assert_dirs_equal(f"{d}/oa-80-big", f"{d}/oa-81-big")


def test_snappy_generate_parity(ca, oracle_bin, tmp_path):
    """GPU snappy WRITE path (generate(snappy=True)) == oracle generator,
    then compact+verify of the result (SnappyCompressor.java:82-86 chunk
    framing through the product writer kernels). Promoted from the round-1
    manual check (VERDICT round-2 item 1)."""
    dg, do = str(tmp_path / "gpu"), str(tmp_path / "cpu")
    os.makedirs(dg), os.makedirs(do)
    _oracle_gen(do, seed=11, n=2, rows=2000, vlen=300, overlap=15, tomb=10, snappy=1)
    ca.generate(dg, seed=11, n_sstables=2, rows_per_sstable=2000, value_len=300,
                overlap_pct=15, tombstone_pct=10, snappy=True)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")
    out = f"{dg}/oa-90-big"
    ca.compact([f"{dg}/oa-1-big", f"{dg}/oa-2-big"], out)
    _oracle_compact(f"{do}/oa-90-big", [f"{do}/oa-1-big", f"{do}/oa-2-big"])
    _assert_dirs_equal(out, f"{do}/oa-90-big")
    ca.verify(out)


def test_ttl_pipeline(ca, oracle_bin, tmp_path):
    """Expiring cells end-to-end (VERDICT round-2 item 5). Generator writes
    ExpiringLivenessInfo rows + expiring cells (LivenessInfo.java:67,
    AbstractCell.java:53-76); three parity cases place `now` on each side of
    expiry: (a) live-expiring (cells stay expiring), (b) expired-unpurgeable
    (expired cells convert to tombstones, ldt -= ttl, retained at
    gcBefore=MIN), (c) expired-purged. Plus GPU-vs-oracle WRITER parity for
    ttl_pct and a wide-partition TTL case."""
    d = str(tmp_path)
    _oracle_gen(d, seed=51, n=3, rows=1200, vlen=200, overlap=20, tomb=10, ttl=35)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    INT64_MIN = -(2 ** 63)
    # lets lie in [base_ldt, base_ldt + 2000)
    cases = [
        (1699999999, INT64_MIN),          # (a) now < every let
        (1800000000, INT64_MIN),          # (b) expired, nothing purgeable
        (1800000000, 1800000000),         # (c) expired + purgeable
    ]
    for i, (now, gcb) in enumerate(cases):
        ca.compact(ins, f"{d}/oa-{60 + i}-big", now_sec=now, gc_before=gcb)
        okw = {"now": now}
        if gcb != INT64_MIN:
            okw["gcbefore"] = gcb
        _oracle_compact(f"{d}/oa-{80 + i}-big", ins, **okw)
        _assert_dirs_equal(f"{d}/oa-{60 + i}-big", f"{d}/oa-{80 + i}-big")
    # writer parity: GPU generate(ttl_pct) == oracle gen ttl=
    dg = d + "/g"
    os.makedirs(dg)
    ca.generate(dg, seed=51, n_sstables=2, rows_per_sstable=800, value_len=150,
                overlap_pct=20, tombstone_pct=10, ttl_pct=35)
    do = d + "/o"
    os.makedirs(do)
    _oracle_gen(do, seed=51, n=2, rows=800, vlen=150, overlap=20, tomb=10, ttl=35)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")
    # wide partitions with TTL'd clustering rows through compaction
    dw = d + "/w"
    os.makedirs(dw)
    _oracle_gen(dw, seed=52, n=2, rows=40, crows=50, vlen=120, rtomb=25, tomb=10,
                overlap=20, ttl=40)
    wins = [f"{dw}/oa-{g}-big" for g in (1, 2)]
    ca.compact(wins, f"{dw}/oa-60-big", now_sec=1800000000, gc_before=INT64_MIN)
    _oracle_compact(f"{dw}/oa-80-big", wins, now=1800000000)
    _assert_dirs_equal(f"{dw}/oa-60-big", f"{dw}/oa-80-big")


def test_complex_columns_pipeline(ca, oracle_bin, tmp_path):
    """Complex (collection) column end-to-end on GPU (VERDICT round-2 item
    4): parse -> ColumnDataReducer complex merge (Row.java:851-884) ->
    ComplexColumnData.purge -> writeComplexColumn serialization, byte-
    identical to the oracle. Covers plain merge, complexDeletion shadowing,
    purge, wide partitions, multi-column schemas, writer parity and
    verify()."""
    d = str(tmp_path)
    INT64_MIN = -(2 ** 63)
    _oracle_gen(d, seed=61, n=3, rows=1200, vlen=150, overlap=25, tomb=10,
                cpx=40, cpxdel=30, pdel=3)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    ca.compact(ins, f"{d}/oa-60-big")
    _oracle_compact(f"{d}/oa-80-big", ins)
    _assert_dirs_equal(f"{d}/oa-60-big", f"{d}/oa-80-big")
    ca.verify(f"{d}/oa-60-big")
    # re-compaction of a compaction output
    ca.compact([f"{d}/oa-60-big", ins[2]], f"{d}/oa-61-big")
    _oracle_compact(f"{d}/oa-81-big", [f"{d}/oa-80-big", ins[2]])
    _assert_dirs_equal(f"{d}/oa-61-big", f"{d}/oa-81-big")
    # purge: gcBefore past every ldt drops shadowed complex data identically
    ca.compact(ins, f"{d}/oa-62-big", now_sec=1800000000, gc_before=1800000000)
    _oracle_compact(f"{d}/oa-82-big", ins, now=1800000000, gcbefore=1800000000)
    _assert_dirs_equal(f"{d}/oa-62-big", f"{d}/oa-82-big")
    # GPU writer parity for the generator's complex knobs
    dg, do = d + "/g", d + "/o"
    os.makedirs(dg), os.makedirs(do)
    ca.generate(dg, seed=61, n_sstables=2, rows_per_sstable=900, value_len=120,
                overlap_pct=20, tombstone_pct=10, complex_pct=45, complex_del_pct=30)
    _oracle_gen(do, seed=61, n=2, rows=900, vlen=120, overlap=20, tomb=10,
                cpx=45, cpxdel=30)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")
    # wide partitions + multi value columns + statics + complex through merge
    dw = d + "/w"
    os.makedirs(dw)
    _oracle_gen(dw, seed=62, n=2, rows=40, crows=40, vlen=100, rtomb=25, tomb=10,
                overlap=20, ncols=3, statics=40, cpx=50, cpxdel=25)
    wins = [f"{dw}/oa-{g}-big" for g in (1, 2)]
    ca.compact(wins, f"{dw}/oa-60-big")
    _oracle_compact(f"{dw}/oa-80-big", wins)
    _assert_dirs_equal(f"{dw}/oa-60-big", f"{dw}/oa-80-big")


def test_purge_bloom_evaluator(ca, oracle_bin, tmp_path):
    """Per-key purge evaluator (VERDICT round-2 item 8): overlap entries can
    carry the overlapping sstable's Filter.db bits, reproducing
    CompactionController.getPurgeEvaluator's overlapIterator + BF.isPresent
    chain (CompactionController.java:247-286,308-329). A disjoint-key
    overlapping source must NOT gate purge when its bloom is provided, and
    MUST gate everything in the conservative interval-only mode."""
    d = str(tmp_path)
    _oracle_gen(d, seed=5, n=2, rows=800, vlen=80, overlap=0, tomb=40)
    os.makedirs(d + "/y")
    # disjoint keys: 9-byte keys never collide with the 8-byte main set
    _oracle_gen(d + "/y", seed=99, n=1, rows=500, vlen=50, keylen=9)
    ins = [f"{d}/oa-1-big", f"{d}/oa-2-big"]
    LO, HI = -(2 ** 63), 2 ** 63 - 1
    flt = f"{d}/y/oa-1-big-Filter.db"
    GCB = 2000000000
    # (a) interval-only: conservative, nothing purged
    ra = ca.compact(ins, f"{d}/oa-90-big", gc_before=GCB, overlaps=[(LO, HI, 1)])
    _oracle_compact(f"{d}/oa-80-big", ins, gcbefore=GCB, ov=f"{LO}:{HI}:1")
    _assert_dirs_equal(f"{d}/oa-90-big", f"{d}/oa-80-big")
    # (b) bloom-checked: the disjoint source gates only its ~1% false positives
    rb = ca.compact(ins, f"{d}/oa-91-big", gc_before=GCB,
                    overlaps=[(LO, HI, 1, flt)])
    _oracle_compact(f"{d}/oa-81-big", ins, gcbefore=GCB, ov=f"{LO}:{HI}:1:{flt}")
    _assert_dirs_equal(f"{d}/oa-91-big", f"{d}/oa-81-big")
    # (c) unconstrained purge as the floor; the divergence is real and bounded
    rc2 = ca.compact(ins, f"{d}/oa-92-big", gc_before=GCB)
    assert ra["partitions_out"] == 1600, ra          # interval-only: all retained
    assert rc2["partitions_out"] <= rb["partitions_out"] <= rc2["partitions_out"] * 1.05, \
        (ra["partitions_out"], rb["partitions_out"], rc2["partitions_out"])  # ~1% bloom FPs


def test_bti_da_pipeline(ca, oracle_bin, tmp_path):
    """BTI (`da`) format through the GPU pipeline (VERDICT round-2 item 6):
    the engine ingests da inputs (partition positions from the Partitions.db
    trie / Rows.db TrieIndexEntry footers) and emits da outputs (index
    post-pass building both tries from the writer kernels' block structure),
    byte-identical to the oracle whose own da writer regenerates the
    reference's legacy_da fixtures byte-for-byte."""
    BTI_COMPONENTS = ["Data.db", "CompressionInfo.db", "Filter.db",
                      "Digest.crc32", "Statistics.db", "Partitions.db",
                      "Rows.db", "TOC.txt"]
    d = str(tmp_path)
    # wide partitions so Rows.db carries real row-index tries
    _oracle_gen(d, seed=81, n=3, rows=30, crows=120, vlen=400, overlap=25,
                tomb=10, rtomb=25, bti=1)
    ins = [f"{d}/da-{g}-bti" for g in (1, 2, 3)]
    ca.compact(ins, f"{d}/da-60-bti")
    _oracle_compact(f"{d}/da-80-bti", ins)
    _assert_dirs_equal(f"{d}/da-60-bti", f"{d}/da-80-bti", BTI_COMPONENTS)
    # simple schema (no clustering): Partitions.db only, empty Rows.db
    ds = d + "/s"
    os.makedirs(ds)
    _oracle_gen(ds, seed=82, n=2, rows=1500, vlen=150, overlap=20, tomb=15, bti=1)
    sins = [f"{ds}/da-{g}-bti" for g in (1, 2)]
    ca.compact(sins, f"{ds}/da-60-bti")
    _oracle_compact(f"{ds}/da-80-bti", sins)
    _assert_dirs_equal(f"{ds}/da-60-bti", f"{ds}/da-80-bti", BTI_COMPONENTS)
    # boundary-kind block keys (fuzz regression): dense range tombstones +
    # large cells force row-index blocks that START at range-tombstone
    # boundary markers, so separatorGt stores the marker's OSS50 terminator
    # (LT 0x20 / GT 0x60; Kind.asByteComparableValue at Walker.java:64's
    # Version.OSS50) — the kinds the first encoder version rejected
    dbb = d + "/bb"
    os.makedirs(dbb)
    _oracle_gen(dbb, seed=138011248, n=4, rows=20, crows=5, vlen=20000,
                overlap=40, tomb=10, rtomb=70, cktext=1, statics=30, bti=1)
    bbins = [f"{dbb}/da-{g}-bti" for g in (1, 2, 3, 4)]
    ca.compact(bbins, f"{dbb}/da-60-bti")
    _oracle_compact(f"{dbb}/da-80-bti", bbins)
    _assert_dirs_equal(f"{dbb}/da-60-bti", f"{dbb}/da-80-bti", BTI_COMPONENTS)
    # GPU writer parity: generate(bti=True) == oracle gen bti=1
    dg, do = d + "/g", d + "/o"
    os.makedirs(dg), os.makedirs(do)
    ca.generate(dg, seed=83, n_sstables=2, rows_per_sstable=25, value_len=300,
                overlap_pct=20, clustering_rows=100, range_tomb_pct=25,
                tombstone_pct=10, bti=True)
    _oracle_gen(do, seed=83, n=2, rows=25, crows=100, vlen=300, overlap=20,
                rtomb=25, tomb=10, bti=1)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/da-{g}-bti", f"{do}/da-{g}-bti", BTI_COMPONENTS)


def test_cross_feature_combinations(ca, oracle_bin, tmp_path):
    """Round-2 features composed: (a) complex columns + TTL + expiry-crossing
    purge in one table; (b) complex columns in a da (BTI) sstable set;
    (c) snappy + wide partitions + TTL; (d) complex + overlap bloom purge —
    each byte-identical to the oracle."""
    INT64_MIN = -(2 ** 63)
    d = str(tmp_path)
    # (a) complex + ttl + purge across expiry
    _oracle_gen(d, seed=91, n=3, rows=900, vlen=120, overlap=25, tomb=10,
                cpx=40, cpxdel=25, ttl=30)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    ca.compact(ins, f"{d}/oa-60-big", now_sec=1800000000, gc_before=1700001000)
    _oracle_compact(f"{d}/oa-80-big", ins, now=1800000000, gcbefore=1700001000)
    _assert_dirs_equal(f"{d}/oa-60-big", f"{d}/oa-80-big")
    # (b) complex columns inside da sstables
    db = d + "/b"
    os.makedirs(db)
    _oracle_gen(db, seed=92, n=2, rows=700, vlen=100, overlap=20, tomb=10,
                cpx=45, cpxdel=25, bti=1)
    bins = [f"{db}/da-{g}-bti" for g in (1, 2)]
    ca.compact(bins, f"{db}/da-60-bti")
    _oracle_compact(f"{db}/da-80-bti", bins)
    _assert_dirs_equal(f"{db}/da-60-bti", f"{db}/da-80-bti",
                       ["Data.db", "CompressionInfo.db", "Filter.db", "Digest.crc32",
                        "Statistics.db", "Partitions.db", "Rows.db", "TOC.txt"])
    # (c) snappy + wide + ttl
    dc = d + "/c"
    os.makedirs(dc)
    _oracle_gen(dc, seed=93, n=2, rows=30, crows=60, vlen=150, rtomb=25,
                tomb=10, overlap=20, ttl=35, snappy=1)
    cins = [f"{dc}/oa-{g}-big" for g in (1, 2)]
    ca.compact(cins, f"{dc}/oa-60-big", now_sec=1800000000, gc_before=INT64_MIN)
    _oracle_compact(f"{dc}/oa-80-big", cins, now=1800000000)
    _assert_dirs_equal(f"{dc}/oa-60-big", f"{dc}/oa-80-big")
    # (d) complex + per-key bloom-gated purge
    dd = d + "/d"
    os.makedirs(dd)
    _oracle_gen(dd, seed=94, n=2, rows=600, vlen=80, overlap=0, tomb=40,
                cpx=40, cpxdel=30)
    os.makedirs(dd + "/y")
    _oracle_gen(dd + "/y", seed=95, n=1, rows=300, vlen=50, keylen=9)
    dins = [f"{dd}/oa-1-big", f"{dd}/oa-2-big"]
    LO, HI = INT64_MIN, 2 ** 63 - 1
    flt = f"{dd}/y/oa-1-big-Filter.db"
    ca.compact(dins, f"{dd}/oa-60-big", gc_before=2000000000,
               overlaps=[(LO, HI, 1, flt)])
    _oracle_compact(f"{dd}/oa-80-big", dins, gcbefore=2000000000,
                    ov=f"{LO}:{HI}:1:{flt}")
    _assert_dirs_equal(f"{dd}/oa-60-big", f"{dd}/oa-80-big")


def test_bti_sharding_and_gc(ca, oracle_bin, tmp_path):
    """da (BTI) inputs through the two paths that were previously rejected:
    UCS-style sharded outputs (each shard gets its own Partitions.db/Rows.db
    post-pass) and garbagecollect with da tombstone sources — byte-identical
    to the oracle per shard / per mode."""
    BTI_COMPONENTS = ["Data.db", "CompressionInfo.db", "Filter.db", "Digest.crc32",
                      "Statistics.db", "Partitions.db", "Rows.db", "TOC.txt"]
    from cassandra_amd.sharding import split_token_range
    d = str(tmp_path)
    _oracle_gen(d, seed=71, n=3, rows=1200, vlen=120, overlap=25, tomb=15, bti=1)
    ins = [f"{d}/da-{g}-bti" for g in (1, 2, 3)]
    S = 2
    ca.compact(ins, f"{d}/da-61-bti", n_output_shards=S)
    for i in range(S):
        lo, hi = split_token_range(S, i)
        _oracle_compact(f"{d}/da-{80 + 10 * i}-bti", ins, shard=f"{lo}:{hi}")
        _assert_dirs_equal(f"{d}/da-{61 + i}-bti", f"{d}/da-{80 + 10 * i}-bti",
                           BTI_COMPONENTS)
    # wide da + shards (row-index tries in every shard)
    dw = d + "/w"
    os.makedirs(dw)
    _oracle_gen(dw, seed=72, n=2, rows=40, crows=60, vlen=150, rtomb=30, tomb=10, bti=1)
    wins = [f"{dw}/da-{g}-bti" for g in (1, 2)]
    ca.compact(wins, f"{dw}/da-61-bti", n_output_shards=S)
    for i in range(S):
        lo, hi = split_token_range(S, i)
        _oracle_compact(f"{dw}/da-{80 + 10 * i}-bti", wins, shard=f"{lo}:{hi}")
        _assert_dirs_equal(f"{dw}/da-{61 + i}-bti", f"{dw}/da-{80 + 10 * i}-bti",
                           BTI_COMPONENTS)
    # gc over da inputs with da tombstone sources, both modes
    dg = d + "/g"
    os.makedirs(dg)
    _oracle_gen(dg, seed=73, n=2, rows=800, vlen=80, overlap=30, tomb=20, bti=1)
    os.makedirs(dg + "/src")
    _oracle_gen(dg + "/src", seed=73, n=2, rows=800, vlen=80, overlap=30, tomb=50,
                ts0=1700000500000000, bti=1)
    gins = [f"{dg}/da-{g}-bti" for g in (1, 2)]
    gsrc = [f"{dg}/src/da-{g}-bti" for g in (1, 2)]
    for cell in (False, True):
        out = f"{dg}/da-{61 + cell}-bti"
        ref = f"{dg}/da-{81 + cell}-bti"
        ca.compact(gins, out, tombstone_sources=gsrc, cell_level_gc=cell)
        kw = dict(tombsrc=",".join(gsrc))
        if cell:
            kw["cellgc"] = 1
        _oracle_compact(ref, gins, **kw)
        _assert_dirs_equal(out, ref, BTI_COMPONENTS)


def test_segmented_serialize_parity(ca, oracle_bin, tmp_path):
    """The serialize-interleaved-with-compress path (NSEG > 1) engages only
    above 256 MiB of output — every other parity test is smaller and runs
    the single-segment path. This case crosses the threshold (~320 MiB
    uncompressed output) so segment boundaries, the in-order issue
    interleave and the post-drain index copy are byte-compared too."""
    d = str(tmp_path)
    _oracle_gen(d, seed=61, n=2, rows=160000, vlen=1024, overlap=15, tomb=10)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2)]
    ca.compact(ins, f"{d}/oa-60-big")
    _oracle_compact(f"{d}/oa-80-big", ins)
    _assert_dirs_equal(f"{d}/oa-60-big", f"{d}/oa-80-big")


def test_gc_with_sharded_outputs(ca, oracle_bin, tmp_path):
    """garbagecollect + UCS sharded outputs in one call: each shard applies
    the GarbageSkipper before its writer (sources are NOT shard-filtered —
    they shadow regardless of the shard), byte-identical to the oracle's
    shard= + tombsrc= runs."""
    from cassandra_amd.sharding import split_token_range
    d = str(tmp_path)
    _oracle_gen(d, seed=75, n=2, rows=900, vlen=70, overlap=30, tomb=20)
    os.makedirs(d + "/src")
    _oracle_gen(d + "/src", seed=75, n=2, rows=900, vlen=70, overlap=30, tomb=50,
                ts0=1700000500000000)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2)]
    srcs = [f"{d}/src/oa-{g}-big" for g in (1, 2)]
    S = 2
    for cell in (False, True):
        ca.compact(ins, f"{d}/oa-{61 + 10 * cell}-big", n_output_shards=S,
                   tombstone_sources=srcs, cell_level_gc=cell)
        for i in range(S):
            lo, hi = split_token_range(S, i)
            kw = dict(tombsrc=",".join(srcs), shard=f"{lo}:{hi}")
            if cell:
                kw["cellgc"] = 1
            ref = f"{d}/oa-{81 + 10 * cell + i * 2}-big"
            _oracle_compact(ref, ins, **kw)
            _assert_dirs_equal(f"{d}/oa-{61 + 10 * cell + i}-big", ref)


def test_gc_complex_counter(ca, oracle_bin, tmp_path):
    """nodetool garbagecollect over complex (collection) and counter tables
    (CompactionIterator.GarbageSkipper with Rows.removeShadowedCells'
    complex branch, Rows.java:298-316, and Cells.addNonShadowed's
    resolveCounter identity rule, Cells.java:121-189) — both TombstoneOption
    ROW and CELL modes, byte-identical to the oracle."""
    d = str(tmp_path)
    # complex columns
    _oracle_gen(d, seed=41, n=2, rows=700, vlen=60, overlap=30, tomb=20,
                cpx=45, cpxdel=25)
    os.makedirs(d + "/src")
    _oracle_gen(d + "/src", seed=41, n=2, rows=700, vlen=60, overlap=30,
                tomb=50, cpx=45, cpxdel=25, ts0=1700000500000000)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2)]
    srcs = [f"{d}/src/oa-{g}-big" for g in (1, 2)]
    for cell in (False, True):
        out = f"{d}/oa-{60 + cell}-big"
        ref = f"{d}/oa-{80 + cell}-big"
        ca.compact(ins, out, tombstone_sources=srcs, cell_level_gc=cell)
        kw = dict(tombsrc=",".join(srcs))
        if cell:
            kw["cellgc"] = 1
        _oracle_compact(ref, ins, **kw)
        _assert_dirs_equal(out, ref)
    # counter columns
    dc = d + "/ctr"
    os.makedirs(dc)
    _oracle_gen(dc, seed=42, n=2, rows=700, overlap=30, tomb=15, counter=1)
    os.makedirs(dc + "/src")
    _oracle_gen(dc + "/src", seed=42, n=1, rows=700, tomb=60, counter=1,
                ts0=1700000500000000)
    cins = [f"{dc}/oa-{g}-big" for g in (1, 2)]
    csrc = [f"{dc}/src/oa-1-big"]
    for cell in (False, True):
        out = f"{dc}/oa-{60 + cell}-big"
        ref = f"{dc}/oa-{80 + cell}-big"
        ca.compact(cins, out, tombstone_sources=csrc, cell_level_gc=cell)
        kw = dict(tombsrc=",".join(csrc))
        if cell:
            kw["cellgc"] = 1
        _oracle_compact(ref, cins, **kw)
        _assert_dirs_equal(out, ref)
    # complex + TTL through gc, with a purge pass after the filter
    dt = d + "/ttl"
    os.makedirs(dt)
    _oracle_gen(dt, seed=43, n=2, rows=500, vlen=50, overlap=25, tomb=15,
                cpx=40, cpxdel=25, ttl=30)
    os.makedirs(dt + "/src")
    _oracle_gen(dt + "/src", seed=43, n=1, rows=500, tomb=55, cpx=40,
                cpxdel=30, ts0=1700000500000000)
    tins = [f"{dt}/oa-{g}-big" for g in (1, 2)]
    tsrc = [f"{dt}/src/oa-1-big"]
    ca.compact(tins, f"{dt}/oa-60-big", tombstone_sources=tsrc, cell_level_gc=True,
               now_sec=1800000000, gc_before=1700001000)
    _oracle_compact(f"{dt}/oa-80-big", tins, tombsrc=tsrc[0], cellgc=1,
                    now=1800000000, gcbefore=1700001000)
    _assert_dirs_equal(f"{dt}/oa-60-big", f"{dt}/oa-80-big")


def test_flush_table_full_schema(ca, oracle_bin, tmp_path):
    """gpuc_flush_table — the REAL memtable flush path (VERDICT round-1 weak
    item: the v1 flush schema was `pk blob, val blob` only): partitions with
    clustering rows, range-tombstone markers, statics, TTL, complex columns,
    counters, multi-column subsets, Snappy and BTI (`da`) outputs, handed
    over the C ABI exactly as Memtable.FlushablePartitionSet would
    (token-unsorted partitions, clustering-sorted unfiltereds). The oracle's
    gen dump=1 writes the logical memtable content (.memdump) next to its
    own written sstable; flushing the dump through gpuc_flush_table must
    reproduce that sstable byte-for-byte."""
    BTI_COMPONENTS = ["Data.db", "CompressionInfo.db", "Filter.db", "Digest.crc32",
                      "Statistics.db", "Partitions.db", "Rows.db", "TOC.txt"]
    cases = [
        ("simple", dict(seed=21, rows=800, vlen=90, tomb=15, pdel=3), COMPONENTS),
        ("wide", dict(seed=22, rows=40, crows=50, vlen=120, rtomb=40, tomb=10,
                      statics=40, cktext=1, ttl=30), COMPONENTS),
        ("cpx", dict(seed=23, rows=500, vlen=60, cpx=45, cpxdel=25), COMPONENTS),
        ("counter", dict(seed=24, rows=500, counter=1), COMPONENTS),
        ("multicol", dict(seed=25, rows=60, crows=12, ckcols=2, ncols=5,
                          colmiss=30, snappy=1), COMPONENTS),
        ("bti", dict(seed=26, rows=30, crows=40, vlen=200, rtomb=30, bti=1),
         BTI_COMPONENTS),
    ]
    for tag, genkw, comps in cases:
        d = os.path.join(str(tmp_path), tag)
        os.makedirs(d)
        _oracle_gen(d, n=1, dump=1, **genkw)
        stem = "da-1-bti" if genkw.get("bti") else "oa-1-big"
        ca.flush_table(f"{d}/{stem}.memdump", f"{d}/out")
        _assert_dirs_equal(f"{d}/out", f"{d}/{stem}", comps)
    # dumpsst round-trip: ANY written sstable (here a compact OUTPUT with
    # pdel-only partitions — shapes the generator never emits — and a
    # compacted counter table) re-flushes byte-identically
    for tag, genkw, job in [
        ("pdel", dict(seed=31, n=2, rows=300, vlen=60, overlap=30, tomb=40, pdel=8),
         ["nevergc=1", "gcbefore=2000000000"]),
        ("rectr", dict(seed=32, n=2, rows=400, overlap=30, tomb=20, counter=1), []),
    ]:
        dr = os.path.join(str(tmp_path), "rt_" + tag)
        os.makedirs(dr)
        _oracle_gen(dr, **genkw)
        subprocess.run([ORACLE, "compact", f"{dr}/oa-90-big",
                        *[f"{dr}/oa-{g}-big" for g in (1, 2)], *job],
                       check=True, capture_output=True)
        subprocess.run([ORACLE, "dumpsst", f"{dr}/oa-90-big", f"{dr}/rt.memdump"],
                       check=True, capture_output=True)
        ca.flush_table(f"{dr}/rt.memdump", f"{dr}/oa-60-big")
        _assert_dirs_equal(f"{dr}/oa-60-big", f"{dr}/oa-90-big")
    # loud failures: out-of-order unfiltereds and duplicate keys
    d = os.path.join(str(tmp_path), "bad")
    os.makedirs(d)
    _oracle_gen(d, n=1, seed=27, rows=50, crows=6, vlen=20, dump=1)
    data = open(f"{d}/oa-1-big.memdump", "rb").read()
    S, parts, n, bufs = ca._parse_memdump(data)
    u = parts[0].unf
    b0, b1 = bytes(u[0]), bytes(u[1])  # swap two clustering-ordered rows
    ctypes.memmove(ctypes.byref(u[0]), b1, len(b1))
    ctypes.memmove(ctypes.byref(u[1]), b0, len(b0))
    with pytest.raises(ca.GpuCompactError, match="clustering order"):
        ca._flush_table_parsed(S, parts, n, f"{d}/bad-out")
    del bufs


def test_counter_columns_pipeline(ca, oracle_bin, tmp_path):
    """Counter columns on GPU (CounterColumnType): the k-way CounterContext
    merge in reconcile (provably equal to the reference's pairwise chain —
    commutative lattice; oracle transcribes pairwise) + Cells.resolveCounter
    tombstone/empty rules, byte-identical to the oracle across 3-way merges,
    re-compaction, purge of counter tombstones, and writer parity. Cell
    format pinned by the reference's legacy_oa_*_counter fixtures."""
    d = str(tmp_path)
    _oracle_gen(d, seed=99, n=3, rows=1200, overlap=30, tomb=10, counter=1)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    ca.compact(ins, f"{d}/oa-60-big")
    _oracle_compact(f"{d}/oa-80-big", ins)
    _assert_dirs_equal(f"{d}/oa-60-big", f"{d}/oa-80-big")
    ca.verify(f"{d}/oa-60-big")
    # re-compaction of a compaction output + purge of counter tombstones
    ca.compact([f"{d}/oa-60-big", ins[2]], f"{d}/oa-61-big",
               now_sec=1800000000, gc_before=1800000000)
    _oracle_compact(f"{d}/oa-81-big", [f"{d}/oa-80-big", ins[2]],
                    now=1800000000, gcbefore=1800000000)
    _assert_dirs_equal(f"{d}/oa-61-big", f"{d}/oa-81-big")
    # writer parity: generate(counter=True) == oracle gen counter=1
    dg, do = d + "/g", d + "/o"
    os.makedirs(dg), os.makedirs(do)
    ca.generate(dg, seed=99, n_sstables=2, rows_per_sstable=800, overlap_pct=30,
                tombstone_pct=10, counter=True)
    _oracle_gen(do, seed=99, n=2, rows=800, overlap=30, tomb=10, counter=1)
    for g in (1, 2):
        _assert_dirs_equal(f"{dg}/oa-{g}-big", f"{do}/oa-{g}-big")
    # wide counter partitions (the clust-counter fixture shape)
    dw = d + "/w"
    os.makedirs(dw)
    _oracle_gen(dw, seed=98, n=2, rows=40, crows=50, overlap=25, tomb=10, counter=1)
    wins = [f"{dw}/oa-{g}-big" for g in (1, 2)]
    ca.compact(wins, f"{dw}/oa-60-big")
    _oracle_compact(f"{dw}/oa-80-big", wins)
    _assert_dirs_equal(f"{dw}/oa-60-big", f"{dw}/oa-80-big")


def test_validation_compaction(ca, oracle_bin, tmp_path):
    """VALIDATION compaction (the repair digest epilogue — the last §8(f)3
    piece): gpuc_validate merges+purges like a compaction and emits per-
    partition (token, 32-byte concat-murmur3_128(1000|2000)) repair digests
    (Validator.rowHash, Digest.forValidator), byte-identical to the oracle's
    transcription across simple, wide+tombstone, complex, TTL and counter
    tables."""
    d = str(tmp_path)
    cases = [
        ("s", dict(seed=5, n=3, rows=900, vlen=100, overlap=25, tomb=15, pdel=2), {}),
        ("w", dict(seed=6, n=2, rows=40, crows=60, vlen=120, rtomb=25, tomb=10,
                   overlap=20, statics=40, cktext=1), {}),
        ("x", dict(seed=7, n=2, rows=600, vlen=80, overlap=25, tomb=10, cpx=40,
                   cpxdel=25), {}),
        ("t", dict(seed=8, n=2, rows=600, vlen=80, overlap=25, ttl=35), dict(now=1800000000)),
        ("c", dict(seed=9, n=2, rows=600, overlap=30, tomb=10, counter=1), {}),
    ]
    for tag, genkw, ckw in cases:
        dd = os.path.join(d, tag)
        os.makedirs(dd)
        _oracle_gen(dd, **genkw)
        ins = [f"{dd}/oa-{g}-big" for g in range(1, genkw["n"] + 1)]
        subprocess.run([ORACLE, "validate", f"{dd}/v_cpu.bin", *ins,
                        *[f"{k}={v}" for k, v in ckw.items()]],
                       check=True, capture_output=True)
        n = ca.validate(ins, f"{dd}/v_gpu.bin",
                        now_sec=ckw.get("now", 1800000000))
        a = open(f"{dd}/v_cpu.bin", "rb").read()
        b = open(f"{dd}/v_gpu.bin", "rb").read()
        assert n == len(a) // 40 and a == b, (
            tag, n, len(a) // 40, len(b) // 40,
            next((i for i in range(min(len(a), len(b))) if a[i] != b[i]), -1))


def test_cancellation(ca, oracle_bin, tmp_path):
    """Cooperative cancel (CompactionIterator.isStopRequested): a set
    cancel_flag aborts the task with GPUC_ERR_CANCELLED; a zero flag is
    inert and the output still matches the oracle byte-for-byte."""
    import ctypes
    d = str(tmp_path)
    _oracle_gen(d, seed=401, n=2, rows=800, vlen=100, overlap=20)
    ins = [f"{d}/oa-1-big", f"{d}/oa-2-big"]
    flag = ctypes.c_int32(1)
    try:
        ca.compact(ins, f"{d}/oa-90-big", cancel_flag=ctypes.pointer(flag))
        raise AssertionError("cancelled compaction did not fail")
    except ca.GpuCompactError as e:
        assert "rc=7" in str(e) and "cancel" in str(e), e
    flag.value = 0
    ca.compact(ins, f"{d}/oa-91-big", cancel_flag=ctypes.pointer(flag))
    _oracle_compact(f"{d}/oa-92-big", ins)
    _assert_dirs_equal(f"{d}/oa-92-big", f"{d}/oa-91-big")


def test_empty_outputs(ca, oracle_bin, tmp_path):
    """Degenerate outputs stay byte-identical: (1) a compaction whose every
    row is purged writes the oracle's empty sstable; (2) a scrub where every
    chunk is corrupt keeps 0 partitions and writes the same empty table."""
    import json
    d = str(tmp_path)
    _oracle_gen(d, seed=5, n=1, rows=200, vlen=50, tomb=100, overlap=0,
                ts0=1000000, ldt0=1000)
    _oracle_compact(f"{d}/oa-90-big", [f"{d}/oa-1-big"], gcbefore=2000000000)
    ca.compact([f"{d}/oa-1-big"], f"{d}/oa-91-big", gc_before=2000000000)
    _assert_dirs_equal(f"{d}/oa-90-big", f"{d}/oa-91-big")
    os.makedirs(d + "/s")
    _oracle_gen(d + "/s", seed=6, n=1, rows=50, vlen=40, overlap=0)
    base = f"{d}/s/oa-1-big"
    sz = os.path.getsize(base + "-Data.db")
    with open(base + "-Data.db", "r+b") as f:
        for off in range(5, sz, 4000):
            f.seek(off)
            b0 = f.read(1)
            f.seek(-1, 1)
            f.write(bytes([b0[0] ^ 0xFF]))
    out = subprocess.run([ORACLE, "scrub", f"{d}/s/oa-80-big", base],
                         capture_output=True, text=True, check=True)
    ores = json.loads(out.stdout.splitlines()[-1])
    kept, dropped = ca.scrub(base, f"{d}/s/oa-81-big")
    assert kept == 0 and (kept, dropped) == (ores["partitions_kept"],
                                             ores["partitions_dropped"])
    _assert_dirs_equal(f"{d}/s/oa-80-big", f"{d}/s/oa-81-big")


def test_compact_rejects_corrupt_input(ca, oracle_bin, tmp_path):
    """A flipped byte in an input chunk must fail the compaction loudly
    (CompressedChunkReader CRC semantics) — no silent bad output."""
    d = str(tmp_path)
    _oracle_gen(d, seed=131, n=2, rows=1000, vlen=300, overlap=20)
    with open(f"{d}/oa-1-big-Data.db", "r+b") as f:
        f.seek(os.path.getsize(f"{d}/oa-1-big-Data.db") // 3)
        b0 = f.read(1)
        f.seek(-1, 1)
        f.write(bytes([b0[0] ^ 0x40]))
    try:
        ca.compact([f"{d}/oa-1-big", f"{d}/oa-2-big"], f"{d}/oa-9-big")
        raise AssertionError("corrupt input accepted")
    except Exception as e:
        assert "gpuc_compact" in str(e), e


def test_purge_overlap_table_parity(ca, oracle_bin, tmp_path):
    """gcBefore purge gated by the token-interval min-timestamp table."""
    d = str(tmp_path)
    _oracle_gen(d, seed=55, n=3, rows=2000, vlen=128, overlap=20, tomb=30)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2, 3)]
    # overlap table: left half of the token space has an old overlapping source
    # (min_timestamp below every generated ts -> nothing there may purge)
    lo, hi, ts = -(2**63), 0, 1
    # oracle CLI has no overlap flag yet -> emulate with two sharded runs:
    # shard A (overlapped: purge denied == nevergc), shard B (purge allowed)
    _oracle_compact(f"{d}/oa-80-big", ins, shard=f"{lo}:{hi}",
                    gcbefore=2000000000, nevergc=1)
    _oracle_compact(f"{d}/oa-81-big", ins, shard=f"{hi+1}:{2**63-1}",
                    gcbefore=2000000000)
    ca.compact(ins, f"{d}/oa-90-big", gc_before=2000000000,
               overlaps=[(lo, hi, ts)], token_range=(lo, hi))
    ca.compact(ins, f"{d}/oa-91-big", gc_before=2000000000,
               overlaps=[(lo, hi, ts)], token_range=(hi + 1, 2**63 - 1))
    _assert_dirs_equal(f"{d}/oa-80-big", f"{d}/oa-90-big", ["Data.db", "Index.db"])
    _assert_dirs_equal(f"{d}/oa-81-big", f"{d}/oa-91-big", ["Data.db", "Index.db"])


def test_token_shards_compose(ca, oracle_bin, tmp_path):
    """Disjoint token shards partition the full output exactly (§8(e))."""
    d = str(tmp_path)
    _oracle_gen(d, seed=61, n=2, rows=3000, vlen=200, overlap=10)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2)]
    full = ca.compact(ins, f"{d}/oa-90-big")
    bounds = [-(2**63), -(2**62), 0, 2**62, 2**63 - 1]
    total = 0
    for i in range(4):
        lo = bounds[i] if i == 0 else bounds[i] + 1
        r = ca.compact(ins, f"{d}/oa-9{i+1}-big", token_range=(lo, bounds[i + 1]))
        total += r["partitions_out"]
    assert total == full["partitions_out"]


def test_reads_reference_written_inputs(ca, oracle_bin, tmp_path):
    """GPU path consumes oracle-written sstables (which byte-round-trip the
    reference's own fixtures) — covers the full read path against the same
    wire format the reference writes."""
    d = str(tmp_path)
    _oracle_gen(d, seed=77, n=2, rows=1000, vlen=100, overlap=0)
    ins = [f"{d}/oa-{g}-big" for g in (1, 2)]
    r = ca.compact(ins, f"{d}/oa-90-big")
    assert r["partitions_in"] == 2000
    assert r["input_uncompressed_bytes"] > 0
