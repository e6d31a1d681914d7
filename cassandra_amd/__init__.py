"""cassandra_amd — MI355X-native SSTable compaction (host bindings).

Thin ctypes wrapper over libcassandra_gpucompact.so (the C-ABI HIP library;
see include/gpucompact.h). All compute is GPU-side; this package only
marshals job descriptions, mirroring what a JNI/Panama shim in the Java host
would do.
"""
import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libcassandra_gpucompact.so")


class GpucPurgeRange(ctypes.Structure):
    _fields_ = [
        ("token_lo", ctypes.c_int64),
        ("token_hi", ctypes.c_int64),
        ("min_timestamp", ctypes.c_int64),
        # optional Filter.db bit payload of the overlapping sstable: enables
        # the per-key bloom-checked purge evaluator (see gpucompact.h)
        ("bloom_bits", ctypes.POINTER(ctypes.c_uint32)),
        ("bloom_bit_len", ctypes.c_uint64),
        ("bloom_hash_count", ctypes.c_int32),
    ]


class GpucJob(ctypes.Structure):
    _fields_ = [
        ("input_bases", ctypes.POINTER(ctypes.c_char_p)),
        ("n_inputs", ctypes.c_int32),
        ("output_base", ctypes.c_char_p),
        ("now_sec", ctypes.c_int64),
        ("gc_before", ctypes.c_int64),
        ("never_purge", ctypes.c_int32),
        ("enforce_strict_liveness", ctypes.c_int32),
        ("overlaps", ctypes.POINTER(GpucPurgeRange)),
        ("n_overlaps", ctypes.c_int32),
        ("has_token_range", ctypes.c_int32),
        ("token_lo", ctypes.c_int64),
        ("token_hi", ctypes.c_int64),
        ("keep_ranges", ctypes.POINTER(GpucPurgeRange)),
        ("n_keep_ranges", ctypes.c_int32),
        ("invert_ranges", ctypes.c_int32),
        ("device", ctypes.c_int32),
        ("tombstone_source_bases", ctypes.POINTER(ctypes.c_char_p)),
        ("n_tomb_sources", ctypes.c_int32),
        ("cell_level_gc", ctypes.c_int32),
        ("n_output_shards", ctypes.c_int32),
        ("cancel_flag", ctypes.POINTER(ctypes.c_int32)),
    ]


class GpucResult(ctypes.Structure):
    _fields_ = [
        ("input_uncompressed_bytes", ctypes.c_uint64),
        ("output_uncompressed_bytes", ctypes.c_uint64),
        ("output_compressed_bytes", ctypes.c_uint64),
        ("partitions_in", ctypes.c_uint64),
        ("partitions_out", ctypes.c_uint64),
        ("rows_in", ctypes.c_uint64),
        ("rows_out", ctypes.c_uint64),
        ("merged_counts", ctypes.c_uint64 * 64),
        ("ms_read_io", ctypes.c_double),
        ("ms_h2d", ctypes.c_double),
        ("ms_decompress", ctypes.c_double),
        ("ms_parse", ctypes.c_double),
        ("ms_merge", ctypes.c_double),
        ("ms_reconcile", ctypes.c_double),
        ("ms_serialize", ctypes.c_double),
        ("ms_compress", ctypes.c_double),
        ("ms_d2h", ctypes.c_double),
        ("ms_write_io", ctypes.c_double),
        ("ms_total", ctypes.c_double),
        ("dominant_kernel", ctypes.c_char * 64),
        ("dominant_kernel_ms", ctypes.c_double),
        ("dominant_kernel_launches", ctypes.c_uint64),
        ("error", ctypes.c_char * 256),
    ]


class GpucGenSpec(ctypes.Structure):
    _fields_ = [
        ("seed", ctypes.c_uint64),
        ("n_sstables", ctypes.c_uint32),
        ("rows_per_sstable", ctypes.c_uint64),
        ("overlap_pct", ctypes.c_uint32),
        ("value_len", ctypes.c_uint32),
        ("value_repeat_pct", ctypes.c_uint32),
        ("tombstone_pct", ctypes.c_uint32),
        ("partition_del_pct", ctypes.c_uint32),
        ("clustering_rows", ctypes.c_uint32),
        ("range_tomb_pct", ctypes.c_uint32),
        ("key_len", ctypes.c_uint32),
        ("ck_text", ctypes.c_uint32),
        ("ck_cols", ctypes.c_uint32),
        ("static_pct", ctypes.c_uint32),
        ("n_value_cols", ctypes.c_uint32),
        ("col_missing_pct", ctypes.c_uint32),
        ("base_ts", ctypes.c_int64),
        ("base_ldt", ctypes.c_int64),
        ("first_generation", ctypes.c_uint64),
        ("device", ctypes.c_int32),
        ("snappy", ctypes.c_int32),
        ("ttl_pct", ctypes.c_uint32),
        ("complex_pct", ctypes.c_uint32),
        ("complex_del_pct", ctypes.c_uint32),
        ("bti", ctypes.c_int32),
        ("counter", ctypes.c_int32),
    ]


class GpucFlushRows(ctypes.Structure):
    _fields_ = [
        ("n_rows", ctypes.c_uint64),
        ("keys", ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        ("key_lens", ctypes.POINTER(ctypes.c_uint16)),
        ("timestamps", ctypes.POINTER(ctypes.c_int64)),
        ("values", ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        ("value_lens", ctypes.POINTER(ctypes.c_uint32)),
        ("del_ldts", ctypes.POINTER(ctypes.c_uint32)),
    ]


INT64_MIN = -(2**63)


class GpuCompactError(RuntimeError):
    pass


_lib = None


def load_library(path=None):
    """Load the HIP library. Raises loudly when absent — there is no fallback."""
    global _lib
    if _lib is not None and path is None:
        return _lib
    p = path or _LIB_PATH
    if not os.path.exists(p):
        raise GpuCompactError(
            f"libcassandra_gpucompact.so not found at {p}; run __graft_entry__.build()"
        )
    lib = ctypes.CDLL(p)
    lib.gpuc_compact.argtypes = [ctypes.POINTER(GpucJob), ctypes.POINTER(GpucResult)]
    lib.gpuc_compact.restype = ctypes.c_int
    lib.gpuc_generate.argtypes = [
        ctypes.POINTER(GpucGenSpec),
        ctypes.c_char_p,
        ctypes.c_char_p,
        ctypes.c_size_t,
    ]
    lib.gpuc_generate.restype = ctypes.c_int
    lib.gpuc_verify.argtypes = [ctypes.c_char_p, ctypes.c_int32, ctypes.c_char_p, ctypes.c_size_t]
    lib.gpuc_verify.restype = ctypes.c_int
    lib.gpuc_scrub.argtypes = [ctypes.c_char_p, ctypes.c_char_p, ctypes.c_int32,
                               ctypes.POINTER(ctypes.c_uint64), ctypes.POINTER(ctypes.c_uint64),
                               ctypes.c_char_p, ctypes.c_size_t]
    lib.gpuc_scrub.restype = ctypes.c_int
    lib.gpuc_flush.argtypes = [ctypes.POINTER(GpucFlushRows), ctypes.c_char_p, ctypes.c_int32,
                               ctypes.c_char_p, ctypes.c_size_t]
    lib.gpuc_flush.restype = ctypes.c_int
    if hasattr(lib, "gpuc_flush_table"):  # absent only in frozen A/B probe builds
        lib.gpuc_flush_table.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_uint64,
                                         ctypes.c_char_p, ctypes.c_int32,
                                         ctypes.c_char_p, ctypes.c_size_t]
        lib.gpuc_flush_table.restype = ctypes.c_int
    lib.gpuc_version.restype = ctypes.c_char_p
    lib.gpuc_device_count.restype = ctypes.c_int
    _lib = lib
    return lib


def flush(rows, output_base, device=0):
    """Memtable-flush analog: rows = list of (key_bytes, timestamp, value_bytes)
    for live rows or (key_bytes, timestamp, None, del_ldt) for row tombstones.
    Keys must be unique; sorting happens on the GPU."""
    lib = load_library()
    n = len(rows)
    keys = (ctypes.POINTER(ctypes.c_uint8) * n)()
    klens = (ctypes.c_uint16 * n)()
    tss = (ctypes.c_int64 * n)()
    vals = (ctypes.POINTER(ctypes.c_uint8) * n)()
    vlens = (ctypes.c_uint32 * n)()
    dldts = (ctypes.c_uint32 * n)()
    bufs = []
    for i, r in enumerate(rows):
        kb = ctypes.create_string_buffer(bytes(r[0]), len(r[0]))
        bufs.append(kb)
        keys[i] = ctypes.cast(kb, ctypes.POINTER(ctypes.c_uint8))
        klens[i] = len(r[0])
        tss[i] = r[1]
        if r[2] is None:
            vals[i] = None
            vlens[i] = 0
            dldts[i] = r[3]
        else:
            vb = ctypes.create_string_buffer(bytes(r[2]), len(r[2]))
            bufs.append(vb)
            vals[i] = ctypes.cast(vb, ctypes.POINTER(ctypes.c_uint8))
            vlens[i] = len(r[2])
            dldts[i] = 0xFFFFFFFF
    fr = GpucFlushRows(n_rows=n, keys=keys, key_lens=klens, timestamps=tss,
                       values=vals, value_lens=vlens, del_ldts=dldts)
    err = ctypes.create_string_buffer(256)
    rc = lib.gpuc_flush(ctypes.byref(fr), output_base.encode(), device, err, 256)
    if rc != 0:
        raise GpuCompactError(f"gpuc_flush rc={rc}: {err.value.decode(errors='replace')}")


# ---- full-schema memtable flush (gpuc_flush_table) ----
LDT_NONE = 0xFFFFFFFF
CELLF_PRESENT, CELLF_HAS_VALUE = 1, 2


class GpucCell(ctypes.Structure):
    _fields_ = [
        ("flags", ctypes.c_uint8),
        ("ts", ctypes.c_int64),
        ("ldt", ctypes.c_uint32),
        ("ttl", ctypes.c_int32),
        ("value", ctypes.POINTER(ctypes.c_uint8)),
        ("value_len", ctypes.c_uint32),
    ]


class GpucCpxCell(ctypes.Structure):
    _fields_ = [
        ("cell", GpucCell),
        ("path", ctypes.POINTER(ctypes.c_uint8)),
        ("path_len", ctypes.c_uint32),
    ]


class GpucUnfiltered(ctypes.Structure):
    _fields_ = [
        ("kind", ctypes.c_uint8),
        ("ck_count", ctypes.c_uint8),
        ("row_flags", ctypes.c_uint8),
        ("ck", ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        ("ck_lens", ctypes.POINTER(ctypes.c_uint32)),
        ("live_ts", ctypes.c_int64),
        ("live_ttl", ctypes.c_int32),
        ("live_let", ctypes.c_int64),
        ("del_mfda", ctypes.c_int64),
        ("del_ldt", ctypes.c_uint32),
        ("open_mfda", ctypes.c_int64),
        ("open_ldt", ctypes.c_uint32),
        ("cells", ctypes.POINTER(GpucCell)),
        ("has_cpx", ctypes.c_uint8),
        ("cpx_del_mfda", ctypes.c_int64),
        ("cpx_del_ldt", ctypes.c_uint32),
        ("n_cpx_cells", ctypes.c_uint32),
        ("cpx_cells", ctypes.POINTER(GpucCpxCell)),
    ]


class GpucFlushPart(ctypes.Structure):
    _fields_ = [
        ("key", ctypes.POINTER(ctypes.c_uint8)),
        ("key_len", ctypes.c_uint16),
        ("pdel_mfda", ctypes.c_int64),
        ("pdel_ldt", ctypes.c_uint32),
        ("static_flags", ctypes.c_uint8),
        ("static_live_ts", ctypes.c_int64),
        ("static_live_ttl", ctypes.c_int32),
        ("static_live_let", ctypes.c_int64),
        ("static_del_mfda", ctypes.c_int64),
        ("static_del_ldt", ctypes.c_uint32),
        ("static_cells", ctypes.POINTER(GpucCell)),
        ("n_unf", ctypes.c_uint64),
        ("unf", ctypes.POINTER(GpucUnfiltered)),
    ]


class GpucFlushSchema(ctypes.Structure):
    _fields_ = [
        ("key_type", ctypes.c_char_p),
        ("n_ck", ctypes.c_uint32),
        ("ck_types", ctypes.POINTER(ctypes.c_char_p)),
        ("n_cols", ctypes.c_uint32),
        ("col_names", ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        ("col_name_lens", ctypes.POINTER(ctypes.c_uint32)),
        ("col_types", ctypes.POINTER(ctypes.c_char_p)),
        ("n_static", ctypes.c_uint32),
        ("static_names", ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        ("static_name_lens", ctypes.POINTER(ctypes.c_uint32)),
        ("static_types", ctypes.POINTER(ctypes.c_char_p)),
        ("n_cpx", ctypes.c_uint32),
        ("column_index_size", ctypes.c_uint32),
        ("snappy", ctypes.c_uint32),
        ("bti", ctypes.c_uint32),
        ("has_stats", ctypes.c_uint8),
        ("stats_min_ts", ctypes.c_int64),
        ("stats_min_ldt", ctypes.c_int64),
        ("stats_min_ttl", ctypes.c_int32),
    ]


# oracle CqlType ordinal -> AbstractType class name (sstable.h:30,
# cql_type_name) — the memdump stores the ordinal
_CQL_TYPE_NAMES = [
    b"org.apache.cassandra.db.marshal.BytesType",
    b"org.apache.cassandra.db.marshal.UTF8Type",
    b"org.apache.cassandra.db.marshal.AsciiType",
    b"org.apache.cassandra.db.marshal.LongType",
    b"org.apache.cassandra.db.marshal.Int32Type",
    b"org.apache.cassandra.db.marshal.MapType"
    b"(org.apache.cassandra.db.marshal.BytesType,"
    b"org.apache.cassandra.db.marshal.BytesType)",
    b"org.apache.cassandra.db.marshal.CounterColumnType",
]


class _MemdumpReader:
    """Parses the oracle's .memdump interchange (write_memdump) into the
    gpuc_flush_table ctypes graph. Keeps every referenced buffer alive via
    self.bufs (ctypes pointers do not own memory)."""

    def __init__(self, data):
        self.d = data
        self.o = 0
        self.bufs = []

    def u8(self):
        v = self.d[self.o]; self.o += 1; return v

    def u32(self):
        v = int.from_bytes(self.d[self.o:self.o + 4], "little"); self.o += 4; return v

    def i32(self):
        v = int.from_bytes(self.d[self.o:self.o + 4], "little", signed=True); self.o += 4; return v

    def u64(self):
        v = int.from_bytes(self.d[self.o:self.o + 8], "little"); self.o += 8; return v

    def i64(self):
        v = int.from_bytes(self.d[self.o:self.o + 8], "little", signed=True); self.o += 8; return v

    def blob(self):
        n = self.u32()
        v = self.d[self.o:self.o + n]; self.o += n; return v

    def ptr(self, b):
        buf = ctypes.create_string_buffer(bytes(b), max(len(b), 1))
        self.bufs.append(buf)
        return ctypes.cast(buf, ctypes.POINTER(ctypes.c_uint8)), len(b)

    def cell_into(self, c):
        cf = self.u8()
        c.flags = cf
        if not cf:
            return
        c.ts = self.i64()
        c.ldt = self.u32()
        c.ttl = self.i32()
        v = self.blob()
        c.value, c.value_len = self.ptr(v)

    def row_head(self):
        f = self.u8()
        return f, self.i64(), self.i32(), self.i64(), self.i64(), self.u32()


def _parse_memdump(data):
    r = _MemdumpReader(data)
    if data[:4] != b"GMD2":
        raise GpuCompactError("bad memdump magic")
    r.o = 4
    key_type = _CQL_TYPE_NAMES[r.u8()]
    n_ck = r.u32()
    ck_types = [_CQL_TYPE_NAMES[r.u8()] for _ in range(n_ck)]
    n_static = r.u32()
    statics = [(r.blob(), _CQL_TYPE_NAMES[r.u8()]) for _ in range(n_static)]
    n_cols = r.u32()
    cols = [(r.blob(), r.u8()) for _ in range(n_cols)]
    n_cpx = sum(1 for _, t in cols if t == 5)
    cols = [(nm, _CQL_TYPE_NAMES[t]) for nm, t in cols]
    snappy = r.u8()
    bti = r.u8()
    cis = r.u32()
    stats_ts, stats_ldt, stats_ttl = r.i64(), r.i64(), r.i32()
    n_parts = r.u64()

    S = GpucFlushSchema()
    S.key_type = key_type
    S.n_ck = n_ck
    S.n_cols = n_cols
    S.n_static = n_static
    cn = [r.ptr(nm) for nm, _ in cols]
    sn = [r.ptr(nm) for nm, _ in statics]
    # ctypes Structure POINTER fields do not keep their targets alive: build
    # the arrays as locals, stash them in r.bufs, THEN point S at them
    a_ckt = (ctypes.c_char_p * max(n_ck, 1))(*ck_types)
    a_cnp = (ctypes.POINTER(ctypes.c_uint8) * n_cols)(*[p for p, _ in cn])
    a_cnl = (ctypes.c_uint32 * n_cols)(*[l for _, l in cn])
    a_ct = (ctypes.c_char_p * n_cols)(*[t for _, t in cols])
    a_snp = (ctypes.POINTER(ctypes.c_uint8) * max(n_static, 1))(*[p for p, _ in sn])
    a_snl = (ctypes.c_uint32 * max(n_static, 1))(*[l for _, l in sn])
    a_st = (ctypes.c_char_p * max(n_static, 1))(*[t for _, t in statics])
    r.bufs.extend((a_ckt, a_cnp, a_cnl, a_ct, a_snp, a_snl, a_st))
    S.ck_types, S.col_names, S.col_name_lens, S.col_types = a_ckt, a_cnp, a_cnl, a_ct
    S.static_names, S.static_name_lens, S.static_types = a_snp, a_snl, a_st
    S.n_cpx = n_cpx
    S.column_index_size = cis
    S.snappy = snappy
    S.bti = bti
    S.has_stats = 1
    S.stats_min_ts, S.stats_min_ldt, S.stats_min_ttl = stats_ts, stats_ldt, stats_ttl
    n_simple = n_cols - n_cpx
    cpx_idx = [i for i, (_, t) in enumerate(cols) if b"MapType" in t]

    parts = (GpucFlushPart * n_parts)()
    for pi in range(n_parts):
        P = parts[pi]
        key = r.blob()
        P.key, P.key_len = r.ptr(key)
        P.pdel_mfda = r.i64()
        P.pdel_ldt = r.u32()
        P.static_flags = 0
        if n_static:
            f = r.u8()
            if f:
                P.static_flags = f
                P.static_live_ts = r.i64()
                P.static_live_ttl = r.i32()
                P.static_live_let = r.i64()
                P.static_del_mfda = r.i64()
                P.static_del_ldt = r.u32()
                sc = (GpucCell * n_static)()
                for c in range(n_static):
                    r.cell_into(sc[c])
                r.bufs.append(sc)
                P.static_cells = sc
        n_unf = r.u32()
        P.n_unf = n_unf
        unf = (GpucUnfiltered * max(n_unf, 1))()
        r.bufs.append(unf)
        P.unf = unf
        for ui in range(n_unf):
            U = unf[ui]
            tag = r.u8()
            if tag == 1:
                U.kind = r.u8()
            else:
                U.kind = 4
            ckc = r.u8()
            U.ck_count = ckc
            ckp = (ctypes.POINTER(ctypes.c_uint8) * max(ckc, 1))()
            ckl = (ctypes.c_uint32 * max(ckc, 1))()
            for c in range(ckc):
                state = r.u8()
                comp = r.blob()
                if state != 0:
                    raise GpuCompactError("EMPTY/NULL clustering components unsupported")
                ckp[c], ckl[c] = r.ptr(comp)
            r.bufs.extend((ckp, ckl))
            U.ck = ckp
            U.ck_lens = ckl
            if tag == 1:
                U.del_mfda = r.i64()
                U.del_ldt = r.u32()
                U.open_mfda = r.i64()
                U.open_ldt = r.u32()
                continue
            f, lts, lttl, llet, dm, dl = r.row_head()
            U.row_flags = f
            U.live_ts, U.live_ttl, U.live_let = lts, lttl, llet
            U.del_mfda, U.del_ldt = dm, dl
            cells = (GpucCell * max(n_simple, 1))()
            r.bufs.append(cells)
            U.cells = cells
            si = 0
            for i in range(n_cols):
                if i in cpx_idx:
                    has = r.u8()
                    U.has_cpx = has
                    if has:
                        U.cpx_del_mfda = r.i64()
                        U.cpx_del_ldt = r.u32()
                        nc = r.u32()
                        U.n_cpx_cells = nc
                        xc = (GpucCpxCell * max(nc, 1))()
                        r.bufs.append(xc)
                        U.cpx_cells = xc
                        for e in range(nc):
                            r.cell_into(xc[e].cell)
                            path = r.blob()
                            xc[e].path, xc[e].path_len = r.ptr(path)
                else:
                    r.cell_into(cells[si])
                    si += 1
    if r.o != len(data):
        raise GpuCompactError(f"memdump trailing bytes: {r.o} != {len(data)}")
    return S, parts, n_parts, r.bufs


def _flush_table_parsed(S, parts, n_parts, output_base, device=0):
    lib = load_library()
    err = ctypes.create_string_buffer(256)
    rc = lib.gpuc_flush_table(ctypes.byref(S), parts, n_parts, output_base.encode(),
                              device, err, 256)
    if rc != 0:
        raise GpuCompactError(f"gpuc_flush_table rc={rc}: {err.value.decode(errors='replace')}")


def flush_table(memdump_path, output_base, device=0):
    """Full-schema memtable flush: marshal an oracle .memdump (the logical
    memtable content) through gpuc_flush_table. A real host would build the
    same structs straight from its memtable."""
    with open(memdump_path, "rb") as f:
        data = f.read()
    S, parts, n_parts, bufs = _parse_memdump(data)
    _flush_table_parsed(S, parts, n_parts, output_base, device)
    del bufs


def scrub(input_base, output_base, device=0):
    """sstablescrub equivalent: salvage partitions untouched by corrupt
    chunks into a clean sstable. Returns (kept, dropped)."""
    lib = load_library()
    kept = ctypes.c_uint64()
    dropped = ctypes.c_uint64()
    err = ctypes.create_string_buffer(256)
    rc = lib.gpuc_scrub(input_base.encode(), output_base.encode(), device,
                        ctypes.byref(kept), ctypes.byref(dropped), err, 256)
    if rc != 0:
        raise GpuCompactError(f"gpuc_scrub rc={rc}: {err.value.decode(errors='replace')}")
    return kept.value, dropped.value


def verify(input_base, device=0):
    """sstableverify --extended equivalent: raises GpuCompactError on any
    corruption (chunk CRC, digest, row format, key order, bloom mismatch)."""
    lib = load_library()
    err = ctypes.create_string_buffer(256)
    rc = lib.gpuc_verify(input_base.encode(), device, err, 256)
    if rc != 0:
        raise GpuCompactError(f"gpuc_verify rc={rc}: {err.value.decode(errors='replace')}")


def version():
    return load_library().gpuc_version().decode()


def device_count():
    return load_library().gpuc_device_count()


def compact(
    input_bases,
    output_base,
    now_sec=1800000000,
    gc_before=INT64_MIN,
    never_purge=False,
    enforce_strict_liveness=False,
    overlaps=None,
    token_range=None,
    device=0,
    n_output_shards=1,
    cancel_flag=None,
    tombstone_sources=None,
    cell_level_gc=False,
    keep_ranges=None,
    invert_ranges=False,
):
    """One compaction task (mirrors CompactionTask.runMayThrow's hot loop).

    input_bases: list of sstable path prefixes ('<dir>/oa-<gen>-big').
    overlaps: list of (token_lo, token_hi, min_timestamp) for the purge
      evaluator (CompactionController.getPurgeEvaluator); None == no overlaps.
    token_range: (lo, hi) inclusive token shard restriction or None.
    """
    lib = load_library()
    job = GpucJob()
    arr = (ctypes.c_char_p * len(input_bases))(*[b.encode() for b in input_bases])
    job.input_bases = arr
    job.n_inputs = len(input_bases)
    job.output_base = output_base.encode()
    job.now_sec = now_sec
    job.gc_before = gc_before
    job.never_purge = 1 if never_purge else 0
    job.enforce_strict_liveness = 1 if enforce_strict_liveness else 0
    if overlaps:
        ovr = (GpucPurgeRange * len(overlaps))()
        _bloom_keep = []  # keep the bit buffers alive for the call
        for i, ov in enumerate(overlaps):
            if len(ov) == 3:
                lo, hi, ts = ov
                blm = None
            else:
                # (lo, hi, ts, filter_db_path): load the sstable's Filter.db
                # for the per-key bloom-checked evaluator
                lo, hi, ts, fpath = ov
                raw = open(fpath, "rb").read()
                k = int.from_bytes(raw[0:4], "big")
                words = int.from_bytes(raw[4:8], "big")
                bits = raw[8:8 + words * 8]
                buf = (ctypes.c_uint32 * (len(bits) // 4)).from_buffer_copy(bits)
                _bloom_keep.append(buf)
                blm = (buf, words * 64, k)
            ovr[i].token_lo, ovr[i].token_hi, ovr[i].min_timestamp = lo, hi, ts
            if blm is not None:
                ovr[i].bloom_bits = ctypes.cast(blm[0], ctypes.POINTER(ctypes.c_uint32))
                ovr[i].bloom_bit_len = blm[1]
                ovr[i].bloom_hash_count = blm[2]
        job.overlaps = ovr
        job.n_overlaps = len(overlaps)
    if token_range is not None:
        job.has_token_range = 1
        job.token_lo, job.token_hi = token_range
    if keep_ranges:
        kr = (GpucPurgeRange * len(keep_ranges))()
        for i, (lo, hi) in enumerate(keep_ranges):
            kr[i].token_lo, kr[i].token_hi = lo, hi
        job.keep_ranges = kr
        job.n_keep_ranges = len(keep_ranges)
        job.invert_ranges = 1 if invert_ranges else 0
    job.device = device
    job.n_output_shards = n_output_shards
    if cancel_flag is not None:
        if isinstance(cancel_flag, ctypes.c_int32):
            cancel_flag = ctypes.pointer(cancel_flag)
        job.cancel_flag = ctypes.cast(cancel_flag, ctypes.POINTER(ctypes.c_int32))
    if tombstone_sources:
        tarr = (ctypes.c_char_p * len(tombstone_sources))(*[b.encode() for b in tombstone_sources])
        job.tombstone_source_bases = tarr
        job.n_tomb_sources = len(tombstone_sources)
    job.cell_level_gc = 1 if cell_level_gc else 0
    res = GpucResult()
    rc = lib.gpuc_compact(ctypes.byref(job), ctypes.byref(res))
    if rc != 0:
        raise GpuCompactError(f"gpuc_compact rc={rc}: {res.error.decode(errors='replace')}")
    return {
        "input_uncompressed_bytes": res.input_uncompressed_bytes,
        "output_uncompressed_bytes": res.output_uncompressed_bytes,
        "output_compressed_bytes": res.output_compressed_bytes,
        "partitions_in": res.partitions_in,
        "partitions_out": res.partitions_out,
        "rows_in": res.rows_in,
        "rows_out": res.rows_out,
        "merged_counts": list(res.merged_counts),
        "ms": {
            "read_io": res.ms_read_io,
            "h2d": res.ms_h2d,
            "decompress": res.ms_decompress,
            "parse": res.ms_parse,
            "merge": res.ms_merge,
            "reconcile": res.ms_reconcile,
            "serialize": res.ms_serialize,
            "compress": res.ms_compress,
            "d2h": res.ms_d2h,
            "write_io": res.ms_write_io,
            "total": res.ms_total,
        },
        "dominant_kernel": res.dominant_kernel.decode(),
        "dominant_kernel_ms": res.dominant_kernel_ms,
    }


def validate(input_bases, out_path, now_sec=1800000000, gc_before=INT64_MIN,
             overlaps=None, device=0):
    """VALIDATION compaction (CompactionManager.doValidationCompaction +
    Validator.rowHash): merge+purge, then per-partition repair digests
    written to out_path as (token i64 BE + 32-byte hash) records."""
    lib = load_library()
    job = GpucJob()
    arr = (ctypes.c_char_p * len(input_bases))(*[b.encode() for b in input_bases])
    job.input_bases = arr
    job.n_inputs = len(input_bases)
    job.output_base = b"/unused"
    job.now_sec = now_sec
    job.gc_before = gc_before
    if overlaps:
        ovr = (GpucPurgeRange * len(overlaps))()
        for i, (lo, hi, ts) in enumerate(overlaps):
            ovr[i].token_lo, ovr[i].token_hi, ovr[i].min_timestamp = lo, hi, ts
        job.overlaps = ovr
        job.n_overlaps = len(overlaps)
    job.device = device
    job.n_output_shards = 1
    n = ctypes.c_uint64(0)
    err = ctypes.create_string_buffer(256)
    rc = lib.gpuc_validate(ctypes.byref(job), out_path.encode(), ctypes.byref(n), err, 256)
    if rc != 0:
        raise GpuCompactError(f"gpuc_validate rc={rc}: {err.value.decode(errors='replace')}")
    return n.value


def generate(
    out_dir,
    snappy=False,
    seed=42,
    n_sstables=4,
    rows_per_sstable=1000,
    overlap_pct=10,
    value_len=1024,
    value_repeat_pct=55,
    tombstone_pct=0,
    partition_del_pct=0,
    clustering_rows=0,
    range_tomb_pct=0,
    key_len=8,
    ck_text=False,
    ck_cols=0,
    static_pct=0,
    n_value_cols=1,
    col_missing_pct=0,
    ttl_pct=0,
    complex_pct=0,
    complex_del_pct=0,
    bti=False,
    counter=False,
    base_ts=1700000000000000,
    base_ldt=1700000000,
    first_generation=1,
    device=0,
):
    """GPU-side synthetic sstable generation (shared contract: oracle/src/gen.h)."""
    lib = load_library()
    spec = GpucGenSpec(
        seed=seed,
        n_sstables=n_sstables,
        rows_per_sstable=rows_per_sstable,
        overlap_pct=overlap_pct,
        value_len=value_len,
        value_repeat_pct=value_repeat_pct,
        tombstone_pct=tombstone_pct,
        partition_del_pct=partition_del_pct,
        clustering_rows=clustering_rows,
        range_tomb_pct=range_tomb_pct,
        key_len=key_len,
        ck_text=1 if ck_text else 0,
        ck_cols=ck_cols,
        static_pct=static_pct,
        n_value_cols=n_value_cols,
        col_missing_pct=col_missing_pct,
        base_ts=base_ts,
        base_ldt=base_ldt,
        first_generation=first_generation,
        device=device,
        snappy=1 if snappy else 0,
        ttl_pct=ttl_pct,
        complex_pct=complex_pct,
        complex_del_pct=complex_del_pct,
        bti=1 if bti else 0,
        counter=1 if counter else 0,
    )
    err = ctypes.create_string_buffer(256)
    rc = lib.gpuc_generate(ctypes.byref(spec), out_dir.encode(), err, 256)
    if rc != 0:
        raise GpuCompactError(f"gpuc_generate rc={rc}: {err.value.decode(errors='replace')}")
