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
