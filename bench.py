#!/usr/bin/env python3
"""Benchmark: SSTable compaction MB/s (input uncompressed bytes merged+compressed).

Driver contract: `python bench.py --gpus N --steps K --warmup W`; for N>1 the
driver launches via torch.distributed.run (one rank per GPU). A "step" is one
full compaction job of the C2 workload (BASELINE.json configs[1]): 8 x 2 GiB
synthetic sstables, 10% key overlap, ~1 KiB values, LZ4 16 KiB chunks.
Multi-GPU: weak scaling — every rank owns an independent token-disjoint job
(seeded per rank); no collectives on the data path.

Inputs are generated ON THE GPU by the product write path before the timed
region; the timed region is gpuc_compact (file read -> H2D -> kernels -> D2H
-> file write). The cpu_baseline leg times the CPU oracle (kind "port") on a
bounded sample at N=1.
"""
import argparse
import json
import os
import shutil
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

HBM_PEAK_GBS = 8000.0  # MI355X spec peak (MI355X_MICROARCH.md)


def build_if_needed():
    so = os.path.join(REPO, "cassandra_amd", "libcassandra_gpucompact.so")
    if not os.path.exists(so):
        subprocess.run(["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
                        "-shared", os.path.join(REPO, "cassandra_amd", "csrc", "gpucompact.cpp"),
                        "-o", so], check=True)


def oracle_bin():
    p = os.path.join(REPO, "oracle", "bin", "oracle_tool")
    if not os.path.exists(p):
        subprocess.run(["make", "-C", os.path.join(REPO, "oracle")], check=True,
                       stdout=subprocess.DEVNULL)
    return p


def algorithmic_bytes(kernel, r):
    """Algorithmic HBM bytes per launch of the dominant kernel.

    Values are per ONE invocation of the whole-job kernel launch.
    """
    in_comp = r["input_uncompressed_bytes"] * 0  # filled by caller knowing file sizes
    out_unc = r["output_uncompressed_bytes"]
    out_comp = r["output_compressed_bytes"]
    in_unc = r["input_uncompressed_bytes"]
    if kernel == "k_lz4_decompress":
        return r["_input_compressed_bytes"] + in_unc
    if kernel == "k_parse":
        return r["partitions_in"] * (48 + 77)  # header read + SoA write per partition
    if kernel == "merge+reconcile":
        import math
        k = max(1, r["_n_inputs"])
        rounds = max(1, math.ceil(math.log2(k)))
        return r["partitions_in"] * 48 * rounds + r["partitions_in"] * 90
    if kernel == "k_serialize":
        return 2 * out_unc  # gather values/meta + write serialized stream
    if kernel == "k_lz4_compress":
        return out_unc + out_comp
    return in_unc


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--sstables", type=int, default=8)
    ap.add_argument("--rows", type=int, default=2_050_000,
                    help="rows per sstable (2,050,000 * ~1048 B = 2 GiB uncompressed)")
    ap.add_argument("--vlen", type=int, default=1024)
    ap.add_argument("--overlap", type=int, default=10)
    ap.add_argument("--clustering-rows", type=int, default=0,
                    help="C4 shape: rows per partition under a bigint clustering column")
    ap.add_argument("--tombstone-pct", type=int, default=0)
    ap.add_argument("--range-tomb-pct", type=int, default=0)
    ap.add_argument("--dir", default=os.environ.get("GPUC_BENCH_DIR", "/tmp/gpuc_bench"),
                    help="sstable dir (page-cache-backed; measured on this pool: overlay "
                         "page cache absorbs ~14.5 GB/s vs tmpfs ~4-8 GB/s for the "
                         "single-file pwrite drain — tools/wbench.c). Inputs, outputs "
                         "and the cpu_baseline leg all use the same dir")
    ap.add_argument("--shards", type=int, default=1,
                    help="n_output_shards per job (UCS-style sharded outputs; "
                         "2 shards run concurrently inside the library)")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--snappy", action="store_true",
                    help="C3 codec: SnappyCompressor chunks end to end")
    ap.add_argument("--dry-run", action="store_true",
                    help="CPU rehearsal of the exact rank path (init, device "
                         "mapping, barriers, MAX/SUM reductions, JSON emit) with "
                         "generate/compact mocked — so the first real multi-GPU "
                         "run is not the first execution of this code")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    dist = None
    if world > 1:
        import torch.distributed as tdist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        tdist.init_process_group("gloo")  # control-plane only; data path has no collectives
        dist = tdist

    if args.dry_run:
        class _MockCa:
            @staticmethod
            def device_count():
                return max(1, world)
            @staticmethod
            def generate(*a, **k):
                pass
            @staticmethod
            def compact(bases, out, device=0, n_output_shards=1):
                os.makedirs(os.path.dirname(out), exist_ok=True)
                return {"input_uncompressed_bytes": 1 << 20,
                        "output_uncompressed_bytes": 1 << 20,
                        "output_compressed_bytes": 1 << 19,
                        "partitions_in": 100, "partitions_out": 100,
                        "dominant_kernel": "k_lz4_compress",
                        "dominant_kernel_ms": 1.0, "dominant_kernel_launches": 1,
                        "ms": {"total": 1.0}}
        ca = _MockCa()
    else:
        build_if_needed()
        import cassandra_amd as ca
    if ca.device_count() < 1:
        print(json.dumps({"error": "no GPU visible"}))
        return 1
    device = local_rank % max(1, ca.device_count())

    # ---- per-rank inputs (GPU-generated, seed disjoint per rank) ----
    # Disk budget at high rank counts: 8 ranks x (8.6 GB inputs + up to
    # 2 x 8.6 GB in-flight outputs) would overflow the pool boxes' ~81 GB
    # /tmp overlay. With >=4 ranks, park the READ-ONLY inputs on /dev/shm
    # (tmpfs reads are fine; only its write path is slow) and keep outputs
    # on the page-cache-backed dir; deletes join before the next write.
    in_root = args.dir
    if world >= 4 and args.dir.startswith("/tmp") and os.path.isdir("/dev/shm"):
        in_root = "/dev/shm/gpuc_bench_in"
    d = os.path.join(in_root, f"r{rank}")
    dout = os.path.join(args.dir, f"r{rank}")
    if dout != d:
        shutil.rmtree(dout, ignore_errors=True)
        os.makedirs(dout, exist_ok=True)
    shutil.rmtree(d, ignore_errors=True)
    os.makedirs(d, exist_ok=True)
    t_gen = time.time()
    ca.generate(d, seed=42 + rank, n_sstables=args.sstables,
                rows_per_sstable=args.rows, overlap_pct=args.overlap,
                value_len=args.vlen, value_repeat_pct=55, device=device,
                clustering_rows=args.clustering_rows,
                tombstone_pct=args.tombstone_pct,
                range_tomb_pct=args.range_tomb_pct,
                snappy=args.snappy)
    t_gen = time.time() - t_gen
    bases = [os.path.join(d, f"oa-{g}-big") for g in range(1, args.sstables + 1)]
    input_compressed = 0 if args.dry_run else sum(os.path.getsize(b + "-Data.db") for b in bases)

    import threading
    cleaners = []

    def one_step(i, prev=[None]):
        out = os.path.join(dout, f"out-{i}", "oa-100-big")
        os.makedirs(os.path.dirname(out), exist_ok=True)
        r = ca.compact(bases, out, device=device, n_output_shards=args.shards)
        # previous step's output is deleted in the background (bounded disk,
        # no serial rmtree inside the measured path); at >=4 ranks the delete
        # must land before the NEXT step's write or /tmp overflows
        if prev[0]:
            t = threading.Thread(target=shutil.rmtree, args=(prev[0],), kwargs={"ignore_errors": True})
            t.start()
            if world >= 4:
                t.join()
            else:
                cleaners.append(t)
        prev[0] = os.path.dirname(out)
        return r

    import torch
    for i in range(args.warmup):
        one_step(f"w{i}")
    if dist:
        dist.barrier()
    torch.cuda.synchronize() if torch.cuda.is_available() else None
    t0 = time.time()
    last = None
    for i in range(args.steps):
        last = one_step(i)
    torch.cuda.synchronize() if torch.cuda.is_available() else None
    if dist:
        dist.barrier()
    elapsed = time.time() - t0
    for t in cleaners:
        t.join()

    my_bytes = float(last["input_uncompressed_bytes"] * args.steps)
    if dist:
        t = torch.tensor([elapsed])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()
        b = torch.tensor([my_bytes])
        dist.all_reduce(b, op=dist.ReduceOp.SUM)
        my_bytes = b.item()

    if rank != 0:
        if dist:
            dist.destroy_process_group()
        return 0

    value_mb_s = my_bytes / elapsed / 1e6

    # ---- roofline for the dominant kernel (HIP-event timed inside the lib) ----
    last["_input_compressed_bytes"] = input_compressed
    last["_n_inputs"] = args.sstables
    dom = last["dominant_kernel"]
    dom_ms = last["dominant_kernel_ms"]
    r_out_unc = last["output_uncompressed_bytes"]
    ab = algorithmic_bytes(dom, last)
    achieved_gbs = (ab / (dom_ms / 1e3)) / 1e9 if dom_ms > 0 else 0.0
    # PMC-measured HBM traffic for the LZ4 compress kernel (separate rocprofv3
    # --pmc FETCH_SIZE / WRITE_SIZE passes, profiles/r02_pmc_c2.txt): full-slab
    # launches move 0.73 GB per 0.537 GB of uncompressed input (1.36 B/B) vs
    # 1.53 B/B algorithmic — traffic ~= algorithmic, the kernel is latency-
    # bound, not HBM-bound. Scaled to this job's uncompressed output bytes.
    traffic = int(1.36 * r_out_unc) if dom == "k_lz4_compress" else None
    roofline = {
        "bound": "hbm",
        "achieved": round(achieved_gbs, 1),
        "peak": HBM_PEAK_GBS,
        "unit": "GB/s",
        "frac": round(achieved_gbs / HBM_PEAK_GBS, 4),
        "traffic": traffic,
    }

    # ---- CPU baseline: oracle compactor, bounded sample, 1 thread ----
    cpu_baseline = None
    if world == 1 and not args.no_cpu_baseline and not args.dry_run:
        ob = oracle_bin()
        sd = os.path.join(args.dir, "cpu_sample")
        shutil.rmtree(sd, ignore_errors=True)
        os.makedirs(sd)
        sample_rows = 120_000  # ~1 GiB in: ~10-20 s of single-core work
        if args.clustering_rows:
            sample_rows = max(1, sample_rows // args.clustering_rows)
        gen_args = [ob, "gen", sd, "seed=42", f"n={args.sstables}",
                    f"rows={sample_rows}", f"vlen={args.vlen}", f"overlap={args.overlap}"]
        if args.clustering_rows:
            gen_args += [f"crows={args.clustering_rows}", f"tomb={args.tombstone_pct}",
                         f"rtomb={args.range_tomb_pct}"]
        subprocess.run(gen_args, check=True, capture_output=True)
        sins = [os.path.join(sd, f"oa-{g}-big") for g in range(1, args.sstables + 1)]
        out = subprocess.run([ob, "compact", os.path.join(sd, "oa-100-big"), *sins],
                             check=True, capture_output=True, text=True)
        st = json.loads(out.stdout.strip().splitlines()[-1])
        cpu_baseline = {
            "value": round(st["mb_per_s"], 2),
            "unit": "MB/s input sstable bytes",
            "cores": 1,
            "kind": "port",
            "sample": f"{args.sstables}x{sample_rows} rows (~{st['input_uncompressed_bytes']//2**20} MiB input), oracle compactor incl. file IO",
        }
        shutil.rmtree(sd, ignore_errors=True)

    out = {
        "metric": "compaction MB/s (input SSTable bytes merged+compressed)",
        "value": round(value_mb_s, 2),
        "unit": "MB/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed * 1000 / args.steps, 2),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # no published compaction MB/s in the reference
        "dtype": "u8",
        "data": "synthetic",
        "config": {
            "workload": ("C4: 8x2GiB wide-partition sstables "
                         f"({args.clustering_rows} clustering rows/partition, "
                         f"{args.tombstone_pct}% row + {args.range_tomb_pct}% range tombstones)"
                         if args.clustering_rows else
                         f"C3: {args.sstables}x sstables, Snappy 16KiB chunks"
                         if args.snappy else
                         f"C5: full major compaction, {args.sstables}x sstables, "
                         f"LZ4, {args.shards} token shards (vnode ranges)"
                         if args.shards > 1 else
                         "C2: 8x2GiB sstables, 10% key overlap, ~1KiB values, LZ4 16KiB chunks"),
            "sstables": args.sstables,
            "rows_per_sstable": args.rows,
            "value_len": args.vlen,
            "overlap_pct": args.overlap,
            "clustering_rows": args.clustering_rows,
            "tombstone_pct": args.tombstone_pct,
            "range_tomb_pct": args.range_tomb_pct,
            "parallelism": f"token-independent shards x{world}, no collectives"
                           + (f"; {args.shards} sharded outputs/job" if args.shards > 1 else ""),
            "storage": args.dir,
            "input_uncompressed_bytes_per_rank": last["input_uncompressed_bytes"],
            "gen_seconds": round(t_gen, 1),
            "phase_ms": last["ms"],
        },
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(out))
    if dist:
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
