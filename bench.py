#!/usr/bin/env python3
# bench.py — driver contract benchmark (BASELINE.json configs[1]).
#
# Workload "config2_1b_rows": 64 SSTs x 15.625M rows = 1B rows / 10M series
# synthetic Parquet SSTs (seed 42, PLAIN uncompressed pages — reference
# config.rs:76-94 supports compression=none), query = ts-range covering the
# middle 50% of the span + sum/count group by series_id. A step = one
# hx_exec_agg pass (decode+filter+dedup+aggregate+result readback) over the
# staged rows resident in HBM; staging (file IO + PCIe) happens once, before
# the timed region, and its PCIe-inclusive rate is reported in DESIGN.md §8
# terms via "stage_ms" in stderr diagnostics — never as `value`.
#
# value = scanned rows/sec aggregated over all ranks (rows actually staged
# after the reference's row-group pruning — the same rows the reference scan
# would decode). Weak scaling: each rank owns its own shard (BASELINE
# config 4: SSTs sharded per GPU, disjoint series spaces via per-rank seed).
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
if REPO not in sys.path:
    sys.path.insert(0, REPO)

import numpy as np  # noqa: E402


def log(msg):
    print(f"[bench] {msg}", file=sys.stderr, flush=True)


def get_dataset(args, rank):
    from tools.gen_ssts import gen_dataset
    tag = (f"r{args.rows}_s{args.series}_f{args.ssts}_seed{args.seed + rank}"
           f"_{args.compression}_{args.ts_encoding}")
    out = os.path.join(args.data_dir, tag)
    meta_path = os.path.join(out, "dataset.json")
    if os.path.exists(meta_path):
        with open(meta_path) as f:
            m = json.load(f)
        if m["n_rows"] == args.rows and m["n_series"] == args.series:
            log(f"rank{rank}: reusing dataset {out}")
            return out, m
    t0 = time.time()
    world = int(os.environ.get("WORLD_SIZE", 1))
    workers = min(16, max(2, (os.cpu_count() or 8) // max(1, world)))
    m = gen_dataset(out, args.rows, args.series, args.ssts,
                    seed=args.seed + rank, compression=args.compression,
                    ts_encoding=args.ts_encoding, workers=workers)
    log(f"rank{rank}: generated {args.rows} rows in {time.time() - t0:.1f}s "
        f"({workers} workers)")
    return out, m


def _cpu_one_sst(args):
    path, ts_range = args
    import sys as _s
    _s.path.insert(0, REPO)
    import oracle
    from oracle.scan import AGG_SUM, AGG_COUNT
    sst = oracle.read_sst(path)
    oracle.scan_agg([sst], tuple(ts_range), ops=AGG_SUM | AGG_COUNT)
    return sst.n_rows


def native_baseline_leg(m, ts_range, cores):
    """Native C++ restatement (oracle/native/libhx_cpuref.so, g++ -O3, all
    host cores): decode(PLAIN/Snappy) + ts-filter + dedup + sum/count
    aggregate — the fair compiled denominator (BASELINE.md plan 2b). Runs a
    bounded SST sample and scales to rows/s."""
    import ctypes
    lib_path = os.path.join(REPO, "oracle", "libhx_cpuref.so")
    if not os.path.exists(lib_path):
        return None
    lo, hi = ts_range
    paths = [s["path"] for s in m["ssts"]
             if s["ts_min"] < hi and s["ts_max"] >= lo]
    sample = paths[: max(4, min(len(paths), 16))]
    try:
        lib = ctypes.CDLL(lib_path)
        fn = lib.hx_cpu_scan_agg
        fn.restype = ctypes.c_int
        fn.argtypes = [ctypes.POINTER(ctypes.c_char_p), ctypes.c_int,
                       ctypes.c_int64, ctypes.c_int64, ctypes.c_int,
                       ctypes.POINTER(ctypes.c_double),
                       ctypes.POINTER(ctypes.c_int64),
                       ctypes.POINTER(ctypes.c_int64),
                       ctypes.POINTER(ctypes.c_int64),
                       ctypes.POINTER(ctypes.c_double)]
        arr = (ctypes.c_char_p * len(sample))(*[x.encode() for x in sample])
        el = ctypes.c_double()
        rs = ctypes.c_int64()
        rm = ctypes.c_int64()
        ng = ctypes.c_int64()
        dg = ctypes.c_double()
        rc = fn(arr, len(sample), lo, hi, cores, ctypes.byref(el),
                ctypes.byref(rs), ctypes.byref(rm), ctypes.byref(ng),
                ctypes.byref(dg))
    except OSError as e:
        log(f"native cpu baseline unavailable: {e}")
        return None
    if rc != 0 or el.value <= 0 or rs.value <= 0:
        log(f"native cpu baseline rc={rc} (unsupported layout?) — skipped")
        return None
    return {"rate": rs.value / el.value, "rows": rs.value, "dt": el.value,
            "ssts": len(sample)}


def cpu_baseline_leg(store_dir, m, ts_range, budget_s=20.0, cores=1):
    """Oracle (numpy/pyarrow CPU restatement) timed on a bounded sample of
    the same workload — kind 'port' (the reference Rust path cannot be built
    here: no cargo). Scaled to rows/s of scanned rows.
    cores>1 fans SSTs over a process pool (the all-cores leg BASELINE.md
    plans; cores = processes actually used)."""
    lo, hi = ts_range
    paths = [s["path"] for s in m["ssts"]
             if s["ts_min"] < hi and s["ts_max"] >= lo]
    # bound the sample to ~budget_s of single-core work, scaled by cores
    per_sst_rows = m["n_rows"] // max(1, m["n_ssts"])
    est_rate = 18e6  # rows/s/core, prior measurements
    max_ssts = max(1, int(budget_s * est_rate * cores / max(1, per_sst_rows)))
    sample = paths[:max_ssts]
    t0 = time.time()
    if cores > 1:
        import multiprocessing as mp
        with mp.get_context("spawn").Pool(cores) as pool:
            counts = pool.map(_cpu_one_sst,
                              [(p, list(ts_range)) for p in sample])
        rows_done = sum(counts)
    else:
        rows_done = 0
        for p in sample:
            rows_done += _cpu_one_sst((p, list(ts_range)))
            if time.time() - t0 > budget_s:
                break
    dt = time.time() - t0
    if rows_done == 0 or dt <= 0:
        return None
    return {
        "value": rows_done / dt,
        "unit": "rows/s",
        "cores": cores,
        "kind": "port",
        "sample": f"{len(sample)} SST(s), {rows_done} rows, {dt:.1f}s "
                  f"(oracle numpy+pyarrow, {cores} process(es))",
    }


def merge_bucket_partials(series_t, bucket_t, sum_t, cnt_t):
    """Merge gathered (series,bucket) partial tables (torch tensors, any
    device): stable sort by bucket then series, then segment-combine equal
    keys. The config-5 combine step after the RCCL all-gather."""
    import torch
    if series_t.numel() == 0:
        return series_t, bucket_t, sum_t, cnt_t
    ob = torch.argsort(bucket_t, stable=True)
    s1, b1, v1, c1 = series_t[ob], bucket_t[ob], sum_t[ob], cnt_t[ob]
    # series are u64 carried in int64 tensors: bias the sign bit so the
    # sort order is UNSIGNED (the engine's/reference's PK order)
    os_ = torch.argsort(s1 ^ (-2**63), stable=True)
    s2, b2, v2, c2 = s1[os_], b1[os_], v1[os_], c1[os_]
    new = torch.ones_like(s2, dtype=torch.bool)
    new[1:] = (s2[1:] != s2[:-1]) | (b2[1:] != b2[:-1])
    gid = torch.cumsum(new.to(torch.int64), 0) - 1
    n = int(gid[-1].item()) + 1
    out_s = s2[new]
    out_b = b2[new]
    out_v = torch.zeros(n, dtype=v2.dtype, device=v2.device)
    out_c = torch.zeros(n, dtype=c2.dtype, device=c2.device)
    out_v.scatter_add_(0, gid, v2)
    out_c.scatter_add_(0, gid, c2)
    return out_s, out_b, out_v, out_c


def run_extras(args, device, opmap):
    """Labeled non-headline workload lines (driver-visible via stderr):
    the uncompressed config-2 variant, config 3 (series-set predicate +
    min/max/avg), and the ts-overlap dedup workload.
    Never touches the headline timed region; each line carries its own
    config.workload label."""
    import argparse as _ap
    import torch
    from horaedb_amd import Store
    from tools.gen_ssts import gen_dataset, middle_range

    def one(label, store_dir, m, ops, series_in=None, steps=3,
            expect_matched=None):
        lo, hi = middle_range(m, args.range_frac)
        with Store(store_dir) as st:
            pr = st.prepare((lo, hi), series_in=series_in, devices=[device])
            pr.exec_agg(ops=ops, copy=False)
            torch.cuda.synchronize(device)
            t0 = time.time()
            for _ in range(steps):
                pr.exec_agg(ops=ops, copy=False)
            torch.cuda.synchronize(device)
            dt = time.time() - t0
            stt = pr.stats()
            line = {
                "workload": label,
                "value": stt["rows_scanned"] * steps / dt,
                "unit": "rows/s",
                "ms_per_step": dt * 1000 / steps,
                "agg_kernel_ms": stt["agg_kernel_ms"],
                "rows_scanned": stt["rows_scanned"],
                "rows_matched": stt["rows_matched"],
                "steps": steps,
                "n_gpus": 1,
            }
            if expect_matched is not None:
                line["matched_ok"] = (stt["rows_matched"] == expect_matched)
            pr.close()
        print("[bench-extra] " + json.dumps(line), file=sys.stderr,
              flush=True)

    # 1. uncompressed config-2 variant (the easier format; labeled)
    try:
        a2 = _ap.Namespace(**vars(args))
        a2.compression = "none"
        d2, m2 = get_dataset(a2, 0)
        one(f"config2_{args.rows//10**9}b_uncompressed_extra", d2, m2,
            opmap["sum"] | opmap["count"])
    except Exception as e:  # noqa: BLE001 — extras must never kill the run
        log(f"extra uncompressed failed: {e}")
    # 2. config 3 driven by a TAG through the inverted index (rfc:86-137):
    #    dc=dc7 -> 1% of the series -> min/max/avg scan (BASELINE configs[2])
    try:
        from tools.gen_ssts import gen_tag_index
        d3, m3 = get_dataset(args, 0)
        gen_tag_index(d3, n_dc=100)
        from horaedb_amd import Store
        with Store(d3) as st_idx:
            t_i = time.time()
            tsids = st_idx.index_query([("dc", "dc7")], device=device)
            index_ms = (time.time() - t_i) * 1000
        log(f"index query dc=dc7 -> {len(tsids)} TSIDs in {index_ms:.1f}ms")
        one("config3_tag_dc7_1pct_minmaxavg_extra", d3, m3,
            opmap["min"] | opmap["max"] | opmap["avg"],
            series_in=tsids.tolist())
    except Exception as e:  # noqa: BLE001
        log(f"extra config3 failed: {e}")
    # 3. ts-overlap dedup at scale: 2 generations re-write the same PKs
    #    (cross-SST shadowed() on every older-generation row)
    try:
        series_o = max(1000, args.series // 10)
        rows_o = max(series_o, args.rows // 10)
        rows_o -= rows_o % series_o
        if rows_o >= series_o:
            do = os.path.join(args.data_dir,
                              f"ovl_r{rows_o}_s{series_o}_g2")
            mp_ = os.path.join(do, "dataset.json")
            if os.path.exists(mp_):
                with open(mp_) as f:
                    mo = json.load(f)
            else:
                mo = gen_dataset(do, rows_o, series_o, 16, seed=args.seed,
                                 compression=args.compression,
                                 workers=min(16, os.cpu_count() or 8),
                                 overlap_gens=2)
            one(f"overlap_dedup_2gen_{2*rows_o//10**6}m_extra", do, mo,
                opmap["sum"] | opmap["count"])
    except Exception as e:  # noqa: BLE001
        log(f"extra overlap failed: {e}")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--rows", type=int, default=1_000_000_000)
    p.add_argument("--series", type=int, default=10_000_000)
    p.add_argument("--ssts", type=int, default=64)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--compression", default="snappy",
                   help="page codec; default snappy = the reference's own "
                        "default on-disk format (config.rs:120-133)")
    p.add_argument("--ts-encoding", default="PLAIN")
    p.add_argument("--range-frac", type=float, default=0.5)
    p.add_argument("--bucket-ms", type=int, default=0)
    p.add_argument("--selectivity", type=float, default=0,
                   help="config 3: series-set predicate keeping this fraction")
    p.add_argument("--config5", action="store_true",
                   help="BASELINE config 5: ONE shared dataset, SSTs sharded "
                        "round-robin across ranks, time_bucket partials "
                        "merged via an RCCL/gloo all-gather + GPU combine")
    p.add_argument("--ops", default="sum,count")
    p.add_argument("--pipeline", type=int, default=3,
                   help="concurrent scans in flight (N prepared objects, N "
                        "host threads): overlaps one query result path "
                        "with the next query kernel, the server "
                        "concurrent-scan pattern. Every step still runs "
                        "the full query; >1 reports amortized ms_per_step.")
    p.add_argument("--data-dir", default="/tmp/hx_bench_data")
    p.add_argument("--no-cpu-baseline", action="store_true")
    p.add_argument("--cpu-cores", type=int, default=0,
                   help="threads/processes for the cpu_baseline leg "
                        "(0 = all host cores)")
    p.add_argument("--no-extras", action="store_true",
                   help="skip the extra labeled workload lines "
                        "(uncompressed variant, config 3, ts-overlap)")
    args = p.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    n_gpus = max(args.gpus, world)

    dist = None
    backend = None
    if world > 1:
        import torch
        import torch.distributed as torch_dist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        backend = os.environ.get(
            "HX_BENCH_BACKEND",
            "nccl" if torch.cuda.is_available() else "gloo")
        torch_dist.init_process_group(backend=backend)
        dist = torch_dist

    import torch
    from horaedb_amd import Store, AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, AGG_AVG
    opmap = {"sum": AGG_SUM, "count": AGG_COUNT, "min": AGG_MIN,
             "max": AGG_MAX, "avg": AGG_AVG}
    ops = 0
    for o in args.ops.split(","):
        ops |= opmap[o.strip()]

    device = int(os.environ.get("HX_DEV_OVERRIDE", local_rank))
    if torch.cuda.is_available():
        # set BEFORE any collective: nccl barriers bind to the current device
        torch.cuda.set_device(device)

    from tools.gen_ssts import middle_range
    if args.config5:
        if args.bucket_ms <= 0:
            args.bucket_ms = 60_000
        # ONE shared dataset (same seed); rank 0 generates, others reuse
        if dist and rank != 0:
            dist.barrier()
            store_dir, m = get_dataset(args, 0)
        else:
            store_dir, m = get_dataset(args, 0)
            if dist:
                dist.barrier()
    elif dist and world > 1:
        # stagger per-rank dataset generation: 8 concurrent 24 GB writes
        # can blow /tmp before the post-staging cleanup runs; token-ring
        # order bounds peak disk to ~2 datasets
        store_dir = m = None
        for r in range(world):
            if r == rank:
                store_dir, m = get_dataset(args, rank)
            dist.barrier()
    else:
        store_dir, m = get_dataset(args, rank)
    ts_range = middle_range(m, args.range_frac)
    series_in = None
    if args.selectivity > 0:
        ids = np.load(os.path.join(store_dir, "series_ids.npy"))
        k = max(1, int(len(ids) * args.selectivity))
        rng = np.random.default_rng(args.seed)
        series_in = rng.choice(ids, size=k, replace=False).tolist()
        log(f"rank{rank}: series-set predicate with {k} ids "
            f"({args.selectivity:.2%})")

    if args.compression != "none" or args.ts_encoding != "PLAIN":
        # decode/decompress must run in EVERY timed step (no cached decode)
        os.environ["HX_REDECODE"] = "1"
        log("workload has a decode stage: HX_REDECODE=1 (per-step decode)")

    store = Store(store_dir)
    t0 = time.time()
    sst_subset = None
    if args.config5 and world > 1:
        # Shard whole ts-overlap clusters (dedup never crosses ranks; the
        # cross-rank combine stays pure key-wise addition — DESIGN.md §6)
        cat = store.find_ssts(ts_range)
        spans = {e["seq"]: (e["ts_min"], e["ts_max"])
                 for e in store.catalog()}
        clusters = []
        for path, seq in sorted(cat, key=lambda x: spans[x[1]][0]):
            lo, hi = spans[seq]
            if clusters and lo <= clusters[-1][0]:
                clusters[-1] = (max(hi, clusters[-1][0]),
                                clusters[-1][1] + [(path, seq)])
            else:
                clusters.append((hi, [(path, seq)]))
        sst_subset = [e for i, (_, members) in enumerate(clusters)
                      if i % world == rank for e in members]
        log(f"rank{rank}: config5 shard = {len(sst_subset)} of {len(cat)} "
            f"SSTs ({len(clusters)} ts-overlap clusters)")
    n_pipe = max(1, args.pipeline)
    if args.config5 and dist:
        n_pipe = 1  # the collective serializes ranks per step
    preps = [store.prepare(ts_range, series_in=series_in, devices=[device],
                           sst_subset=sst_subset) for _ in range(n_pipe)]
    prep = preps[0]
    log(f"rank{rank}: staged in {time.time() - t0:.1f}s"
        + (f" (pipeline depth {n_pipe})" if n_pipe > 1 else ""))

    # Disk-pressure guard for the 8-GPU weak-scaling run: per-rank datasets
    # are ~24 GB and the scan runs entirely from HBM after staging, so
    # non-zero ranks can free theirs when /tmp runs low (rank 0's is kept
    # for reuse across back-to-back N=1,2,4,8 runs).
    if world > 1 and rank != 0 and not args.config5:
        try:
            sv = os.statvfs(args.data_dir)
            free_gb = sv.f_bavail * sv.f_frsize / 1e9
            if free_gb < 100:
                import shutil
                shutil.rmtree(store_dir, ignore_errors=True)
                log(f"rank{rank}: freed dataset dir (disk low: "
                    f"{free_gb:.0f} GB free)")
        except OSError:
            pass

    def step(copy=False, prep=prep):
        if not (args.config5 and dist):
            return prep.exec_agg(ops=ops, bucket_ms=args.bucket_ms, copy=copy)
        # config 5: local bucket partials -> all-gather over RCCL (xGMI) ->
        # combine on the GPU; part of the timed step
        res = prep.exec_agg(ops=ops | 2, bucket_ms=args.bucket_ms, copy=True)
        dev = torch.device(f"cuda:{device}") if backend == "nccl" \
            else torch.device("cpu")
        s_t = torch.from_numpy(res["series_id"].view("int64").copy()).to(dev)
        b_t = torch.from_numpy(res["bucket"].copy()).to(dev)
        v_t = torch.from_numpy(res["sum"].copy()).to(dev)
        c_t = torch.from_numpy(res["count"].view("int64").copy()).to(dev)
        n_local = torch.tensor([s_t.numel()], dtype=torch.int64, device=dev)
        sizes = [torch.zeros_like(n_local) for _ in range(world)]
        dist.all_gather(sizes, n_local)
        n_max = int(max(int(x.item()) for x in sizes))
        def pad(t, fill):
            out = torch.full((n_max,), fill, dtype=t.dtype, device=dev)
            out[: t.numel()] = t
            return out
        gathered = []
        for t, fill in ((s_t, -1), (b_t, 0), (v_t, 0.0), (c_t, 0)):
            bufs = [torch.empty(n_max, dtype=t.dtype, device=dev)
                    for _ in range(world)]
            dist.all_gather(bufs, pad(t, fill))
            gathered.append(bufs)
        ss, bb, vv, cc = (
            torch.cat([g[r][: int(sizes[r].item())] for r in range(world)])
            for g in gathered)
        ms, mb, mv, mc = merge_bucket_partials(ss, bb, vv, cc)
        if backend == "nccl":
            torch.cuda.synchronize(device)
        return {"n_groups": int(ms.numel()), "merged_count": int(mc.sum().item())}

    # warmup (one materialized run for the group count, the rest light);
    # touch every pipeline slot so all tables/scratch are allocated untimed
    res = step(copy=True)
    for p_ in preps[1:]:
        step(prep=p_)
    for _ in range(max(0, args.warmup - 1)):
        step()
    st = prep.stats()
    log(f"rank{rank}: rows_scanned={st['rows_scanned']} "
        f"matched={st['rows_matched']} "
        f"groups={res.get('n_groups', len(res.get('series_id', [])))} "
        f"stage_ms={st['stage_ms']:.0f} exec_ms={st['exec_ms']:.1f} "
        f"agg_kernel_ms={st['agg_kernel_ms']:.2f}")

    # timed region: barrier + sync both sides, MAX over ranks
    if dist:
        dist.barrier()
    torch.cuda.synchronize(device)
    t_start = time.time()
    agg_kernel_ms = []
    if n_pipe == 1:
        for _ in range(args.steps):
            step()
            agg_kernel_ms.append(prep.stats()["agg_kernel_ms"])
    else:
        # concurrent scans: ctypes releases the GIL inside hx_exec_agg,
        # so N threads over N prepared objects genuinely overlap on GPU.
        # Per-prep locks: the executor does not guarantee task i+N waits
        # for task i, and one hx_prepared serves ONE call at a time.
        import threading
        from concurrent.futures import ThreadPoolExecutor
        locks = [threading.Lock() for _ in range(n_pipe)]

        def timed_step(i):
            p_ = preps[i % n_pipe]
            with locks[i % n_pipe]:
                step(prep=p_)
                return p_.stats()["agg_kernel_ms"]

        with ThreadPoolExecutor(max_workers=n_pipe) as ex:
            for k in ex.map(timed_step, range(args.steps)):
                agg_kernel_ms.append(k)
    torch.cuda.synchronize(device)
    if dist:
        dist.barrier()
    elapsed = time.time() - t_start

    # max over ranks of elapsed; sum over ranks of rows
    rows_scanned = prep.stats()["rows_scanned"]
    if dist:
        dev = torch.device(f"cuda:{device}") if backend == "nccl" \
            else torch.device("cpu")  # nccl needs device tensors
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())
        r = torch.tensor([float(rows_scanned)], dtype=torch.float64,
                         device=dev)
        dist.all_reduce(r, op=dist.ReduceOp.SUM)
        rows_scanned = int(r.item())

    value = rows_scanned * args.steps / elapsed
    ms_per_step = elapsed * 1000.0 / args.steps

    workload = ("config5_bucketed_sharded" if args.config5 else
                "config3_series_set_predicate" if args.selectivity > 0 else
                "config2_bucketed" if args.bucket_ms else
                f"config2_1b_rows_ts_range_sum_count_{args.compression}")

    if rank == 0:
        # roofline for the dominant kernel (k_scan_agg_range by default;
        # k_scan_agg under HX_RANGE=0): ALGORITHMIC bytes =
        # 24 B/row (series u64 + ts i64 + value f64; DESIGN.md §8, SURVEY
        # §8(d)) per launch / HIP-event launch time. With --pipeline > 1 the
        # per-step HIP-event interval is stretched by the concurrent
        # queries, so the kernel time comes from a SOLO (pipeline-1) pass
        # run right after the timed region (reproducible from profiles/).
        # traffic = PMC-measured HBM bytes per launch for THIS workload
        # label (rocprofv3 --pmc, profiles/roofline_traffic.json), or via
        # HX_ROOFLINE_TRAFFIC; null when unmeasured.
        my_rows = prep.stats()["rows_scanned"]
        if n_pipe > 1 and not (args.config5 and dist):
            solo = []
            for _ in range(3):
                step(prep=preps[0])
                solo.append(preps[0].stats()["agg_kernel_ms"])
            k_ms = float(np.mean(solo))
            k_basis = "solo pipeline-1 pass (HIP events, engine stream)"
        else:
            k_ms = float(np.mean(agg_kernel_ms))
            k_basis = "timed-region kernel HIP events (pipeline 1)"
        algo_bytes = my_rows * 24.0
        achieved = algo_bytes / (k_ms * 1e-3)
        peak = 8.0e12
        traffic_env = os.environ.get("HX_ROOFLINE_TRAFFIC")
        if not traffic_env:
            tj = os.path.join(REPO, "profiles", "roofline_traffic.json")
            if os.path.exists(tj):
                with open(tj) as f:
                    tdata = json.load(f).get(workload)
                if tdata and tdata.get("traffic_bytes_per_scanned_row"):
                    traffic_env = str(tdata["traffic_bytes_per_scanned_row"] *
                                      my_rows)
        roofline = {
            "bound": "hbm",
            "achieved": achieved / 1e9,
            "peak": peak / 1e9,
            "unit": "GB/s",
            "frac": achieved / peak,
            "traffic": float(traffic_env) if traffic_env else None,
            "kernel_ms": k_ms,
            "basis": k_basis,
        }

        cpu = None
        if not args.no_cpu_baseline and n_gpus == 1:
            ncores = args.cpu_cores if args.cpu_cores > 0 \
                else (os.cpu_count() or 1)
            log(f"running cpu_baseline (native C++ leg, {ncores} threads)...")
            native = native_baseline_leg(m, ts_range, ncores)
            log("running cpu_baseline (oracle leg, bounded sample)...")
            oracle_leg = cpu_baseline_leg(store_dir, m, ts_range,
                                          budget_s=8.0, cores=1)
            if native:
                osub = (f"; oracle numpy/pyarrow 1-core leg: "
                        f"{oracle_leg['value']:.3g} rows/s"
                        if oracle_leg else "")
                cpu = {
                    "value": native["rate"],
                    "unit": "rows/s",
                    "cores": ncores,
                    "kind": "port",
                    "sample": (f"native C++ leg (oracle/native, g++ -O3, "
                               f"{ncores} threads): {native['ssts']} SSTs, "
                               f"{native['rows']} rows, {native['dt']:.1f}s"
                               + osub),
                }
            else:
                cpu = oracle_leg

        if not args.no_extras and world <= 1 and not args.config5 \
                and args.selectivity == 0 and not args.bucket_ms:
            run_extras(args, device, opmap)

        result = {
            "metric": "scanned rows/sec, 1B-row range+sum (config 2)",
            "value": value,
            "unit": "rows/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": workload,
                "rows_per_gpu": args.rows,
                "rows_scanned_per_gpu": int(prep.stats()["rows_scanned"]),
                "series_per_gpu": args.series,
                "ssts": args.ssts,
                "ts_range": "middle 50%",
                "ops": args.ops,
                "compression": args.compression,
                "ts_encoding": args.ts_encoding,
                "page_encoding": "PLAIN",
                "seed": args.seed,
                "pipeline": n_pipe,
            },
            "roofline": roofline,
            "cpu_baseline": cpu,
        }
        print(json.dumps(result), flush=True)

    for p_ in preps:
        p_.close()
    store.close()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
