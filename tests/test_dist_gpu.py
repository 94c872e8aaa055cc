# Multi-rank semantics with the HIP LIBRARY under each rank (VERDICT r01
# weak #5: the CPU dist test exercised the oracle's math, not the product's
# distributed code path). Two gloo ranks share ONE GPU (HX_DEV_OVERRIDE=0);
# each runs Store.prepare/exec_agg (libhoraedb_hx.so) on its shard:
#   - config 4: disjoint per-rank datasets (per-rank seed), result set =
#     union of shard-local tables; verified against one oracle pass.
#   - config 5: one shared dataset, SSTs sharded by rank, time_bucket
#     partials all-gathered and key-combined (bench.merge_bucket_partials);
#     verified against the oracle's bucketed aggregate over ALL SSTs.
# DESIGN.md §6; the gloo exchange here is the same code path bench.py
# --config5 drives over RCCL on an 8-GPU node.
import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

pytestmark = pytest.mark.gpu


def _rank4(rank, world, tmpdir, q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = os.environ.get("HX_TEST_PORT", "29531")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    sys.path.insert(0, REPO)
    from horaedb_amd import Store, AGG_SUM, AGG_COUNT
    from tools.gen_ssts import gen_dataset, middle_range

    shard_dir = os.path.join(tmpdir, f"shard{rank}")
    m = gen_dataset(shard_dir, n_rows=40_000, n_series=400, n_ssts=4,
                    seed=300 + rank)
    ts_range = middle_range(m)
    with Store(shard_dir) as st:
        res = st.scan_agg(ts_range, ops=AGG_SUM | AGG_COUNT, devices=[0])
    dist.barrier()
    # O(1) digest exchange (the config-4 timed-region exchange)
    digest = torch.tensor([float(len(res["series_id"])),
                           float(res["count"].sum())], dtype=torch.float64)
    dist.all_reduce(digest, op=dist.ReduceOp.SUM)
    dist.barrier()
    q.put((rank, {
        "series": res["series_id"].tolist(),
        "sum": res["sum"].tolist(),
        "count": res["count"].tolist(),
        "digest": digest.tolist(),
        "shard_dir": shard_dir,
        "ts_range": list(ts_range),
    }))
    dist.destroy_process_group()


def test_two_rank_config4_library(tmp_path):
    os.environ["HX_DEV_OVERRIDE"] = "0"
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank4, args=(r, 2, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = dict(q.get(timeout=300) for _ in range(2))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    import oracle
    from oracle.scan import AGG_SUM, AGG_COUNT
    # union of shard tables == one oracle pass over both shards' SSTs
    union = {}
    for r in (0, 1):
        for s, v, c in zip(outs[r]["series"], outs[r]["sum"],
                           outs[r]["count"]):
            assert s not in union, "shards must be disjoint"
            union[s] = (v, c)
    import json
    all_ssts = []
    for r in (0, 1):
        with open(os.path.join(outs[r]["shard_dir"], "dataset.json")) as f:
            mm = json.load(f)
        all_ssts += [oracle.read_sst(x["path"]) for x in mm["ssts"]]
    lo, hi = outs[0]["ts_range"]
    exp = oracle.scan_agg(all_ssts, (lo, hi), ops=AGG_SUM | AGG_COUNT)
    assert len(union) == len(exp["series_id"])
    got_sum = np.array([union[s][0] for s in exp["series_id"]])
    got_cnt = np.array([union[s][1] for s in exp["series_id"]])
    np.testing.assert_array_equal(got_cnt, exp["count"])
    np.testing.assert_allclose(got_sum, exp["sum"], rtol=1e-9)
    # both ranks saw the same summed digest
    assert outs[0]["digest"] == outs[1]["digest"]


def _rank5(rank, world, shared_dir, q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = os.environ.get("HX_TEST_PORT2", "29532")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    sys.path.insert(0, REPO)
    import json
    from horaedb_amd import Store, AGG_SUM, AGG_COUNT
    from tools.gen_ssts import middle_range
    from bench import merge_bucket_partials

    with open(os.path.join(shared_dir, "dataset.json")) as f:
        m = json.load(f)
    ts_range = middle_range(m)
    bucket_ms = 60_000
    with Store(shared_dir) as st:
        cat = st.find_ssts(ts_range)
        mine = [e for i, e in enumerate(sorted(cat, key=lambda x: x[1]))
                if i % world == rank]
        pr = st.prepare(ts_range, devices=[0], sst_subset=mine)
        res = pr.exec_agg(ops=AGG_SUM | AGG_COUNT, bucket_ms=bucket_ms,
                          copy=True)
        pr.close()
    dev = torch.device("cpu")
    s_t = torch.from_numpy(res["series_id"].view("int64").copy())
    b_t = torch.from_numpy(res["bucket"].copy())
    v_t = torch.from_numpy(res["sum"].copy())
    c_t = torch.from_numpy(res["count"].view("int64").copy())
    n_local = torch.tensor([s_t.numel()], dtype=torch.int64)
    sizes = [torch.zeros_like(n_local) for _ in range(world)]
    dist.all_gather(sizes, n_local)
    n_max = int(max(int(x.item()) for x in sizes))

    def pad(t, fill):
        out = torch.full((n_max,), fill, dtype=t.dtype, device=dev)
        out[: t.numel()] = t
        return out

    gathered = []
    for t, fill in ((s_t, -1), (b_t, 0), (v_t, 0.0), (c_t, 0)):
        bufs = [torch.empty(n_max, dtype=t.dtype) for _ in range(world)]
        dist.all_gather(bufs, pad(t, fill))
        gathered.append(bufs)
    ss, bb, vv, cc = (
        torch.cat([g[r][: int(sizes[r].item())] for r in range(world)])
        for g in gathered)
    ms, mb, mv, mc = merge_bucket_partials(ss, bb, vv, cc)
    q.put((rank, {
        "series": ms.numpy().view(np.uint64).tolist(),
        "bucket": mb.tolist(),
        "sum": mv.tolist(),
        "count": mc.tolist(),
        "ts_range": list(ts_range),
        "n_ssts_mine": len(mine),
        "n_ssts_cat": len(cat),
    }))
    dist.destroy_process_group()


def test_two_rank_config5_library(tmp_path):
    from tools.gen_ssts import gen_dataset
    os.environ["HX_DEV_OVERRIDE"] = "0"
    shared = str(tmp_path / "shared")
    m = gen_dataset(shared, n_rows=40_000, n_series=400, n_ssts=6, seed=77)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank5, args=(r, 2, shared, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = dict(q.get(timeout=300) for _ in range(2))
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # shards partition exactly the SSTs overlapping the scan range
    assert outs[0]["n_ssts_cat"] == outs[1]["n_ssts_cat"] > 0
    assert (outs[0]["n_ssts_mine"] + outs[1]["n_ssts_mine"]
            == outs[0]["n_ssts_cat"])

    import oracle
    from oracle.scan import AGG_SUM, AGG_COUNT
    ssts = [oracle.read_sst(x["path"]) for x in m["ssts"]]
    lo, hi = outs[0]["ts_range"]
    exp = oracle.scan_agg(ssts, (lo, hi), ops=AGG_SUM | AGG_COUNT,
                          bucket_ms=60_000)
    for r in (0, 1):  # every rank holds the identical merged table
        got_s = np.array(outs[r]["series"], dtype=np.uint64)
        got_b = np.array(outs[r]["bucket"], dtype=np.int64)
        # merged table is sorted by (series, bucket) — same as the oracle
        np.testing.assert_array_equal(got_s, exp["series_id"])
        np.testing.assert_array_equal(got_b, exp["bucket"])
        np.testing.assert_array_equal(
            np.array(outs[r]["count"]), exp["count"].astype(np.int64))
        np.testing.assert_allclose(np.array(outs[r]["sum"]), exp["sum"],
                                   rtol=1e-9)
