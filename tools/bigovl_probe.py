import sys, os, time, json
sys.path.insert(0, ".")
from tools.gen_ssts import gen_dataset, middle_range
import torch
from horaedb_amd import Store, AGG_SUM, AGG_COUNT


def main():
    d = "/tmp/bigovl"
    mp = os.path.join(d, "dataset.json")
    m = json.load(open(mp)) if os.path.exists(mp) else gen_dataset(
        d, 500_000_000, 5_000_000, 32, seed=42, workers=16, overlap_gens=2)
    lo, hi = middle_range(m)
    st = Store(d)
    pr = st.prepare((lo, hi), devices=[0])
    pr.exec_agg(ops=AGG_SUM | AGG_COUNT, copy=False)
    torch.cuda.synchronize(0)
    t0 = time.time()
    for _ in range(4):
        pr.exec_agg(ops=AGG_SUM | AGG_COUNT, copy=False)
    torch.cuda.synchronize(0)
    dt = (time.time() - t0) / 4
    s = pr.stats()
    print("big-overlap 1B rows (2 gens x 500M, cross-SST shadowing on every "
          "older in-range row): ms_per_step=%.1f scanned=%d matched=%d "
          "agg_kernel_ms=%.2f rows_per_s=%.3g" %
          (dt * 1000, s["rows_scanned"], s["rows_matched"],
           s["agg_kernel_ms"], s["rows_scanned"] / dt))



if __name__ == "__main__":   # mp spawn re-imports this module
    main()