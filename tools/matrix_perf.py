"""MatrixTable perf harness — the rebuild of Test/test_matrix_perf.cpp
(TestmatrixPerformance, :33-171): a 1e6x50 float matrix; sweeps adding
10%..100% of rows, times get-all before/after each sweep point, and
prints the Dashboard.

Run (CPU or GPU):  python tools/matrix_perf.py [--rows N] [--cols N]
Multi-rank:        python -m torch.distributed.run --nproc-per-node N \
                       --master-addr 127.0.0.1 tools/matrix_perf.py
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=1_000_000)
    p.add_argument("--cols", type=int, default=50)
    p.add_argument("--iters", type=int, default=3)
    p.add_argument("--sparse", action="store_true",
                   help="TestSparsePerf equivalent: SparseMatrixTable "
                        "stale-filtered get_into instead of dense get")
    args = p.parse_args()

    import multiverso_amd as mv
    mv.init(sync=True)
    device = mv.Zoo.get().device
    if device.type != "cuda":
        args.rows = min(args.rows, 50_000)

    if args.sparse:
        t = mv.SparseMatrixTable(args.rows, args.cols,
                                 updater_type="default")
    else:
        t = mv.MatrixTable(args.rows, args.cols, updater_type="default")
    out = torch.empty(args.rows, args.cols, device=device)

    def timed(fn):
        mv.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        fn()
        if device.type == "cuda":
            torch.cuda.synchronize()
        mv.barrier()
        return (time.perf_counter() - t0) * 1e3

    rows_all = torch.arange(args.rows, device=device)
    results = []
    for pct in range(10, 101, 10):
        k = args.rows * pct // 100
        ids = rows_all[torch.randperm(args.rows, device=device)[:k]]
        vals = torch.rand(k, args.cols, device=device)
        if args.sparse:
            got = [0]

            def sget():
                got[0] = t.get_into(out)

            get_before = timed(sget)
            n_before = got[0]
            add_ms = timed(lambda: t.add_rows(ids, vals))
            get_after = timed(sget)
            results.append((pct, add_ms, get_before, get_after))
            if mv.rank() == 0:
                print(f"add {pct:3d}% rows ({k}): add {add_ms:8.2f} ms | "
                      f"stale-get before {get_before:8.2f} ms "
                      f"({n_before} rows) after {get_after:8.2f} ms "
                      f"({got[0]} rows)", flush=True)
        else:
            get_before = timed(lambda: t.get(out=out))
            add_ms = timed(lambda: t.add_rows(ids, vals))
            get_after = timed(lambda: t.get(out=out))
            results.append((pct, add_ms, get_before, get_after))
            if mv.rank() == 0:
                print(f"add {pct:3d}% rows ({k}): add {add_ms:8.2f} ms | "
                      f"get-all before {get_before:8.2f} ms after "
                      f"{get_after:8.2f} ms", flush=True)

    if mv.rank() == 0:
        print(mv.Dashboard.display())
    mv.shutdown()


if __name__ == "__main__":
    main()
