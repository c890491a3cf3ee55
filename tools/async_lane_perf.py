"""Async host-lane vs sync collective-plane round-trip cost.

Measures, at world_size N (torchrun), the per-op wall time of
ArrayTable add+get in (a) true-async mode (host gloo p2p served by the
peer's server thread) and (b) BSP collective mode (reduce-scatter +
all-gather). The async lane is the latency/independence lane — this
quantifies what a worker pays per op so the trade is documented, not
assumed (docs/ENGINEERING_NOTES.md).

Run: python -m torch.distributed.run --nproc-per-node 2 \
     --master-addr 127.0.0.1 tools/async_lane_perf.py [--size 65536]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def bench_mode(sync: bool, size: int, iters: int) -> float:
    import multiverso_amd as mv
    mv.init(sync=sync)
    t = mv.ArrayTable(size)
    delta = torch.ones(size)
    for _ in range(5):
        t.add(delta)
        t.get()
    mv.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        t.add(delta)
        t.get()
    dt = (time.perf_counter() - t0) / iters
    mv.barrier()
    rank = mv.rank()
    mv.shutdown()
    if rank == 0:
        mode = "sync-collective" if sync else "async-host-lane"
        print(f"{mode:16s} size={size:8d} ({size * 4 / 1e6:7.2f} MB): "
              f"{dt * 1e6:9.1f} us/(add+get)", flush=True)
    return dt


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    args = p.parse_args()
    for size in (1024, 65536, 1 << 20, 1 << 23):
        bench_mode(False, size, args.iters)
        bench_mode(True, size, args.iters)


if __name__ == "__main__":
    main()
