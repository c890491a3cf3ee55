"""RCCL bucket-size sweep for the 4 GB sync-SGD stress (BASELINE.json
config 5: "test_allreduce-style 4 GB MatrixTable sync-SGD stress").

Times MV_Aggregate (all-reduce) over a large fp32 buffer at a range of
bucket sizes — the rebuild's tunable replacing the reference
AllreduceEngine's 4096-byte small-message switch
(src/net/allreduce_engine.cpp:35). xGMI is 7 point-to-point links per GPU
(~153 GB/s each), so ring all-reduce is per-link bound; the sweep finds
the bucket size where pipelining stops paying.

Run on an 8-GPU node:
  python -m torch.distributed.run --nproc-per-node 8 \
      --master-addr 127.0.0.1 tools/bucket_sweep.py --gb 4
(N=1 degenerates to a no-op all-reduce; the numbers only mean something
at N>=2.)
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gb", type=float, default=4.0)
    p.add_argument("--iters", type=int, default=5)
    args = p.parse_args()

    import multiverso_amd as mv
    mv.init()
    device = mv.Zoo.get().device
    if device.type != "cuda":
        args.gb = min(args.gb, 0.125)

    n = int(args.gb * (1 << 30) / 4)
    buf = torch.ones(n, dtype=torch.float32, device=device)
    results = []
    for bucket_mb in [0, 4, 16, 32, 64, 128, 256, 512, 1024]:
        mv.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            mv.aggregate(buf, bucket_mb=bucket_mb)
        if device.type == "cuda":
            torch.cuda.synchronize()
        mv.barrier()
        dt = (time.perf_counter() - t0) / args.iters
        # ring all-reduce busbw = 2*(n-1)/n * bytes / time
        w = mv.size()
        busbw = (2 * (w - 1) / max(w, 1)) * n * 4 / dt / 1e9
        results.append((bucket_mb, dt * 1e3, busbw))
        if mv.rank() == 0:
            label = "monolithic" if bucket_mb == 0 else f"{bucket_mb} MiB"
            print(f"bucket {label:>10}: {dt * 1e3:8.2f} ms  "
                  f"busbw {busbw:7.1f} GB/s", flush=True)
    mv.shutdown()


if __name__ == "__main__":
    main()
