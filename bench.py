"""Flagship benchmark: MatrixTable Add+Get updates/sec (whole node).

BASELINE.json headline metric on the named config "MatrixTable 1e6x128
dense SGD updater, N workers on N x MI355X": every worker performs, per
step, one whole-table Add (its own 1e6x128 fp32 delta; reduce-scatter over
xGMI + fused SGD updater kernel on the owned shard) and one whole-table
Get (all-gather of shards into its buffer). Synthetic random deltas,
random-init weights. ``value`` is the whole-job aggregate: element-updates
applied per second = N_workers * rows * cols / step_time.

Launch: python bench.py --gpus N --steps K --warmup W
(N>1 via torch.distributed.run; reads RANK/WORLD_SIZE/MASTER_* from env.)
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--rows", type=int, default=1_000_000)
    p.add_argument("--cols", type=int, default=128)
    p.add_argument("--updater", type=str, default="sgd")
    p.add_argument("--app", type=str, default="matrix",
                   choices=["matrix", "wordembedding", "logreg"])
    p.add_argument("--vocab", type=int, default=1_000_000)
    p.add_argument("--dim", type=int, default=200)
    p.add_argument("--block-words", dest="block_words", type=int,
                   default=500_000)
    p.add_argument("--use-adagrad", dest="use_adagrad", action="store_true",
                   help="wordembedding: per-element AdaGrad mode")
    return p.parse_args()


def main():
    args = parse_args()
    cuda = torch.cuda.is_available()
    if not cuda:
        # CPU debug only — the driver always runs this on MI355X.
        args.rows = min(args.rows, 20_000)

    import multiverso_amd as mv
    if args.app == "wordembedding":
        from multiverso_amd.apps.wordembedding.bench import run_bench
        run_bench(args)
        return
    if args.app == "logreg":
        from multiverso_amd.apps.logreg.bench import run_bench
        run_bench(args)
        return

    mv.init(sync=True)
    n = mv.size()
    rank = mv.rank()
    device = mv.Zoo.get().device

    table = mv.MatrixTable(args.rows, args.cols, updater_type=args.updater,
                           random_init=(-0.1, 0.1))
    if cuda:
        from multiverso_amd import ops
        ops.module(required=True)  # native kernels must be loaded

    gen = torch.Generator(device="cpu").manual_seed(1234 + rank)
    delta = (torch.rand(args.rows * args.cols, generator=gen) * 1e-4).to(device)
    out = torch.empty(args.rows, args.cols, dtype=torch.float32, device=device)

    def step():
        table.add(delta)        # reduce-scatter + fused SGD kernel
        table.get(out=out)      # all-gather into user buffer

    for _ in range(args.warmup):
        step()

    mv.barrier()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if cuda:
        torch.cuda.synchronize()
    mv.barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if dist.is_initialized() and n > 1:
        if mv.Zoo.get().backend == "nccl":
            t = t.to(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t[0])

    ms_per_step = elapsed / args.steps * 1e3
    updates_per_sec = n * args.rows * args.cols / (elapsed / args.steps)

    if rank == 0:
        print(json.dumps({
            "metric": "MatrixTable Add+Get updates/sec (whole node)",
            "value": updates_per_sec,
            "unit": "updates/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": f"MatrixTable {args.rows}x{args.cols} dense "
                         f"{args.updater} updater",
                "global_batch": n,
                "seq_len": None,
                "parallelism": f"ps-sharded dp{n} (reduce-scatter/all-gather "
                               "over xGMI)",
            },
        }), flush=True)
    mv.shutdown()


if __name__ == "__main__":
    main()
