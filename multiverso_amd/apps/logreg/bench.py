"""LogReg samples/sec benchmark (BASELINE.json config:
"Applications/LogisticRegression 1e9 sparse features, AdaGrad updater,
N workers"). Synthetic sparse stream; weights in a 1e9-row sharded table
(500 MB/rank fp32 at 8 ranks — HBM-resident); per-chunk
pull/push over xGMI with the keyed adagrad kernel on the owner."""

import json
import time

import torch
import torch.distributed as dist


def run_bench(args):
    import multiverso_amd as mv
    from .config import LogRegConfig
    from .model import PSModel
    from .reader import synthetic_batches

    mv.init(sync=True)
    n = mv.size()
    rank = mv.rank()
    device = mv.Zoo.get().device
    cuda = device.type == "cuda"

    input_size = getattr(args, "features", 1_000_000_000)
    minibatch = getattr(args, "minibatch", 4096)
    nnz = getattr(args, "nnz", 64)
    sync_freq = 4
    if not cuda:
        input_size, minibatch = min(input_size, 100_000), min(minibatch, 256)

    objective = getattr(args, "objective", "sigmoid")
    classes = getattr(args, "classes", 10) if objective == "softmax" else 1
    dense = bool(getattr(args, "dense", False))
    if dense:
        # reference mnist.config shape class: dense softmax; input_size
        # is the dense width (incl. bias), far below the sparse 1e9
        input_size = getattr(args, "features", 2048)
        if input_size > 1_000_000:
            input_size = 2048
        if not cuda:
            input_size = min(input_size, 256)
    cfg = LogRegConfig(input_size=input_size, minibatch_size=minibatch,
                       use_ps=True, sparse=not dense,
                       updater_type="sgd" if (objective == "ftrl" or dense)
                       else "adagrad",
                       objective_type=objective, output_size=classes,
                       sync_frequency=sync_freq, learning_rate=0.05,
                       show_time_per_sample=0)
    model = PSModel(cfg, device)

    n_chunks = args.warmup + args.steps
    if dense:
        from .objective import DenseBatch
        g = torch.Generator(device=device).manual_seed(31 + rank)
        hidden = torch.randn(input_size, classes, generator=g,
                             device=device)
        batches = []
        for _ in range(n_chunks * sync_freq):
            x = torch.randn(minibatch, input_size, generator=g,
                            device=device)
            x[:, -1] = 1.0   # bias column
            lab = ((x @ hidden).argmax(1).float() if classes > 1
                   else ((x @ hidden).squeeze(1) > 0).float())
            batches.append(DenseBatch(x, lab))
    else:
        batches, _ = synthetic_batches(input_size, n_chunks * sync_freq,
                                       minibatch, nnz=nnz,
                                       output_size=classes,
                                       seed=31 + rank, device=device)
    chunks = [batches[i * sync_freq:(i + 1) * sync_freq]
              for i in range(n_chunks)]

    for c in chunks[:args.warmup]:
        model.train_chunk(c)

    mv.barrier()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for c in chunks[args.warmup:]:
        model.train_chunk(c)
    if cuda:
        torch.cuda.synchronize()
    mv.barrier()
    elapsed = time.perf_counter() - t0

    t = torch.tensor([elapsed], dtype=torch.float64)
    if dist.is_initialized() and n > 1:
        if mv.Zoo.get().backend == "nccl":
            t = t.to(device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t[0])
    samples = args.steps * sync_freq * minibatch
    sps = n * samples / elapsed

    if rank == 0:
        print(json.dumps({
            "metric": ("LogReg dense samples/sec (whole node)" if dense
                       else "LogReg sparse samples/sec (whole node)"),
            "value": sps,
            "unit": "samples/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": (f"dense logistic regression {input_size} "
                          f"features {objective}"
                          if dense else
                          f"sparse logistic regression {input_size} "
                          f"features nnz={nnz} {objective}")
                         + (f" K={classes}" if classes > 1 else ""),
                "global_batch": n * minibatch * sync_freq,
                "seq_len": None,
                "parallelism": f"ps-sharded dp{n} (row all-to-all over xGMI)",
            },
        }), flush=True)
    mv.shutdown()
