"""WordEmbedding words/sec benchmark (BASELINE.json config:
"Applications/WordEmbedding skip-gram dim=200, 1M vocab, neg=5, N workers
over xGMI"). Synthetic Zipf corpus, random-init weights. ``value`` is
whole-job words/sec (summed sentence words consumed per second across all
workers) — the GPU generation of the reference's words/thread/sec metric
(trainer.cpp:45)."""

import json
import time

import torch
import torch.distributed as dist


def run_bench(args):
    import multiverso_amd as mv
    from .data import synthetic_block, zipf_counts
    from .model import WordEmbedding, WordEmbeddingOption

    mv.init(sync=True)
    n = mv.size()
    rank = mv.rank()
    device = mv.Zoo.get().device
    cuda = device.type == "cuda"

    vocab = getattr(args, "vocab", 1_000_000)
    dim = getattr(args, "dim", 200)
    block_words = getattr(args, "block_words", 500_000)
    if not cuda:
        vocab, block_words = min(vocab, 20_000), min(block_words, 20_000)

    opt = WordEmbeddingOption(embedding_size=dim, window=5, negative_num=5,
                              use_adagrad=getattr(args, "use_adagrad",
                                                  False),
                              total_words=block_words * args.steps * n,
                              seed=17)
    counts = zipf_counts(vocab, opt.total_words)
    model = WordEmbedding(opt, counts, device=device)

    # pre-generate blocks (data prep excluded from the timed region is NOT
    # allowed — generation is cheap torch ops; we still pre-generate the
    # random ids and count full train_block time including group building,
    # table pull/push and the kernel)
    blocks = [synthetic_block(vocab, block_words, seed=1000 * rank + i,
                              device=device)
              for i in range(args.warmup + args.steps)]

    for i in range(args.warmup):
        model.train_block(*blocks[i])
        model.sync_word_count()

    mv.barrier()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    total_words = 0
    for i in range(args.warmup, args.warmup + args.steps):
        total_words += model.train_block(*blocks[i])
        model.sync_word_count()
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
    words_per_sec = n * total_words / elapsed

    if rank == 0:
        print(json.dumps({
            "metric": "WordEmbedding words/sec (whole node)",
            "value": words_per_sec,
            "unit": "words/s",
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
                "model": f"word2vec skip-gram dim={dim} vocab={vocab} neg=5"
                         + (" adagrad" if getattr(args, 'use_adagrad', False) else ''),
                "global_batch": n * block_words,
                "seq_len": None,
                "parallelism": f"ps-sharded dp{n} (row all-to-all over xGMI)",
            },
        }), flush=True)
    mv.shutdown()
