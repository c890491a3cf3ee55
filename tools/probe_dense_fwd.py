"""Isolate k_lr_dense_fwd vs rocBLAS GEMM + k_lr_dense_post timing
(B=4096, d=2048, K=10 — the dense bench minibatch shape)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from multiverso_amd import ops

hip = ops.module(required=True)
dev = "cuda:0"
B, d, K = 4096, 2048, 10
x = torch.randn(B, d, device=dev)
w = torch.randn(d, K, device=dev) * 0.05
labels = torch.randint(0, K, (B,), device=dev).float()
diff = torch.empty(B, K, device=dev)
loss = torch.zeros((), device=dev)


def bench(name, fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters * 1e6
    print(f"{name:28s} {dt:8.1f} us")
    return dt


bench("fused lr_dense_fwd nt", lambda: hip.lr_dense_fwd(
    x, w, labels, None, diff, loss, 1.0 / B, True))
bench("fused lr_dense_fwd plain", lambda: hip.lr_dense_fwd(
    x, w, labels, None, diff, loss, 1.0 / B, False))
logits = torch.empty(B, K, device=dev)


def gemm_post():
    torch.matmul(x, w, out=logits)
    hip.lr_dense_post(logits, labels, None, loss, 1.0 / B)


bench("rocBLAS gemm + dense_post", gemm_post)
bench("gemm alone", lambda: torch.matmul(x, w, out=logits))
bench("post alone", lambda: hip.lr_dense_post(logits, labels, None, loss,
                                              1.0 / B))
bench("bwd addmm (X^T@diff)", lambda: w.addmm_(x.t(), diff, alpha=-1e-6))
print("peak-est: X bytes", B * d * 4 / 1e6, "MB")

rows = torch.randint(0, 10_000_000, (1_000_000,), device=dev).unique()
shard = torch.randn(10_000_000, 10, device=dev)
gsq = torch.rand(10_000_000, 10, device=dev)
vals = torch.randn(rows.numel(), 10, device=dev)
bench("row_gather 1M rows K=10", lambda: hip.row_gather(shard, rows))
bench("scatter_adagrad_u 1M rows", lambda: hip.row_scatter_adagrad(
    shard, gsq, rows, vals, 0.05, 0.01, 1e-10, True))
