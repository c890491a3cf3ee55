"""Torch logistic regression with per-batch ASGD sync — the rebuild of the
reference's theano example
(binding/python/examples/theano/logistic_regression.py:339-477): shared
W,b through mv_shared tensors, per-batch sync_all, master-only
validation, per-epoch barrier.

  python -m torch.distributed.run --nproc-per-node N \
      --master-addr 127.0.0.1 examples/logistic_regression_asgd.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import multiverso_amd as mv
from multiverso_amd.torch_ext import mv_shared, sync_all_mv_shared


def main() -> None:
    mv.init()
    device = mv.Zoo.get().device
    torch.manual_seed(123)  # identical init on every rank

    n_in, n_out, n_train, batch = 784, 10, 6000, 100
    # synthetic separable data (no network for MNIST)
    gw = torch.randn(n_in, n_out)
    gx = torch.randn(n_train, n_in)
    gy = (gx @ gw).argmax(1)
    # each rank takes an interleaved shard
    x = gx[mv.rank()::mv.size()].to(device)
    y = gy[mv.rank()::mv.size()].to(device)

    W = torch.zeros(n_in, n_out, device=device)
    b = torch.zeros(n_out, device=device)
    sw, sb = mv_shared(W), mv_shared(b)
    lr = 0.5

    for epoch in range(5):
        perm = torch.randperm(x.size(0), device=device)
        for i in range(0, x.size(0) - batch + 1, batch):
            idx = perm[i:i + batch]
            xb, yb = x[idx], y[idx]
            logits = xb @ W + b
            p = torch.softmax(logits, 1)
            grad = p.clone()
            grad[torch.arange(batch, device=device), yb] -= 1.0
            W -= lr / batch * (xb.T @ grad)
            b -= lr / batch * grad.sum(0)
            sync_all_mv_shared()       # per-batch ASGD delta merge
        if mv.is_master_worker():      # master-only validation
            acc = ((gx.to(device) @ W + b).argmax(1) == gy.to(device)) \
                .float().mean()
            print(f"epoch {epoch}: train-set acc {float(acc):.4f}",
                  flush=True)
        mv.barrier()                   # per-epoch barrier
    mv.shutdown()


if __name__ == "__main__":
    main()
