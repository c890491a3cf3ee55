"""Small CNN with per-batch ASGD sync via an MVSharedTensor per parameter
— the rebuild of the reference's theano cnn example
(binding/python/examples/theano/cnn.py, wired through
theano_ext.sharedvar.mv_shared + sync_all_mv_shared_vars): this one uses
the sharedvar protocol (per-tensor delta add, then set to merged value)
rather than the single concatenated param-manager table, exercising the
second binding pattern. Synthetic MNIST-shaped data (no network for
datasets).

Launch: python -m torch.distributed.run --nproc-per-node N \
            --master-addr 127.0.0.1 examples/cnn_mnist_asgd.py
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn
import torch.nn.functional as F


class SmallCNN(nn.Module):
    def __init__(self):
        super().__init__()
        self.c1 = nn.Conv2d(1, 20, 5)
        self.c2 = nn.Conv2d(20, 50, 5)
        self.f1 = nn.Linear(50 * 4 * 4, 500)
        self.f2 = nn.Linear(500, 10)

    def forward(self, x):
        x = F.max_pool2d(F.relu(self.c1(x)), 2)
        x = F.max_pool2d(F.relu(self.c2(x)), 2)
        x = F.relu(self.f1(x.flatten(1)))
        return self.f2(x)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batches", type=int, default=200)
    p.add_argument("--batch", type=int, default=256)
    args = p.parse_args()

    import multiverso_amd as mv
    from multiverso_amd.torch_ext import MVSharedTensor

    mv.init()
    device = mv.Zoo.get().device
    torch.manual_seed(11)
    model = SmallCNN().to(device)
    shared = [MVSharedTensor(q) for q in model.parameters()]
    opt = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    gen = torch.Generator().manual_seed(500 + mv.rank())

    t0 = time.perf_counter()
    for step in range(args.batches):
        x = torch.randn(args.batch, 1, 28, 28, generator=gen).to(device)
        y = torch.randint(0, 10, (args.batch,), generator=gen).to(device)
        opt.zero_grad(set_to_none=True)
        loss = F.cross_entropy(model(x), y)
        loss.backward()
        opt.step()
        for s in shared:               # sync_all_mv_shared_vars
            s.mv_sync()
        if mv.rank() == 0 and (step + 1) % 100 == 0:
            print(f"step {step + 1}: loss {float(loss):.3f}", flush=True)
    mv.barrier()
    if mv.rank() == 0:
        print(f"done in {time.perf_counter() - t0:.1f}s "
              f"({mv.size()} workers)", flush=True)
    mv.shutdown()


if __name__ == "__main__":
    main()
