"""CIFAR-10 ResNet-32 ASGD via multiverso_amd — the rebuild of the
reference's headline binding benchmark (binding/lua/docs/BENCHMARK.md:37-39
and binding/python/docs/BENCHMARK.md:57-59: 8 proc x 1 GPU ASGD via a
Multiverso ArrayTable, 11.37 s/epoch on 8x K40m / 34.1 s/epoch
Theano+Lasagne).

Each rank trains its shard of the (synthetic, random — no network for
datasets) 50,000-image epoch locally and syncs all 464k-ish parameters
through one sharded ArrayTable after every batch (the keras/lasagne
per-batch callback protocol). Launch:

  python -m torch.distributed.run --nproc-per-node N \
      --master-addr 127.0.0.1 examples/resnet_cifar_asgd.py --epochs 2
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn
import torch.nn.functional as F


class BasicBlock(nn.Module):
    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, cout, 3, stride, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(cout)
        self.conv2 = nn.Conv2d(cout, cout, 3, 1, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(cout)
        self.short = None
        if stride != 1 or cin != cout:
            self.short = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride, bias=False),
                nn.BatchNorm2d(cout))

    def forward(self, x):
        out = F.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        out = out + (self.short(x) if self.short else x)
        return F.relu(out)


class ResNet32(nn.Module):
    """ResNet-32 for CIFAR (3 stages x 5 blocks; ~466k params, the
    reference quotes 464,154 for its Lasagne variant)."""

    def __init__(self, num_classes=10):
        super().__init__()
        self.conv = nn.Conv2d(3, 16, 3, 1, 1, bias=False)
        self.bn = nn.BatchNorm2d(16)
        layers = []
        cin = 16
        for stage, cout in enumerate([16, 32, 64]):
            for b in range(5):
                layers.append(BasicBlock(cin, cout,
                                         2 if (stage > 0 and b == 0) else 1))
                cin = cout
        self.blocks = nn.Sequential(*layers)
        self.fc = nn.Linear(64, num_classes)

    def forward(self, x):
        out = F.relu(self.bn(self.conv(x)))
        out = self.blocks(out)
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--batch", type=int, default=128)
    p.add_argument("--epoch-images", type=int, default=50_000)
    p.add_argument("--lr", type=float, default=0.1)
    args = p.parse_args()

    import multiverso_amd as mv
    from multiverso_amd.torch_ext import MVTorchParamManager

    mv.init()
    device = mv.Zoo.get().device
    torch.manual_seed(42)  # same init on every rank
    model = ResNet32().to(device)
    nparams = sum(p.numel() for p in model.parameters())
    manager = MVTorchParamManager(model)
    opt = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9,
                          weight_decay=1e-4)

    n = mv.size()
    per_rank = args.epoch_images // n
    batches = per_rank // args.batch
    g = torch.Generator().manual_seed(1000 + mv.rank())

    for epoch in range(args.epochs):
        t0 = time.perf_counter()
        total_loss = 0.0
        for _ in range(batches):
            x = torch.randn(args.batch, 3, 32, 32, generator=g).to(device)
            y = torch.randint(0, 10, (args.batch,), generator=g).to(device)
            opt.zero_grad(set_to_none=True)
            loss = F.cross_entropy(model(x), y)
            loss.backward()
            opt.step()
            manager.sync_all_param()   # per-batch ASGD sync
            total_loss += float(loss)
        if device.type == "cuda":
            torch.cuda.synchronize()
        mv.barrier()
        dt = time.perf_counter() - t0
        if mv.rank() == 0:
            print(f"epoch {epoch}: {dt:.2f} s/epoch "
                  f"({args.epoch_images} imgs, {n} workers, "
                  f"{nparams} params, loss {total_loss / batches:.3f})",
                  flush=True)
    mv.shutdown()


if __name__ == "__main__":
    main()
