"""Sequence-to-sequence addition RNN trained with per-batch ASGD sync —
the rebuild of the reference's keras example
(binding/python/examples/keras/addition_rnn.py, wired through
keras_ext.MVCallback / keras_ext.param_manager per-batch sync): learn to
map "123+58" -> "181" character-by-character. Data is generated, so no
dataset download is needed.

Launch (N workers):
  python -m torch.distributed.run --nproc-per-node N \
      --master-addr 127.0.0.1 examples/addition_rnn_asgd.py
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn as nn

CHARS = "0123456789+ "
C2I = {c: i for i, c in enumerate(CHARS)}
DIGITS = 3
MAXLEN = DIGITS * 2 + 1          # "123+456"
OUTLEN = DIGITS + 1              # "579 "


def encode(s, length):
    s = s.ljust(length)
    return torch.tensor([C2I[c] for c in s], dtype=torch.int64)


def make_batch(batch, gen):
    a = torch.randint(0, 10 ** DIGITS, (batch,), generator=gen)
    b = torch.randint(0, 10 ** DIGITS, (batch,), generator=gen)
    x = torch.stack([encode(f"{int(ai)}+{int(bi)}", MAXLEN)
                     for ai, bi in zip(a, b)])
    y = torch.stack([encode(str(int(ai + bi)), OUTLEN)
                     for ai, bi in zip(a, b)])
    return x, y


class AdditionRNN(nn.Module):
    """Encoder LSTM -> repeated context -> decoder LSTM -> softmax per
    output char (the keras example's architecture)."""

    def __init__(self, hidden=128):
        super().__init__()
        self.emb = nn.Embedding(len(CHARS), len(CHARS))
        self.enc = nn.LSTM(len(CHARS), hidden, batch_first=True)
        self.dec = nn.LSTM(hidden, hidden, batch_first=True)
        self.out = nn.Linear(hidden, len(CHARS))

    def forward(self, x):
        _, (h, _) = self.enc(self.emb(x))
        ctx = h[-1].unsqueeze(1).expand(-1, OUTLEN, -1)
        d, _ = self.dec(ctx)
        return self.out(d)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batches", type=int, default=300)
    p.add_argument("--batch", type=int, default=128)
    args = p.parse_args()

    import multiverso_amd as mv
    from multiverso_amd.torch_ext import MVTorchParamManager

    mv.init()
    device = mv.Zoo.get().device
    torch.manual_seed(7)            # same init on every rank
    model = AdditionRNN().to(device)
    manager = MVTorchParamManager(model)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    gen = torch.Generator().manual_seed(100 + mv.rank())

    t0 = time.perf_counter()
    for step in range(args.batches):
        x, y = make_batch(args.batch, gen)
        x, y = x.to(device), y.to(device)
        opt.zero_grad(set_to_none=True)
        logits = model(x)
        loss = nn.functional.cross_entropy(
            logits.reshape(-1, len(CHARS)), y.reshape(-1))
        loss.backward()
        opt.step()
        manager.sync_all_param()     # per-batch ASGD (MVCallback protocol)
        if mv.rank() == 0 and (step + 1) % 100 == 0:
            xq, yq = make_batch(256, torch.Generator().manual_seed(1))
            with torch.no_grad():
                pred = model(xq.to(device)).argmax(-1).cpu()
            acc = (pred == yq).all(1).float().mean()
            print(f"step {step + 1}: loss {float(loss):.3f} "
                  f"seq-acc {float(acc):.3f}", flush=True)
    mv.barrier()
    if mv.rank() == 0:
        print(f"done in {time.perf_counter() - t0:.1f}s "
              f"({mv.size()} workers)", flush=True)
    mv.shutdown()


if __name__ == "__main__":
    main()
