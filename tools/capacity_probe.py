"""HBM-capacity / 64-bit-indexing probe: a 133 GiB MatrixTable shard on
one MI355X (35.8e9 fp32 elements — beyond int32 indexing), whole-table Add
through the fused SGD kernel, then row-keyed Get/Add at the far end of
the table. Validates the "server shards resident in 288 GB HBM3E, no
host staging" design point at a size the reference
could only reach with 24+ machines.

Run: python tools/capacity_probe.py  (needs a GPU with >= 270 GB free)
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    assert torch.cuda.is_available(), "capacity probe needs the GPU"
    import multiverso_amd as mv
    mv.init()
    rows, cols = 280_000_000, 128      # 35.8e9 elements, 143 GB fp32
    t = mv.MatrixTable(rows, cols, updater_type="sgd")
    n = rows * cols
    print(f"table: {rows}x{cols} = {n/1e9:.1f}G elements "
          f"({n*4/2**30:.0f} GiB)", flush=True)

    delta = torch.ones(rows, cols, device="cuda:0")
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    t.add(delta)
    t.flush()      # materialize the (deferred) add: timed region is real
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"whole-table Add (sgd): {dt*1e3:.1f} ms = "
          f"{3*n*4/dt/1e12:.2f} TB/s effective", flush=True)
    del delta
    torch.cuda.empty_cache()

    # far-end rows exercise offsets > 2^31 elements
    far = [0, rows // 2, rows - 2, rows - 1]
    got = t.get_rows(far)
    torch.cuda.synchronize()
    assert torch.equal(got, torch.full((4, cols), -1.0, device="cuda:0")), \
        got[:, :4]
    t.add_rows([rows - 1], torch.full((1, cols), 5.0, device="cuda:0"))
    got = t.get_rows([rows - 1, rows - 2])
    torch.cuda.synchronize()
    assert torch.equal(got[0], torch.full((cols,), -6.0, device="cuda:0"))
    assert torch.equal(got[1], torch.full((cols,), -1.0, device="cuda:0"))
    print("row ops at >2^31 element offsets: OK", flush=True)

    free, total = torch.cuda.mem_get_info()
    print(f"HBM: {free/2**30:.0f} GiB free of {total/2**30:.0f} GiB",
          flush=True)
    mv.shutdown()
    print("capacity probe OK", flush=True)


if __name__ == "__main__":
    main()
