"""Probe: can RCCL run world_size=2 with BOTH ranks on cuda:0?

Strict NCCL refuses duplicate devices in one communicator; RCCL has
historically allowed it (it serializes the ranks on the device).  If
this passes, the whole nccl-backend collective plane (the exact calls
comm.py issues: all_reduce, reduce_scatter_tensor,
all_gather_into_tensor, all_to_all_single, barrier(device_ids)) becomes
testable on the 1-GPU gpurun boxes instead of waiting for the driver's
8-GPU node.

Run:  python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 --master-port 2961 \
        tools/probe_rccl_ws2_1gpu.py
"""

import os
import sys

import torch
import torch.distributed as dist


def main():
    rank = int(os.environ["RANK"])
    torch.cuda.set_device(0)  # BOTH ranks on cuda:0
    dist.init_process_group("nccl", device_id=torch.device("cuda", 0))
    ws = dist.get_world_size()
    dev = torch.device("cuda", 0)

    t = torch.full((1024,), float(rank + 1), device=dev)
    dist.all_reduce(t)
    assert torch.all(t == 3.0), t[:4]

    # reduce_scatter_tensor: input ws*shard, output shard
    inp = torch.full((2 * 512,), float(rank + 1), device=dev)
    out = torch.empty(512, device=dev)
    dist.reduce_scatter_tensor(out, inp)
    assert torch.all(out == 3.0)

    # all_gather_into_tensor
    gat = torch.empty(2 * 512, device=dev)
    dist.all_gather_into_tensor(gat, out)
    assert torch.all(gat == 3.0)

    # all_to_all_single with uneven splits
    send = torch.arange(6, dtype=torch.float32, device=dev) + 10 * rank
    recv = torch.empty(6 if rank == 0 else 6, dtype=torch.float32,
                       device=dev)
    in_splits = [4, 2] if rank == 0 else [2, 4]
    out_splits = [4, 2] if rank == 0 else [2, 4]
    dist.all_to_all_single(recv, send,
                           output_split_sizes=out_splits,
                           input_split_sizes=in_splits)
    torch.cuda.synchronize()

    dist.barrier(device_ids=[0])
    if rank == 0:
        print("RCCL_WS2_1GPU_OK", ws, file=sys.stderr)
        print("RCCL_WS2_1GPU_OK")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
