"""SparseFilter — sparsity compressor for keyed table traffic.

Capability parity with the reference SparseFilter<data,index>
(include/multiverso/util/quantization_util.h:25-158): per-payload, if more
than half the values are zero, encode as (index, value) pairs with a size
header; otherwise pass through dense. ``OneBitsFilter`` is an empty stub
in the reference (:160-161) and is intentionally not reproduced.

MI355X placement: dense whole-table traffic goes
through reduce-scatter/all-gather and is never filtered (collectives need
dense payloads — and xGMI bandwidth makes dense cheaper than host-side
repacking for those). The filter applies to row-keyed exchange payloads;
the bytes saved/spent are recorded on the Dashboard either way so the
keep-dense decision is measured, not assumed."""

from __future__ import annotations

from typing import Tuple

import torch

from .dashboard import Dashboard


def try_compress(values: torch.Tensor) -> Tuple[bool, torch.Tensor]:
    """If >50% of ``values`` are zero, return (True, packed) where packed =
    [count | idx...| val...] in one flat float32 tensor. The count and the
    indices are int32 values REINTERPRETED as float32 bits (torch .view),
    not converted — exact for any payload up to 2^31 elements; larger
    payloads pass through dense (they would overflow int32 indexing)."""
    flat = values.reshape(-1)
    m = Dashboard.get("sparse_filter.bytes_in")
    m.count += 1
    m.elapsed_ms += flat.numel() * 4
    if flat.numel() >= (1 << 31):
        Dashboard.get("sparse_filter.bytes_out").elapsed_ms += flat.numel() * 4
        return False, flat
    nz = torch.nonzero(flat, as_tuple=False).reshape(-1)
    if nz.numel() * 2 + 1 >= flat.numel():
        Dashboard.get("sparse_filter.bytes_out").elapsed_ms += flat.numel() * 4
        return False, flat
    packed = torch.empty(1 + 2 * nz.numel(), dtype=torch.float32,
                         device=flat.device)
    packed[0] = torch.tensor([nz.numel()],
                             dtype=torch.int32).view(torch.float32)[0]
    packed[1:1 + nz.numel()] = nz.to(torch.int32).view(torch.float32)
    packed[1 + nz.numel():] = flat[nz]
    Dashboard.get("sparse_filter.bytes_out").elapsed_ms += packed.numel() * 4
    return True, packed


def decompress(packed: torch.Tensor, out_numel: int) -> torch.Tensor:
    """Inverse of try_compress for a compressed payload. The nonzero
    count is derived from the payload size (numel = 1 + 2k), so no
    device read of the count header is needed — the header stays on the
    wire for format parity and as a consistency check on CPU."""
    k = (packed.numel() - 1) // 2
    out = torch.zeros(out_numel, dtype=torch.float32, device=packed.device)
    idx = packed[1:1 + k].view(torch.int32).long()
    out[idx] = packed[1 + k:1 + 2 * k]
    return out


def filter_in(values: torch.Tensor) -> Tuple[torch.Tensor, bool]:
    """FilterIn equivalent: returns (payload, is_compressed)."""
    compressed, payload = try_compress(values)
    return payload, compressed


def filter_out(payload: torch.Tensor, is_compressed: bool,
               out_numel: int) -> torch.Tensor:
    """FilterOut equivalent."""
    if is_compressed:
        return decompress(payload, out_numel)
    return payload
