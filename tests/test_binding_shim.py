"""The `multiverso` drop-in package (binding/python) must expose the
reference binding's API surface (binding/python/multiverso/api.py,
tables.py) backed by multiverso_amd."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "binding", "python"))

import torch


def test_shim_surface_and_roundtrip():
    import multiverso as mv
    for name in ("init", "shutdown", "barrier", "workers_num", "worker_id",
                 "server_id", "is_master_worker", "ArrayTableHandler",
                 "MatrixTableHandler"):
        assert hasattr(mv, name), name
    mv.init()
    t = mv.ArrayTableHandler(12, init_value=torch.arange(12).float())
    got = t.get()
    assert torch.allclose(torch.as_tensor(got),
                          torch.arange(12).float())
    m = mv.MatrixTableHandler(4, 3)
    m.add(torch.ones(4, 3))
    assert torch.equal(torch.as_tensor(m.get()), torch.ones(4, 3))
    mv.shutdown()
