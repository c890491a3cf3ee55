"""torch_ext tests: sharedvar protocol, param manager ASGD math (exact
delta-merge oracle, reference test_multiverso.py:74-105 pattern), and the
ResNet example wiring."""

import pytest
import torch

import multiverso_amd as mv
from conftest import run_dist


@pytest.fixture()
def env():
    mv.init()
    yield
    from multiverso_amd.torch_ext.sharedvar import clear_registry
    clear_registry()
    mv.shutdown()


def test_sharedvar_single(env):
    from multiverso_amd.torch_ext import mv_shared, sync_all_mv_shared
    t = torch.arange(6, dtype=torch.float32)
    sv = mv_shared(t)
    assert torch.equal(sv.tensor, torch.arange(6, dtype=torch.float32))
    t += 1.0
    sync_all_mv_shared()
    assert torch.equal(t, torch.arange(6, dtype=torch.float32) + 1)


def test_param_manager_single(env):
    from multiverso_amd.torch_ext import MVTorchParamManager
    m = torch.nn.Linear(4, 3)
    pm = MVTorchParamManager(m)
    before = [p.detach().clone() for p in m.parameters()]
    with torch.no_grad():
        for p in m.parameters():
            p += 0.5
    pm.sync_all_param()
    for p, b in zip(m.parameters(), before):
        assert torch.allclose(p.detach(), b + 0.5)


def _param_manager_dist(rank, world):
    import multiverso_amd as mv
    import torch
    from multiverso_amd.torch_ext import MVTorchParamManager
    mv.init(sync=True)
    torch.manual_seed(7)  # same init everywhere
    m = torch.nn.Linear(5, 2, bias=False)
    pm = MVTorchParamManager(m)
    w0 = m.weight.detach().clone()
    # each rank applies a different additive step; after sync everyone
    # must hold w0 + sum of all steps (delta-merge oracle)
    with torch.no_grad():
        m.weight += (rank + 1)
    pm.sync_all_param()
    expect = w0 + sum(r + 1 for r in range(world))
    assert torch.allclose(m.weight.detach(), expect), (rank, m.weight)
    # second round: deltas accumulate from the merged state
    with torch.no_grad():
        m.weight += 1.0
    pm.sync_all_param()
    assert torch.allclose(m.weight.detach(), expect + world)
    mv.shutdown()


def test_param_manager_dist():
    run_dist(_param_manager_dist, 2)


def test_resnet32_shape(env):
    import sys, os
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "examples"))
    from resnet_cifar_asgd import ResNet32
    m = ResNet32()
    nparams = sum(p.numel() for p in m.parameters())
    assert 400_000 < nparams < 600_000, nparams
    out = m(torch.randn(2, 3, 32, 32))
    assert out.shape == (2, 10)
