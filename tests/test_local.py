"""Single-process runtime tests (reference Test/unittests tier: a 1-process
run degenerates to 1 worker + 1 server through the same code path,
SURVEY.md §4)."""

import numpy as np
import pytest
import torch

import multiverso_amd as mv


@pytest.fixture()
def env():
    mv.init()
    yield
    mv.shutdown()


@pytest.fixture()
def sync_env():
    mv.init(sync=True)
    yield
    mv.set_flag("sync", False)
    mv.shutdown()


def test_array_add_get(env):
    t = mv.ArrayTable(100)
    t.add(torch.ones(100))
    out = t.get()
    assert torch.equal(out, torch.ones(100))
    t.add(torch.full((100,), 2.0))
    assert torch.equal(t.get(), torch.full((100,), 3.0))


def test_array_async(env):
    t = mv.ArrayTable(50)
    h = t.add(torch.ones(50), async_op=True)
    h.wait()
    out, h2 = t.get(async_op=True)
    h2.wait()
    assert torch.equal(out, torch.ones(50))


def test_array_handler_init_value(env):
    h = mv.ArrayTableHandler(10, init_value=np.arange(10, dtype=np.float32))
    got = h.get()
    assert torch.equal(got, torch.arange(10, dtype=torch.float32))


def test_matrix_whole(env):
    t = mv.MatrixTable(8, 4)
    delta = torch.arange(32, dtype=torch.float32).reshape(8, 4)
    t.add(delta)
    assert torch.equal(t.get(), delta)


def test_matrix_rows(env):
    t = mv.MatrixTable(10, 3)
    vals = torch.ones(2, 3)
    t.add_rows([1, 7], vals)
    got = t.get_rows([7, 1, 0])
    expect = torch.stack([torch.ones(3), torch.ones(3), torch.zeros(3)])
    assert torch.equal(got, expect)


def test_matrix_rows_duplicate(env):
    t = mv.MatrixTable(10, 2)
    t.add_rows([3, 3], torch.ones(2, 2))
    assert torch.equal(t.get_rows([3]), torch.full((1, 2), 2.0))
    # duplicate ids in a GET return one row per occurrence, caller order
    # (exercises the ws=1 direct-gather fast path)
    got = t.get_rows([3, 0, 3])
    assert torch.equal(got, torch.stack([torch.full((2,), 2.0),
                                         torch.zeros(2),
                                         torch.full((2,), 2.0)]))


def test_matrix_random_init(env):
    t = mv.MatrixTable(16, 4, random_init=(-0.5, 0.5))
    got = t.get()
    assert got.abs().max() <= 0.5
    assert got.std() > 0


def test_kv_table(env):
    t = mv.KVTable()
    t.add([1, 5, 9], [1.0, 2.0, -3.0])
    got = t.get([1, 5, 9, 100])
    assert got == {1: 1.0, 5: 2.0, 9: -3.0, 100: 0}
    assert t.raw()[5] == 2.0


def test_sgd_updater(env):
    t = mv.ArrayTable(20, updater_type="sgd")
    t.add(torch.ones(20))  # sgd: data -= delta
    assert torch.equal(t.get(), -torch.ones(20))


def test_momentum_updater(env):
    t = mv.ArrayTable(4, updater_type="momentum")
    opt = mv.AddOption(momentum=0.5)
    d = torch.ones(4)
    t.add(d, option=opt)   # m = 0.5*0 + 0.5*1 = .5 ; data = -0.5
    t.add(d, option=opt)   # m = 0.25 + 0.5 = .75 ; data = -1.25
    assert torch.allclose(t.get(), torch.full((4,), -1.25))


def test_adagrad_updater(env):
    t = mv.ArrayTable(4, updater_type="adagrad")
    opt = mv.AddOption(learning_rate=0.1, rho=0.1)
    t.add(torch.ones(4), option=opt)
    # g = 1/0.1 = 10; G = 100; data -= 0.1 * 10 / sqrt(100 + 1e-6)
    expect = -0.1 * 10 / np.sqrt(100 + 1e-6)
    assert torch.allclose(t.get(), torch.full((4,), float(expect)))


def test_dcasgd_updater(env):
    t = mv.ArrayTable(4, updater_type="dcasgd")
    opt = mv.AddOption(learning_rate=0.1, lambda_=0.5)
    g = torch.full((4,), 2.0)
    # backup lazily = current shard (0); first step: w -= 0.1*(2 + 0.5*4*0)
    t.add(g, option=opt)
    assert torch.allclose(t.get(), torch.full((4,), -0.2))
    # second step from a different worker: its backup initializes to the
    # CURRENT w (-0.2), so the compensation term is again 0
    opt2 = mv.AddOption(worker_id=1, learning_rate=0.1, lambda_=0.5)
    t.add(g, option=opt2)
    assert torch.allclose(t.get(), torch.full((4,), -0.4))
    # third step from worker 0: bak_0 = -0.2, w = -0.4
    # w -= 0.1 * (2 + 0.5*4*(-0.4 - -0.2)) = -0.4 - 0.1*(2 - 0.4) = -0.56
    t.add(g, option=opt)
    assert torch.allclose(t.get(), torch.full((4,), -0.56))


def test_dcasgda_updater(env):
    t = mv.ArrayTable(4, updater_type="dcasgda")
    opt = mv.AddOption(learning_rate=0.1, lambda_=0.5, rho=0.9)
    g = torch.full((4,), 2.0)
    t.add(g, option=opt)
    # m = 0.1*4 = 0.4; lam_t = 0.5/sqrt(0.4+eps); bak=w0=0 -> comp term 0
    assert torch.allclose(t.get(), torch.full((4,), -0.2))
    t.add(g, option=opt)
    # m = 0.9*0.4 + 0.1*4 = 0.76; lam_t = 0.5/sqrt(0.76); bak=-0.2, w=-0.2
    # comp = lam_t*4*(w - bak) = 0 -> w = -0.4
    assert torch.allclose(t.get(), torch.full((4,), -0.4))


def test_checkpoint_roundtrip(env, tmp_path):
    t = mv.ArrayTable(32)
    t.add(torch.arange(32, dtype=torch.float32))
    p = str(tmp_path / "arr.bin")
    t.store(p)
    # byte-format check: raw little-endian fp32, whole table
    raw = np.fromfile(p, dtype=np.float32)
    assert np.array_equal(raw, np.arange(32, dtype=np.float32))
    t2 = mv.ArrayTable(32)
    t2.load(p)
    assert torch.equal(t2.get(), torch.arange(32, dtype=torch.float32))


def test_matrix_checkpoint(env, tmp_path):
    t = mv.MatrixTable(6, 5)
    vals = torch.arange(30, dtype=torch.float32).reshape(6, 5)
    t.add(vals)
    p = str(tmp_path / "mat.bin")
    t.store(p)
    t2 = mv.MatrixTable(6, 5)
    t2.load(p)
    assert torch.equal(t2.get(), vals)


def test_kv_checkpoint(env, tmp_path):
    t = mv.KVTable()
    t.add([2, 4], [1.0, 2.0])
    p = str(tmp_path / "kv.bin")
    t.store(p)
    t2 = mv.KVTable()
    t2.load(p)
    assert t2.get([2, 4]) == {2: 1.0, 4: 2.0}


def test_aggregate_single(env):
    x = torch.ones(4)
    mv.aggregate(x)
    assert torch.equal(x, torch.ones(4))


def test_dashboard(env):
    mv.Dashboard.reset()
    with mv.monitor("unit.test"):
        pass
    assert mv.Dashboard.get("unit.test").count == 1
    assert "unit.test" in mv.Dashboard.display()


def test_table_smaller_than_world_checks(env):
    from multiverso_amd.log import FatalError
    try:
        mv.ArrayTable(0)
        assert False, "expected CHECK failure"
    except FatalError:
        pass


def test_get_with_user_buffer(env):
    t = mv.ArrayTable(16, updater_type="default")
    t.add(torch.arange(16, dtype=torch.float32))
    buf = torch.empty(16)
    out = t.get(out=buf)
    assert out is buf
    assert torch.equal(buf, torch.arange(16, dtype=torch.float32))


def test_handler_numpy_inputs(env):
    h = mv.ArrayTableHandler(8)
    h.add(np.ones(8))
    got = h.get()
    assert torch.equal(got.cpu(), torch.ones(8))
    m = mv.MatrixTableHandler(3, 4)
    m.add(np.full((3, 4), 2.0))
    assert torch.equal(m.get().cpu(), torch.full((3, 4), 2.0))
    m.add(np.ones((2, 4)), row_ids=[0, 2])
    assert torch.equal(m.get(row_ids=[0]).cpu(), torch.full((1, 4), 3.0))


def test_updater_default_options(env):
    # option=None must behave like a default AddOption for every updater
    for u in ("momentum", "adagrad", "dcasgd", "dcasgda"):
        t = mv.ArrayTable(4, updater_type=u)
        t.add(torch.ones(4))
        got = t.get()
        assert torch.isfinite(got).all(), u


def test_deterministic_keyed_scatter(env):
    mv.set_flag("deterministic", True)
    try:
        t = mv.MatrixTable(10, 2)
        t.add_rows([3, 3, 5], torch.tensor([[1.0, 2.0], [3.0, 4.0],
                                            [5.0, 6.0]]))
        assert torch.equal(t.get_rows([3, 5]),
                           torch.tensor([[4.0, 6.0], [5.0, 6.0]]))
    finally:
        mv.set_flag("deterministic", False)


def test_lifecycle_robustness():
    # double init is idempotent; shutdown twice is safe; re-init gives a
    # fresh table registry (reference MV_ShutDown + re-Init semantics)
    mv.init()
    mv.init()
    t = mv.ArrayTable(4)
    t.add(torch.ones(4))
    mv.shutdown()
    mv.shutdown()
    mv.init()
    t2 = mv.ArrayTable(4)
    assert torch.equal(t2.get(), torch.zeros(4))
    mv.shutdown()
    mv.barrier()   # no-op outside a session
