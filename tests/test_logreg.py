"""LogisticRegression app tests: config parsing, readers, objectives,
local + PS models (sigmoid/softmax/ftrl/adagrad), model io, and a
2-process PS run."""

import numpy as np
import pytest
import torch

import multiverso_amd as mv
from conftest import run_dist
from multiverso_amd.apps.logreg import LogReg, LogRegConfig
from multiverso_amd.apps.logreg.objective import Batch, create_objective
from multiverso_amd.apps.logreg.reader import (SampleReader, parse_text_lines,
                                               read_bsparse_batches,
                                               synthetic_batches,
                                               write_bsparse)


@pytest.fixture()
def env():
    mv.init()
    yield
    mv.shutdown()


def test_config_parse(tmp_path):
    p = tmp_path / "c.cfg"
    p.write_text("input_size=100\nobjective_type=softmax\n"
                 "use_ps=true\nlearning_rate=0.5\n# comment\nbadkey=1\n")
    cfg = LogRegConfig.from_file(str(p))
    assert cfg.input_size == 100
    assert cfg.objective_type == "softmax"
    assert cfg.use_ps is True
    assert cfg.learning_rate == 0.5


def test_text_reader(tmp_path):
    p = tmp_path / "train.txt"
    p.write_text("1 0:0.5 3:1.5\n0 2:1.0\n1 1:2.0 4:0.1\n")
    batches = list(SampleReader(str(p), 2).batches())
    assert len(batches) == 2
    b = batches[0]
    assert b.size == 2
    assert b.keys.tolist() == [0, 3, 2]
    assert b.labels.tolist() == [1.0, 0.0]
    # with input_size the bias feature (row_size-1, value 1) is appended
    # to every sample (reference reader.cpp:195-196,216)
    b2 = list(SampleReader(str(p), 2, input_size=100).batches())[0]
    assert b2.keys.tolist() == [0, 3, 99, 2, 99]
    assert b2.vals.tolist() == [0.5, 1.5, 1.0, 1.0, 1.0]


def test_weighted_reader(tmp_path):
    # WeightedSampleReader folds the weight into the values
    # (reader.cpp:261: value * weight)
    p = tmp_path / "w.txt"
    p.write_text("1 2.0 0:1 3:0.5\n")
    b = list(SampleReader(str(p), 4, "weight", input_size=10).batches())[0]
    assert b.keys.tolist() == [0, 3, 9]
    assert b.vals.tolist() == [2.0, 1.0, 1.0]


def test_bsparse_roundtrip(tmp_path):
    # reference layout: [size_t nnz][int32 label][double weight]
    # [size_t keys...]; value = weight everywhere incl. bias
    p = str(tmp_path / "b.bin")
    write_bsparse(p, [(1, 0.5, [5, 9]), (0, 2.0, [2])])
    batches = list(read_bsparse_batches(p, 10, bias_key=99))
    assert batches[0].size == 2
    assert batches[0].keys.tolist() == [5, 9, 99, 2, 99]
    assert batches[0].vals.tolist() == [0.5, 0.5, 0.5, 2.0, 2.0]
    # byte-layout check against the reference struct sizes
    import struct, os
    raw = open(p, "rb").read()
    assert len(raw) == (20 + 16) + (20 + 8)
    nnz, label, weight = struct.unpack_from("<qid", raw, 0)
    assert (nnz, label, weight) == (2, 1, 0.5)


def _train_local(cfg, n_batches=60):
    batches, _ = synthetic_batches(cfg.input_size, n_batches,
                                   cfg.minibatch_size, nnz=16,
                                   output_size=max(cfg.output_size, 1)
                                   if cfg.objective_type == "softmax" else 1,
                                   seed=7)
    lr = LogReg(cfg)
    lr.train(iter(batches))
    acc, loss = lr.test(iter(batches[:10]))
    return lr, acc


def test_local_sigmoid_learns(env):
    cfg = LogRegConfig(input_size=4096, minibatch_size=64,
                       learning_rate=0.05, learning_rate_coef=1e6,
                       train_epoch=1, show_time_per_sample=0)
    _, acc = _train_local(cfg)
    assert acc > 0.75, acc


def test_local_softmax_learns(env):
    cfg = LogRegConfig(input_size=4096, output_size=4, minibatch_size=64,
                       objective_type="softmax", learning_rate=0.05,
                       learning_rate_coef=1e6, show_time_per_sample=0)
    _, acc = _train_local(cfg)
    assert acc > 0.5, acc


def test_local_ftrl_learns(env):
    cfg = LogRegConfig(input_size=4096, minibatch_size=64,
                       objective_type="ftrl", show_time_per_sample=0)
    _, acc = _train_local(cfg)
    assert acc > 0.7, acc


def test_l2_regular_shrinks(env):
    cfg = LogRegConfig(input_size=512, minibatch_size=32,
                       regular_type="l2", regular_coef=0.1,
                       learning_rate=0.05, learning_rate_coef=1e6,
                       show_time_per_sample=0)
    lr, _ = _train_local(cfg, 30)
    cfg2 = LogRegConfig(input_size=512, minibatch_size=32,
                        learning_rate=0.05, learning_rate_coef=1e6,
                        show_time_per_sample=0)
    lr2, _ = _train_local(cfg2, 30)
    assert lr.model.weight.norm() < lr2.model.weight.norm()


def test_model_io_dense(env, tmp_path):
    cfg = LogRegConfig(input_size=256, minibatch_size=32,
                       show_time_per_sample=0)
    lr, _ = _train_local(cfg, 10)
    p = str(tmp_path / "model.bin")
    lr.save_model(p)
    lr2 = LogReg(LogRegConfig(input_size=256, show_time_per_sample=0))
    lr2.load_model(p)
    assert torch.equal(lr.model.weight, lr2.model.weight)


def test_model_io_sparse(env, tmp_path):
    cfg = LogRegConfig(input_size=256, sparse=True, minibatch_size=32,
                       show_time_per_sample=0)
    lr, _ = _train_local(cfg, 10)
    p = str(tmp_path / "model.bin")
    lr.save_model(p)
    lr2 = LogReg(LogRegConfig(input_size=256, sparse=True,
                              show_time_per_sample=0))
    lr2.load_model(p)
    assert torch.equal(lr.model.weight, lr2.model.weight)


def test_ps_model_single(env):
    cfg = LogRegConfig(input_size=4096, minibatch_size=64, use_ps=True,
                       sync_frequency=4, learning_rate=0.05,
                       learning_rate_coef=1e6, show_time_per_sample=0)
    _, acc = _train_local(cfg)
    assert acc > 0.75, acc


def test_ps_model_adagrad_single(env):
    cfg = LogRegConfig(input_size=4096, minibatch_size=64, use_ps=True,
                       updater_type="adagrad", sync_frequency=4,
                       learning_rate=0.05, show_time_per_sample=0)
    _, acc = _train_local(cfg)
    assert acc > 0.7, acc


def test_ps_model_ftrl_single(env):
    cfg = LogRegConfig(input_size=4096, minibatch_size=64, use_ps=True,
                       objective_type="ftrl", sync_frequency=4,
                       show_time_per_sample=0)
    _, acc = _train_local(cfg)
    assert acc > 0.7, acc


def _ps_dist(rank, world):
    import multiverso_amd as mv
    from multiverso_amd.apps.logreg import LogReg, LogRegConfig
    from multiverso_amd.apps.logreg.reader import synthetic_batches
    mv.init(sync=True)
    cfg = LogRegConfig(input_size=2048, minibatch_size=32, use_ps=True,
                       sync_frequency=2, learning_rate=0.05,
                       learning_rate_coef=1e6, show_time_per_sample=0)
    batches, _ = synthetic_batches(cfg.input_size, 40, cfg.minibatch_size,
                                   nnz=16, seed=100 + rank)
    lr = LogReg(cfg)
    lr.train(iter(batches))
    acc, _ = lr.test(iter(batches[:10]))
    assert acc > 0.7, (rank, acc)
    mv.shutdown()


def test_ps_dist():
    run_dist(_ps_dist, 2)


def _logreg_uneven_files(rank, world, tmpdir):
    """Per-rank train files with different sample counts (the reference's
    multi-file deployment): the collective chunk loop must not
    desynchronize."""
    import os
    import torch
    import multiverso_amd as mv
    from multiverso_amd.apps.logreg.config import LogRegConfig
    from multiverso_amd.apps.logreg.logreg import LogReg
    mv.init(sync=True)
    path = os.path.join(tmpdir, f"r{rank}.txt")
    n = 40 if rank == 0 else 12
    with open(path, "w") as f:
        for i in range(n):
            f.write(f"{i % 2} {i % 50}:1 {(i * 7) % 50}:0.5\n")
    cfg = LogRegConfig(input_size=64, output_size=1,
                       objective_type="sigmoid", updater_type="sgd",
                       learning_rate=0.1, minibatch_size=8,
                       train_epoch=1, reader_type="default", use_ps=True,
                       sync_frequency=2, train_file=path, test_file=path)
    lr = LogReg(cfg)
    lr.train()
    acc, _ = lr.test()
    assert 0.0 <= acc <= 1.0
    mv.shutdown()


def test_logreg_uneven_files_dist(tmp_path):
    import functools
    from conftest import run_dist
    run_dist(functools.partial(_logreg_uneven_files, tmpdir=str(tmp_path)), 2)


# ---- dense data mode (reference sparse=false, its own mnist.config) ----

def test_dense_reader_parse(tmp_path):
    from multiverso_amd.apps.logreg.objective import DenseBatch
    p = tmp_path / "dense.txt"
    p.write_text("1 0.5 2.0 -1.0\n0 1.0 0.0 3.0\n")
    b = list(SampleReader(str(p), 2, input_size=4,
                          sparse=False).batches())[0]
    assert isinstance(b, DenseBatch)
    assert b.x.shape == (2, 4)       # 3 features + bias
    assert torch.equal(b.x[0], torch.tensor([0.5, 2.0, -1.0, 1.0]))
    assert torch.equal(b.labels, torch.tensor([1.0, 0.0]))


def _synthetic_dense(n, d, K, seed=0):
    """Separable dense stream: x ~ N(0,1), label = argmax(x @ hidden)."""
    from multiverso_amd.apps.logreg.objective import DenseBatch
    g = torch.Generator().manual_seed(seed)
    hidden = torch.randn(d, K, generator=g)
    x = torch.randn(n, d, generator=g)
    scores = x @ hidden
    labels = (scores.argmax(1).float() if K > 1
              else (scores.squeeze(1) > 0).float())
    x = torch.cat([x, torch.ones(n, 1)], dim=1)   # bias column
    return x, labels


def test_dense_local_softmax_learns(env):
    from multiverso_amd.apps.logreg.objective import DenseBatch
    cfg = LogRegConfig(input_size=33, output_size=4,
                       objective_type="softmax", sparse=False,
                       minibatch_size=32, learning_rate=0.2,
                       learning_rate_coef=1e6, show_time_per_sample=0)
    lr = LogReg(cfg)
    x, labels = _synthetic_dense(960, 32, 4, seed=3)
    batches = [DenseBatch(x[i:i + 32], labels[i:i + 32])
               for i in range(0, 960, 32)]
    lr.train(iter(batches))
    acc, _ = lr.test(iter(batches[:10]))
    assert acc > 0.8, acc


def _dense_ps_dist(rank, world):
    """Dense PS mode at ws2: whole-table pull/push on the collective
    plane; per-rank test sets of DIFFERENT sizes exercise the dense
    empty-batch participation."""
    import torch
    import multiverso_amd as mv
    from multiverso_amd.apps.logreg import LogReg, LogRegConfig
    from multiverso_amd.apps.logreg.objective import DenseBatch
    from test_logreg import _synthetic_dense
    mv.init(sync=True)
    cfg = LogRegConfig(input_size=17, output_size=3,
                       objective_type="softmax", sparse=False,
                       use_ps=True, sync_frequency=2, minibatch_size=16,
                       learning_rate=0.2, learning_rate_coef=1e6,
                       show_time_per_sample=0)
    # SAME hidden label function on every rank (seed fixed); each rank
    # trains its own interleaved sample shard
    x, labels = _synthetic_dense(640, 16, 3, seed=11)
    x, labels = x[rank::world], labels[rank::world]
    batches = [DenseBatch(x[i:i + 16], labels[i:i + 16])
               for i in range(0, 320, 16)]
    lr = LogReg(cfg)
    lr.train(iter(batches))
    n_test = 6 if rank == 0 else 3       # uneven participation
    acc, _ = lr.test(iter(batches[:n_test]))
    assert acc > 0.7, (rank, acc)
    mv.shutdown()


def test_dense_ps_dist():
    run_dist(_dense_ps_dist, 2)


def test_dense_cli_end_to_end(tmp_path):
    """Reference-style dense deployment: a sparse=false config file +
    'label value value ...' text through the CLI main."""
    import subprocess, sys, os
    d, K = 8, 3
    x, labels = _synthetic_dense(240, d, K, seed=9)
    train = tmp_path / "train.data"
    with open(train, "w") as f:
        for i in range(240):
            vals = " ".join(f"{v:.4f}" for v in x[i, :d].tolist())
            f.write(f"{int(labels[i])} {vals}\n")
    cfgf = tmp_path / "dense.config"
    cfgf.write_text(
        f"input_size={d + 1}\noutput_size={K}\nobjective_type=softmax\n"
        "sparse=false\ntrain_epoch=6\nminibatch_size=24\n"
        "learning_rate=0.2\nlearning_rate_coef=1000000\n"
        f"train_file={train}\ntest_file={train}\n"
        "show_time_per_sample=0\n")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env2 = dict(os.environ)
    env2["PYTHONPATH"] = repo + ":" + env2.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, "-m", "multiverso_amd.apps.logreg.main",
         str(cfgf)], capture_output=True, text=True, timeout=300,
        env=env2)
    assert r.returncode == 0, r.stderr[-2000:]
    import re
    m = re.findall(r"correct \(([0-9.]+)\)", r.stdout + r.stderr)
    assert m and float(m[-1]) > 0.8, (m, r.stdout[-800:])


def test_dense_weighted_reader_parse(tmp_path):
    """Weighted dense lines: 'label weight value ...' (reference
    WeightedSampleReader over dense data)."""
    from multiverso_amd.apps.logreg.objective import DenseBatch
    p = tmp_path / "wd.txt"
    p.write_text("1 0.5 2.0 3.0\n0 2.0 -1.0 1.0\n")
    b = list(SampleReader(str(p), 2, "weight", input_size=3,
                          sparse=False).batches())[0]
    assert isinstance(b, DenseBatch)
    assert torch.equal(b.x, torch.tensor([[2.0, 3.0, 1.0],
                                          [-1.0, 1.0, 1.0]]))
    assert torch.equal(b.weights, torch.tensor([0.5, 2.0]))
