"""End-to-end CLI tests for the two applications (the reference's
distributed_word_embedding and LogisticRegression binaries,
SURVEY.md §2.11): corpus/config in, trained artifacts out."""

import random
import subprocess
import sys
import os

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
REPO = ROOT


def test_wordembedding_cli(tmp_path):
    corpus = tmp_path / "corpus.txt"
    out = tmp_path / "emb.txt"
    rng = random.Random(3)
    words = [f"w{i}" for i in range(100)]
    with open(corpus, "w") as f:
        for _ in range(400):
            f.write(" ".join(rng.choices(words, k=10)) + "\n")
    r = subprocess.run(
        [sys.executable, "-m", "multiverso_amd.apps.wordembedding.main",
         "-train_file", str(corpus), "-output", str(out), "-size", "16",
         "-epoch", "1", "-min_count", "1", "-negative", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    lines = open(out).read().strip().split("\n")
    vocab, dim = map(int, lines[0].split())
    assert vocab == 100 and dim == 16
    assert len(lines) == vocab + 1
    first = lines[1].split()
    assert len(first) == dim + 1 and first[0].startswith("w")


def test_wordembedding_cli_binary_output(tmp_path):
    corpus = tmp_path / "c.txt"
    out = tmp_path / "emb.bin"
    with open(corpus, "w") as f:
        for _ in range(50):
            f.write("a b c d e f g h\n")
    r = subprocess.run(
        [sys.executable, "-m", "multiverso_amd.apps.wordembedding.main",
         "-train_file", str(corpus), "-output", str(out), "-size", "8",
         "-epoch", "1", "-min_count", "1", "-negative", "2", "-binary", "1"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    blob = open(out, "rb").read()
    header = blob.split(b"\n", 1)[0]
    vocab, dim = map(int, header.split())
    assert vocab == 8 and dim == 8


def test_logreg_cli(tmp_path):
    train = tmp_path / "train.txt"
    model = tmp_path / "model.bin"
    rng = random.Random(5)
    with open(train, "w") as f:
        for _ in range(400):
            keys = rng.sample(range(500), 10)
            label = 1 if sum(keys) > 2500 else 0
            f.write(f"{label} " + " ".join(f"{k}:1" for k in sorted(keys))
                    + "\n")
    cfg = tmp_path / "t.config"
    cfg.write_text(
        "input_size=500\noutput_size=1\nobjective_type=sigmoid\n"
        "updater_type=sgd\nlearning_rate=0.1\nminibatch_size=32\n"
        "train_epoch=2\nreader_type=default\nuse_ps=true\n"
        f"sync_frequency=4\ntrain_file={train}\ntest_file={train}\n"
        f"output_model_file={model}\n")
    r = subprocess.run(
        [sys.executable, "-m", "multiverso_amd.apps.logreg.main", str(cfg)],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    assert model.exists() and model.stat().st_size == 500 * 4
    # training must beat chance on this separable-ish problem
    import re
    m = re.search(r"\((0\.\d+)\)", r.stdout + r.stderr)
    assert m and float(m.group(1)) > 0.55, (r.stdout + r.stderr)[-400:]


def test_logreg_cli_softmax(tmp_path):
    """Multiclass softmax through the CLI (separable 5-class synthetic)."""
    import re
    train = tmp_path / "sm.txt"
    rng = random.Random(9)
    with open(train, "w") as f:
        for i in range(400):
            label = i % 5
            keys = sorted(rng.sample(range(40 * label, 40 * label + 40), 8))
            f.write(f"{label} " + " ".join(f"{k}:1" for k in keys) + "\n")
    cfg = tmp_path / "sm.config"
    model = tmp_path / "sm.bin"
    cfg.write_text(
        "input_size=200\noutput_size=5\nobjective_type=softmax\n"
        "updater_type=sgd\nlearning_rate=0.1\nminibatch_size=32\n"
        "train_epoch=3\nreader_type=default\nuse_ps=true\n"
        f"sync_frequency=2\ntrain_file={train}\ntest_file={train}\n"
        f"output_model_file={model}\n")
    r = subprocess.run(
        [sys.executable, "-m", "multiverso_amd.apps.logreg.main", str(cfg)],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    assert model.stat().st_size == 200 * 5 * 4
    m = re.search(r"\((0\.\d+|1\.0000)\)", r.stdout + r.stderr)
    assert m and float(m.group(1)) > 0.9, (r.stdout + r.stderr)[-300:]


def test_logreg_cli_ftrl(tmp_path):
    """FTRL objective+updater through the CLI on a separable problem."""
    import re
    train = tmp_path / "ftrl.txt"
    rng = random.Random(4)
    with open(train, "w") as f:
        for i in range(400):
            pos = i % 2
            lo = 0 if pos else 100
            keys = sorted(rng.sample(range(lo, lo + 100), 8))
            f.write(f"{pos} " + " ".join(f"{k}:1" for k in keys) + "\n")
    cfg = tmp_path / "f.config"
    model = tmp_path / "f.bin"
    cfg.write_text(
        "input_size=200\noutput_size=1\nobjective_type=ftrl\n"
        "updater_type=ftrl\nalpha=0.1\nbeta=1\nlambda1=0.01\nlambda2=0.01\n"
        "minibatch_size=32\ntrain_epoch=3\nreader_type=default\nuse_ps=true\n"
        f"sync_frequency=2\ntrain_file={train}\ntest_file={train}\n"
        f"output_model_file={model}\n")
    r = subprocess.run(
        [sys.executable, "-m", "multiverso_amd.apps.logreg.main", str(cfg)],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-800:]
    import re as _re
    m = _re.search(r"\((0\.\d+|1\.0000)\)", r.stdout + r.stderr)
    assert m and float(m.group(1)) > 0.9, (r.stdout + r.stderr)[-300:]


def test_embedding_server(tmp_path):
    """Serving example: word2vec file -> PS MatrixTable -> HTTP lookups
    (fastapi TestClient, no network)."""
    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "serve_embeddings", os.path.join(ROOT, "examples",
                                         "serve_embeddings.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)

    # tiny text-format embedding file
    p = tmp_path / "emb.txt"
    with open(p, "w") as f:
        f.write("4 3\n")
        f.write("king 1.0 0.0 0.0\n")
        f.write("queen 0.9 0.1 0.0\n")
        f.write("apple 0.0 1.0 0.0\n")
        f.write("pear 0.0 0.9 0.1\n")
    words, vecs = mod.load_word2vec(str(p))
    assert words == ["king", "queen", "apple", "pear"]
    assert vecs.shape == (4, 3)

    import multiverso_amd as mv
    mv.init()
    table = mv.MatrixTable(4, 3)
    table.add(vecs)
    table.flush()
    from fastapi.testclient import TestClient
    client = TestClient(mod.build_app(table, words, 3))
    assert client.get("/healthz").json()["vocab"] == 4
    v = client.get("/vec/king").json()
    assert v["vector"] == [1.0, 0.0, 0.0]
    nn = client.get("/nn/king?k=1").json()
    assert nn["neighbors"][0]["word"] == "queen"
    nn2 = client.get("/nn/apple?k=1").json()
    assert nn2["neighbors"][0]["word"] == "pear"
    assert client.get("/vec/zzz").status_code == 404
    mv.shutdown()


@pytest.mark.parametrize("name", ["array", "kv", "net", "matrix",
                                  "allreduce"])
def test_selftest_dispatcher_ws2(name, tmp_path):
    """The reference's `mpirun -np N ./multiverso.test <name>` CLI tests
    (Test/main.cpp:12-24) as a torchrun dispatcher."""
    import subprocess, sys, os
    from conftest import free_port
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + ":" + env.get("PYTHONPATH", "")
    env["HIP_VISIBLE_DEVICES"] = env["CUDA_VISIBLE_DEVICES"] = ""
    env["MV_BACKEND"] = "gloo"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()),
         "-m", "multiverso_amd.selftest", name],
        capture_output=True, text=True, timeout=300, env=env)
    assert r.returncode == 0, (name, r.stdout[-500:], r.stderr[-1500:])
    assert "PASS" in r.stdout


@pytest.mark.parametrize("appargs", [
    ["--rows", "4000", "--cols", "8", "--steps", "3", "--warmup", "1",
     "--phase-steps", "2"],
    ["--app", "logreg", "--steps", "2", "--warmup", "1"],
    ["--app", "wordembedding", "--steps", "2", "--warmup", "1",
     "--block-words", "2000", "--vocab", "200", "--dim", "16"],
    ["--app", "sweep", "--gb", "0.001", "--steps", "2", "--warmup", "1"],
])
def test_bench_rehearsal_ws2(appargs, tmp_path):
    """Rehearse the EXACT driver invocation shape (torchrun --nnodes=1
    --nproc-per-node N bench.py ...) at world_size 2 on gloo/CPU: every
    N>1-only branch of every bench app must run and rank 0 must print a
    parseable JSON contract line.  The driver's SCALE run gets no second
    chance — this is its CI net."""
    import json as _json
    import subprocess, sys, os
    from conftest import free_port
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + ":" + env.get("PYTHONPATH", "")
    env["HIP_VISIBLE_DEVICES"] = env["CUDA_VISIBLE_DEVICES"] = ""
    env["MV_BACKEND"] = "gloo"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()),
         os.path.join(REPO, "bench.py"), "--gpus", "2"] + appargs,
        capture_output=True, text=True, timeout=600, env=env)
    assert r.returncode == 0, (r.stdout[-800:], r.stderr[-1500:])
    line = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert line, r.stdout[-800:]
    rec = _json.loads(line[-1])
    assert rec["n_gpus"] == 2 and "value" in rec and "metric" in rec
    if "--app" not in appargs:
        assert "phases" in rec and rec["phases"], rec
