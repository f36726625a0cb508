"""Example-script smoke tests (CPU, tiny configs).

The reference ships its examples untested; here each example is exercised
end-to-end with tiny arguments so the scripts cannot bit-rot (reference
parity targets: examples/dlrm/main.py, examples/benchmarks/synthetic_models/
main.py, examples/criteo/main.py, examples/benchmarks/benchmark.py).
"""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, args, timeout=240):
    proc = subprocess.run(
        [sys.executable, os.path.join(ROOT, "examples", script)] + args,
        capture_output=True, text=True, timeout=timeout, cwd=ROOT)
    assert proc.returncode == 0, (
        f"{script} failed:\n{proc.stdout[-2000:]}\n{proc.stderr[-2000:]}")
    return proc.stdout


def test_dlrm_example_smoke():
    out = _run("dlrm_main.py", ["--batch-size", "128", "--num-batches", "3",
                                "--embedding-dim", "16",
                                "--table-size-cap", "1000"])
    assert "loss" in out


def test_synthetic_benchmark_smoke():
    out = _run("synthetic_benchmark.py",
               ["--model", "tiny", "--batch-size", "256",
                "--num-steps", "3", "--warmup", "1"])
    assert "ms/iteration" in out


def test_integer_lookup_example_smoke():
    out = _run("criteo_integer_lookup.py", ["--rows", "500", "--epochs", "1"])
    assert "vocab" in out


def test_lookup_benchmark_smoke():
    out = _run("lookup_benchmark.py",
               ["--vocab", "2000", "--batch", "64", "--width", "16",
                "--max-hotness", "5"])
    assert "fwd" in out


def test_bench_contract_help():
    """bench.py is the driver contract — it must always parse its args."""
    proc = subprocess.run([sys.executable, os.path.join(ROOT, "bench.py"),
                           "--help"], capture_output=True, text=True,
                          timeout=120, cwd=ROOT)
    assert proc.returncode == 0
    for flag in ("--gpus", "--steps", "--warmup"):
        assert flag in proc.stdout


def test_synthetic_benchmark_torchrun_world2():
    """The torchrun env-rendezvous path (exactly how the driver launches
    bench.py at N>1): WORLD_SIZE/RANK from env, init_process_group('gloo')."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(ROOT, "examples", "synthetic_benchmark.py"),
         "--model", "tiny", "--batch-size", "128",
         "--num-steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=280, cwd=ROOT)
    assert proc.returncode == 0, proc.stderr[-2000:]
    assert "world=2" in proc.stdout and "ms/iteration" in proc.stdout


def test_bench_torchrun_world2_json_contract():
    """The EXACT driver launch (torch.distributed.run, env rendezvous) at
    world 2 on CPU with capped tables: one JSON line from rank 0 with every
    contract field."""
    import json
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(ROOT, "bench.py"), "--gpus", "2",
         "--table-size-cap", "1000", "--batch-per-gpu", "64",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=280, cwd=ROOT)
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [l for l in proc.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, proc.stdout  # ONE JSON line, rank 0 only
    rec = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["n_gpus"] == 2 and rec["steps"] == 2
    assert rec["scaling"] == "weak"
    assert rec["config"]["global_batch"] == 128


def test_auc_matches_sklearn():
    import importlib.util
    import torch as t
    spec = importlib.util.spec_from_file_location(
        "dlrm_main", os.path.join(ROOT, "examples", "dlrm_main.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    from sklearn.metrics import roc_auc_score
    g = t.Generator().manual_seed(0)
    scores = t.rand(1000, generator=g)
    labels = (t.rand(1000, generator=g) < scores).float()  # correlated
    ours = mod.auc(scores, labels)
    ref = roc_auc_score(labels.numpy(), scores.numpy())
    assert abs(ours - ref) < 1e-6


def test_dlrm_example_eval_and_dump(tmp_path):
    """--eval (AUC over allgathered predictions) and --dump-embeddings
    (np.savez via get_weights) end-to-end."""
    import numpy as np
    dump = str(tmp_path / "emb.npz")
    out = _run("dlrm_main.py", ["--batch-size", "64", "--num-batches", "2",
                                "--embedding-dim", "8",
                                "--table-size-cap", "200",
                                "--eval", "--dump-embeddings", dump])
    assert "AUC:" in out and "dumped" in out
    tables = np.load(dump)
    assert len(tables.files) == 26


def test_dlrm_example_with_binary_dataset(tmp_path):
    """The example's real-dataset path: tiny split-binary Criteo layout on
    disk, 26 capped tables, 2 training steps."""
    import numpy as np
    n, nnum, cap = 64, 13, 200
    sizes = [min(s, cap) for s in
             [39884407, 39043, 17289, 7420, 20263, 3, 7120, 1543, 63,
              38532952, 2953546, 403346, 10, 2208, 11938, 155, 4, 976, 14,
              39979772, 25641295, 39664985, 585935, 12972, 108, 36]]
    d = tmp_path / "train"
    d.mkdir()
    rng = np.random.RandomState(0)
    (d / "label.bin").write_bytes(rng.randint(0, 2, n).astype(np.int8).tobytes())
    (d / "numerical.bin").write_bytes(
        rng.rand(n, nnum).astype(np.float16).tobytes())
    for i, s in enumerate(sizes):
        dt = (np.int8 if s < np.iinfo(np.int8).max else
              np.int16 if s < np.iinfo(np.int16).max else np.int32)
        (d / f"cat_{i}.bin").write_bytes(
            rng.randint(0, s, n).astype(dt).tobytes())
    out = _run("dlrm_main.py", ["--batch-size", "32", "--num-batches", "2",
                                "--embedding-dim", "8",
                                "--table-size-cap", str(cap),
                                "--dataset-path", str(tmp_path)])
    assert "loss" in out


def test_serving_example_smoke(tmp_path):
    """Serving demo runs on CPU with capped tables (latency path + shapes)."""
    import subprocess
    import sys
    import pathlib
    ex = pathlib.Path(__file__).parent.parent / "examples" / "serving.py"
    r = subprocess.run(
        [sys.executable, str(ex), "--table-size-cap", "500",
         "--batch-size", "32", "--iters", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stderr
    assert "p50" in r.stdout
