"""CLI contract tests: main.py / ensemble.py run end-to-end on CPU with
the reference flag surface (tiny synthetic configs)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_cli(args, timeout=300):
    return subprocess.run([sys.executable] + args, cwd=REPO, timeout=timeout,
                          capture_output=True, text=True)


@pytest.mark.timeout(300)
def test_main_cli_cpu_tiny(tmp_path):
    ck = str(tmp_path / "m.pt")
    jl = str(tmp_path / "m.jsonl")
    r = run_cli(["main.py", "--device", "cpu", "--data",
                 "synthetic:vocab=50,tokens=6000", "--hidden_size", "32",
                 "--layer_num", "2", "--seq_length", "10", "--batch_size",
                 "4", "--total_epochs", "2", "--dropout", "0.2", "--seed",
                 "1", "--save", ck, "--jsonl", jl, "--lstm_type", "custom"])
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Model will be training on the CPU." in r.stdout
    assert "Validation set perplexity" in r.stdout
    assert "Test set perplexity" in r.stdout
    assert "Training is over." in r.stdout
    assert os.path.exists(ck)
    events = [json.loads(l) for l in open(jl)]
    assert any(e["event"] == "final" for e in events)
    # resume from the checkpoint (epoch 2 of 3)
    r2 = run_cli(["main.py", "--device", "cpu", "--data",
                  "synthetic:vocab=50,tokens=6000", "--hidden_size", "32",
                  "--layer_num", "2", "--seq_length", "10", "--batch_size",
                  "4", "--total_epochs", "3", "--seed", "1",
                  "--resume", ck])
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert "Resumed from" in r2.stdout


@pytest.mark.timeout(300)
def test_ensemble_cli_cpu_tiny(tmp_path):
    r = run_cli(["ensemble.py", "--device", "cpu", "--data",
                 "synthetic:vocab=40,tokens=4000", "--hidden_size", "24",
                 "--layer_num", "1", "--seq_length", "8", "--batch_size",
                 "4", "--total_epochs", "1", "--ensemble_num", "2",
                 "--seed", "2", "--save_dir", str(tmp_path)])
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Test set perplexity of 2 averaged models" in r.stdout
    assert os.path.exists(str(tmp_path / "model_2.pt"))


@pytest.mark.timeout(120)
def test_main_cli_missing_ptb_message(tmp_path):
    r = run_cli(["main.py", "--device", "cpu", "--data_dir", str(tmp_path)])
    assert r.returncode != 0
    assert "synthetic" in (r.stderr + r.stdout)


@pytest.mark.timeout(120)
def test_fp32_with_hip_engine_rejected():
    """--dtype fp32 --engine hip must fail loudly (the HIP path computes
    in bf16 with fp32 masters); round-1 bug: it silently ran bf16."""
    for script in ("main.py", "ensemble.py"):
        r = run_cli([script, "--device", "cpu", "--engine", "hip",
                     "--dtype", "fp32", "--data", "synthetic:vocab=20,tokens=500"])
        assert r.returncode != 0, script
        assert "fp32 is not supported" in (r.stderr + r.stdout), script


@pytest.mark.timeout(120)
def test_bench_gpus_flag_is_authoritative():
    """bench.py --gpus N run single-process (world=1) must fail loudly
    instead of silently benchmarking dp1 (protects the driver's scaling
    runs)."""
    r = run_cli(["bench.py", "--gpus", "2", "--steps", "1", "--warmup", "0"])
    assert r.returncode != 0
    assert "world size" in (r.stderr + r.stdout)


@pytest.mark.timeout(300)
def test_bench_torchrun_dp2_contract():
    """The EXACT launch the driver uses for the scaling bench, world=2
    (CPU/gloo here; RCCL on GPU nodes): one JSON line from rank 0 with
    the whole-job aggregate and the dp2 config."""
    import json as _json

    r = run_cli(["-m", "torch.distributed.run", "--nnodes=1",
                 "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
                 "--master-port", "29575", "bench.py", "--gpus", "2",
                 "--steps", "2", "--warmup", "1", "--hidden_size", "64",
                 "--vocab", "200", "--batch_size", "4", "--seq_length", "8"])
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout[-1500:]  # rank 0 only
    d = _json.loads(lines[0])
    assert d["n_gpus"] == 2 and d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 8  # world * per-rank B
    assert d["value"] > 0 and d["scaling"] == "weak"


def test_ensemble_jsonl_per_model_suffix(monkeypatch):
    """Multi-rank ensemble mode must not share one JSONL file across
    concurrently-training ranks (advisor finding): the path is suffixed
    per model; single-process mode keeps the user's path."""
    import importlib
    import sys as _sys
    _sys.path.insert(0, REPO)
    ens = importlib.import_module("ensemble")
    from zaremba_amd.parallel import dist as zdist

    class A:
        jsonl = "/tmp/run.jsonl"

    monkeypatch.setattr(zdist, "world_size", lambda: 1)
    assert ens.jsonl_for_model(A, 3) == "/tmp/run.jsonl"
    monkeypatch.setattr(zdist, "world_size", lambda: 4)
    assert ens.jsonl_for_model(A, 3) == "/tmp/run.model3.jsonl"
    A.jsonl = None
    assert ens.jsonl_for_model(A, 1) is None
