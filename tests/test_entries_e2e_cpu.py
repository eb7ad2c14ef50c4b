"""End-to-end entry-script runs on CPU (gloo): each CLI trains 2 capped steps
of one epoch on synthetic data, evaluates 2 batches, and writes a final
checkpoint — the L5->L1 vertical slice of SURVEY.md §1 without a GPU."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

COMMON = ["--epochs", "1", "--batch_size", "64", "--num_workers", "0",
          "--synthetic", "--max_train_steps", "2", "--max_eval_steps", "2",
          "--log_interval", "1", "--save_epoch", "0", "--seed", "3"]


def _run(tmp_path, script, extra, timeout=420):
    env = dict(os.environ, PYTHONPATH=ROOT, MASTER_ADDR="127.0.0.1")
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, script)] + COMMON + extra,
        cwd=tmp_path, env=env, capture_output=True, text=True, timeout=timeout)
    assert r.returncode == 0, f"{script} failed:\n{r.stdout[-2000:]}\n{r.stderr[-2000:]}"
    return r


def test_distributed_single_process(tmp_path, free_port):
    r = _run(tmp_path, "distributed.py", ["--port", str(free_port)])
    assert "Epoch: [0][0/" in r.stdout
    assert "Acc@1" in r.stdout
    assert (tmp_path / "ckpts").exists()  # final checkpoint
    metrics = tmp_path / "runs" / "metrics.jsonl"
    assert metrics.exists()
    kinds = {json.loads(l)["kind"] for l in metrics.read_text().splitlines()}
    assert {"train", "epoch", "val"} <= kinds


def test_distributed_mp_two_ranks(tmp_path, free_port):
    r = _run(tmp_path, "distributed_mp.py",
             ["--nprocs", "2", "--port", str(free_port)])
    assert "Acc@1" in r.stdout


def test_grad_accumulation_entry(tmp_path, free_port):
    r = _run(tmp_path, "distributed_gradient_accumulation.py",
             ["--grad_accu_steps", "2", "--port", str(free_port)])
    assert "Acc@1" in r.stdout


def test_dataparallel_entry(tmp_path):
    r = _run(tmp_path, "dataparallel.py", [])
    assert "Acc@1" in r.stdout


def test_apex_equivalent_entry_fp16_scaler(tmp_path, free_port):
    """distributed_apex.py on CPU: fp32 fallback for autocast-free CPU mode is
    exercised via --amp fp32; the scaler path is covered by unit tests."""
    r = _run(tmp_path, "distributed_apex.py",
             ["--amp", "fp32", "--port", str(free_port)])
    assert "Acc@1" in r.stdout


def test_bench_multirank_contract(tmp_path, free_port):
    """The driver's exact multi-rank launch: torchrun x2 on CPU (gloo),
    one JSON line from rank 0 with the whole-job aggregate."""
    env = dict(os.environ, PYTHONPATH=ROOT)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port),
         os.path.join(ROOT, "bench.py"), "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--global-batch", "32"],
        cwd=tmp_path, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2 and rec["config"]["parallelism"] == "dp2"
    assert rec["config"]["sync_bn"] is True
    assert rec["value"] > 0 and rec["unit"] == "images/sec"


def test_resume_from_checkpoint(tmp_path, free_port):
    """Train 1 epoch, then resume into a 2-epoch run from the final
    checkpoint (reference only documented the rank-0 save pattern;
    resume never existed there)."""
    _run(tmp_path, "distributed.py", ["--port", str(free_port)])
    ckpts = sorted((tmp_path / "ckpts").glob("*.pt"))
    assert ckpts, "no checkpoint written"
    final = [c for c in ckpts if "final" in c.name][0]
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "distributed.py")] + COMMON +
        ["--epochs", "2", "--resume", str(final), "--port", str(free_port + 1)],
        cwd=tmp_path, env=dict(os.environ, PYTHONPATH=ROOT),
        capture_output=True, text=True, timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "resumed from" in r.stdout
    assert "Epoch: [1][0/" in r.stdout  # continued at epoch 1, not 0


def test_apex_entry_o2_mode(tmp_path, free_port):
    """distributed_apex.py --amp bf16_o2: the apex-O2-equivalent path runs
    end to end (bf16 model, fp32 masters, train + eval + checkpoint)."""
    r = _run(tmp_path, "distributed_apex.py",
             ["--amp", "bf16_o2", "--port", str(free_port)])
    assert "Epoch: [0][0/" in r.stdout
    assert "Acc@1" in r.stdout
    assert (tmp_path / "ckpts").exists()
