"""Entry-script surface tests: all six reference-compatible CLIs parse and
import (reference scripts at repo root)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
ENTRIES = ["distributed.py", "distributed_mp.py", "distributed_apex.py",
           "distributed_gradient_accumulation.py", "dataparallel.py",
           "dataparallel_apex.py"]


@pytest.mark.parametrize("script", ENTRIES)
def test_entry_help(script):
    r = subprocess.run([sys.executable, os.path.join(ROOT, script), "--help"],
                       capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    assert "--batch_size" in r.stdout or "--batch-size" in r.stdout


def test_reference_flag_names_accepted():
    """The reference's exact flag spellings must parse (distributed.py:18-25)."""
    import argparse
    from mi355x_ddp.config import add_common_args, config_from_args
    p = argparse.ArgumentParser()
    p.add_argument("--local_rank", default=None, type=int)
    add_common_args(p)
    args = p.parse_args(["--seed", "1", "--batch_size", "128", "--epochs", "2",
                         "--lr", "0.05", "--ip", "127.0.0.1", "--port", "9999"])
    cfg = config_from_args(args)
    assert cfg.batch_size == 128 and cfg.epochs == 2 and cfg.lr == 0.05
    assert cfg.port == 9999 and cfg.seed == 1


def test_grad_accu_flag():
    import argparse
    from mi355x_ddp.config import add_common_args, config_from_args
    p = argparse.ArgumentParser()
    p.add_argument("--grad_accu_steps", default=1, type=int)
    add_common_args(p)
    cfg = config_from_args(p.parse_args(["--grad_accu_steps", "4"]))
    assert cfg.grad_accu_steps == 4
