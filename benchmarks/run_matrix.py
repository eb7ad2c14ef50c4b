#!/usr/bin/env python3
"""Benchmark matrix: sweep the framework's modes on one node and print a
table. Run on an MI355X box:

    python benchmarks/run_matrix.py --steps 30 --warmup 10

Each row launches bench.py as a subprocess (multi-rank rows via torchrun)
and reports images/sec + seconds/epoch, mirroring the reference README's
comparison table (reference README.md:59-77) across this framework's modes.
"""
import argparse
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

MATRIX = [
    # (label, extra bench.py args)
    ("bf16 NHWC native conv (default)", []),
    ("bf16 NCHW (MIOpen conv)", ["--no-channels-last"]),
    ("bf16 + hipGraph step", ["--hip-graph"]),
    ("fp32", ["--amp", "fp32"]),
    ("fp16 + loss scaler", ["--amp", "fp16"]),
    ("bf16 + grad_accu 4", ["--grad-accu-steps", "4"]),
    ("bf16, torch-DDP reducer A/B", ["--mode", "torchddp"]),
    ("bf16_o2 (pure-bf16 model, fp32 masters)", ["--amp", "bf16_o2"]),
    ("bf16 + bf16 gradient comm", ["--comm-bf16"]),
    ("ResNet50", ["--arch", "resnet50"]),
    ("ResNet34", ["--arch", "resnet34"]),
    ("ResNet50-ImageNet @224 b64",
     ["--arch", "resnet50_imagenet", "--image-size", "224",
      "--global-batch", "64"]),
]


def run_one(args, extra, gpus):
    cmd = [sys.executable]
    if gpus > 1:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={gpus}", "--master-addr", "127.0.0.1"]
    cmd += [os.path.join(ROOT, "bench.py"), "--gpus", str(gpus),
            "--steps", str(args.steps), "--warmup", str(args.warmup)] + extra
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=900)
    if r.returncode != 0:
        return None, r.stderr.strip().splitlines()[-1:] or ["failed"]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    return json.loads(line), None


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--gpus", type=int, default=1)
    args = p.parse_args()
    print(f"{'mode':>34} {'img/s':>10} {'ms/step':>9} {'s/epoch':>9}")
    for label, extra in MATRIX:
        rec, err = run_one(args, extra, args.gpus)
        if rec is None:
            print(f"{label:>34} {'FAILED':>10}  {err[0][:60]}")
            continue
        print(f"{label:>34} {rec['value']:>10.1f} {rec['ms_per_step']:>9.3f} "
              f"{rec['config'].get('seconds_per_epoch', 0):>9.3f}")


if __name__ == "__main__":
    main()
