"""Multi-process harness coverage on CPU (gloo, world_size 2): the bench's
rendezvous, sharded stepping and max-over-ranks reduction must work without a
GPU (the GPU work itself is stubbed by --dry-run)."""
import json
import os
import subprocess
import sys

from conftest import ROOT


def test_bench_dry_run_world2():
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", os.path.join(ROOT, "bench.py"),
         "--dry-run", "--gpus", "2", "--steps", "2", "--warmup", "1"],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, timeout=180)
    assert out.returncode == 0, out.stdout.decode()[-2000:]
    lines = [l for l in out.stdout.decode().splitlines() if l.startswith("{")]
    assert len(lines) == 1, "expected exactly one JSON line from rank 0"
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["scaling"] == "weak"


def test_bench_dry_run_single():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--dry-run",
         "--steps", "1", "--warmup", "0"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, timeout=120)
    assert out.returncode == 0
    rec = json.loads([l for l in out.stdout.decode().splitlines() if l.startswith("{")][0])
    assert rec["dry_run"] is True
