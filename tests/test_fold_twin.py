"""Flat-array (device-layout) fold core vs the pointer-based graph fold.

abpoa_amd_foldtwin replays oracle CIGARs into both representations and
compares the complete structure (bases, adjacency order + weights, read-id
bitsets, aligned groups, n_read, qpos->node maps) after every read. This is
the round-2 groundwork proving the on-device fold's data layout reproduces
the sequential mutation order bit-exactly on CPU."""
import os
import subprocess

import pytest

from conftest import ROOT, GOLDEN, ORACLE_SO

TWIN = os.path.join(ROOT, "abpoa_amd", "csrc", "abpoa_amd_foldtwin")


def _run(fa, extra=()):
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    out = subprocess.run([TWIN, fa] + list(extra), env=env,
                         stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    assert out.returncode == 0, out.stderr.decode()[-500:]
    assert b"twin OK" in out.stdout


def test_twin_golden():
    _run(os.path.join(GOLDEN, "seq.fa"))
    _run(os.path.join(GOLDEN, "seq.fa"), ["-r1"])


def test_twin_synthetic(tmp_path):
    # depth 80 exercises multi-word per-edge read-id bitsets (rid_n = 2)
    for seed, length, depth in ((1, 300, 12), (2, 900, 20), (3, 1800, 30), (4, 4000, 40),
                                (5, 600, 80)):
        fa = str(tmp_path / ("s%d.fa" % seed))
        subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), fa,
                        "--seed", str(seed), "--len", str(length), "--depth", str(depth)],
                       check=True, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
        _run(fa)
        _run(fa, ["-r1"])


@pytest.mark.gpu
def test_device_fold_kernel(tmp_path):
    """abamd_fold_kernel on a real GPU: one launch per read mutates the
    device-resident flat graph; the full state (chains, pools, read-id
    bitsets, topo index, remain) must equal the host twin after every read."""
    bin_ = os.path.join(ROOT, "abpoa_amd", "csrc", "abpoa_amd_foldgpu")
    for fa in ("seq.fa", "fgt.fa"):
        out = subprocess.run([bin_, os.path.join(GOLDEN, fa)],
                             stdout=subprocess.PIPE, stderr=subprocess.PIPE)
        assert out.returncode == 0, out.stderr.decode()[-500:]
        assert b"device fold OK" in out.stdout
