"""Drop-in proof: the reference's own host code (compiled unmodified from
/root/reference by oracle/Makefile) linked against OUR seam library
(libabpoa_amd_align.so exporting simd_abpoa_align_sequence_to_graph/subgraph)
must reproduce the pure reference binary byte-exactly.

CPU tests inject the oracle through the seam's test hook; the -m gpu test runs
the hybrid's real GPU path against the committed goldens."""
import os
import subprocess
import pytest

from conftest import ROOT, GOLDEN, run_stdout

HYBRID = os.path.join(ROOT, "oracle", "_ref", "abpoa_hybrid")
REF = os.path.join(ROOT, "oracle", "_ref", "abpoa")
ORACLE_SO = os.path.join(ROOT, "oracle", "liboracle.so")


@pytest.fixture(scope="module")
def hybrid_bin():
    if not os.path.exists(HYBRID):
        if os.path.isdir("/root/reference"):
            subprocess.run(["make", "align-shim"], cwd=os.path.join(ROOT, "abpoa_amd", "csrc"),
                           check=True, stdout=subprocess.DEVNULL)
            subprocess.run(["make", "hybrid",
                            "ALIGNER_LIB=../abpoa_amd/csrc/libabpoa_amd_align.so"],
                           cwd=os.path.join(ROOT, "oracle"), check=True,
                           stdout=subprocess.DEVNULL)
    if not os.path.exists(HYBRID):
        pytest.skip("hybrid binary unavailable")
    return HYBRID


@pytest.mark.parametrize("opts", [[], ["-r1"], ["-O", "4", "-E", "2"], ["-m1"]],
                         ids=["cons", "msa", "affine", "local"])
def test_hybrid_oracle_vs_reference(hybrid_bin, ref_bin, opts):
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    fa = os.path.join(GOLDEN, "seq.fa")
    got = run_stdout([hybrid_bin, fa] + opts, env=env)
    want = run_stdout([ref_bin, fa] + opts)
    assert got == want


@pytest.mark.gpu
def test_hybrid_gpu_golden(hybrid_bin):
    """The hybrid's GPU path (no oracle injected) vs the committed golden."""
    fa = os.path.join(GOLDEN, "seq.fa")
    out = run_stdout([hybrid_bin, fa])
    want = open(os.path.join(GOLDEN, "expected_seq_cons.txt"), "rb").read()
    assert out == want
