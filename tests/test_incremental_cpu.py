"""Incremental MSA (-i) parity: restoring a graph from a previous run's MSA
FASTA or GFA output, then aligning new reads into it, must match the live
reference byte-for-byte (abpoa_seq.c:340-673)."""
import os
import subprocess

import pytest

from conftest import ROOT, CPUTEST_BIN, ORACLE_SO, REF_BIN, run_stdout

pytestmark = pytest.mark.skipif(
    not os.path.exists(REF_BIN), reason="reference binary not built here")


def _synth(tmp_path, name, seed, length, depth, prefix):
    fa = str(tmp_path / name)
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), fa,
                    "--seed", str(seed), "--len", str(length), "--depth", str(depth)],
                   check=True, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    txt = open(fa).read().replace(">r", ">" + prefix)
    open(fa, "w").write(txt)
    return fa


@pytest.mark.parametrize("restore_fmt", ["-r1", "-r3"], ids=["msa-fasta", "gfa"])
@pytest.mark.parametrize("opts", [[], ["-r1"], ["-d2"]],
                         ids=["cons", "rc-msa", "multicons"])
def test_incremental_restore(tmp_path, restore_fmt, opts):
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    first = _synth(tmp_path, "a.fa", 51, 400, 10, "r")
    new = _synth(tmp_path, "b.fa", 52, 400, 8, "s")
    restore = str(tmp_path / "restore.out")
    with open(restore, "wb") as f:
        f.write(subprocess.run([REF_BIN, first, restore_fmt],
                               stdout=subprocess.PIPE,
                               stderr=subprocess.DEVNULL).stdout)
    ref = subprocess.run([REF_BIN, "-i", restore, new] + opts,
                         stdout=subprocess.PIPE, stderr=subprocess.DEVNULL).stdout
    amd = run_stdout([CPUTEST_BIN, "-i", restore, new] + opts, env=env)
    assert ref == amd


def test_restore_rejects_unequal_msa_rows(tmp_path):
    """A longer-than-first MSA row in a restore file must fail cleanly (the
    reference reads out of bounds here; we refuse with an error instead)."""
    bad = tmp_path / "bad.fa"
    bad.write_text(">a\nACGT\n>b\nAC-T-EXTRA-LONGER-ROW\n")
    new = tmp_path / "new.fa"
    new.write_text(">r0\nACGTACGT\n")
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    r = subprocess.run([CPUTEST_BIN, "-i", str(bad), str(new)], env=env,
                       stdout=subprocess.DEVNULL, stderr=subprocess.PIPE)
    assert r.returncode == 1 and b"unequal length" in r.stderr
