"""Minimizer seeding / progressive POA (-S / -p) parity vs the live reference
(abpoa_seed.c, abpoa_align.c:209-310): sketch, guide tree, anchor chaining
and anchor-windowed alignment must be byte-identical, including klib
radix-sort tie order, reverse-complement handling under -s, and aa k-mers."""
import os
import random
import subprocess

import pytest

from conftest import ROOT, CPUTEST_BIN, ORACLE_SO, REF_BIN

pytestmark = pytest.mark.skipif(
    not os.path.exists(REF_BIN), reason="reference binary not built here")

COMP = {"A": "T", "C": "G", "G": "C", "T": "A"}


def _gen(path, seed, L, depth, rc_frac=0.0, aa=False):
    rng = random.Random(seed)
    alpha = "ARNDCQEGHILKMFPSTWYV" if aa else "ACGT"
    ref = "".join(rng.choice(alpha) for _ in range(L))
    with open(path, "w") as f:
        for i in range(depth):
            out = []
            for ch in ref:
                r = rng.random()
                if r < 0.03:
                    out.append(rng.choice(alpha))
                elif r < 0.05:
                    pass
                elif r < 0.07 and not aa:
                    out.extend((ch, rng.choice(alpha)))
                else:
                    out.append(ch)
            s = "".join(out)
            if not aa and i > 0 and rng.random() < rc_frac:
                s = "".join(COMP[c] for c in reversed(s))
            f.write(">r%d\n%s\n" % (i, s))


def _cmp(fa, opts):
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    ref = subprocess.run([REF_BIN, fa] + opts, stdout=subprocess.PIPE,
                         stderr=subprocess.DEVNULL).stdout
    amd = subprocess.run([CPUTEST_BIN, fa] + opts, env=env,
                         stdout=subprocess.PIPE, stderr=subprocess.DEVNULL).stdout
    assert ref == amd, "seeding divergence opts=%r" % (opts,)


@pytest.mark.parametrize("opts", [["-S"], ["-p"], ["-S", "-p"], ["-S", "-r1"],
                                  ["-S", "-n", "100"], ["-S", "-O", "4", "-E", "2"],
                                  ["-S", "-d2"]],
                         ids=["seed", "prog", "seed-prog", "seed-msa",
                              "seed-n100", "seed-affine", "seed-multicons"])
def test_seeding_modes(tmp_path, opts):
    fa = str(tmp_path / "s.fa")
    _gen(fa, 4, 5000, 8)
    _cmp(fa, opts)


def test_seeding_amb_strand_rc(tmp_path):
    fa = str(tmp_path / "rc.fa")
    _gen(fa, 11, 4000, 10, rc_frac=0.5)
    for opts in (["-S", "-s"], ["-S", "-s", "-r1"]):
        _cmp(fa, opts)


def test_seeding_aa(tmp_path):
    fa = str(tmp_path / "aa.fa")
    _gen(fa, 13, 1500, 10, aa=True)
    mtx = os.path.join(ROOT, "tests", "golden", "BLOSUM62.mtx")
    for opts in (["-c", "-t", mtx, "-S"], ["-c", "-t", mtx, "-S", "-n", "100"]):
        _cmp(fa, opts)
