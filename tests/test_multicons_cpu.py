"""Multi-consensus (-d >= 2) parity: k-medoids read clustering + per-cluster
consensus must be byte-identical to the live reference binary.

Covers HB and MF algorithms, RC-MSA and FASTQ outputs, ploidy 1-3 (ploidy 1
must collapse back to a single consensus), and a non-default gap mode.
Restates abpoa_output.c:650-1182; our CLI runs with the oracle injected so
the whole host pipeline (alignment results included) is compared."""
import os
import random
import subprocess

import pytest

from conftest import CPUTEST_BIN, ORACLE_SO, REF_BIN

pytestmark = pytest.mark.skipif(
    not os.path.exists(REF_BIN), reason="reference binary not built here")


def _gen(path, seed, n_hap, L=500, depth=24, snps=6):
    rng = random.Random(seed)
    base = "".join(rng.choice("ACGT") for _ in range(L))
    haps = [base]
    for _h in range(1, n_hap):
        hb = list(base)
        for p in rng.sample(range(10, L - 10), snps):
            hb[p] = rng.choice([c for c in "ACGT" if c != hb[p]])
        if rng.random() < 0.5:
            d = rng.randrange(20, L - 30)
            hb = hb[:d] + hb[d + 3:]
        haps.append("".join(hb))

    def noisy(h):
        out = []
        for ch in h:
            r = rng.random()
            if r < 0.02:
                out.append(rng.choice("ACGT"))
            elif r < 0.03:
                pass
            elif r < 0.04:
                out.extend((ch, rng.choice("ACGT")))
            else:
                out.append(ch)
        return "".join(out)

    with open(path, "w") as f:
        for i in range(depth):
            f.write(">r%d\n%s\n" % (i, noisy(haps[i % n_hap])))


MODES = [["-d2"], ["-d2", "-r1"], ["-d3"], ["-d3", "-r1"],
         ["-d2", "-a1"], ["-d2", "-r4"], ["-d2", "-O", "4", "-E", "2"]]


@pytest.mark.parametrize("seed", [1, 2, 7])
@pytest.mark.parametrize("n_hap", [1, 2, 3])
def test_multicons_matches_reference(tmp_path, seed, n_hap):
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    fa = str(tmp_path / "dip.fa")
    _gen(fa, seed, n_hap)
    for opts in MODES:
        ref = subprocess.run([REF_BIN, fa] + opts, stdout=subprocess.PIPE,
                             stderr=subprocess.DEVNULL).stdout
        amd = subprocess.run([CPUTEST_BIN, fa] + opts, env=env,
                             stdout=subprocess.PIPE,
                             stderr=subprocess.DEVNULL).stdout
        assert ref == amd, "mismatch seed=%d n_hap=%d opts=%r" % (seed, n_hap, opts)
