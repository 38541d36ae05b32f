"""Batched driver parity (-m gpu): the lockstep multi-set batch path must
produce exactly the consensus the sequential CLI path produces."""
import os
import subprocess
import sys
import pytest

from conftest import ROOT, GPU_BIN, run_stdout

pytestmark = pytest.mark.gpu


def test_batch_matches_cli(tmp_path):
    sys.path.insert(0, ROOT)
    import abpoa_amd
    import numpy as np
    rng = np.random.default_rng(123)
    sys.path.insert(0, ROOT)
    import importlib
    bench = importlib.import_module("bench")
    sets = bench.gen_sets(rng, 4, depth=20, qlen=800)
    # batched GPU consensus
    cons = abpoa_amd.msa_batch_consensus(sets, n_threads=2)
    # sequential CLI on the same sets
    code2ch = "ACGT"
    for i, s in enumerate(sets):
        fa = tmp_path / ("s%d.fa" % i)
        with open(fa, "w") as f:
            for r_i, r in enumerate(s):
                f.write(">r%d\n%s\n" % (r_i, "".join(code2ch[b] for b in r)))
        out = run_stdout([GPU_BIN, str(fa)]).decode()
        seq = "".join(out.splitlines()[1:])
        assert cons[i] == seq, "batch/CLI consensus mismatch on set %d" % i
