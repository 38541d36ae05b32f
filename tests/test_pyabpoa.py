"""pyabpoa-compatible binding (abpoa_amd.pyabpoa): same API as the
reference's python package (python/pyabpoa.pyx); results must equal the CLI
product path on identical input."""
import os
import random
import subprocess
import sys

import pytest

from conftest import ROOT, GPU_BIN, run_stdout

sys.path.insert(0, ROOT)


def test_api_surface_cpu():
    """Constructor, parameter mirror validation and error paths run without
    a GPU (no alignment is performed)."""
    import abpoa_amd.pyabpoa as pa
    a = pa.msa_aligner(aln_mode='g', match=2, mismatch=4)
    assert bool(a)
    with pytest.raises(Exception):
        pa.msa_aligner(aln_mode='x')
    with pytest.raises(Exception):
        pa.msa_aligner(cons_algrm='zz')
    r = pa.msa_result(2, 1, [2], [[0, 1]], [3], ["ACG"], [[2, 2, 2]], ["II"], 0, [])
    assert r.n_cons == 1 and r.cons_seq == ["ACG"]


@pytest.mark.gpu
def test_pyabpoa_matches_cli(tmp_path):
    import abpoa_amd.pyabpoa as pa
    rng = random.Random(21)
    ref = "".join(rng.choice("ACGT") for _ in range(600))
    seqs = []
    for _ in range(12):
        out = []
        for ch in ref:
            r = rng.random()
            if r < 0.03: out.append(rng.choice("ACGT"))
            elif r < 0.05: pass
            elif r < 0.07: out.extend((ch, rng.choice("ACGT")))
            else: out.append(ch)
        seqs.append("".join(out))
    fa = tmp_path / "s.fa"
    with open(fa, "w") as f:
        for i, s in enumerate(seqs):
            f.write(">r%d\n%s\n" % (i, s))

    a = pa.msa_aligner()
    res = a.msa(seqs, out_cons=True, out_msa=True)
    cli_cons = "".join(run_stdout([GPU_BIN, str(fa)]).decode().splitlines()[1:])
    assert res.n_cons == 1
    assert res.cons_seq[0] == cli_cons
    cli_msa = run_stdout([GPU_BIN, str(fa), "-r1"]).decode().splitlines()
    cli_rows = [cli_msa[i] for i in range(1, len(cli_msa), 2)]
    # CLI -r1 prints the read rows only; the binding adds n_cons consensus rows
    assert res.msa_seq[:len(seqs)] == cli_rows
    assert len(res.msa_seq) == len(seqs) + res.n_cons
    assert res.msa_seq[len(seqs)].replace("-", "") == cli_cons

    # multi-consensus through the binding
    a2 = pa.msa_aligner()
    res2 = a2.msa(seqs, out_cons=True, out_msa=False, max_n_cons=2)
    assert 1 <= res2.n_cons <= 2
    # incremental add + output
    a3 = pa.msa_aligner()
    a3.msa_align(seqs[:6], out_cons=True, out_msa=False)
    a3.msa_add(seqs[6:])
    res3 = a3.msa_output()
    assert res3.cons_seq[0] == cli_cons


def test_pyabpoa_msa_cpu_oracle(tmp_path):
    """Full msa() through the binding against the CPU-test library with the
    oracle injected (subprocess: the library choice is per-process)."""
    script = r'''
import sys, random
sys.path.insert(0, %r)
import abpoa_amd.pyabpoa as pa
rng = random.Random(5)
ref = "".join(rng.choice("ACGT") for _ in range(300))
seqs = []
for _ in range(8):
    out = []
    for ch in ref:
        r = rng.random()
        if r < 0.04: out.append(rng.choice("ACGT"))
        elif r < 0.07: pass
        else: out.append(ch)
    seqs.append("".join(out))
a = pa.msa_aligner()
res = a.msa(seqs, out_cons=True, out_msa=True)
assert res.n_cons == 1 and res.cons_len[0] == len(res.cons_seq[0]) > 0
assert len(res.msa_seq) == len(seqs) + 1
assert all(len(s) == res.msa_len for s in res.msa_seq)
assert res.msa_seq[-1].replace("-", "") == res.cons_seq[0]
assert len(res.cons_qv[0]) == res.cons_len[0]
a2 = pa.msa_aligner(cons_algrm="MF")
res2 = a2.msa(seqs, out_cons=True, out_msa=False)
assert res2.n_cons == 1 and res2.cons_seq[0]
print("OK")
'''.replace("%r", repr(ROOT))
    env = dict(os.environ)
    env["ABPOA_AMD_LIB"] = os.path.join(ROOT, "abpoa_amd", "csrc", "libabpoa_amd_cputest.so")
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = os.path.join(ROOT, "oracle", "liboracle.so")
    out = subprocess.run([sys.executable, "-c", script], env=env,
                         stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    assert out.returncode == 0 and b"OK" in out.stdout, out.stderr.decode()[-2000:]
