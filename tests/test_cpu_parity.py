"""CPU-side parity: host pipeline + scalar oracle DP vs the reference's
committed golden outputs, and (when the reference tree is present) vs the
live reference binary on fresh synthetic inputs.

These tests pin the oracle per SURVEY.md §8(c): every golden fixture the
reference's own regression suite checks, plus synthetic read sets at several
shapes. The oracle is injected explicitly (ABPOA_AMD_TEST_ALIGNER_SO); the
product GPU path is covered by tests/test_gpu_parity.py.
"""
import os
import subprocess
import pytest

from conftest import GOLDEN, ROOT, REF_BIN, run_stdout

CASES = [
    ("seq.fa", [], "expected_seq_cons.txt"),
    ("seq.fa", ["-a1"], "expected_seq_msa.txt"),
    ("seq.fa", ["-r1"], "expected_seq_r1.txt"),
    ("seq.fa", ["-r2"], "expected_seq_r2.txt"),
    ("seq.fa", ["-m1"], "expected_seq_local.txt"),
    ("seq.fa", ["-m2"], "expected_seq_extend.txt"),
    ("seq.fa", ["-r3"], "expected_seq_gfa.txt"),
    ("seq.fa", ["-r5"], "expected_seq_fq.txt"),
    ("test.fa", [], "expected_test_cons.txt"),
    ("heter.fa", [], "expected_heter_cons.txt"),
]


@pytest.mark.parametrize("fa,opts,expected", CASES,
                         ids=["%s%s" % (c[0], "".join(c[1])) for c in CASES])
def test_golden(cputest_bin, oracle_env, fa, opts, expected):
    out = run_stdout([cputest_bin, os.path.join(GOLDEN, fa)] + opts, env=oracle_env)
    want = open(os.path.join(GOLDEN, expected), "rb").read()
    assert out == want


@pytest.mark.parametrize("seed,length,depth", [(1, 500, 20), (2, 1000, 30), (3, 2000, 10)])
def test_synthetic_vs_reference(cputest_bin, oracle_env, ref_bin, tmp_path, seed, length, depth):
    fa = tmp_path / "s.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", str(seed), "--len", str(length), "--depth", str(depth)],
                   check=True, stderr=subprocess.DEVNULL)
    for opts in ([], ["-r1"]):
        ref = run_stdout([ref_bin, str(fa)] + opts)
        got = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
        assert got == ref, "divergence at seed=%d len=%d opts=%r" % (seed, length, opts)


def test_product_fails_loudly_without_aligner(cputest_bin):
    """The CPU test binary must refuse to align without an injected oracle
    (mirrors the product's no-silent-fallback guarantee)."""
    p = subprocess.run([cputest_bin, os.path.join(GOLDEN, "seq.fa")],
                       stdout=subprocess.PIPE, stderr=subprocess.PIPE)
    assert p.returncode != 0
    assert b"no GPU aligner" in p.stderr or b"CPU-only TEST build" in p.stderr


MODE_CASES = [
    ["-O", "4", "-E", "2"],          # affine
    ["-O", "0", "-E", "2"],          # linear
    ["-m1"],                          # local (convex)
    ["-m2"],                          # extension (convex)
    ["-O", "4", "-E", "2", "-m1"],   # local affine
    ["-O", "0", "-E", "2", "-m2"],   # extension linear
]


@pytest.mark.parametrize("opts", MODE_CASES, ids=lambda o: "".join(o))
def test_modes_vs_reference(cputest_bin, oracle_env, ref_bin, tmp_path, opts):
    fa = tmp_path / "s.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", "7", "--len", "1200", "--depth", "25"],
                   check=True, stderr=subprocess.DEVNULL)
    ref = run_stdout([ref_bin, str(fa)] + opts)
    got = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
    assert got == ref


def test_aa_mode_vs_reference(cputest_bin, oracle_env, ref_bin, tmp_path):
    """Amino-acid alphabet (m=27) with BLOSUM62, global affine (configs[3])."""
    import random
    random.seed(5)
    aa = "ARNDCQEGHILKMFPSTWYV"
    refseq = "".join(random.choice(aa) for _ in range(400))
    reads = []
    for _ in range(10):
        s = []
        for ch in refseq:
            r = random.random()
            if r < 0.04: s.append(random.choice(aa))
            elif r < 0.07: pass
            elif r < 0.09: s.extend((ch, random.choice(aa)))
            else: s.append(ch)
        reads.append("".join(s))
    fa = tmp_path / "aa.fa"
    with open(fa, "w") as f:
        for i, r in enumerate(reads):
            f.write(">r%d\n%s\n" % (i, r))
    mtx = os.path.join(GOLDEN, "BLOSUM62.mtx")
    for opts in (["-c", "-t", mtx, "-O", "4", "-E", "2"], ["-c", "-t", mtx]):
        ref = run_stdout([ref_bin, str(fa)] + opts)
        got = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
        assert got == ref


def test_pog_dot_output(cputest_bin, oracle_env, tmp_path):
    """--out-pog DOT file is byte-identical to the reference's (the graphviz
    render step itself needs `dot`, absent here; both sides fail alike)."""
    import subprocess
    ref_png = tmp_path / "ref.png"
    amd_png = tmp_path / "amd.png"
    fa = os.path.join(GOLDEN, "seq.fa")
    subprocess.run([REF_BIN, fa, "-g", str(ref_png)], stdout=subprocess.DEVNULL,
                   stderr=subprocess.DEVNULL)
    subprocess.run([cputest_bin, fa, "-g", str(amd_png)], env=oracle_env,
                   stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    ref_dot = open(str(ref_png) + ".dot", "rb").read()
    amd_dot = open(str(amd_png) + ".dot", "rb").read()
    assert ref_dot == amd_dot


def test_list_input_mode(cputest_bin, oracle_env, tmp_path):
    """-l: input file is a list of FASTA paths, one MSA per file with
    batch_index-numbered consensus headers (abpoa.c:152-161)."""
    lst = tmp_path / "list.txt"
    fas = []
    for seed in (61, 62):
        fa = tmp_path / ("s%d.fa" % seed)
        subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                        "--seed", str(seed), "--len", "400", "--depth", "8"],
                       check=True, stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
        fas.append(str(fa))
    lst.write_text("".join(p + "\n" for p in fas))
    for opts in ([], ["-r1"]):
        ref = subprocess.run([REF_BIN, "-l", str(lst)] + opts, stdout=subprocess.PIPE,
                             stderr=subprocess.DEVNULL).stdout
        amd = run_stdout([cputest_bin, "-l", str(lst)] + opts, env=oracle_env)
        assert ref == amd


def test_quality_weights_fastq(cputest_bin, oracle_env, tmp_path):
    """-Q: FASTQ qualities become per-base edge weights; weighted consensus,
    MSA, FASTQ output and weighted multi-consensus must match."""
    import random
    rng = random.Random(31)
    ref_seq = "".join(rng.choice("ACGT") for _ in range(500))
    fq = tmp_path / "q.fq"
    with open(fq, "w") as f:
        for i in range(12):
            out = []
            for ch in ref_seq:
                r = rng.random()
                if r < 0.04: out.append(rng.choice("ACGT"))
                elif r < 0.07: pass
                else: out.append(ch)
            s = "".join(out)
            qual = "".join(chr(33 + rng.randrange(5, 40)) for _ in s)
            f.write("@r%d\n%s\n+\n%s\n" % (i, s, qual))
    for opts in (["-Q"], ["-Q", "-r1"], ["-Q", "-d2"], ["-Q", "-r4"], ["-Q", "-a1"]):
        ref = subprocess.run([REF_BIN, str(fq)] + opts, stdout=subprocess.PIPE,
                             stderr=subprocess.DEVNULL).stdout
        amd = run_stdout([cputest_bin, str(fq)] + opts, env=oracle_env)
        assert ref == amd, "mismatch opts=%r" % (opts,)
