"""GPU parity: the HIP/CDNA4 hot path must reproduce the reference outputs
byte-exactly. Runs only on a machine with an AMD GPU (-m gpu).

The comparison anchors are committed golden fixtures (generated from the
unmodified reference binary in the dev container) and the CPU oracle run live
next to the GPU on identical synthetic inputs — /root/reference itself is not
needed at run time."""
import os
import subprocess
import pytest

from conftest import GOLDEN, ROOT, GPU_BIN, run_stdout

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu_bin():
    assert os.path.exists(GPU_BIN), "abpoa_amd not built (run __graft_entry__.build())"
    return GPU_BIN


def test_gpu_consensus_golden(gpu_bin):
    out = run_stdout([gpu_bin, os.path.join(GOLDEN, "seq.fa")])
    want = open(os.path.join(GOLDEN, "expected_seq_cons.txt"), "rb").read()
    assert out == want


def test_gpu_msa_golden(gpu_bin):
    out = run_stdout([gpu_bin, os.path.join(GOLDEN, "seq.fa"), "-r1"])
    want = open(os.path.join(GOLDEN, "expected_seq_r1.txt"), "rb").read()
    assert out == want


def test_gpu_mf_consensus_golden(gpu_bin):
    out = run_stdout([gpu_bin, os.path.join(GOLDEN, "seq.fa"), "-a1"])
    want = open(os.path.join(GOLDEN, "expected_seq_msa.txt"), "rb").read()
    assert out == want


def test_gpu_test_fa_golden(gpu_bin):
    out = run_stdout([gpu_bin, os.path.join(GOLDEN, "test.fa")])
    want = open(os.path.join(GOLDEN, "expected_test_cons.txt"), "rb").read()
    assert out == want


@pytest.mark.parametrize("seed,length,depth", [(11, 500, 20), (12, 1000, 50), (13, 3000, 30)])
def test_gpu_vs_oracle_synthetic(gpu_bin, cputest_bin, oracle_env, tmp_path, seed, length, depth):
    """GPU hot path vs CPU oracle on identical fresh inputs; both consensus
    and RC-MSA bytes (the RC-MSA exercises every read's CIGAR)."""
    fa = tmp_path / "s.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", str(seed), "--len", str(length), "--depth", str(depth)],
                   check=True, stderr=subprocess.DEVNULL)
    for opts in ([], ["-r1"]):
        gpu = run_stdout([gpu_bin, str(fa)] + opts)
        cpu = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
        assert gpu == cpu, "GPU/oracle divergence seed=%d len=%d opts=%r" % (seed, length, opts)


def test_gpu_10kbp_set(gpu_bin, cputest_bin, oracle_env, tmp_path):
    """Full north-star shape: one 50x10kbp set, GPU vs oracle consensus."""
    fa = tmp_path / "s10k.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", "42", "--len", "10000", "--depth", "50"],
                   check=True, stderr=subprocess.DEVNULL)
    gpu = run_stdout([gpu_bin, str(fa)])
    cpu = run_stdout([cputest_bin, str(fa)], env=oracle_env)
    assert gpu == cpu


@pytest.mark.parametrize("opts", [["-O", "4", "-E", "2"], ["-O", "0", "-E", "2"]],
                         ids=["affine", "linear"])
def test_gpu_gap_modes_vs_oracle(gpu_bin, cputest_bin, oracle_env, tmp_path, opts):
    """Affine and linear gap GPU kernels vs the CPU oracle (consensus + RC-MSA)."""
    fa = tmp_path / "s.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", "21", "--len", "1500", "--depth", "30"],
                   check=True, stderr=subprocess.DEVNULL)
    for extra in ([], ["-r1"]):
        gpu = run_stdout([gpu_bin, str(fa)] + opts + extra)
        cpu = run_stdout([cputest_bin, str(fa)] + opts + extra, env=oracle_env)
        assert gpu == cpu, "GPU/oracle divergence opts=%r extra=%r" % (opts, extra)


def test_gpu_aa_mode_vs_oracle(gpu_bin, cputest_bin, oracle_env, tmp_path):
    """Amino-acid alphabet (m=27, BLOSUM62) on the GPU vs the oracle."""
    import random
    random.seed(6)
    aa = "ARNDCQEGHILKMFPSTWYV"
    refseq = "".join(random.choice(aa) for _ in range(600))
    reads = []
    for _ in range(12):
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
    for opts in (["-c", "-t", mtx], ["-c", "-t", mtx, "-O", "4", "-E", "2"]):
        gpu = run_stdout([gpu_bin, str(fa)] + opts)
        cpu = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
        assert gpu == cpu


@pytest.mark.parametrize("opts", [["-m1"], ["-m2"], ["-m1", "-O", "4", "-E", "2"],
                                  ["-m2", "-O", "0", "-E", "2"], ["-m2", "-z", "100"]],
                         ids=["local", "extend", "local-affine", "extend-linear", "extend-zdrop"])
def test_gpu_align_modes_vs_oracle(gpu_bin, cputest_bin, oracle_env, tmp_path, opts):
    """Local and extension alignment GPU kernels vs the CPU oracle."""
    fa = tmp_path / "s.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", "31", "--len", "1200", "--depth", "25"],
                   check=True, stderr=subprocess.DEVNULL)
    for extra in ([], ["-r1"]):
        gpu = run_stdout([gpu_bin, str(fa)] + opts + extra)
        cpu = run_stdout([cputest_bin, str(fa)] + opts + extra, env=oracle_env)
        assert gpu == cpu, "GPU/oracle divergence opts=%r extra=%r" % (opts, extra)


def test_gpu_multicons_vs_oracle(gpu_bin, cputest_bin, oracle_env, tmp_path):
    """Multi-consensus (-d >= 2): GPU DP + k-medoids clustering end-to-end
    must match the oracle-injected host pipeline byte-for-byte."""
    import random
    rng = random.Random(11)
    L = 500
    base = "".join(rng.choice("ACGT") for _ in range(L))
    hapB = list(base)
    for p in rng.sample(range(10, L - 10), 6):
        hapB[p] = rng.choice([c for c in "ACGT" if c != hapB[p]])
    haps = [base, "".join(hapB)]

    def noisy(h):
        out = []
        for ch in h:
            r = rng.random()
            if r < 0.02: out.append(rng.choice("ACGT"))
            elif r < 0.03: pass
            elif r < 0.04: out.extend((ch, rng.choice("ACGT")))
            else: out.append(ch)
        return "".join(out)

    fa = tmp_path / "dip.fa"
    with open(fa, "w") as f:
        for i in range(24):
            f.write(">r%d\n%s\n" % (i, noisy(haps[i % 2])))
    for opts in (["-d2"], ["-d2", "-r1"], ["-d3"], ["-d2", "-a1"]):
        gpu = run_stdout([gpu_bin, str(fa)] + opts)
        cpu = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
        assert gpu == cpu, "GPU/oracle multicons divergence opts=%r" % (opts,)


def test_gpu_seeding_vs_oracle(gpu_bin, cputest_bin, oracle_env, tmp_path):
    """Seeded (-S) and progressive (-p) POA through the GPU subgraph kernels
    end-to-end vs the oracle-injected host pipeline."""
    fa = tmp_path / "s.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", "4", "--len", "5000", "--depth", "8"],
                   check=True, stderr=subprocess.DEVNULL)
    for opts in (["-S"], ["-S", "-p"], ["-S", "-n", "100", "-r1"]):
        gpu = run_stdout([gpu_bin, str(fa)] + opts)
        cpu = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
        assert gpu == cpu, "GPU/oracle seeding divergence opts=%r" % (opts,)


@pytest.mark.parametrize("opts", [["-O", "4", "-E", "2"], ["-O", "0", "-E", "2"]],
                         ids=["affine-10k", "linear-10k"])
def test_gpu_gap_modes_full_shape(gpu_bin, cputest_bin, oracle_env, tmp_path, opts):
    """Affine/linear kernels at the full north-star shape (50 x 10 kbp):
    graphs large enough that most rounds run the int32 rescore width."""
    fa = tmp_path / "s10k.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", "43", "--len", "10000", "--depth", "50"],
                   check=True, stderr=subprocess.DEVNULL)
    gpu = run_stdout([gpu_bin, str(fa)] + opts)
    cpu = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
    assert gpu == cpu


def test_gpu_forced_int32_overflow(gpu_bin, cputest_bin, oracle_env, tmp_path):
    """-M9 on 10 kbp reads puts the score bound (qlen x max_mat = 90k) past
    int16 from the FIRST read: the whole set runs the int32 kernel path,
    including its width-specific inf_min clamps (abpoa_align_simd.c:1299)."""
    fa = tmp_path / "s.fa"
    subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), str(fa),
                    "--seed", "44", "--len", "10000", "--depth", "12"],
                   check=True, stderr=subprocess.DEVNULL)
    for opts in (["-M", "9", "-X", "12"], ["-M", "9", "-X", "12", "-O", "4", "-E", "2"]):
        gpu = run_stdout([gpu_bin, str(fa)] + opts)
        cpu = run_stdout([cputest_bin, str(fa)] + opts, env=oracle_env)
        assert gpu == cpu, "int32-forced divergence opts=%r" % (opts,)
