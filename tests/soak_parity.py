#!/usr/bin/env python3
"""Parity soak: many random shapes/modes, GPU CLI vs injected-oracle CLI.

Run on a GPU box:  python tests/soak_parity.py [n_cases] [base_seed]
Every case compares full stdout (consensus, and RC-MSA for a subset) between
the product GPU path and the CPU oracle on identical inputs. Exits non-zero
on the first divergence.
"""
import os
import random
import subprocess
import sys
import tempfile

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
GPU = os.path.join(ROOT, "abpoa_amd", "csrc", "abpoa_amd")
CPU = os.path.join(ROOT, "abpoa_amd", "csrc", "abpoa_amd_cputest")
ORACLE = os.path.join(ROOT, "oracle", "liboracle.so")
MTX = os.path.join(ROOT, "tests", "golden", "BLOSUM62.mtx")

MODE_POOL = [
    [],
    ["-O", "4", "-E", "2"],
    ["-O", "0", "-E", "2"],
    ["-m1"],
    ["-m2"],
    ["-m2", "-z", "150"],
    ["-O", "2,30", "-E", "3,1"],
    ["-M", "4", "-X", "6"],
    ["-b", "50", "-f", "0.02"],
    ["-R"],        # put_gap_on_right
    ["-J"],        # put_gap_at_end
]


def gen(path, rng, length, depth, aa=False):
    alpha = "ARNDCQEGHILKMFPSTWYV" if aa else "ACGT"
    ref = "".join(rng.choice(alpha) for _ in range(length))
    with open(path, "w") as f:
        for i in range(depth):
            s = []
            for ch in ref:
                r = rng.random()
                if r < 0.045:
                    s.append(rng.choice(alpha))
                elif r < 0.075:
                    pass
                elif r < 0.1:
                    s.extend((ch, rng.choice(alpha)))
                else:
                    s.append(ch)
            f.write(">r%d\n%s\n" % (i, "".join(s)))


def run(binary, fa, opts, env=None):
    return subprocess.run([binary, fa] + opts, env=env, check=True,
                          stdout=subprocess.PIPE, stderr=subprocess.DEVNULL).stdout


def main():
    n_cases = int(sys.argv[1]) if len(sys.argv) > 1 else 40
    base = int(sys.argv[2]) if len(sys.argv) > 2 else 1000
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE
    fails = 0
    with tempfile.TemporaryDirectory() as td:
        for c in range(n_cases):
            rng = random.Random(base + c)
            aa = rng.random() < 0.15
            length = rng.choice([300, 700, 1500, 2500, 4000])
            depth = rng.choice([8, 15, 25, 40])
            opts = list(rng.choice(MODE_POOL))
            if aa:
                opts = ["-c", "-t", MTX] + [o for o in opts if o not in ("-m1", "-m2")]
            out_modes = [[]]
            if rng.random() < 0.4 and "-m1" not in opts and "-m2" not in opts:
                out_modes.append(["-r1"])
            fa = os.path.join(td, "case%d.fa" % c)
            gen(fa, rng, length, depth, aa)
            for om in out_modes:
                g = run(GPU, fa, opts + om)
                o = run(CPU, fa, opts + om, env=env)
                tag = "case %d len=%d depth=%d aa=%d opts=%r out=%r" % (c, length, depth, aa, opts, om)
                if g != o:
                    print("DIVERGENCE: " + tag)
                    fails += 1
                else:
                    print("ok " + tag, flush=True)
    print("soak done: %d cases, %d divergences" % (n_cases, fails))
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
