#!/usr/bin/env python3
"""Randomized full-CLI-surface soak vs the live reference binary.

Usage:  python tests/soak_cli.py [n_cases] [base_seed] [binary]
binary defaults to the CPU test build with the oracle injected; pass the
GPU CLI path on a GPU box to soak the product path instead. Every case
draws a random shape (length/depth/ploidy/alphabet) and a random flag
combination spanning alignment modes, gap modes, band overrides, outputs,
multi-consensus, seeding/progressive, amb-strand and incremental restore,
then byte-compares full stdout. Exits non-zero on the first divergence.
"""
import os
import random
import subprocess
import sys
import tempfile

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CPU = os.path.join(ROOT, "abpoa_amd", "csrc", "abpoa_amd_cputest")
REF = os.path.join(ROOT, "oracle", "_ref", "abpoa")
ORACLE = os.path.join(ROOT, "oracle", "liboracle.so")
MTX = os.path.join(ROOT, "tests", "golden", "BLOSUM62.mtx")
COMP = {"A": "T", "C": "G", "G": "C", "T": "A"}


def gen(path, rng, length, depth, n_hap=1, aa=False, rc_frac=0.0, fastq=False):
    alpha = "ARNDCQEGHILKMFPSTWYV" if aa else "ACGT"
    base = "".join(rng.choice(alpha) for _ in range(length))
    haps = [base]
    for _h in range(1, n_hap):
        hb = list(base)
        for p in rng.sample(range(5, length - 5), max(2, length // 80)):
            hb[p] = rng.choice([c for c in alpha if c != hb[p]])
        haps.append("".join(hb))
    with open(path, "w") as f:
        for i in range(depth):
            out = []
            for ch in haps[i % n_hap]:
                r = rng.random()
                if r < 0.04:
                    out.append(rng.choice(alpha))
                elif r < 0.07:
                    pass
                elif r < 0.09 and not aa:
                    out.extend((ch, rng.choice(alpha)))
                else:
                    out.append(ch)
            s = "".join(out)
            if not aa and i > 0 and rng.random() < rc_frac:
                s = "".join(COMP[c] for c in reversed(s))
            if fastq:
                qual = "".join(chr(33 + rng.randrange(5, 40)) for _ in s)
                f.write("@r%d\n%s\n+\n%s\n" % (i, s, qual))
            else:
                f.write(">r%d\n%s\n" % (i, s))


def draw_opts(rng, aa, has_rc, length):
    opts = []
    mode = rng.choice(["", "", "", "-m1", "-m2"])
    if mode:
        opts.append(mode)
    gap = rng.choice([[], [], ["-O", "4", "-E", "2"], ["-O", "0", "-E", "2"],
                      ["-O", "2,30", "-E", "3,1"]])
    opts += gap
    if rng.random() < 0.2:
        opts += ["-M", str(rng.choice([1, 2, 4])), "-X", str(rng.choice([2, 4, 6]))]
    if rng.random() < 0.2:
        opts += ["-b", str(rng.choice([10, 50, 200])), "-f", rng.choice(["0.01", "0.05"])]
    if rng.random() < 0.15 and mode == "-m2":
        opts += ["-z", str(rng.choice([100, 200]))]
    out = rng.choice(["", "", "-r1", "-r2", "-r3", "-r4"])
    if out:
        opts.append(out)
    if rng.random() < 0.25 and not mode and out not in ("-r3",):
        opts.append(rng.choice(["-d2", "-d3"]))
    if rng.random() < 0.3 and not mode and not aa:
        opts.append("-S")
        if rng.random() < 0.4:
            opts += ["-n", str(rng.choice([100, 250]))]
        if rng.random() < 0.3:
            opts.append("-p")
    if has_rc:
        opts.append("-s")
    if rng.random() < 0.15:
        opts.append(rng.choice(["-R", "-J"]))
    if rng.random() < 0.1:
        opts.append("-A")  # sort by length, if supported
    return opts


def main():
    n_cases = int(sys.argv[1]) if len(sys.argv) > 1 else 120
    base = int(sys.argv[2]) if len(sys.argv) > 2 else 9000
    binary = sys.argv[3] if len(sys.argv) > 3 else CPU
    env = dict(os.environ)
    # inject the oracle for any CPU-test build; the product GPU CLI (named
    # exactly abpoa_amd) must run its own aligner
    if os.path.basename(binary) != "abpoa_amd":
        env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE
    fails = 0
    with tempfile.TemporaryDirectory() as td:
        for c in range(n_cases):
            rng = random.Random(base + c)
            aa = rng.random() < 0.12
            length = rng.choice([200, 400, 800, 1500, 2500])
            depth = rng.choice([6, 10, 16, 24])
            n_hap = rng.choice([1, 1, 2, 3])
            has_rc = (not aa) and rng.random() < 0.15
            fastq = (not aa) and rng.random() < 0.15
            fa = os.path.join(td, "c%d.%s" % (c, "fq" if fastq else "fa"))
            gen(fa, rng, length, depth, n_hap, aa, 0.5 if has_rc else 0.0, fastq)
            opts = draw_opts(rng, aa, has_rc, length)
            if fastq and rng.random() < 0.7:
                opts.append("-Q")
            if aa:
                opts = ["-c", "-t", MTX] + [o for o in opts if o not in ("-m1", "-m2", "-s")]
            # incremental restore on a fraction of cases
            if rng.random() < 0.15 and "-d2" not in opts and "-d3" not in opts:
                first = os.path.join(td, "c%d_a.fa" % c)
                gen(first, rng, length, max(4, depth // 2), n_hap, aa)
                rfmt = rng.choice(["-r1", "-r3"]) if not aa else "-r1"
                restore = os.path.join(td, "c%d_restore" % c)
                with open(restore, "wb") as f:
                    f.write(subprocess.run([REF, first] + (["-c", "-t", MTX] if aa else []) + [rfmt],
                                           stdout=subprocess.PIPE, stderr=subprocess.DEVNULL).stdout)
                opts = ["-i", restore] + opts
            r = subprocess.run([REF, fa] + opts, stdout=subprocess.PIPE,
                               stderr=subprocess.DEVNULL)
            a = subprocess.run([binary, fa] + opts, env=env, stdout=subprocess.PIPE,
                               stderr=subprocess.DEVNULL)
            tag = "case %d len=%d depth=%d hap=%d aa=%d opts=%r rc(%d,%d)" % (
                c, length, depth, n_hap, aa, opts, r.returncode, a.returncode)
            if r.returncode == a.returncode and r.stdout == a.stdout:
                print("ok " + tag, flush=True)
            elif r.returncode != 0:
                # the reference itself crashes or errors on this input (e.g.
                # SEGV from the -d3 partition-count integer overflow, or
                # "Error in lg_backtrack" dead-ends under amb-strand); we do
                # not reproduce reference crashes — skip the comparison
                print("ref-fails(skipped) " + tag, flush=True)
            else:
                print("DIVERGENCE: " + tag)
                fails += 1
    print("cli soak done: %d cases, %d divergences" % (n_cases, fails))
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
