#!/usr/bin/env python3
"""Randomized fold-twin soak: random synthetic shapes through
abpoa_amd_foldtwin (flat fold + derived passes + DP-row CSR + graph
importer + flat HB consensus, all compared against the live pointer-graph
path after EVERY read). Usage: python tools/soak_twin.py [n_cases] [seed]"""
import os
import random
import subprocess
import sys
import tempfile

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
TWIN = os.path.join(ROOT, "abpoa_amd", "csrc", "abpoa_amd_foldtwin")
ORACLE = os.path.join(ROOT, "oracle", "liboracle.so")


def main():
    n_cases = int(sys.argv[1]) if len(sys.argv) > 1 else 20
    rng = random.Random(int(sys.argv[2]) if len(sys.argv) > 2 else 5)
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE
    for c in range(n_cases):
        length = rng.randrange(100, 4000)
        depth = rng.randrange(3, 100)  # > 64 exercises multi-word read-id bitsets
        seed = rng.randrange(1, 10**6)
        with tempfile.TemporaryDirectory() as td:
            fa = os.path.join(td, "s.fa")
            subprocess.run(["python3", os.path.join(ROOT, "tests", "make_synth.py"), fa,
                            "--seed", str(seed), "--len", str(length), "--depth", str(depth)],
                           check=True, stderr=subprocess.DEVNULL)
            for extra in ([], ["-r1"]):
                out = subprocess.run([TWIN, fa] + extra, env=env,
                                     stdout=subprocess.PIPE, stderr=subprocess.PIPE)
                if out.returncode != 0 or b"twin OK" not in out.stdout:
                    print("FAIL case %d len=%d depth=%d seed=%d extra=%r\n%s"
                          % (c, length, depth, seed, extra, out.stderr.decode()[-400:]))
                    sys.exit(1)
        print("ok case %d len=%d depth=%d seed=%d" % (c, length, depth, seed), flush=True)
    print("twin soak done: %d cases, 0 divergences" % n_cases)


if __name__ == "__main__":
    main()
