#!/usr/bin/env python3
"""Kernel phase breakdown (diagnostic; GPU box).

Runs the batched driver against libabpoa_amd_kprof.so (make kprof), which
carries s_memtime counters around each phase of the convex DP kernel's row
loop and each pass of the per-round fold kernel, then prints where the
cycles go. Relative shares are the point; the absolute clock of s_memtime
is not calibrated here.

Usage: python tools/kprof_run.py [n_sets] [depth] [qlen]
"""
import ctypes
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
os.environ["ABPOA_AMD_LIB"] = os.path.join(
    ROOT, "abpoa_amd", "csrc", "libabpoa_amd_kprof.so")

import abpoa_amd  # noqa: E402
import bench  # noqa: E402
import numpy as np  # noqa: E402


def fetch(lib, name):
    out = (ctypes.c_uint64 * 16)()
    getattr(lib, name)(out)
    return list(out)


def main():
    n_sets = int(sys.argv[1]) if len(sys.argv) > 1 else 24
    depth = int(sys.argv[2]) if len(sys.argv) > 2 else 50
    qlen = int(sys.argv[3]) if len(sys.argv) > 3 else 10000
    L = abpoa_amd.lib()
    rng = np.random.default_rng(7)
    sets = bench.gen_sets(rng, n_sets, depth=depth, qlen=qlen)
    L.abamd_kprof_reset()
    L.abamd_kprof_fold_reset()
    abpoa_amd.msa_batch_consensus(sets, n_threads=8)
    k = fetch(L, "abamd_kprof_fetch")
    f = fetch(L, "abamd_kprof_fold_fetch")

    mwa, mwb, mwc, mwtot, mwrows = k[11], k[12], k[13], k[14], k[15]
    if mwrows:
        print("== cg MULTI-WAVE kernel (%d rows, %.0f cyc/row total) ==" %
              (mwrows, mwtot / mwrows))
        for name, v in (("gather+hpre", mwa), ("B1+scan+B2+fold", mwb),
                        ("carry+B3", mwc), ("epilogue", mwtot - mwa - mwb - mwc)):
            print("  %-16s %12d cyc  (%.0f cyc/row)" % (name, v, v / mwrows))

    fjobs, fapply, findeg, fbfs, fsort, fremspan, fbuild = f[:7]
    ftot = fapply + findeg + fbfs + fsort + fremspan + fbuild
    print("== fold kernel (%d jobs) ==" % fjobs)
    for name, v in (("apply", fapply), ("in_deg", findeg), ("bfs", fbfs),
                    ("sort", fsort), ("remain+nspan", fremspan), ("build_rows", fbuild)):
        print("  %-14s %12d cyc  %5.1f%%  (%.0f cyc/job)" %
              (name, v, 100.0 * v / max(ftot, 1), v / max(fjobs, 1)))


if __name__ == "__main__":
    main()
