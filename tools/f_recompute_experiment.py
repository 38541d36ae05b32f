#!/usr/bin/env python3
"""Round-2 design experiment: can the backtrack recompute the F1/F2 planes
from the STORED (post-F) H plane instead of reading stored F planes?

The DP stores H after H = max(H, F1, F2), so a row's F planes recomputed
from stored H pick up extra candidates (F1/F2 feeding back through H).
This script compares, over randomized rows and several (o1,e1,o2,e2)
regimes including inverted-convex:
  - the F-entry decision (H[j] == F1[j] / F2[j], tested in order), and
  - every decision of the in-run walk (open: H[j-1]-oe == F[j], first;
    else extend: F[j-1]-e == F[j]) — the exact reference order
    (abpoa_align_simd.c backtrack; oracle/ref_core.c:397-420).

Result (seed 11, 5 regimes x 2000 rows, ~560k entries): raw F VALUES differ
~40% of the time, but every DECISION is identical and no walk dead-ends.
Conclusion: the arena can drop the F1/F2 planes (40% of plane bytes) and
recompute F per visited row during backtrack; end-to-end bit-parity must
still be confirmed by the full suite when implemented.
"""
import random

NEG = -10**6


def row_f(h_pre, oe1, e1, oe2, e2):
    n = len(h_pre)
    f1 = [NEG] * n
    f2 = [NEG] * n
    for j in range(1, n):
        f1[j] = max(f1[j - 1] - e1, h_pre[j - 1] - oe1)
        f2[j] = max(f2[j - 1] - e2, h_pre[j - 1] - oe2)
    hp = [max(h_pre[j], f1[j], f2[j]) for j in range(n)]
    return f1, f2, hp


def recompute_f(hp, oe1, e1, oe2, e2):
    n = len(hp)
    f1 = [NEG] * n
    f2 = [NEG] * n
    for j in range(1, n):
        f1[j] = max(f1[j - 1] - e1, hp[j - 1] - oe1)
        f2[j] = max(f2[j - 1] - e2, hp[j - 1] - oe2)
    return f1, f2


def walk(hp, f, oe, e, j):
    out = []
    while j > 0:
        if hp[j - 1] - oe == f[j]:
            out.append('O')
            break
        elif f[j - 1] - e == f[j]:
            out.append('E')
            j -= 1
        else:
            out.append('X')
            break
    return ''.join(out)


def main():
    rng = random.Random(11)
    dec_mism = entries = dead_true = dead_rec = 0
    for (o1, e1, o2, e2) in ((4, 2, 24, 1), (2, 1, 12, 2), (0, 2, 24, 1),
                             (6, 2, 48, 1), (2, 30, 3, 1)):
        oe1, oe2 = o1 + e1, o2 + e2
        for _ in range(2000):
            n = rng.randrange(8, 120)
            h, base = [], rng.randrange(-50, 400)
            for _j in range(n):
                r = rng.random()
                if r < 0.08:
                    base += rng.randrange(20, 120)
                elif r < 0.30:
                    base -= rng.randrange(10, 80)
                else:
                    base += rng.randrange(-4, 6)
                h.append(base)
            f1, f2, hp = row_f(h, oe1, e1, oe2, e2)
            r1, r2 = recompute_f(hp, oe1, e1, oe2, e2)
            for j in range(1, n):
                for (f, rr, oe, e) in ((f1, r1, oe1, e1), (f2, r2, oe2, e2)):
                    if hp[j] == f[j]:
                        entries += 1
                        a = walk(hp, f, oe, e, j)
                        b = walk(hp, rr, oe, e, j)
                        if a != b:
                            dec_mism += 1
                        dead_true += 'X' in a
                        dead_rec += 'X' in b
    print("entries", entries, "decision mismatches", dec_mism,
          "deadends true/recomputed", dead_true, dead_rec)
    assert dec_mism == 0 and dead_true == 0 and dead_rec == 0


if __name__ == "__main__":
    main()
