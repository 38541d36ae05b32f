#!/usr/bin/env python3
"""Synthetic read-set generator for the north-star workload:
per set, one random ACGT reference of --len bases; --depth reads derived with
~10% ONT-style error (4.5% substitution / 3% deletion / 2.5% insertion per
base). Deterministic for a given --seed (default 42).

Usage: make_synth.py out.fa [--sets N] [--depth 50] [--len 10000] [--seed 42]
       [--sub 0.045 --del 0.03 --ins 0.025] [--outdir DIR  (one file per set)]
"""
import argparse, random, sys, os

BASES = "ACGT"

def mutate(rng, ref, p_sub, p_del, p_ins):
    out = []
    for b in ref:
        r = rng.random()
        if r < p_sub:
            out.append(rng.choice([x for x in BASES if x != b]))
        elif r < p_sub + p_del:
            pass
        elif r < p_sub + p_del + p_ins:
            out.append(b)
            out.append(rng.choice(BASES))
        else:
            out.append(b)
    return "".join(out)

def gen_set(rng, depth, length, p_sub, p_del, p_ins):
    ref = "".join(rng.choice(BASES) for _ in range(length))
    return ref, [mutate(rng, ref, p_sub, p_del, p_ins) for _ in range(depth)]

def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("out", help="output FASTA (or prefix with --outdir)")
    ap.add_argument("--sets", type=int, default=1)
    ap.add_argument("--depth", type=int, default=50)
    ap.add_argument("--len", type=int, default=10000, dest="length")
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--sub", type=float, default=0.045)
    ap.add_argument("--del", type=float, default=0.03, dest="p_del")
    ap.add_argument("--ins", type=float, default=0.025)
    ap.add_argument("--outdir", default=None,
                    help="write one FASTA per set into DIR (out is the name prefix)")
    args = ap.parse_args()
    rng = random.Random(args.seed)
    if args.outdir:
        os.makedirs(args.outdir, exist_ok=True)
    for s in range(args.sets):
        ref, reads = gen_set(rng, args.depth, args.length, args.sub, args.p_del, args.ins)
        if args.outdir:
            path = os.path.join(args.outdir, "%s_%05d.fa" % (args.out, s))
        else:
            path = args.out
        mode = "w" if (args.outdir or s == 0) else "a"
        with open(path, mode) as f:
            for i, r in enumerate(reads):
                f.write(">set%d_read%d\n%s\n" % (s, i, r))
    print("wrote %d set(s)" % args.sets, file=sys.stderr)

if __name__ == "__main__":
    main()
