#!/bin/bash
# Regenerates the golden parity fixtures committed under tests/golden/.
#
# Inputs are the reference's own test data (test data, not code) copied from
# /root/reference/test_data; expected outputs are produced by the reference
# binary built by oracle/Makefile from the reference's unmodified sources.
# /root/reference exists only in the dev container, so the fixtures are
# committed and the GPU-side tests read them from here.
#
# Usage: bash tests/golden/make_golden.sh   (from anywhere)
set -e
cd "$(dirname "$0")"
REF=/root/reference
BIN=../../oracle/_ref/abpoa
[ -x "$BIN" ] || { echo "build oracle/_ref first: (cd oracle && make)"; exit 1; }

for f in seq.fa heter.fa 3alleles.fa test.fa; do
    cp "$REF/test_data/$f" .
done

run() { # run <outfile> <args...>
    out=$1; shift
    "$BIN" "$@" 2>/dev/null > "$out"
}

# the three upstream regression goldens (tests/run_all.sh:33-56 in the reference)
run expected_seq_cons.txt       seq.fa
run expected_seq_msa.txt        seq.fa -a1
run expected_heter_d2.txt       heter.fa -d2
# extra configurations covering the north-star variants on the small input
run expected_seq_r1.txt         seq.fa -r1          # RC-MSA output
run expected_seq_r2.txt         seq.fa -r2          # cons + MSA
run expected_seq_affine.txt     seq.fa -O 4 -E 2    # affine gap
run expected_seq_linear.txt     seq.fa -O 0 -E 2    # linear gap
run expected_seq_local.txt      seq.fa -m 1         # local mode
run expected_seq_extend.txt     seq.fa -m 2         # extension mode
run expected_seq_gfa.txt        seq.fa -r3          # GFA output
run expected_seq_fq.txt         seq.fa -r5          # FASTQ consensus
run expected_test_cons.txt      test.fa
run expected_3alleles_d3.txt    3alleles.fa -d3
run expected_heter_cons.txt     heter.fa
echo "golden fixtures regenerated"
