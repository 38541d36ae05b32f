/* Test tool: run the batched multi-set driver over N FASTA files (one read
 * set per file) and print each set's consensus, one ">set_N" record per set.
 *
 * Exists so the pipeline driver's item ordering (groups, lookahead, big-item
 * fallback) is exercised on CPU-only machines too: built against gpu_stub.c
 * the alignments route through the dispatch seam, where tests inject the
 * oracle via ABPOA_AMD_TEST_ALIGNER_SO. Parity requirement: output must be
 * byte-identical across ABPOA_AMD_GROUPS=1/2/3 and equal to the sequential
 * CLI consensus on each file. */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

/* forward-declared file-local seq API (abamd_seq.c) */
typedef struct abamd_fx_t abamd_fx_t;
abpoa_seq_t *abamd_seq_new(void);
void abamd_seq_destroy(abpoa_seq_t *abs);
abamd_fx_t *abamd_fx_open(const char *fn);
void abamd_fx_close(abamd_fx_t *x);
int abamd_read_seq(abpoa_seq_t *abs, abamd_fx_t *x);

typedef struct { char **cons; int n; } out_t;

static void cb(int set_idx, const abpoa_cons_t *cons, void *user) {
    out_t *o = (out_t*)user;
    if (cons->n_cons < 1) { o->cons[set_idx] = strdup(""); return; }
    char *s = (char*)malloc((size_t)cons->cons_len[0] + 1);
    for (int i = 0; i < cons->cons_len[0]; ++i)
        s[i] = "ACGTN"[cons->cons_base[0][i]];
    s[cons->cons_len[0]] = 0;
    o->cons[set_idx] = s;
}

int main(int argc, char **argv) {
    if (argc < 2) { fprintf(stderr, "usage: %s set1.fa [set2.fa ...]\n", argv[0]); return 2; }
    int n_sets = argc - 1;
    abpoa_para_t *abpt = abpoa_init_para();
    abpoa_post_set_para(abpt);

    int *n_seqs = (int*)abamd_calloc(n_sets, sizeof(int));
    int **lens = (int**)abamd_calloc(n_sets, sizeof(int*));
    uint8_t ***seqs = (uint8_t***)abamd_calloc(n_sets, sizeof(uint8_t**));
    for (int s = 0; s < n_sets; ++s) {
        abpoa_seq_t *abs = abamd_seq_new();
        abamd_fx_t *fx = abamd_fx_open(argv[s + 1]);
        if (!fx) { fprintf(stderr, "cannot open %s\n", argv[s + 1]); return 2; }
        n_seqs[s] = abamd_read_seq(abs, fx);
        abamd_fx_close(fx);
        lens[s] = (int*)abamd_calloc(n_seqs[s], sizeof(int));
        seqs[s] = (uint8_t**)abamd_calloc(n_seqs[s], sizeof(uint8_t*));
        for (int i = 0; i < n_seqs[s]; ++i) {
            int l = abs->seq[i].l;
            lens[s][i] = l;
            seqs[s][i] = (uint8_t*)abamd_malloc(l);
            for (int j = 0; j < l; ++j)
                seqs[s][i][j] = ab_amd_nt4_table[(uint8_t)abs->seq[i].s[j]];
        }
        abamd_seq_destroy(abs);
    }
    out_t out; out.n = n_sets;
    out.cons = (char**)abamd_calloc(n_sets, sizeof(char*));
    abpoa_amd_msa_batch(abpt, n_sets, n_seqs,
                        (const int *const *)lens,
                        (const uint8_t *const *const *)seqs, cb, &out, 4);
    for (int s = 0; s < n_sets; ++s)
        printf(">set_%d\n%s\n", s, out.cons[s] ? out.cons[s] : "");
    return 0;
}
