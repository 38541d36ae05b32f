/* Parameter handling: defaults, derived settings, scoring matrices.
 * Semantics mirror abpoa_align.c:13-193 and abpoa_init_para (:101-158). */
#include <ctype.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

/* simple match/mismatch matrix with a zero 'N' row/column
 * (gen_simple_mat, abpoa_align.c:13-26) */
static void gen_simple_mat(abpoa_para_t *p) {
    int m = p->m, i, j;
    int match = p->match < 0 ? -p->match : p->match;
    int mismatch = p->mismatch > 0 ? -p->mismatch : p->mismatch;
    for (i = 0; i < m - 1; ++i) {
        for (j = 0; j < m - 1; ++j)
            p->mat[i * m + j] = (i == j) ? match : mismatch;
        p->mat[i * m + m - 1] = 0;
    }
    for (j = 0; j < m; ++j) p->mat[(m - 1) * m + j] = 0;
    p->max_mat = match;
    p->min_mis = -mismatch;
}

void abpoa_set_mat_from_file(abpoa_para_t *p, char *mat_fn) {
    FILE *fp = fopen(mat_fn, "r");
    if (!fp) abamd_fatal("abpoa_set_mat_from_file", "cannot open scoring matrix '%s'", mat_fn);
    char line[1024];
    int *order = (int*)abamd_malloc((size_t)p->m * sizeof(int));
    int first = 1, i, n_hdr = 0;
    while (fgets(line, sizeof(line), fp)) {
        if (line[0] == '#') continue;
        if (first) {
            int n = 0;
            for (i = 0; line[i]; ++i) {
                if (isspace((unsigned char)line[i])) continue;
                if (n >= p->m)
                    abamd_fatal("abpoa_set_mat_from_file", "too many residues in matrix header");
                order[n] = ab_amd_char26_table[(int)line[i]];
                if (order[n] >= p->m)
                    abamd_fatal("abpoa_set_mat_from_file", "unknown residue '%c' in matrix header", line[i]);
                n++;
            }
            n_hdr = n;
            first = 0;
        } else {
            char *s = line, *end; int row = -1, n = 0, is_base = 1;
            while (*s) {
                if (!isalpha((unsigned char)*s) && !isdigit((unsigned char)*s) && *s != '+' && *s != '-') { ++s; continue; }
                if (is_base) {
                    row = ab_amd_char26_table[(int)*s];
                    if (row >= p->m) abamd_fatal("abpoa_set_mat_from_file", "unknown residue '%c'", *s);
                    is_base = 0; ++s;
                } else {
                    if (n >= n_hdr) abamd_fatal("abpoa_set_mat_from_file", "more scores than header residues");
                    long v = strtol(s, &end, 10);
                    s = end;
                    p->mat[row * p->m + order[n]] = (int)v;
                    n++;
                }
            }
        }
    }
    p->min_mis = 0; p->max_mat = 0;
    for (i = 0; i < p->m * p->m; ++i) {
        if (p->mat[i] > p->max_mat) p->max_mat = p->mat[i];
        if (-p->mat[i] > p->min_mis) p->min_mis = -p->mat[i];
    }
    free(order); fclose(fp);
}

static void set_gap_mode(abpoa_para_t *p) {
    if (p->match < 0 || p->mismatch < 0 || p->gap_open1 < 0 || p->gap_open2 < 0 || p->gap_ext1 < 0 || p->gap_ext2 < 0)
        abamd_fatal("abpoa_set_gap_mode", "negative scoring parameters are not allowed");
    if (p->gap_ext1 == 0 && p->gap_ext2 == 0)
        abamd_fatal("abpoa_set_gap_mode", "at least one gap extension penalty must be positive");
    if (p->gap_open1 == 0) p->gap_mode = ABPOA_LINEAR_GAP;
    else if (p->gap_open1 > 0 && p->gap_open2 == 0) p->gap_mode = ABPOA_AFFINE_GAP;
    else p->gap_mode = ABPOA_CONVEX_GAP;
}

abpoa_para_t *abpoa_init_para(void) {
    ab_amd_init_tables();
    abpoa_para_t *p = (abpoa_para_t*)abamd_calloc(1, sizeof(abpoa_para_t));
    p->align_mode = ABPOA_GLOBAL_MODE;
    p->gap_mode = ABPOA_CONVEX_GAP;
    p->zdrop = -1;
    p->end_bonus = -1;
    p->wb = ABPOA_EXTRA_B;
    p->wf = ABPOA_EXTRA_F;
    p->ret_cigar = 1;
    p->out_cons = 1;
    p->cons_algrm = ABPOA_HB;
    p->max_n_cons = 1;
    p->min_freq = 0.25;
    p->m = 5;
    p->mat = (int*)abamd_malloc((size_t)p->m * p->m * sizeof(int));
    p->match = 2; p->mismatch = 4;
    p->gap_open1 = 4; p->gap_open2 = 24;
    p->gap_ext1 = 2; p->gap_ext2 = 1;
    p->disable_seeding = 1;
    p->k = 19; p->w = 10; p->min_w = 500;
    p->verbose = ABPOA_NONE_VERBOSE;
    return p;
}

void abpoa_post_set_para(abpoa_para_t *p) {
    ab_amd_init_tables();
    set_gap_mode(p);
    if (p->out_msa || p->out_gfa || p->max_n_cons > 1 || p->cons_algrm == ABPOA_MF) {
        p->use_read_ids = 1;
        if (p->out_msa || p->out_gfa || p->max_n_cons > 1) ab_amd_set_65536_table();
        if (p->max_n_cons > 1 || p->cons_algrm == ABPOA_MF) ab_amd_set_bit_table16();
    }
    if (p->align_mode == ABPOA_LOCAL_MODE) p->wb = -1;
    int i;
    if (p->m > 5) {
        for (i = 0; i < 256; ++i) {
            ab_amd_char26_table[i] = (char)ab_amd_aa26_table[i];
            ab_amd_char256_table[i] = ab_amd_aa256_table[i];
        }
        if (p->k > 11) { p->k = 7; p->w = 4; }
    } else {
        for (i = 0; i < 256; ++i) {
            ab_amd_char26_table[i] = (char)ab_amd_nt4_table[i];
            ab_amd_char256_table[i] = ab_amd_nt256_table[i];
        }
    }
    if (p->use_score_matrix == 0) gen_simple_mat(p);
    else abpoa_set_mat_from_file(p, p->mat_fn);
}

void abpoa_free_para(abpoa_para_t *p) {
    free(p->mat);
    free(p->mat_fn);
    free(p->out_pog);
    free(p->incr_fn);
    free(p);
}
