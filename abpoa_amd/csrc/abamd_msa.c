/* MSA drivers: per-set orchestration of align -> fold -> consensus.
 * Mirrors abpoa_align.c:313-540 (abpoa_poa / abpoa_msa / abpoa_msa1),
 * including incremental graph restore (-i, abamd_restore.c) and the
 * minimizer-seeded / progressive guide-tree path (-S / -p, abamd_seed.c). */
#include "abpoa_amd.h"
#include "abamd_util.h"

abpoa_seq_t *abamd_realloc_seq(abpoa_seq_t *abs);
void abamd_cpy_str(abpoa_str_t *dst, const char *s, int l);
typedef struct abamd_fx_t abamd_fx_t;
abamd_fx_t *abamd_fx_open(const char *fn);
void abamd_fx_close(abamd_fx_t *x);
int abamd_read_seq(abpoa_seq_t *abs, abamd_fx_t *x);

int abpoa_align_sequence_to_graph(abpoa_t *ab, abpoa_para_t *abpt, uint8_t *query, int qlen, abpoa_res_t *res);

/* sequential POA over one read set (abpoa_poa, abpoa_align.c:313-353);
 * ambiguous-strand re-alignment included */
static int abamd_poa(abpoa_t *ab, abpoa_para_t *abpt, uint8_t **seqs, int **weights, int *seq_lens,
                     int exist_n_seq, int n_seq) {
    abpoa_seq_t *abs = ab->abs;
    abpoa_res_t res;
    int i, j, read_id, qlen, tot_n_seq = exist_n_seq + n_seq;
    uint8_t *qseq; int *weight;
    for (i = 0; i < n_seq; ++i) {
        qlen = seq_lens[i]; qseq = seqs[i]; weight = weights[i]; read_id = exist_n_seq + i;
        res.graph_cigar = 0; res.n_cigar = 0; res.n_aln_bases = res.n_matched_bases = 0;
        if (abpoa_align_sequence_to_graph(ab, abpt, qseq, qlen, &res) >= 0) {
            if (abpt->amb_strand && (res.best_score < AB_MIN2(qlen, ab->abg->node_n - 2) * abpt->max_mat * .3333)) {
                uint8_t *rc_qseq = (uint8_t*)abamd_malloc((size_t)qlen);
                int *rc_weight = (int*)abamd_malloc((size_t)qlen * sizeof(int));
                for (j = 0; j < qlen; ++j) {
                    rc_qseq[j] = qseq[qlen - j - 1] < 4 ? (uint8_t)(3 - qseq[qlen - j - 1]) : 4;
                    rc_weight[j] = weight[qlen - j - 1];
                }
                abpoa_res_t rc_res; rc_res.n_cigar = 0; rc_res.graph_cigar = 0;
                rc_res.n_aln_bases = rc_res.n_matched_bases = 0;
                simd_abpoa_align_sequence_to_graph(ab, abpt, rc_qseq, qlen, &rc_res);
                if (rc_res.best_score > res.best_score) {
                    if (res.n_cigar) free(res.graph_cigar);
                    res = rc_res;
                    res.graph_cigar = (abpoa_cigar_t*)abamd_malloc((size_t)rc_res.n_cigar * sizeof(abpoa_cigar_t));
                    memcpy(res.graph_cigar, rc_res.graph_cigar, (size_t)rc_res.n_cigar * sizeof(abpoa_cigar_t));
                    qseq = rc_qseq; weight = rc_weight;
                    abs->is_rc[read_id] = 1;
                } else { free(rc_qseq); free(rc_weight); }
                if (rc_res.n_cigar) free(rc_res.graph_cigar);
            }
        }
        abpoa_add_graph_alignment(ab, abpt, qseq, weight, qlen, NULL, res, read_id, tot_n_seq, 1);
        if (abs->is_rc[read_id]) { free(qseq); free(weight); }
        if (res.n_cigar) free(res.graph_cigar);
    }
    return 0;
}

/* seeded/progressive path: guide tree + per-pair anchors, then
 * anchor-windowed POA (abpoa_align.c:443-465) */
static void abamd_seeded_poa(abpoa_t *ab, abpoa_para_t *abpt, uint8_t **seqs, int **weights,
                             int *seq_lens, int exist_n_seq, int n_seq, int max_len) {
    int *tpos_to_node_id = (int*)abamd_calloc((size_t)(max_len > 0 ? max_len : 1), sizeof(int));
    int *qpos_to_node_id = (int*)abamd_calloc((size_t)(max_len > 0 ? max_len : 1), sizeof(int));
    int *read_id_map = (int*)abamd_malloc((size_t)n_seq * sizeof(int));
    abamd_u64v_t par_anchors = {0, 0, 0};
    int *par_c = (int*)abamd_calloc(n_seq, sizeof(int));
    abamd_build_guide_tree_partition(seqs, seq_lens, n_seq, abpt, read_id_map, &par_anchors, par_c);
    abamd_anchor_poa(ab, abpt, seqs, weights, seq_lens, par_anchors, par_c,
                     tpos_to_node_id, qpos_to_node_id, read_id_map, exist_n_seq, n_seq);
    free(read_id_map); free(tpos_to_node_id); free(qpos_to_node_id); free(par_c);
    if (par_anchors.m > 0) free(par_anchors.a);
}

/* library entry: align a set supplied as arrays (abpoa_msa, abpoa_align.c:402-472) */
int abpoa_msa(abpoa_t *ab, abpoa_para_t *abpt, int n_seq, char **seq_names, int *seq_lens,
              uint8_t **seqs, int **qual_weights, FILE *out_fp) {
    if (n_seq <= 0) return 0;
    abpoa_seq_t *abs = ab->abs;
    if (abs->n_seq <= 0) {
        abpoa_reset(ab, abpt, 1024);
        if (abpt->incr_fn) abpoa_restore_graph(ab, abpt); /* abpoa_align.c:406-412 */
    } else if (abpt->incr_fn) {
        fprintf(stderr, "[abpoa_msa] Graph already exists, but incr_fn is also provided. Not restoring graph from file.\n");
    }
    int i, j, exist_n_seq = abs->n_seq;
    abs->n_seq += n_seq; abamd_realloc_seq(abs);
    if (seq_names)
        for (i = 0; i < n_seq; ++i)
            abamd_cpy_str(&abs->name[exist_n_seq + i], seq_names[i], (int)strlen(seq_names[i]));
    int max_len = 0;
    for (i = 0; i < n_seq; ++i) if (seq_lens[i] > max_len) max_len = seq_lens[i];
    int **weights = (int**)abamd_malloc((size_t)n_seq * sizeof(int*));
    for (i = 0; i < n_seq; ++i) {
        weights[i] = (int*)abamd_malloc((size_t)seq_lens[i] * sizeof(int));
        if (abpt->use_qv && qual_weights && qual_weights[i]) {
            for (j = 0; j < seq_lens[i]; ++j) weights[i][j] = qual_weights[i][j];
        } else {
            for (j = 0; j < seq_lens[i]; ++j) weights[i][j] = 1;
        }
    }
    if ((abpt->disable_seeding && abpt->progressive_poa == 0) || abpt->align_mode != ABPOA_GLOBAL_MODE)
        abamd_poa(ab, abpt, seqs, weights, seq_lens, exist_n_seq, n_seq);
    else
        abamd_seeded_poa(ab, abpt, seqs, weights, seq_lens, exist_n_seq, n_seq, max_len);
    abpoa_output(ab, abpt, out_fp);
    for (i = 0; i < n_seq; ++i) free(weights[i]);
    free(weights);
    return 0;
}

/* length-descending insertion sort of the freshly read sequences
 * (abpoa_sort_seq_by_length, abpoa_align.c:374-391) */
static void sort_seq_by_length(abpoa_seq_t *abs, int exist_n_seq, int n_seq) {
    int i, j;
    for (i = 0; i < n_seq - 1; ++i)
        for (j = i + 1; j < n_seq; ++j)
            if (abs->seq[exist_n_seq + i].l < abs->seq[exist_n_seq + j].l) {
                abpoa_str_t t;
                t = abs->seq[exist_n_seq+i]; abs->seq[exist_n_seq+i] = abs->seq[exist_n_seq+j]; abs->seq[exist_n_seq+j] = t;
                t = abs->name[exist_n_seq+i]; abs->name[exist_n_seq+i] = abs->name[exist_n_seq+j]; abs->name[exist_n_seq+j] = t;
                t = abs->comment[exist_n_seq+i]; abs->comment[exist_n_seq+i] = abs->comment[exist_n_seq+j]; abs->comment[exist_n_seq+j] = t;
                t = abs->qual[exist_n_seq+i]; abs->qual[exist_n_seq+i] = abs->qual[exist_n_seq+j]; abs->qual[exist_n_seq+j] = t;
            }
}

/* CLI entry: one input file -> one MSA/consensus (abpoa_msa1, abpoa_align.c:474-540) */
int abpoa_msa1(abpoa_t *ab, abpoa_para_t *abpt, char *read_fn, FILE *out_fp) {
    if (!abpt->out_msa && !abpt->out_cons && !abpt->out_gfa) return 0;
    abpoa_reset(ab, abpt, 1024);
    if (abpt->incr_fn) abpoa_restore_graph(ab, abpt); /* abpoa_align.c:477 */
    abpoa_seq_t *abs = ab->abs;
    int exist_n_seq = abs->n_seq;

    abamd_fx_t *fx = abamd_fx_open(read_fn);
    int i, j, n_seq = abamd_read_seq(abs, fx);
    abamd_fx_close(fx);

    if (abpt->sort_input_seq) sort_seq_by_length(abs, exist_n_seq, n_seq);

    int max_len = 0;
    for (i = 0; i < abs->n_seq; ++i)
        if (abs->seq[i].l > max_len) max_len = abs->seq[i].l;

    uint8_t **seqs = (uint8_t**)abamd_malloc((size_t)n_seq * sizeof(uint8_t*));
    int *seq_lens = (int*)abamd_malloc((size_t)n_seq * sizeof(int));
    int **weights = (int**)abamd_malloc((size_t)n_seq * sizeof(int*));
    for (i = 0; i < n_seq; ++i) {
        seq_lens[i] = abs->seq[exist_n_seq + i].l;
        seqs[i] = (uint8_t*)abamd_malloc((size_t)seq_lens[i]);
        weights[i] = (int*)abamd_malloc((size_t)seq_lens[i] * sizeof(int));
        for (j = 0; j < seq_lens[i]; ++j)
            seqs[i][j] = (uint8_t)ab_amd_char26_table[(int)abs->seq[exist_n_seq + i].s[j]];
        if (abpt->use_qv && abs->qual[exist_n_seq + i].l > 0) {
            for (j = 0; j < seq_lens[i]; ++j) weights[i][j] = (int)abs->qual[exist_n_seq + i].s[j] - 32;
        } else {
            for (j = 0; j < seq_lens[i]; ++j) weights[i][j] = 1;
        }
    }
    if ((abpt->disable_seeding && abpt->progressive_poa == 0) || abpt->align_mode != ABPOA_GLOBAL_MODE)
        abamd_poa(ab, abpt, seqs, weights, seq_lens, exist_n_seq, n_seq);
    else
        abamd_seeded_poa(ab, abpt, seqs, weights, seq_lens, exist_n_seq, n_seq, max_len);
    abpoa_output(ab, abpt, out_fp);
    for (i = 0; i < n_seq; ++i) { free(seqs[i]); free(weights[i]); }
    free(seqs); free(weights); free(seq_lens);
    return 0;
}

