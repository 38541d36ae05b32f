/* abpoa_amd — MI355X-native partial order alignment library.
 *
 * Public C ABI. This header mirrors the reference abPOA public interface
 * (yangao07/abPOA include/abpoa.h) so that the library is a drop-in for the
 * hot path: struct layouts and function signatures are field-for-field
 * compatible with the reference (each declaration cites the reference line it
 * replaces), while every implementation behind them is written from scratch
 * for MI355X (HIP/CDNA4 aligner core; clean C host).
 *
 * The aligner seam exported here is exactly the one the reference resolves
 * per-ISA at runtime (abpoa_align_simd.h:11-12, abpoa_dispatch_simd.c:59-82):
 *   simd_abpoa_align_sequence_to_graph / _to_subgraph
 * so reference host code links against this library unchanged.
 */
#ifndef ABPOA_AMD_H
#define ABPOA_AMD_H

#include <stdint.h>
#include <stdio.h>

/* alignment modes (abpoa.h:6-8) */
#define ABPOA_GLOBAL_MODE 0
#define ABPOA_LOCAL_MODE  1
#define ABPOA_EXTEND_MODE 2

/* gap cost models (abpoa.h:12-14) */
#define ABPOA_LINEAR_GAP 0
#define ABPOA_AFFINE_GAP 1
#define ABPOA_CONVEX_GAP 2

/* adaptive band defaults: band half-width = wb + wf*qlen (abpoa.h:16-17) */
#define ABPOA_EXTRA_B 10
#define ABPOA_EXTRA_F 0.01

/* graph-CIGAR operations (abpoa.h:19-25) */
#define ABPOA_CIGAR_STR "MIDXSH"
#define ABPOA_CMATCH     0
#define ABPOA_CINS       1
#define ABPOA_CDEL       2
#define ABPOA_CDIFF      3
#define ABPOA_CSOFT_CLIP 4
#define ABPOA_CHARD_CLIP 5

/* the two virtual terminals of every partial order graph (abpoa.h:27-28) */
#define ABPOA_SRC_NODE_ID  0
#define ABPOA_SINK_NODE_ID 1

/* output selection (abpoa.h:30-35) */
#define ABPOA_OUT_CONS     0
#define ABPOA_OUT_MSA      1
#define ABPOA_OUT_CONS_MSA 2
#define ABPOA_OUT_GFA      3
#define ABPOA_OUT_CONS_GFA 4
#define ABPOA_OUT_CONS_FQ  5

/* consensus algorithms (abpoa.h:37-38) */
#define ABPOA_HB 0
#define ABPOA_MF 1

/* verbosity (abpoa.h:40-43) */
#define ABPOA_NONE_VERBOSE 0
#define ABPOA_INFO_VERBOSE 1
#define ABPOA_DEBUG_VERBOSE 2
#define ABPOA_LONG_DEBUG_VERBOSE 3

/* 64-bit packed graph-CIGAR word (abpoa.h:45-50):
 *   M/X: node_id<<34 | query_id<<4 | op
 *   I/S/H: query_id<<34 | len<<4 | op
 *   D:   node_id<<34 | len<<4 | op            */
#define abpoa_cigar_t uint64_t

#ifdef __cplusplus
extern "C" {
#endif

/* alignment result (abpoa.h:57-64) */
typedef struct {
    int n_cigar, m_cigar; abpoa_cigar_t *graph_cigar;
    int node_s, node_e, query_s, query_e;
    int n_aln_bases, n_matched_bases;
    int32_t best_score;
} abpoa_res_t;

/* parameters (abpoa.h:66-89); field order preserved for ABI compatibility */
typedef struct {
    int m; int *mat; char *mat_fn;
    int use_score_matrix;
    int match, max_mat, mismatch, min_mis, gap_open1, gap_open2, gap_ext1, gap_ext2; int inf_min;
    int sort_input_seq;
    int inc_path_score;
    int k, w, min_w;                 /* minimizer seeding */
    int wb; float wf;                /* adaptive band extra width */
    int zdrop, end_bonus;
    uint8_t ret_cigar:1, rev_cigar:1, out_msa:1, out_cons:1, out_gfa:1, out_fq:1, use_read_ids:1, amb_strand:1;
    uint8_t sub_aln:1, use_qv:1, disable_seeding:1, progressive_poa:1, put_gap_on_right:1, put_gap_at_end:1;
    char *incr_fn, *out_pog;
    int align_mode, gap_mode, max_n_cons, cons_algrm;
    double min_freq;
    int verbose;
    int batch_index;
} abpoa_para_t;

/* one graph node (abpoa.h:91-105) */
typedef struct {
    int node_id;
    int in_edge_n, in_edge_m, *in_id; int *in_edge_weight;
    int out_edge_n, out_edge_m, *out_id; int *out_edge_weight;
    int *read_weight, n_read, m_read, n_span_read;
    uint64_t **read_ids; int read_ids_n;
    int aligned_node_n, aligned_node_m, *aligned_node_id;
    uint8_t base;
} abpoa_node_t;

/* the DAG (abpoa.h:107-112) */
typedef struct {
    abpoa_node_t *node; int node_n, node_m, index_rank_m;
    int *index_to_node_id;
    int *node_id_to_index, *node_id_to_max_pos_left, *node_id_to_max_pos_right, *node_id_to_max_remain, *node_id_to_msa_rank;
    uint8_t is_topological_sorted:1, is_called_cons:1, is_set_msa_rank:1;
} abpoa_graph_t;

/* consensus / MSA results (abpoa.h:114-124) */
typedef struct {
    int n_cons, n_seq, msa_len;
    int *clu_n_seq;
    int **clu_read_ids;
    int *cons_len;
    int **cons_node_ids;
    uint8_t **cons_base;
    uint8_t **msa_base;
    int **cons_cov;
    int **cons_phred_score;
} abpoa_cons_t;

typedef struct { int l, m; char *s; } abpoa_str_t;           /* abpoa.h:126-128 */

typedef struct {
    int n_seq, m_seq;
    abpoa_str_t *seq, *name, *comment, *qual;
    uint8_t *is_rc;
} abpoa_seq_t;                                               /* abpoa.h:130-134 */

/* device-side DP arena; opaque to callers (abpoa.h:136) */
typedef struct abpoa_simd_matrix_t abpoa_simd_matrix_t;

typedef struct {
    abpoa_graph_t *abg;
    abpoa_seq_t *abs;
    abpoa_simd_matrix_t *abm;
    abpoa_cons_t *abc;
} abpoa_t;                                                   /* abpoa.h:138-143 */

/* ---- public API, 1:1 with the reference (abpoa.h:146-226) ---- */
abpoa_para_t *abpoa_init_para(void);
void abpoa_set_mat_from_file(abpoa_para_t *abpt, char *mat_fn);
void abpoa_post_set_para(abpoa_para_t *abpt);
void abpoa_free_para(abpoa_para_t *abpt);

abpoa_t *abpoa_init(void);
void abpoa_free(abpoa_t *ab);

int abpoa_msa(abpoa_t *ab, abpoa_para_t *abpt, int n_seqs, char **seq_names, int *seq_lens, uint8_t **seqs, int **qual_weights, FILE *out_fp);
void abpoa_clean_msa_cons(abpoa_t *ab);
int abpoa_msa1(abpoa_t *ab, abpoa_para_t *abpt, char *read_fn, FILE *out_fp);
void abpoa_reset(abpoa_t *ab, abpoa_para_t *abpt, int qlen);
abpoa_t *abpoa_restore_graph(abpoa_t *ab, abpoa_para_t *abpt);

int abpoa_align_sequence_to_graph(abpoa_t *ab, abpoa_para_t *abpt, uint8_t *query, int qlen, abpoa_res_t *res);
void abpoa_subgraph_nodes(abpoa_t *ab, abpoa_para_t *abpt, int inc_beg, int inc_end, int *exc_beg, int *exc_end);
int abpoa_align_sequence_to_subgraph(abpoa_t *ab, abpoa_para_t *abpt, int beg_node_id, int end_node_id, uint8_t *query, int qlen, abpoa_res_t *res);

int abpoa_add_graph_node(abpoa_graph_t *abg, uint8_t base);
int abpoa_add_graph_edge(abpoa_graph_t *abg, int from_id, int to_id, int check_edge, int w, uint8_t add_read_id, uint8_t add_read_weight, int read_id, int read_ids_n, int tot_read_n);
int abpoa_add_graph_alignment(abpoa_t *ab, abpoa_para_t *abpt, uint8_t *query, int *weight, int qlen, int *qpos_to_node_id, abpoa_res_t res, int read_id, int tot_read_n, int inc_both_ends);
int abpoa_add_subgraph_alignment(abpoa_t *ab, abpoa_para_t *abpt, int beg_node_id, int end_node_id, uint8_t *query, int *weight, int qlen, int *qpos_to_node_id, abpoa_res_t res, int read_id, int tot_read_n, int inc_both_ends);

void abpoa_BFS_set_node_index(abpoa_graph_t *abg, int src_id, int sink_id);
void abpoa_BFS_set_node_remain(abpoa_graph_t *abg, int src_id, int sink_id);
void abpoa_topological_sort(abpoa_graph_t *abg, abpoa_para_t *abpt);

void abpoa_generate_consensus(abpoa_t *ab, abpoa_para_t *abpt);
void abpoa_output_fx_consensus(abpoa_t *ab, abpoa_para_t *abpt, FILE *out_fp);
void abpoa_generate_rc_msa(abpoa_t *ab, abpoa_para_t *abpt);
void abpoa_output_rc_msa(abpoa_t *ab, abpoa_para_t *abpt, FILE *out_fp);
void abpoa_generate_gfa(abpoa_t *ab, abpoa_para_t *abpt, FILE *out_fp);
void abpoa_output(abpoa_t *ab, abpoa_para_t *abpt, FILE *out_fp);
void abpoa_dump_pog(abpoa_t *ab, abpoa_para_t *abpt);

/* ---- the drop-in aligner seam (abpoa_align_simd.h:11-12) ----
 * On a machine with an AMD GPU these run the HIP/CDNA4 core; with no GPU they
 * abort with a clear message (no silent CPU fallback). */
int simd_abpoa_align_sequence_to_graph(abpoa_t *ab, abpoa_para_t *abpt, uint8_t *query, int qlen, abpoa_res_t *res);
int simd_abpoa_align_sequence_to_subgraph(abpoa_t *ab, abpoa_para_t *abpt, int beg_node_id, int end_node_id, uint8_t *query, int qlen, abpoa_res_t *res);

/* ---- abpoa_amd extensions ---- */

/* Signature of an aligner implementation for the seam above. */
typedef int (*abpoa_amd_aligner_fn)(abpoa_t *ab, abpoa_para_t *abpt,
        int beg_node_id, int end_node_id, uint8_t *query, int qlen, abpoa_res_t *res);

/* TEST HOOK: replace the aligner implementation (used by tests to inject the
 * CPU oracle from oracle/liboracle.so on GPU-less machines). Prints a loud
 * notice; never called on the product path. Pass NULL to restore the GPU core. */
void abpoa_amd_set_test_aligner(abpoa_amd_aligner_fn fn);

/* Batched driver: run POA for many independent read-sets concurrently so the
 * GPU sees thousands of in-flight alignments. Processes sets[0..n_sets) where
 * each set is n_seqs sequences of seq_lens[s][i] bases (0..m-1 codes); writes
 * per-set consensus through the callback. Used by bench.py and the multi-GPU
 * sharding layer. */
typedef void (*abpoa_amd_cons_cb)(int set_idx, const abpoa_cons_t *cons, void *user);
int abpoa_amd_msa_batch(abpoa_para_t *abpt, int n_sets, const int *n_seqs,
        const int *const *seq_lens, const uint8_t *const *const *seqs,
        abpoa_amd_cons_cb cb, void *user, int n_host_threads);

/* Aggregate counters since last reset (for roofline reporting):
 * total DP cells computed by the device core and total device-kernel
 * nanoseconds measured with HIP events on the library's stream. */
void abpoa_amd_get_stats(uint64_t *dp_cells, uint64_t *kernel_ns, uint64_t *n_launches);
/* algorithmic HBM bytes of the plane streams (stored planes x score width x
 * cells: convex 3, affine 3, linear 1), summed per launch at the launch's
 * actual score width */
void abpoa_amd_get_stats2(uint64_t *alg_bytes);
void abpoa_amd_reset_stats(void);

#ifdef __cplusplus
}
#endif

#endif /* ABPOA_AMD_H */
