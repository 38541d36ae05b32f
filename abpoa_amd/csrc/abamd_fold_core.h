/* Flat-array graph fold core (see abamd_fold_core.c). Device-layout graph
 * mutation, bit-exact with the pointer-based abamd_graph.c fold. */
#ifndef ABAMD_FOLD_CORE_H
#define ABAMD_FOLD_CORE_H

#include <stdint.h>

typedef struct flat_graph_t {
    int node_n, node_cap;
    uint8_t *base;               /* [node_cap] */
    int *n_read, *n_span_read;   /* [node_cap] */
    /* append-only per-direction edge pools; per-node head/tail chains keep
     * the pointer graph's append order */
    int edge_n_in, edge_n_out, edge_cap;
    int *in_head, *in_tail, *out_head, *out_tail;   /* [node_cap] */
    int *in_to, *in_w, *in_next;                    /* [edge_cap] */
    int *out_to, *out_w, *out_next;                 /* [edge_cap] */
    uint64_t *rid_pool; int rid_n;  /* per-OUT-edge read-id bitsets (words) */
    /* aligned (mismatch-column) groups as linked lists */
    int aln_n, aln_cap;
    int *aln_head;                                  /* [node_cap] */
    int *aln_id, *aln_next;                         /* [aln_cap] */
} flat_graph_t;

/* Device translation units (gpu_fold.hip) include this header for the
 * flat_graph_t layout only: the function bodies there are __device__ builds
 * of abamd_fold_core.inc, and host declarations of the same names would
 * clash with them. */
#ifndef ABAMD_FC_NO_HOST_DECLS
#ifdef __cplusplus
extern "C" {
#endif

void abamd_flat_init(flat_graph_t *fg, int node_cap, int edge_cap, int aln_cap, int rid_n);
void abamd_flat_free(flat_graph_t *fg);
void abamd_flat_sort_adjacency(flat_graph_t *fg);
void abamd_flat_topo_index(const flat_graph_t *fg, int *index_to_node_id,
                           int *node_id_to_index, int *scratch);
void abamd_flat_remain(const flat_graph_t *fg, int *max_remain, int *scratch);
void abamd_flat_update_n_span(flat_graph_t *fg, const int *index_to_node_id,
                              const int *node_id_to_index, int inc_both_ends);
void abamd_flat_msa_rank(const flat_graph_t *fg, int *msa_rank_out, int *scratch);
int abamd_flat_build_rows(const flat_graph_t *fg, const int *index_to_node_id,
                          const int *node_id_to_index, const int *max_remain,
                          int use_remain, uint8_t *row_base, int *row_node_id,
                          int *pre_off, int *out_off, int *remain,
                          int *pre_idx, int *out_idx);
void abamd_flat_apply_alignment(flat_graph_t *fg, int beg_node_id, int end_node_id,
                                const uint8_t *seq, const int *weight, int seq_l,
                                int *qpos_to_node_id, int n_cigar, const abpoa_cigar_t *cig,
                                int read_id, int add_read_id, int inc_both_ends);

/* single-cluster heaviest-bundle consensus over the flat layout
 * (abamd_cons_core.inc); returns cons_len */
int abamd_flat_hb_consensus(const flat_graph_t *fg, int n_seq,
                            int *scratch, int *score, int *max_out,
                            int *cons_id, uint8_t *cons_base,
                            int *cons_cov, int *cons_phred);

#ifdef ABPOA_AMD_H
/* Rebuild a pointer graph from a (host copy of a) flat graph and topo-sort
 * it — the device-resident batch driver's hand-off to the host consensus
 * path (abamd_graph.c). `ab` must be fresh. All node arrays are carved from
 * ONE returned slab; call abamd_graph_arena_release(ab, slab) before
 * abpoa_free(ab). */
void *abamd_graph_from_flat(abpoa_t *ab, const flat_graph_t *fg, abpoa_para_t *abpt, int read_ids_n,
                            const int *i2n, const int *n2i, const int *remain);
void abamd_graph_arena_release(abpoa_t *ab, void *slab);
#endif

#ifdef __cplusplus
}
#endif
#endif /* ABAMD_FC_NO_HOST_DECLS */

#endif
