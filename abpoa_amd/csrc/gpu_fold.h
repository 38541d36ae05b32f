/* Shared host<->device layout for the per-round device graph fold
 * (gpu_fold.hip :: abamd_fold_round_kernel).
 *
 * One job = one read set: consume the DP kernel's device-resident CIGAR,
 * mutate the set's device-resident flat graph (abamd_fold_core.inc — the
 * algorithm is CPU-twin-proven bit-equal to the pointer graph fold,
 * reference abpoa_graph.c:689-774), then re-derive topo index
 * (abpoa_graph.c:322-357), weight-sorted adjacency, remain BFS, n_span, and
 * materialize the next round's DP-row CSR (mirrors pack_job with the
 * identity row map). The batch driver (gpu_batch_resident.cpp) keeps every
 * graph resident in HBM across all rounds: no host fold, no per-round
 * repack, no per-round graph H2D.
 */
#ifndef ABAMD_GPU_FOLD_H
#define ABAMD_GPU_FOLD_H

#include <stdint.h>
#include "abamd_fold_core.h"   /* flat_graph_t */
#include "gpu_core.h"          /* abamd_gpu_res_t */

#ifdef __cplusplus
extern "C" {
#endif

enum {
    ABAMD_FOLD_OK = 0,
    ABAMD_FOLD_SKIPPED_DP = 1,   /* DP job failed (e.g. arena overflow): graph untouched */
    ABAMD_FOLD_POOL_OVERFLOW = 2,/* pre-checked pool exhaustion: graph untouched */
    ABAMD_FOLD_NOOP = 3          /* empty CIGAR on an existing graph: reference
                                    folds nothing (graph + derived state keep) */
};

/* per-set record the host reads back after each fold launch */
typedef struct {
    int32_t status;
    int32_t node_n, edge_n_in, edge_n_out, aln_n;
    int32_t n_rows, n_pre, n_out; /* DP-row CSR sizes for the next round */
} abamd_fold_out_t;

/* one fold job; all pointers are device pointers into the set's slab */
typedef struct {
    flat_graph_t g;              /* counters = host mirror at launch time */
    uint64_t *cigar;             /* DP output, BACKTRACK order (kernel reverses) */
    const abamd_gpu_res_t *dp_res; /* NULL for the first read (chain build) */
    const uint8_t *seq;
    const int *weight;           /* shared all-ones buffer on this path */
    int seq_l;
    int read_id, add_read_id;
    int *index_to_node_id, *node_id_to_index, *max_remain, *scratch;
    /* DP-row CSR outputs for the next round's aligner launch */
    uint8_t *row_base;
    int *row_node_id, *pre_off, *out_off, *row_remain, *pre_idx, *out_idx;
    int use_remain;              /* abpt->wb >= 0 || abpt->zdrop > 0 */
    int m;                       /* alphabet size: bounds the aligned-pool pre-check */
    abamd_fold_out_t *out;
} abamd_fold_round_job_t;

/* launcher (gpu_fold.hip); stream is a hipStream_t */
void abamd_launch_fold_round(const abamd_fold_round_job_t *dev_jobs, int n_jobs, void *stream);

#ifdef __cplusplus
}
#endif

#endif
