/* Device graph fold: the order-sensitive POA graph mutation + derived passes
 * on the GPU, over the flat-array layout whose algorithm is CPU-twin-proven
 * (abamd_fold_core.inc / tests/test_fold_twin.py, validated byte-exact on
 * MI355X by tests/test_gpu_parity.py::test_device_fold_kernel).
 *
 * Two kernels:
 *   - abamd_fold_kernel: the round-1 validation kernel (one explicit job
 *     descriptor per launch; used by abamd_fold_gpu_test.cpp).
 *   - abamd_fold_round_kernel: the production per-round kernel
 *     (gpu_batch_resident.cpp): reads the DP result + CIGAR straight from
 *     device memory, pre-checks pool capacity (no partial mutation on
 *     overflow), folds, re-derives topo/sort/remain/n_span, and materializes
 *     the next round's DP-row CSR — the whole read->graph->next-read cycle
 *     stays on the GPU.
 *
 * Mutation semantics match abpoa_add_graph_alignment (abpoa_graph.c:689-774)
 * + abpoa_topological_sort (:322-357); the serial passes run on lane 0 (the
 * graph walk is pointer-chasing over an L2-resident working set; cross-set
 * parallelism — one workgroup per read set, hundreds of sets per launch —
 * fills the chip).
 */
#include <hip/hip_runtime.h>
#include <stdint.h>

#define ABAMD_FC_NO_HOST_DECLS
#include "gpu_fold.h"

typedef uint64_t abpoa_cigar_t;
#define ABPOA_CMATCH 0
#define ABPOA_CINS 1
#define ABPOA_CDEL 2
#define ABPOA_CDIFF 3
#define ABPOA_CSOFT_CLIP 4
#define ABPOA_CHARD_CLIP 5
#define ABPOA_SRC_NODE_ID 0
#define ABPOA_SINK_NODE_ID 1

/* device build of the shared bodies: pool exhaustion / non-DAG graphs are
 * driver bugs (capacity is pre-checked before any mutation) — trap the
 * device, matching the host build's abort */
#define ABAMD_FC_FN __device__
#define ABAMD_FC_FAIL(msg) do { abort(); } while (0)
#define ABAMD_FC_FAIL_RET(val, msg) do { abort(); } while (0)
#include "abamd_fold_core.inc"

/* ------------------------------------------------------------------ */
/* Round-1 validation kernel (kept: the GPU twin test drives it).      */
/* ------------------------------------------------------------------ */

typedef struct {
    flat_graph_t g;
    const uint64_t *cigar; int n_cigar;
    const uint8_t *seq; const int *weight; int seq_l;
    int *qpos_to_node_id;
    int read_id, add_read_id;
    int *index_to_node_id, *node_id_to_index, *max_remain, *scratch;
    int *msa_rank; /* optional; skipped when NULL */
} fold_job_t;

extern "C" __global__ void abamd_fold_kernel(fold_job_t *jobs, int n_jobs) {
    int j = blockIdx.x;
    if (j >= n_jobs) return;
    fold_job_t *job = &jobs[j];
    if (threadIdx.x == 0) {
        abamd_flat_apply_alignment(&job->g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID,
                                   job->seq, job->weight, job->seq_l, job->qpos_to_node_id,
                                   job->n_cigar, job->cigar, job->read_id,
                                   job->add_read_id, 1);
        abamd_flat_topo_index(&job->g, job->index_to_node_id, job->node_id_to_index,
                              job->scratch);
        abamd_flat_sort_adjacency(&job->g);
        abamd_flat_remain(&job->g, job->max_remain, job->scratch);
        abamd_flat_update_n_span(&job->g, job->index_to_node_id,
                                 job->node_id_to_index, 1);
        if (job->msa_rank)
            abamd_flat_msa_rank(&job->g, job->msa_rank, job->scratch);
    }
}

/* ------------------------------------------------------------------ */
/* Production per-round kernel.                                        */
/* ------------------------------------------------------------------ */

extern "C" __global__ __launch_bounds__(64)
void abamd_fold_round_kernel(abamd_fold_round_job_t *jobs, int n_jobs) {
    const int j = blockIdx.x;
    if (j >= n_jobs) return;
    abamd_fold_round_job_t *job = &jobs[j];
    flat_graph_t *g = &job->g;
    const int lane = threadIdx.x;

    int n_cigar = 0;
    if (job->dp_res) {
        if (job->dp_res->status != ABAMD_JOB_OK) {
            /* DP failed (arena overflow, retried by the host): leave the
             * graph untouched; the host refolds after the DP retry */
            if (lane == 0) {
                job->out->status = ABAMD_FOLD_SKIPPED_DP;
                job->out->node_n = g->node_n;
                job->out->edge_n_in = g->edge_n_in; job->out->edge_n_out = g->edge_n_out;
                job->out->aln_n = g->aln_n;
            }
            return;
        }
        n_cigar = job->dp_res->n_cigar;
        if (n_cigar == 0 && g->node_n > 2) {
            /* empty alignment on an existing graph: the reference fold
             * returns without touching the graph OR its derived state
             * (abpoa_graph.c: n_cigar==0 early return, before the topo
             * re-sort and the n_span update) — the host keeps its previous
             * CSR mirror */
            if (lane == 0) {
                job->out->status = ABAMD_FOLD_NOOP;
                job->out->node_n = g->node_n;
                job->out->edge_n_in = g->edge_n_in; job->out->edge_n_out = g->edge_n_out;
                job->out->aln_n = g->aln_n;
            }
            return;
        }
    }

    /* capacity pre-check BEFORE any mutation, using per-fold worst cases:
     * <= seq_l new nodes, <= seq_l+2 new edges per direction, and per new
     * node <= 2*(m-1)+2 aligned-pair entries. On failure the graph is
     * untouched and the host re-allocates bigger pools and relaunches. */
    {
        long need_nodes = (long)g->node_n + job->seq_l + 2;
        long need_edges_in = (long)g->edge_n_in + job->seq_l + 2;
        long need_edges_out = (long)g->edge_n_out + job->seq_l + 2;
        long need_aln = (long)g->aln_n + 2L * job->m * job->seq_l + 2;
        if (need_nodes > g->node_cap || need_edges_in > g->edge_cap ||
            need_edges_out > g->edge_cap || need_aln > g->aln_cap) {
            if (lane == 0) {
                job->out->status = ABAMD_FOLD_POOL_OVERFLOW;
                job->out->node_n = g->node_n;
                job->out->edge_n_in = g->edge_n_in; job->out->edge_n_out = g->edge_n_out;
                job->out->aln_n = g->aln_n;
            }
            return;
        }
    }

    /* the DP backtrack emits the CIGAR back-to-front; the fold consumes it
     * front-to-back — block-parallel in-place reversal */
    if (n_cigar > 1) {
        for (int a = lane; a < n_cigar / 2; a += 64) {
            uint64_t t = job->cigar[a];
            job->cigar[a] = job->cigar[n_cigar - 1 - a];
            job->cigar[n_cigar - 1 - a] = t;
        }
    }
    __syncthreads();

    if (lane != 0) return;

    abamd_flat_apply_alignment(g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID,
                               job->seq, job->weight, job->seq_l, NULL,
                               n_cigar, job->cigar, job->read_id,
                               job->add_read_id, 1);
    abamd_flat_topo_index(g, job->index_to_node_id, job->node_id_to_index, job->scratch);
    abamd_flat_sort_adjacency(g);
    if (job->use_remain)
        abamd_flat_remain(g, job->max_remain, job->scratch);
    abamd_flat_update_n_span(g, job->index_to_node_id, job->node_id_to_index, 1);

    int n_rows = abamd_flat_build_rows(g, job->index_to_node_id, job->node_id_to_index,
                                       job->max_remain, job->use_remain,
                                       job->row_base, job->row_node_id,
                                       job->pre_off, job->out_off, job->row_remain,
                                       job->pre_idx, job->out_idx);
    abamd_fold_out_t *o = job->out;
    o->status = ABAMD_FOLD_OK;
    o->node_n = g->node_n;
    o->edge_n_in = g->edge_n_in; o->edge_n_out = g->edge_n_out;
    o->aln_n = g->aln_n;
    o->n_rows = n_rows;
    o->n_pre = job->pre_off[n_rows];
    o->n_out = job->out_off[n_rows];
}

extern "C" void abamd_launch_fold_round(const abamd_fold_round_job_t *dev_jobs,
                                        int n_jobs, void *stream) {
    hipLaunchKernelGGL(abamd_fold_round_kernel, dim3(n_jobs), dim3(64), 0,
                       (hipStream_t)stream, (abamd_fold_round_job_t*)dev_jobs, n_jobs);
}
