/* Device fold kernel skeleton (round-2): one block per read set, the
 * order-sensitive graph mutation on lane 0 over the flat-array layout whose
 * algorithm is CPU-twin-proven (abamd_fold_core.inc / tests/test_fold_twin).
 *
 * NOT linked into the product yet — `make fold-kernel-check` cross-compiles
 * it for gfx950 so the device build of the shared bodies stays green from
 * round 1 onward. Round 2 wires it into the batch driver: after each DP
 * launch the per-set CIGARs are already in device memory, so the fold, the
 * weight sort, the topo-index BFS and the remain BFS all run here, and the
 * per-round host fold + repack + ~200 MB/launch H2D disappear.
 */
#include <hip/hip_runtime.h>
#include <stdint.h>

typedef uint64_t abpoa_cigar_t;
#define ABPOA_CMATCH 0
#define ABPOA_CINS 1
#define ABPOA_CDEL 2
#define ABPOA_CDIFF 3
#define ABPOA_CSOFT_CLIP 4
#define ABPOA_CHARD_CLIP 5
#define ABPOA_SRC_NODE_ID 0
#define ABPOA_SINK_NODE_ID 1

/* flat_graph_t mirrors abamd_fold_core.h; device pointers into a per-set
 * slab carved by the (round-2) driver */
typedef struct {
    int node_n, node_cap;
    uint8_t *base;
    int *n_read, *n_span_read;
    int edge_n_in, edge_n_out, edge_cap;
    int *in_head, *in_tail, *out_head, *out_tail;
    int *in_to, *in_w, *in_next;
    int *out_to, *out_w, *out_next;
    uint64_t *rid_pool; int rid_n;
    int aln_n, aln_cap;
    int *aln_head;
    int *aln_id, *aln_next;
} flat_graph_t;

/* device build of the shared bodies: pool exhaustion / non-DAG graphs are
 * driver bugs — trap the device, matching the host build's abort */
#define ABAMD_FC_FN __device__
#define ABAMD_FC_FAIL(msg) do { abort(); } while (0)
#define ABAMD_FC_FAIL_RET(val, msg) do { abort(); } while (0)
#include "abamd_fold_core.inc"

/* per-set job descriptor the round-2 driver fills */
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
