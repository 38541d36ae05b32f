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
#include "abamd_cons_core.inc"

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

#ifdef ABAMD_KPROF
__device__ unsigned long long abamd_kprof_fold[16];
extern "C" void abamd_kprof_fold_fetch(unsigned long long *out) {
    (void)hipMemcpyFromSymbol(out, HIP_SYMBOL(abamd_kprof_fold), sizeof(abamd_kprof_fold));
}
extern "C" void abamd_kprof_fold_reset(void) {
    unsigned long long z[16] = {0};
    (void)hipMemcpyToSymbol(HIP_SYMBOL(abamd_kprof_fold), z, sizeof(z));
}
#define FKPROF_T(v) unsigned long long v = __builtin_readcyclecounter()
#define FKPROF_ACC(slot, a, b) if (lane == 0) atomicAdd(&abamd_kprof_fold[slot], (b) - (a))
#else
#define FKPROF_T(v)
#define FKPROF_ACC(slot, a, b)
#endif

/* ------------------------------------------------------------------ */
/* Wave-parallel variants of the order-insensitive derived passes.     */
/*                                                                     */
/* The mutation (apply) and the Kahn BFS stay serial on lane 0 — their */
/* ORDER is observable (queue order defines the topo index; the fold   */
/* order defines edge append order). Everything else is a pure         */
/* function of the graph whose outputs the serial twin provably        */
/* produces; these versions compute the same values with all 64 lanes  */
/* (per-node independence, wave prefix scans), cutting the per-set     */
/* fold latency that otherwise matches the DP kernel's whole wall.     */
/* Byte-parity with the host path is enforced end-to-end by            */
/* tests/test_batch_gpu.py (resident vs host-fold vs sequential CLI).  */
/* ------------------------------------------------------------------ */

#define FOLD_WAVE 64

/* per-node in-degrees into in_deg[]; replaces the serial count loop of
 * abamd_flat_topo_index so the lane-0 BFS starts from precomputed degrees */
__device__ static void dev_par_in_deg(const flat_graph_t *fg, int *in_deg, int lane) {
    for (int i = lane; i < fg->node_n; i += FOLD_WAVE) {
        int d = 0;
        for (int e = fg->in_head[i]; e != -1; e = fg->in_next[e]) ++d;
        in_deg[i] = d;
    }
}

/* Kahn BFS with whole-aligned-group queue entry (identical order to
 * abamd_flat_topo_index, which the twin test pins); in_deg precomputed */
__device__ static void dev_topo_bfs(const flat_graph_t *fg, int *index_to_node_id,
                                    int *node_id_to_index, int *in_deg, int *q) {
    int e, a, cur, index = 0, qh = 0, qt = 0;
    q[qt++] = 0; /* SRC */
    while (qh < qt) {
        cur = q[qh++];
    chain:
        index_to_node_id[index] = cur;
        node_id_to_index[cur] = index++;
        if (cur == 1 /* SINK */) return;
        /* chain fast path: when the queue is otherwise drained and cur has a
         * single out-edge whose target becomes ready, the general code would
         * push exactly that node and dequeue it next — skip the queue */
        e = fg->out_head[cur];
        if (qh == qt && e != -1 && fg->out_next[e] == -1) {
            int out = fg->out_to[e];
            if (in_deg[out] == 1 && fg->aln_head[out] == -1) {
                in_deg[out] = 0;
                cur = out;
                goto chain;
            }
        }
        for (e = fg->out_head[cur]; e != -1; e = fg->out_next[e]) {
            int out = fg->out_to[e];
            if (--in_deg[out] == 0) {
                int ready = 1;
                for (a = fg->aln_head[out]; a != -1; a = fg->aln_next[a])
                    if (in_deg[fg->aln_id[a]] != 0) { ready = 0; break; }
                if (!ready) continue;
                q[qt++] = out;
                for (a = fg->aln_head[out]; a != -1; a = fg->aln_next[a])
                    q[qt++] = fg->aln_id[a];
            }
        }
    }
    abort(); /* not a connected DAG: driver bug */
}

/* weight-descending adjacency sort — per-node work is independent, so nodes
 * stride across lanes. The host reference's pairwise-swap pass with strict <
 * (abamd_flat_sort_adjacency) is a STABLE selection sort: position j ends
 * with the first-occurring maximum of the remaining edges. Typical degrees
 * (<= 64: at most one edge per read plus merges) sort via a small private
 * array; larger degrees fall back to an in-place stable selection over the
 * chain — identical order, no O(KB) per-lane scratch (a 1024-int array here
 * forced a multi-GB device scratch pool that OOM'd full-HBM runs). */
#define SORT_LOC 64
__device__ static void dev_sort_chain(int *head, int *tail, int *next,
                                      const int *w) {
    int loc[SORT_LOC];
    int e, n = 0;
    for (e = *head; e != -1; e = next[e]) {
        if (n < SORT_LOC) loc[n] = e;
        ++n;
    }
    if (n <= 1) return;
    if (n <= SORT_LOC) {
        int j, k;
        for (j = 0; j < n - 1; ++j)
            for (k = j + 1; k < n; ++k)
                if (w[loc[j]] < w[loc[k]]) {
                    int t = loc[j]; loc[j] = loc[k]; loc[k] = t;
                }
        *head = loc[0];
        for (j = 0; j + 1 < n; ++j) next[loc[j]] = loc[j + 1];
        next[loc[n - 1]] = -1;
        *tail = loc[n - 1];
        return;
    }
    /* rare big-degree fallback: stable selection by chain re-linking */
    int out_head = -1, out_tail = -1;
    while (*head != -1) {
        int best = *head, prev_best = -1, prev = *head;
        for (e = next[*head]; e != -1; prev = e, e = next[e])
            if (w[e] > w[best]) { best = e; prev_best = prev; }
        if (prev_best == -1) *head = next[best];
        else next[prev_best] = next[best];
        if (out_tail == -1) out_head = best;
        else next[out_tail] = best;
        out_tail = best;
    }
    next[out_tail] = -1;
    *head = out_head;
    *tail = out_tail;
}

__device__ static void dev_par_sort_adjacency(flat_graph_t *fg, int lane) {
    for (int i = lane; i < fg->node_n; i += FOLD_WAVE) {
        dev_sort_chain(&fg->in_head[i], &fg->in_tail[i], fg->in_next, fg->in_w);
        dev_sort_chain(&fg->out_head[i], &fg->out_tail[i], fg->out_next, fg->out_w);
    }
}

/* max_remain: every successor has a HIGHER topo index, so scanning indices
 * high->low makes each node's value a pure function of already-final
 * successors — the same values the twin's reverse BFS computes (max-weight
 * out-edge, strict >, chain-order tie-break). Wave-parallel: 64 indices per
 * chunk; the (parallel) chain walks find each node's max-weight successor,
 * then a shuffle wavefront resolves intra-chunk dependence (successor lanes
 * are strictly higher, so it converges in <= 64 steps; out-of-chunk
 * successors were finalized by earlier chunks). */
__device__ static void dev_par_remain(const flat_graph_t *fg, const int *index_to_node_id,
                                      const int *node_id_to_index, int sink_index,
                                      int *max_remain, int lane) {
    for (int base = (sink_index / FOLD_WAVE) * FOLD_WAVE; base >= 0; base -= FOLD_WAVE) {
        const int k = base + lane;
        int id = -1, src = 0, val = 0;
        bool have = true;
        if (k <= sink_index) {
            id = index_to_node_id[k];
            if (id == 1 /* SINK */) val = -1;
            else {
                int max_w = INT32_MIN, max_id = -1;
                for (int e = fg->out_head[id]; e != -1; e = fg->out_next[e])
                    if (fg->out_w[e] > max_w) { max_w = fg->out_w[e]; max_id = fg->out_to[e]; }
                const int sidx = node_id_to_index[max_id];
                if (sidx > base + (FOLD_WAVE - 1)) val = max_remain[max_id] + 1;
                else { src = sidx - base; have = false; }
            }
        }
        unsigned long long done = __ballot(have ? 1 : 0);
        int guard = 0;
        while (done != ~0ull) {
            const int sval = __shfl(val, src);
            if (!have && ((done >> src) & 1)) { val = sval + 1; have = true; }
            done = __ballot(have ? 1 : 0);
            /* each step resolves at least the highest unresolved lane
             * (successor indices are strictly higher); > 64 steps = bug */
            if (++guard > FOLD_WAVE) abort();
        }
        if (k <= sink_index) max_remain[id] = val;
    }
}

/* span counts: disjoint increments across lanes */
__device__ static void dev_par_n_span(flat_graph_t *fg, const int *index_to_node_id,
                                      int src_index, int sink_index, int lane) {
    for (int i = src_index + 1 + lane; i < sink_index; i += FOLD_WAVE)
        fg->n_span_read[index_to_node_id[i]] += 1;
    if (lane == 0) {
        fg->n_span_read[0] += 1;
        fg->n_span_read[1] += 1;
    }
}

/* DP-row CSR materialization (mirrors abamd_flat_build_rows / pack_job):
 * lane-parallel degree count, wave prefix scan into the offset arrays,
 * lane-parallel fill. deg scratch: scratch[0..nc) = pre, scratch[nc..2nc) =
 * out (the BFS queue is done with it by now). Returns n_rows on lane 0. */
__device__ static int dev_par_build_rows(const flat_graph_t *fg,
                                         const int *index_to_node_id,
                                         const int *node_id_to_index,
                                         const int *max_remain, int use_remain,
                                         uint8_t *row_base, int *row_node_id,
                                         int *pre_off, int *out_off, int *remain,
                                         int *pre_idx, int *out_idx,
                                         int *scratch, int node_cap, int lane) {
    const int n_rows = node_id_to_index[1 /* SINK */] + 1;
    int *dpre = scratch, *dout = scratch + node_cap;
    for (int r = lane; r < n_rows; r += FOLD_WAVE) {
        int nid = index_to_node_id[r];
        row_base[r] = fg->base[nid];
        row_node_id[r] = nid;
        remain[r] = use_remain ? max_remain[nid] : 0;
        int e, dp = 0, dq = 0;
        if (r > 0)
            for (e = fg->in_head[nid]; e != -1; e = fg->in_next[e]) ++dp;
        for (e = fg->out_head[nid]; e != -1; e = fg->out_next[e]) ++dq;
        dpre[r] = dp; dout[r] = dq;
    }
    __syncthreads();
    /* wave prefix scan: lane l owns rows [l*B, (l+1)*B) */
    const int B = (n_rows + FOLD_WAVE) / FOLD_WAVE; /* +1 slot for the totals */
    int sum_pre = 0, sum_out = 0;
    {
        const int r0 = lane * B, r1 = min(r0 + B, n_rows);
        for (int r = r0; r < r1; ++r) { sum_pre += dpre[r]; sum_out += dout[r]; }
        /* exclusive scan of per-lane sums */
        int ex_pre = sum_pre, ex_out = sum_out;
        for (int s = 1; s < FOLD_WAVE; s <<= 1) {
            int vp = __shfl_up(ex_pre, s), vo = __shfl_up(ex_out, s);
            if (lane >= s) { ex_pre += vp; ex_out += vo; }
        }
        ex_pre -= sum_pre; ex_out -= sum_out; /* inclusive -> exclusive */
        int ap = ex_pre, ao = ex_out;
        for (int r = r0; r < r1; ++r) {
            pre_off[r] = ap; out_off[r] = ao;
            ap += dpre[r]; ao += dout[r];
        }
        if (r1 == n_rows && r0 <= n_rows) { pre_off[n_rows] = ap; out_off[n_rows] = ao; }
    }
    __syncthreads();
    for (int r = lane; r < n_rows; r += FOLD_WAVE) {
        int nid = index_to_node_id[r];
        int e, np = pre_off[r], no = out_off[r];
        if (r > 0)
            for (e = fg->in_head[nid]; e != -1; e = fg->in_next[e])
                pre_idx[np++] = node_id_to_index[fg->in_to[e]];
        for (e = fg->out_head[nid]; e != -1; e = fg->out_next[e])
            out_idx[no++] = node_id_to_index[fg->out_to[e]];
    }
    return n_rows;
}

/* ------------------------------------------------------------------ */
/* Production per-round kernel.                                        */
/* ------------------------------------------------------------------ */

extern "C" __global__ __launch_bounds__(64)
void abamd_fold_round_kernel(abamd_fold_round_job_t *jobs, int n_jobs) {
    const int j = blockIdx.x;
    if (j >= n_jobs) return;
    abamd_fold_round_job_t *job = &jobs[j];
    /* by-VALUE graph descriptor: pool pointers and counters live in
     * registers — through job->g every chain step would re-load the pool
     * base pointer from global memory (stores to the pools could alias the
     * jobs array in the compiler's view) */
    flat_graph_t gl = job->g;
    flat_graph_t *g = &gl;
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

    FKPROF_T(ft0);
    if (lane == 0) {
        abamd_flat_apply_alignment(g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID,
                                   job->seq, job->weight, job->seq_l, NULL,
                                   n_cigar, job->cigar, job->read_id,
                                   job->add_read_id, 1);
        /* publish the mutated counters so every lane's local copy agrees */
        job->g.node_n = gl.node_n;
        job->g.edge_n_in = gl.edge_n_in;
        job->g.edge_n_out = gl.edge_n_out;
        job->g.aln_n = gl.aln_n;
    }
    __syncthreads();
    gl.node_n = job->g.node_n;
    gl.edge_n_in = job->g.edge_n_in;
    gl.edge_n_out = job->g.edge_n_out;
    gl.aln_n = job->g.aln_n;
    FKPROF_T(ft1);
    FKPROF_ACC(1, ft0, ft1);
    dev_par_in_deg(g, job->scratch, lane);
    __syncthreads();
    FKPROF_T(ft2);
    FKPROF_ACC(2, ft1, ft2);
    if (lane == 0)
        dev_topo_bfs(g, job->index_to_node_id, job->node_id_to_index,
                     job->scratch, job->scratch + g->node_cap);
    __syncthreads();
    FKPROF_T(ft3);
    FKPROF_ACC(3, ft2, ft3);
    dev_par_sort_adjacency(g, lane);
    __syncthreads();
    FKPROF_T(ft4);
    FKPROF_ACC(4, ft3, ft4);
    const int sink_index = job->node_id_to_index[1 /* SINK */];
    if (job->use_remain)
        dev_par_remain(g, job->index_to_node_id, job->node_id_to_index, sink_index,
                       job->max_remain, lane);
    __syncthreads();
    dev_par_n_span(g, job->index_to_node_id, job->node_id_to_index[0 /* SRC */],
                   sink_index, lane);
    __syncthreads();
    FKPROF_T(ft5);
    FKPROF_ACC(5, ft4, ft5);
    int n_rows = dev_par_build_rows(g, job->index_to_node_id, job->node_id_to_index,
                                    job->max_remain, job->use_remain,
                                    job->row_base, job->row_node_id,
                                    job->pre_off, job->out_off, job->row_remain,
                                    job->pre_idx, job->out_idx,
                                    job->scratch, g->node_cap, lane);
    __syncthreads();
#ifdef ABAMD_KPROF
    {
        unsigned long long ft6 = __builtin_readcyclecounter();
        if (lane == 0) {
            atomicAdd(&abamd_kprof_fold[6], ft6 - ft5);
            atomicAdd(&abamd_kprof_fold[0], 1ull);
        }
    }
#endif
    if (lane == 0) {
        abamd_fold_out_t *o = job->out;
        o->status = ABAMD_FOLD_OK;
        o->node_n = g->node_n;
        o->edge_n_in = g->edge_n_in; o->edge_n_out = g->edge_n_out;
        o->aln_n = g->aln_n;
        o->n_rows = n_rows;
        o->n_pre = job->pre_off[n_rows];
        o->n_out = job->out_off[n_rows];
    }
}


/* Round-3 skeleton: per-set single-cluster consensus on device (compile-
 * checked; not wired — the resident driver still downloads graphs and runs
 * host consensus). One block per set, the twin-proven core on lane 0;
 * phred parity (double pow/log10 on device libm) must be validated on
 * hardware before wiring. */
typedef struct {
    flat_graph_t g;
    int n_seq;
    int *scratch, *score, *max_out;
    int *cons_id; uint8_t *cons_base; int *cons_cov, *cons_phred;
    int *cons_len_out;
} abamd_cons_job_t;

extern "C" __global__ void abamd_cons_kernel(abamd_cons_job_t *jobs, int n_jobs) {
    int j = blockIdx.x;
    if (j >= n_jobs) return;
    abamd_cons_job_t *job = &jobs[j];
    if (threadIdx.x == 0) {
        flat_graph_t gl = job->g;
        *job->cons_len_out = abamd_flat_hb_consensus(&gl, job->n_seq,
                                                     job->scratch, job->score, job->max_out,
                                                     job->cons_id, job->cons_base,
                                                     job->cons_cov, job->cons_phred);
    }
}

extern "C" void abamd_launch_fold_round(const abamd_fold_round_job_t *dev_jobs,
                                        int n_jobs, void *stream) {
    hipLaunchKernelGGL(abamd_fold_round_kernel, dim3(n_jobs), dim3(64), 0,
                       (hipStream_t)stream, (abamd_fold_round_job_t*)dev_jobs, n_jobs);
}
