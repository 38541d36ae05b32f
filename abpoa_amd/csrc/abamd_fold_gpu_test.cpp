/* Functional GPU test for abamd_fold_kernel (gpu_fold.hip).
 *
 * Maintains the SAME flat graph twice — on the device (mutated only by the
 * fold kernel, one launch per read) and on the host (abamd_fold_core.c) —
 * while the live pointer graph feeds the product GPU aligner for CIGARs.
 * After every read the whole device flat state (counts, chains, pools,
 * read-id bitsets, topo index, remain) is downloaded and compared
 * byte-for-byte against the host twin, which tests/test_fold_twin.py has
 * already proven equal to the pointer graph. Passing here means the
 * round-2 device fold produces bit-identical graphs.
 *
 * Usage (GPU box): abpoa_amd_foldgpu reads.fa
 */
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include "abpoa_amd.h"
#include "abamd_util.h"
#include "abamd_fold_core.h"

#define HIP_CHECK(x) do { hipError_t _e = (x); if (_e != hipSuccess) { \
    fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorName(_e), __FILE__, __LINE__); \
    exit(1); } } while (0)

extern "C" {
typedef struct abamd_fx_t abamd_fx_t;
abpoa_seq_t *abamd_seq_new(void);
abamd_fx_t *abamd_fx_open(const char *fn);
void abamd_fx_close(abamd_fx_t *x);
int abamd_read_seq(abpoa_seq_t *abs, abamd_fx_t *x);
}

/* must match gpu_fold.hip's fold_job_t layout exactly */
typedef struct {
    flat_graph_t g;
    const uint64_t *cigar; int n_cigar;
    const uint8_t *seq; const int *weight; int seq_l;
    int *qpos_to_node_id;
    int read_id, add_read_id;
    int *index_to_node_id, *node_id_to_index, *max_remain, *scratch;
    int *msa_rank;
} fold_job_t;

extern "C" __global__ void abamd_fold_kernel(fold_job_t *jobs, int n_jobs);

static void die(const char *what, int read_i) {
    fprintf(stderr, "DEVICE FOLD MISMATCH after read %d: %s\n", read_i, what);
    exit(1);
}

int main(int argc, char **argv) {
    if (argc < 2) { fprintf(stderr, "usage: %s reads.fa\n", argv[0]); return 2; }
    abpoa_para_t *abpt = abpoa_init_para();
    abpt->out_msa = 1; /* enable read-id tracking so bitsets are exercised */
    abpoa_post_set_para(abpt);
    abpoa_t *ab = abpoa_init();
    abpoa_reset(ab, abpt, 1024);

    abpoa_seq_t *abs = abamd_seq_new();
    abamd_fx_t *fx = abamd_fx_open(argv[1]);
    int n_seq = abamd_read_seq(abs, fx);
    abamd_fx_close(fx);
    ab->abs->n_seq = n_seq;

    int total_len = 0, max_len = 0, i, j;
    for (i = 0; i < n_seq; ++i) {
        total_len += abs->seq[i].l;
        if (abs->seq[i].l > max_len) max_len = abs->seq[i].l;
    }
    int rid_n = 1 + ((n_seq - 1) >> 6);
    int node_cap = total_len + 2, edge_cap = 4 * total_len + 64, aln_cap = 8 * total_len + 1024;

    /* host twin */
    flat_graph_t hg;
    abamd_flat_init(&hg, node_cap, edge_cap, aln_cap, rid_n);
    int *h_i2n = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    int *h_n2i = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    int *h_rem = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    int *h_msa = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    int *h_scr = (int*)abamd_malloc((size_t)2 * node_cap * sizeof(int));
    int *h_qmap = (int*)abamd_malloc((size_t)max_len * sizeof(int));

    /* device twin: same capacities, device pools */
    flat_graph_t dgh; /* host-side mirror holding device pointers */
    memset(&dgh, 0, sizeof(dgh));
    dgh.node_cap = node_cap; dgh.edge_cap = edge_cap; dgh.aln_cap = aln_cap;
    dgh.rid_n = rid_n; dgh.node_n = 2;
    #define DALLOC(p, n, T) HIP_CHECK(hipMalloc((void**)&(p), (size_t)(n) * sizeof(T)))
    DALLOC(dgh.base, node_cap, uint8_t);
    DALLOC(dgh.n_read, node_cap, int); DALLOC(dgh.n_span_read, node_cap, int);
    DALLOC(dgh.in_head, node_cap, int); DALLOC(dgh.in_tail, node_cap, int);
    DALLOC(dgh.out_head, node_cap, int); DALLOC(dgh.out_tail, node_cap, int);
    DALLOC(dgh.aln_head, node_cap, int);
    DALLOC(dgh.in_to, edge_cap, int); DALLOC(dgh.in_w, edge_cap, int); DALLOC(dgh.in_next, edge_cap, int);
    DALLOC(dgh.out_to, edge_cap, int); DALLOC(dgh.out_w, edge_cap, int); DALLOC(dgh.out_next, edge_cap, int);
    DALLOC(dgh.rid_pool, (size_t)edge_cap * rid_n, uint64_t);
    DALLOC(dgh.aln_id, aln_cap, int); DALLOC(dgh.aln_next, aln_cap, int);
    HIP_CHECK(hipMemset(dgh.base, 0, node_cap));
    HIP_CHECK(hipMemset(dgh.n_read, 0, (size_t)node_cap * sizeof(int)));
    HIP_CHECK(hipMemset(dgh.n_span_read, 0, (size_t)node_cap * sizeof(int)));
    HIP_CHECK(hipMemset(dgh.rid_pool, 0, (size_t)edge_cap * rid_n * sizeof(uint64_t)));
    {
        int *minus1 = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
        for (i = 0; i < node_cap; ++i) minus1[i] = -1;
        HIP_CHECK(hipMemcpy(dgh.in_head, minus1, (size_t)node_cap * sizeof(int), hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dgh.in_tail, minus1, (size_t)node_cap * sizeof(int), hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dgh.out_head, minus1, (size_t)node_cap * sizeof(int), hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dgh.out_tail, minus1, (size_t)node_cap * sizeof(int), hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dgh.aln_head, minus1, (size_t)node_cap * sizeof(int), hipMemcpyHostToDevice));
        free(minus1);
    }
    uint64_t *d_cigar; uint8_t *d_seq; int *d_w, *d_qmap, *d_i2n, *d_n2i, *d_rem, *d_scr;
    DALLOC(d_cigar, 4 * (size_t)max_len + 16, uint64_t);
    DALLOC(d_seq, max_len, uint8_t);
    DALLOC(d_w, max_len, int);
    DALLOC(d_qmap, max_len, int);
    DALLOC(d_i2n, node_cap, int); DALLOC(d_n2i, node_cap, int);
    DALLOC(d_rem, node_cap, int); DALLOC(d_scr, 2 * (size_t)node_cap, int);
    int *d_msa_rank;
    DALLOC(d_msa_rank, node_cap, int);
    fold_job_t *d_job;
    DALLOC(d_job, 1, fold_job_t);

    int *w = (int*)abamd_malloc((size_t)max_len * sizeof(int));
    for (i = 0; i < max_len; ++i) w[i] = 1;
    HIP_CHECK(hipMemcpy(d_w, w, (size_t)max_len * sizeof(int), hipMemcpyHostToDevice));
    uint8_t *codes = (uint8_t*)abamd_malloc((size_t)max_len);

    /* scratch for downloads */
    int *buf = (int*)abamd_malloc((size_t)2 * node_cap * sizeof(int));
    int *buf2 = (int*)abamd_malloc((size_t)edge_cap * sizeof(int));
    uint64_t *bufr = (uint64_t*)abamd_malloc((size_t)edge_cap * rid_n * sizeof(uint64_t));

    for (i = 0; i < n_seq; ++i) {
        int qlen = abs->seq[i].l;
        for (j = 0; j < qlen; ++j) codes[j] = (uint8_t)ab_amd_char26_table[(int)abs->seq[i].s[j]];
        abpoa_res_t res; memset(&res, 0, sizeof(res));
        abpoa_align_sequence_to_graph(ab, abpt, codes, qlen, &res);

        /* host twin update (pass order proven by tests/test_fold_twin) */
        abamd_flat_apply_alignment(&hg, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID, codes, w, qlen,
                                   h_qmap, res.n_cigar, res.graph_cigar, i, 1, 1);
        abamd_flat_topo_index(&hg, h_i2n, h_n2i, h_scr);
        abamd_flat_sort_adjacency(&hg);
        abamd_flat_remain(&hg, h_rem, h_scr);
        abamd_flat_update_n_span(&hg, h_i2n, h_n2i, 1);
        abamd_flat_msa_rank(&hg, h_msa, h_scr);

        /* device fold */
        if (res.n_cigar)
            HIP_CHECK(hipMemcpy(d_cigar, res.graph_cigar, (size_t)res.n_cigar * 8, hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(d_seq, codes, (size_t)qlen, hipMemcpyHostToDevice));
        fold_job_t job;
        memset(&job, 0, sizeof(job));
        job.g = dgh;
        job.cigar = d_cigar; job.n_cigar = res.n_cigar;
        job.seq = d_seq; job.weight = d_w; job.seq_l = qlen;
        job.qpos_to_node_id = d_qmap;
        job.read_id = i; job.add_read_id = 1;
        job.index_to_node_id = d_i2n; job.node_id_to_index = d_n2i;
        job.max_remain = d_rem; job.scratch = d_scr;
        job.msa_rank = d_msa_rank;
        HIP_CHECK(hipMemcpy(d_job, &job, sizeof(job), hipMemcpyHostToDevice));
        hipLaunchKernelGGL(abamd_fold_kernel, dim3(1), dim3(64), 0, 0, d_job, 1);
        HIP_CHECK(hipGetLastError());
        HIP_CHECK(hipDeviceSynchronize());
        HIP_CHECK(hipMemcpy(&job, d_job, sizeof(job), hipMemcpyDeviceToHost));
        dgh = job.g; /* counts mutated on device */

        /* feed the live graph for the next read's DP */
        abpoa_add_graph_alignment(ab, abpt, codes, w, qlen, NULL, res, i, n_seq, 1);
        if (res.n_cigar) free(res.graph_cigar);

        /* compare device state vs host twin */
        if (dgh.node_n != hg.node_n) die("node_n", i);
        if (dgh.edge_n_in != hg.edge_n_in || dgh.edge_n_out != hg.edge_n_out) die("edge counts", i);
        if (dgh.aln_n != hg.aln_n) die("aln count", i);
        int n = hg.node_n;
        #define CMP_ARR(dptr, hptr, count, T, what) do { \
            HIP_CHECK(hipMemcpy(buf, dptr, (size_t)(count) * sizeof(T), hipMemcpyDeviceToHost)); \
            if (memcmp(buf, hptr, (size_t)(count) * sizeof(T))) die(what, i); } while (0)
        CMP_ARR(dgh.base, hg.base, n, uint8_t, "base");
        CMP_ARR(dgh.n_read, hg.n_read, n, int, "n_read");
        CMP_ARR(dgh.n_span_read, hg.n_span_read, n, int, "n_span_read");
        CMP_ARR(dgh.in_head, hg.in_head, n, int, "in_head");
        CMP_ARR(dgh.out_head, hg.out_head, n, int, "out_head");
        CMP_ARR(dgh.aln_head, hg.aln_head, n, int, "aln_head");
        CMP_ARR(d_i2n, h_i2n, hg.node_n ? 1 + h_n2i[1] : 0, int, "topo index"); /* prefix up to sink */
        #define CMP_POOL(dptr, hptr, count, what) do { \
            HIP_CHECK(hipMemcpy(buf2, dptr, (size_t)(count) * sizeof(int), hipMemcpyDeviceToHost)); \
            if (memcmp(buf2, hptr, (size_t)(count) * sizeof(int))) die(what, i); } while (0)
        CMP_POOL(dgh.in_to, hg.in_to, hg.edge_n_in, "in_to");
        CMP_POOL(dgh.in_w, hg.in_w, hg.edge_n_in, "in_w");
        CMP_POOL(dgh.in_next, hg.in_next, hg.edge_n_in, "in_next");
        CMP_POOL(dgh.out_to, hg.out_to, hg.edge_n_out, "out_to");
        CMP_POOL(dgh.out_w, hg.out_w, hg.edge_n_out, "out_w");
        CMP_POOL(dgh.out_next, hg.out_next, hg.edge_n_out, "out_next");
        CMP_POOL(dgh.aln_id, hg.aln_id, hg.aln_n, "aln_id");
        CMP_POOL(dgh.aln_next, hg.aln_next, hg.aln_n, "aln_next");
        HIP_CHECK(hipMemcpy(bufr, dgh.rid_pool, (size_t)hg.edge_n_out * rid_n * sizeof(uint64_t), hipMemcpyDeviceToHost));
        if (memcmp(bufr, hg.rid_pool, (size_t)hg.edge_n_out * rid_n * sizeof(uint64_t))) die("rid_pool", i);
        {   /* msa rank: every node reachable by the rank DFS */
            HIP_CHECK(hipMemcpy(buf, d_msa_rank, (size_t)n * sizeof(int), hipMemcpyDeviceToHost));
            int k, sink_idx = h_n2i[ABPOA_SINK_NODE_ID];
            for (k = 0; k <= sink_idx; ++k) {
                int id = h_i2n[k];
                if (buf[id] != h_msa[id]) die("msa rank", i);
            }
        }
        {   /* remain over the topo prefix */
            HIP_CHECK(hipMemcpy(buf, d_rem, (size_t)n * sizeof(int), hipMemcpyDeviceToHost));
            int k, sink_idx = h_n2i[ABPOA_SINK_NODE_ID];
            for (k = 0; k <= sink_idx; ++k) {
                int id = h_i2n[k];
                if (buf[id] != h_rem[id]) die("max_remain", i);
            }
        }
    }
    printf("device fold OK (%d reads, %d nodes, %d out-edges)\n", n_seq, hg.node_n, hg.edge_n_out);
    return 0;
}
