/* Shared host<->device job layout for the CDNA4 aligner core.
 *
 * One job = one sequence-to-(sub)graph alignment: the adaptive-banded DP over
 * the topologically-sorted reachable rows of the partial order graph
 * (reference semantics: abpoa_align_simd.c; clean cell-granularity
 * formulation as in oracle/ref_core.c — the two are checked bit-equal by
 * tests/). Row CSRs come either from the host pack (seam path: only
 * reachable index_map rows travel, predecessor lists remapped to compact
 * row indices) or straight from the device fold (resident batch path).
 *
 * Device execution (round 2): ONE 512-THREAD BLOCK (8 wavefronts) per job
 * walks the rows in topological order; the band of each row is computed on
 * device from the adaptive band state (max_left/max_right/max_remain) with
 * the per-row scalars rolled one row ahead, all 8 waves cover the band at
 * once (per-wave DPP F-scans + a cross-wave carry recurrence), and the
 * banded planes stream to the HBM arena for the backtrack (convex: 3
 * planes H/E1/E2, F recomputed at backtrack; affine: H/E1/F1; linear: H).
 * Thread 0 backtracks in-kernel, pausing for block-wide F-window
 * recomputes on the convex path.
 */
#ifndef ABAMD_GPU_CORE_H
#define ABAMD_GPU_CORE_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* per-row metadata written by the kernel as the band is discovered */
typedef struct {
    int32_t beg, end;
    int64_t off;
} abamd_row_meta_t;

enum {
    ABAMD_JOB_OK = 0,
    ABAMD_JOB_ARENA_OVERFLOW = 1,
    ABAMD_JOB_CIGAR_OVERFLOW = 2,
    ABAMD_JOB_BT_DEAD_END = 3,
    ABAMD_JOB_UNSUPPORTED = 4,
};

typedef struct {
    /* device pointers (all in the job's slab) */
    const uint8_t *query;        /* [qlen] residue codes 0..m-1 */
    const uint8_t *row_base;     /* [n_rows] node residue per compact row */
    const int *row_node_id;      /* [n_rows] original graph node id */
    const int *pre_off;          /* [n_rows+1] CSR into pre_idx/pre_ps */
    const int *pre_idx;          /* compact row index of each predecessor */
    const int *pre_ps;           /* per-edge path score (all 0 unless inc_path_score) */
    const int *out_off;          /* [n_rows+1] CSR into out_idx (mapped out-edges) */
    const int *out_idx;
    const int *max_remain;       /* [n_rows] remaining-path metadata */
    int *max_left;               /* [n_rows] adaptive band state (mutated) */
    int *max_right;
    void *row_meta;              /* [n_rows] abamd_row_meta_t: band bounds +
                                    arena offset, one 16-byte load per row */
    void *arena;                 /* score_t[arena_cap * 5] */
    int64_t arena_cap;           /* capacity in cells per plane */
    uint64_t *cigar;             /* packed graph-CIGAR output (backtrack order) */
    int cigar_cap;

    /* scalars */
    int n_rows, qlen, m;
    int w;                       /* band half-width: wb + wf*qlen, or qlen if wb<0 */
    int banded;                  /* abpt->wb >= 0 */
    int o1, e1, o2, e2, oe1, oe2;
    int inf_min;
    int align_mode;
    int put_gap_on_right, put_gap_at_end;
    int zdrop, inc_path_score;
    int node_n_init;             /* graph node_n: initial value of max_left */
    int ret_cigar;
    const int *mat;              /* [m*m] scoring matrix */
} abamd_gpu_job_t;

typedef struct {
    int32_t best_score;
    int best_i, best_j;          /* compact row / query column of the best cell */
    int n_cigar;
    int status;                  /* ABAMD_JOB_* */
    int n_aln_bases, n_matched_bases;
    int node_s, node_e, query_s, query_e;
    int64_t cells;               /* total banded cells computed (roofline) */
} abamd_gpu_res_t;

/* kernel launchers (gpu_kernels.hip); stream is a hipStream_t */
void abamd_launch_cg_i16(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                         int n_jobs, void *stream);
void abamd_launch_cg_i32(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                         int n_jobs, void *stream);
void abamd_launch_ag_i16(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                         int n_jobs, void *stream);
void abamd_launch_ag_i32(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                         int n_jobs, void *stream);
void abamd_launch_lg_i16(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                         int n_jobs, void *stream);
void abamd_launch_lg_i32(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                         int n_jobs, void *stream);

#ifdef __cplusplus
}
#endif

#endif
