/* CDNA4 (gfx950) aligner core: adaptive-banded sequence-to-graph DP.
 *
 * One wavefront (64 lanes) per alignment job, 4 jobs per 256-thread
 * workgroup. The wave sweeps the graph's topologically-sorted rows; within a
 * row the band is processed in 64-cell register chunks:
 *   - M/E gather from predecessor rows. The common predecessor (in-degree is
 *     ~1.05) is the immediately preceding row, whose H/E planes are kept in a
 *     double-buffered LDS cache (BMAX cells) — a ~50-cycle LDS read
 *     replaces a ~900-cycle HBM round-trip on the row-to-row dependent chain.
 *     Other predecessors (and bands wider than the cache) read the banded
 *     HBM arena with coalesced 2/4-byte loads.
 *   - the F (insertion) recurrence is a wavefront max-plus log-scan
 *     (__shfl_up, 6 doubling steps) with a sequential carry between chunks,
 *   - the banded planes stream to the HBM arena with coalesced stores
 *     (convex: H,E1,E2 only — the F planes are recomputed from stored H at
 *     backtrack time; affine: H,E1,F1; linear: H),
 *   - the row argmax (adaptive band steering) is a wavefront reduction,
 *   - per-row band metadata is one 16-byte abamd_row_meta_t load.
 * Integer max-plus throughout: MFMA does not apply; the target bound is the
 * 6 B/cell (convex int16) HBM plane traffic.
 *
 * Numerics are bit-identical to oracle/ref_core.c (and the reference x86
 * build): int16 arithmetic wraps (non-saturating _mm*_add_epi16 semantics),
 * reproduced by truncating to the score type after every add/sub. Recurrence
 * and backtrack follow abpoa_align_simd.c:935-1074 / :309-458; band formulas
 * abpoa_align.h:34-35.
 *
 * This file covers global alignment with convex gaps (the north-star path),
 * int16 with int32 overflow rescore. Affine/linear gap and local/extension
 * variants are host-dispatched and tracked in DESIGN.md §7.
 */
#include <hip/hip_runtime.h>
#include "gpu_core.h"

/* Compile-time phase profiler (make kprof): accumulates s_memtime cycles per
 * row-loop phase of the convex kernel across all jobs. Diagnostic build
 * only — never part of the product library. */
#ifdef ABAMD_KPROF
__device__ unsigned long long abamd_kprof_acc[16];
extern "C" void abamd_kprof_fetch(unsigned long long *out) {
    (void)hipMemcpyFromSymbol(out, HIP_SYMBOL(abamd_kprof_acc), sizeof(abamd_kprof_acc));
}
extern "C" void abamd_kprof_reset(void) {
    unsigned long long z[16] = {0};
    (void)hipMemcpyToSymbol(HIP_SYMBOL(abamd_kprof_acc), z, sizeof(z));
}
#define KPROF_T(v) unsigned long long v = __builtin_readcyclecounter()
#define KPROF_ACC(slot, val) if (lane == 0) atomicAdd(&abamd_kprof_acc[slot], (unsigned long long)(val))
#else
#define KPROF_T(v)
#define KPROF_ACC(slot, val)
#endif

#define WAVE 64
#define JOBS_PER_BLOCK 4
/* LDS previous-row cache width (cells). Default bands are ~230-450 cells
 * (w = wb + wf*qlen = 110 at the north-star shape); wider rows fall back to
 * arena reads. */
#define ABAMD_BMAX 512
/* int32 rounds halve the cache width to keep LDS within 4 resident blocks/CU */
template <typename S> struct BmaxOf { static constexpr int v = ABAMD_BMAX; };
template <> struct BmaxOf<int32_t> { static constexpr int v = 384; };

template <typename S>
__device__ __forceinline__ S smax(S a, S b) { return a > b ? a : b; }

/* Wave-wide inclusive max-plus scan with decay `e` per lane step:
 *   out[l] = max_{k<=l} (in[k] - (l-k)*e)
 * Intra-row (16-lane) steps use DPP row_shr (VALU latency) instead of
 * ds_bpermute; only the two cross-row steps pay the LDS-unit latency.
 * Arithmetic wraps at the score width after every op (reference int16
 * semantics). `neutral` fills lanes with no source. */
template <typename S>
__device__ __forceinline__ S scan_maxplus(S f, int e, S neutral, int lane) {
    (void)neutral;
    /* lanes with no in-row source keep their own value (old = f), exactly
     * like the guarded shuffle formulation — a synthetic floor could exceed
     * the reference's stored out-of-band values and break bit-parity */
    int cand;
    cand = __builtin_amdgcn_update_dpp((int)f, (int)f, 0x111, 0xf, 0xf, false);
    if ((lane & 15) >= 1) f = smax(f, (S)((S)cand - (S)(1 * e)));
    cand = __builtin_amdgcn_update_dpp((int)f, (int)f, 0x112, 0xf, 0xf, false);
    if ((lane & 15) >= 2) f = smax(f, (S)((S)cand - (S)(2 * e)));
    cand = __builtin_amdgcn_update_dpp((int)f, (int)f, 0x114, 0xf, 0xf, false);
    if ((lane & 15) >= 4) f = smax(f, (S)((S)cand - (S)(4 * e)));
    cand = __builtin_amdgcn_update_dpp((int)f, (int)f, 0x118, 0xf, 0xf, false);
    if ((lane & 15) >= 8) f = smax(f, (S)((S)cand - (S)(8 * e)));
    /* cross-row propagation: each row's full prefix ends at its last lane;
     * broadcast it (readlane, SALU) and decay by the per-lane distance.
     * r after each step covers all rows up to that boundary, so coverage is
     * complete (a plain +16/+32 jump after a row-local scan is NOT). */
    S r = (S)__builtin_amdgcn_readlane((int)f, 15);
    if (lane >= 16) f = smax(f, (S)(r - (S)((lane - 15) * e)));
    r = (S)__builtin_amdgcn_readlane((int)f, 31);
    if (lane >= 32) f = smax(f, (S)(r - (S)((lane - 31) * e)));
    r = (S)__builtin_amdgcn_readlane((int)f, 47);
    if (lane >= 48) f = smax(f, (S)(r - (S)((lane - 47) * e)));
    return f;
}

/* Wave-uniform max/min reductions: 4 DPP row_shr steps collect each 16-lane
 * row's reduction into its last lane, 4 readlanes + scalar ops finish — the
 * result is identical to a 6-step __shfl_xor tree but costs ~4 VALU + 4
 * SALU-ish ops instead of 18 DS-unit permutes (measured ~3.3k cycles per row
 * in the epilogue before this). bound_ctrl=false keeps the lane's own value
 * when the DPP source is out of range: the identity for max/min. */
__device__ __forceinline__ int wave_red_max_i32(int v) {
    int t;
    t = __builtin_amdgcn_update_dpp(v, v, 0x111, 0xf, 0xf, false); v = v > t ? v : t;
    t = __builtin_amdgcn_update_dpp(v, v, 0x112, 0xf, 0xf, false); v = v > t ? v : t;
    t = __builtin_amdgcn_update_dpp(v, v, 0x114, 0xf, 0xf, false); v = v > t ? v : t;
    t = __builtin_amdgcn_update_dpp(v, v, 0x118, 0xf, 0xf, false); v = v > t ? v : t;
    int r0 = __builtin_amdgcn_readlane(v, 15), r1 = __builtin_amdgcn_readlane(v, 31);
    int r2 = __builtin_amdgcn_readlane(v, 47), r3 = __builtin_amdgcn_readlane(v, 63);
    r0 = r0 > r1 ? r0 : r1; r2 = r2 > r3 ? r2 : r3;
    return r0 > r2 ? r0 : r2;
}
__device__ __forceinline__ int wave_red_min_i32(int v) {
    int t;
    t = __builtin_amdgcn_update_dpp(v, v, 0x111, 0xf, 0xf, false); v = v < t ? v : t;
    t = __builtin_amdgcn_update_dpp(v, v, 0x112, 0xf, 0xf, false); v = v < t ? v : t;
    t = __builtin_amdgcn_update_dpp(v, v, 0x114, 0xf, 0xf, false); v = v < t ? v : t;
    t = __builtin_amdgcn_update_dpp(v, v, 0x118, 0xf, 0xf, false); v = v < t ? v : t;
    int r0 = __builtin_amdgcn_readlane(v, 15), r1 = __builtin_amdgcn_readlane(v, 31);
    int r2 = __builtin_amdgcn_readlane(v, 47), r3 = __builtin_amdgcn_readlane(v, 63);
    r0 = r0 < r1 ? r0 : r1; r2 = r2 < r3 ? r2 : r3;
    return r0 < r2 ? r0 : r2;
}

/* device push_cigar, matching abpoa_align.h:54-73 (run-length merge for I/S/H) */
__device__ static int dev_push_cigar(uint64_t *cig, int *n_c, int cap, int op, int len,
                                     int node_id, int query_id, int *status) {
    uint64_t l = (uint64_t)len;
    if (*n_c == 0 || (op != 1 /*I*/ && op != 4 && op != 5) || op != (int)(cig[*n_c - 1] & 0xf)) {
        if (*n_c >= cap) { *status = ABAMD_JOB_CIGAR_OVERFLOW; return -1; }
        uint64_t n_id = (uint64_t)(uint32_t)node_id, q_id = (uint64_t)(uint32_t)query_id;
        if (op == 0 || op == 3) cig[(*n_c)++] = n_id << 34 | q_id << 4 | (uint64_t)op;
        else if (op == 1 || op == 4 || op == 5) cig[(*n_c)++] = q_id << 34 | l << 4 | (uint64_t)op;
        else cig[(*n_c)++] = n_id << 34 | l << 4 | (uint64_t)op; /* D */
    } else cig[*n_c - 1] += l << 4;
    return 0;
}

/* ------------------------------------------------------------------ */
/* Multi-wave convex kernel: ONE JOB PER 512-THREAD BLOCK.              */
/*                                                                      */
/* The one-wave kernel above is instruction-ISSUE bound: a lone wave    */
/* on a SIMD issues one VALU op per 4 cycles, and a ~350-instruction    */
/* chunk body makes a typical 340-cell row cost ~21k cycles (kprof).    */
/* Here 8 waves cover the whole band at once: the M/E gather, H fold,   */
/* stores and argmax are embarrassingly lane-parallel, and the F        */
/* (insertion) max-plus chain splits into per-wave local scans plus a   */
/* cross-wave carry recurrence over the 8 published wave edges — the    */
/* regrouped decays are bit-identical because wrapping addition is      */
/* associative and the inf_min headroom (+512*ext_max, the same bound   */
/* the one-wave scan relies on) keeps every comparison un-wrapped.      */
/* Three block barriers per row replace ~5 serial chunk iterations.     */
/* ------------------------------------------------------------------ */
#define MWAVES 8
#define MWT (WAVE * MWAVES)
#define BMW 512 /* prev-row LDS cache width (cells); wider bands re-read the arena */

template <typename S>
__global__ __launch_bounds__(MWT, 1)
void cg_global_mw_kernel(const abamd_gpu_job_t *__restrict__ jobs,
                         abamd_gpu_res_t *__restrict__ results, int n_jobs) {
    const int jid = blockIdx.x;
    if (jid >= n_jobs) return;
    const int tid = threadIdx.x;
    const int wv = tid / WAVE, lane = tid % WAVE;

    __shared__ int mat_lds[27 * 27];
    __shared__ S prev_lds[2][3 * BMW];
    __shared__ S sc_h[MWAVES], sc_f1[MWAVES], sc_f2[MWAVES];
    __shared__ int sc_red[3 * MWAVES];
    __shared__ S sc_carry[3];

    const abamd_gpu_job_t jb = jobs[jid];
    abamd_gpu_res_t *res = &results[jid];
    abamd_row_meta_t *__restrict__ meta = (abamd_row_meta_t*)jb.row_meta;

    const int qlen = jb.qlen, n_rows = jb.n_rows, w = jb.w, m = jb.m;
    const S inf_min = (S)jb.inf_min;
    const S e1 = (S)jb.e1, e2 = (S)jb.e2;
    const S oe1 = (S)jb.oe1, oe2 = (S)jb.oe2;
    const int local_mode = jb.align_mode == 1, extend_mode = jb.align_mode == 2;
    int32_t run_best = jb.inf_min;
    int run_best_i = 0, run_best_j = 0, run_best_remain = jb.max_remain[0];
    int zdropped = 0;
    const int end_remain = jb.max_remain[n_rows - 1];
    S *arena = (S*)jb.arena;
    const uint8_t *__restrict__ query = jb.query;

    for (int i = tid; i < m * m; i += MWT) mat_lds[i] = jb.mat[i];
    if (tid == 0) { res->status = ABAMD_JOB_OK; res->n_cigar = 0; }

    /* adaptive band state init (abpoa_topological_sort:347-353 + first_dp) */
    for (int i = tid; i < n_rows; i += MWT) {
        jb.max_left[i] = jb.node_n_init;
        jb.max_right[i] = 0;
    }
    if (tid == 0) { jb.max_left[0] = 0; jb.max_right[0] = 0; }
    for (int k = jb.out_off[0] + tid; k < jb.out_off[1]; k += MWT) {
        int o = jb.out_idx[k];
        jb.max_left[o] = 1; jb.max_right[o] = 1;
    }
    __syncthreads();

    int buf_cur = 0;
    int prev_ok = 0, prev_row = -1, prev_beg = 0, prev_end = -1;

    /* ---- first row (simd_abpoa_cg_first_dp) ---- */
    int64_t used;
    {
        int mr = jb.max_remain[0] - end_remain - 1;
        int end0;
        if (jb.banded) {
            int t = jb.max_right[0] > qlen - mr ? jb.max_right[0] : qlen - mr;
            end0 = (qlen < t + w) ? qlen : t + w;
        } else end0 = qlen;
        if (tid == 0) { meta[0].beg = 0; meta[0].end = end0; meta[0].off = 0; }
        int64_t bw = end0 + 1;
        used = bw;
        /* 3 planes only (H,E1,E2): the F planes are recomputed from stored H
         * at backtrack time (validated: tools/f_recompute_experiment.py — a
         * 40% cut of plane bytes and arena capacity) */
        S *H = arena, *E1 = arena + bw, *E2 = arena + 2 * bw;
        S *c = &prev_lds[buf_cur][0];
        const int fits = end0 + 1 <= BMW;
        for (int j = tid; j <= end0; j += MWT) {
            S hv, e1v2, e2v2;
            if (local_mode) {
                hv = 0; e1v2 = 0; e2v2 = 0;
            } else if (j == 0) {
                hv = 0; e1v2 = (S)(0 - oe1); e2v2 = (S)(0 - oe2);
            } else {
                S f1 = (S)(-(jb.o1 + jb.e1 * j));
                S f2 = (S)(-(jb.o2 + jb.e2 * j));
                hv = smax(f1, f2);
                e1v2 = inf_min; e2v2 = inf_min;
            }
            H[j] = hv; E1[j] = e1v2; E2[j] = e2v2;
            if (fits) {
                c[j] = hv; c[BMW + j] = e1v2; c[2 * BMW + j] = e2v2;
            }
        }
        if (fits) { prev_ok = 1; prev_row = 0; prev_beg = 0; prev_end = end0; }
        buf_cur ^= 1;
        __syncthreads();
    }

    /* ---- main row loop (row scalars rolled one row ahead; the common
     * band push r -> r+1 carried in registers — every thread keeps the
     * same copy, and skip-edge pushes are idempotent same-value writes
     * done by every thread so each thread's later prefetch of the pushed
     * row is self-ordered) ---- */
    int cur_pk0 = jb.pre_off[1];
    int cur_pk1 = jb.pre_off[2 <= n_rows ? 2 : n_rows];
    int cur_oo0 = jb.out_off[1];
    int cur_oo1 = jb.out_off[2 <= n_rows ? 2 : n_rows];
    int cur_remain = jb.max_remain[1];
    int cur_ml_mem = jb.max_left[1];
    int cur_mr_mem = jb.max_right[1];
    int cur_pidx0 = jb.pre_idx[cur_pk0];
    int cur_ps0 = jb.pre_ps[cur_pk0];
    int push_ml = 0x7fffffff, push_mr = -0x7fffffff;
#ifdef ABAMD_KPROF
    unsigned long long mw_a = 0, mw_b = 0, mw_c = 0, mw_d = 0, mw_rows = 0;
#endif
    for (int r = 1; r < n_rows - 1; ++r) {
        KPROF_T(mt0);
        const int pk0 = cur_pk0, pk1 = cur_pk1;
        const int oo0 = cur_oo0, oo1 = cur_oo1;
        const int row_remain = cur_remain;
        const int pidx0 = cur_pidx0;
        const S ps0 = (S)cur_ps0;
        const int ml_eff = cur_ml_mem < push_ml ? cur_ml_mem : push_ml;
        const int mr_eff = cur_mr_mem > push_mr ? cur_mr_mem : push_mr;
        {
            const int nr = r + 1;
            cur_pk0 = pk1;
            cur_pk1 = jb.pre_off[nr + 1];
            cur_oo0 = oo1;
            cur_oo1 = jb.out_off[nr + 1];
            cur_remain = jb.max_remain[nr];
            cur_ml_mem = jb.max_left[nr];
            cur_mr_mem = jb.max_right[nr];
            cur_pidx0 = jb.pre_idx[cur_pk0];
            cur_ps0 = jb.pre_ps[cur_pk0];
        }
        int beg, end;
        {
            int mr = row_remain - end_remain - 1;
            if (jb.banded) {
                int lo = ml_eff < qlen - mr ? ml_eff : qlen - mr;
                beg = lo - w; if (beg < 0) beg = 0;
                int hi = mr_eff > qlen - mr ? mr_eff : qlen - mr;
                end = hi + w; if (end > qlen) end = qlen;
                int min_pre_beg;
                if (pk1 - pk0 == 1) {
                    min_pre_beg = (prev_ok && pidx0 == prev_row) ? prev_beg : meta[pidx0].beg;
                } else {
                    min_pre_beg = 0x7fffffff;
                    for (int k = pk0; k < pk1; ++k) {
                        const int pidx = jb.pre_idx[k];
                        int pb = (prev_ok && pidx == prev_row) ? prev_beg : meta[pidx].beg;
                        if (pb < min_pre_beg) min_pre_beg = pb;
                    }
                }
                if (beg < min_pre_beg) beg = min_pre_beg;
            } else { beg = 0; end = qlen; }
        }
        const int64_t bw = end - beg + 1;
        if (used + bw > jb.arena_cap) { if (tid == 0) res->status = ABAMD_JOB_ARENA_OVERFLOW; return; }
        const int64_t off = used;
        if (tid == 0) { meta[r].beg = beg; meta[r].end = end; meta[r].off = off; }
        used += bw;
        S *H = arena + off * 3, *E1r = H + bw, *E2r = E1r + bw;
        const uint8_t base = jb.row_base[r];
        const int *mrow = &mat_lds[base * m];
        const int cache_fits = bw <= BMW;
        S *cw = &prev_lds[buf_cur][0];
        const S *cr = &prev_lds[buf_cur ^ 1][0];
        const bool fast1 = (pk1 - pk0 == 1) && prev_ok && (pidx0 == prev_row);

        S carry_h = inf_min, f1c = inf_min, f2c = inf_min; /* superchunk carries */
        S lmax = inf_min; int lleft = -1, lright = -1;

        for (int ss = beg; ss <= end; ss += MWT) {
            const int j = ss + tid;
            const bool act = j <= end;
            S h = inf_min, e1v = inf_min, e2v = inf_min;
            if (fast1) {
                if (act) {
                    if (local_mode && j == 0) { if (ps0 > h) h = ps0; }
                    if (j - 1 >= prev_beg && j - 1 <= prev_end) {
                        S v = (S)(cr[j - 1 - prev_beg] + ps0);
                        if (v > h) h = v;
                    }
                    if (j >= prev_beg && j <= prev_end) {
                        S v1 = (S)(cr[BMW + j - prev_beg] + ps0);
                        S v2 = (S)(cr[2 * BMW + j - prev_beg] + ps0);
                        if (v1 > e1v) e1v = v1;
                        if (v2 > e2v) e2v = v2;
                    }
                }
            } else for (int k = pk0; k < pk1; ++k) {
                const int p = jb.pre_idx[k];
                const S ps = (S)jb.pre_ps[k];
                if (prev_ok && p == prev_row) {
                    if (act) {
                        if (local_mode && j == 0) { if (ps > h) h = ps; }
                        if (j - 1 >= prev_beg && j - 1 <= prev_end) {
                            S v = (S)(cr[j - 1 - prev_beg] + ps);
                            if (v > h) h = v;
                        }
                        if (j >= prev_beg && j <= prev_end) {
                            S v1 = (S)(cr[BMW + j - prev_beg] + ps);
                            S v2 = (S)(cr[2 * BMW + j - prev_beg] + ps);
                            if (v1 > e1v) e1v = v1;
                            if (v2 > e2v) e2v = v2;
                        }
                    }
                    continue;
                }
                const abamd_row_meta_t pm = meta[p];
                const int64_t pbw = pm.end - pm.beg + 1;
                const S *__restrict__ pH = arena + pm.off * 3;
                const S *__restrict__ pE1 = pH + pbw;
                const S *__restrict__ pE2 = pE1 + pbw;
                if (act) {
                    if (local_mode && j == 0) { if (ps > h) h = ps; }
                    if (j - 1 >= pm.beg && j - 1 <= pm.end) {
                        S v = (S)(pH[j - 1 - pm.beg] + ps);
                        if (v > h) h = v;
                    }
                    if (j >= pm.beg && j <= pm.end) {
                        S v1 = (S)(pE1[j - pm.beg] + ps);
                        S v2 = (S)(pE2[j - pm.beg] + ps);
                        if (v1 > e1v) e1v = v1;
                        if (v2 > e2v) e2v = v2;
                    }
                }
            }
            const S q = (S)((j == 0 || !act) ? 0 : mrow[query[j - 1]]);
            S hpre = (S)(h + q);
            hpre = smax(hpre, smax(e1v, e2v));
            if (!act) hpre = inf_min;

            /* wave edges of hpre, then per-wave local F scans */
            KPROF_T(mt1);
            if (lane == WAVE - 1) sc_h[wv] = hpre;
            __syncthreads();
            S hshift = (S)__shfl_up((int)hpre, 1);
            S c1, c2;
            if (lane == 0) {
                if (wv == 0) {
                    if (ss == beg) { c1 = (S)(inf_min - oe1); c2 = (S)(inf_min - oe2); }
                    else {
                        c1 = smax((S)(carry_h - oe1), (S)(f1c - e1));
                        c2 = smax((S)(carry_h - oe2), (S)(f2c - e2));
                    }
                } else {
                    S hp = sc_h[wv - 1];
                    c1 = (S)(hp - oe1);
                    c2 = (S)(hp - oe2);
                }
            } else {
                c1 = (S)(hshift - oe1);
                c2 = (S)(hshift - oe2);
            }
            S L1 = scan_maxplus(c1, jb.e1, inf_min, lane);
            S L2 = scan_maxplus(c2, jb.e2, inf_min, lane);
            if (lane == WAVE - 1) { sc_f1[wv] = L1; sc_f2[wv] = L2; }
            __syncthreads();
            /* cross-wave carry: C = true F at the last lane of wave wv-1
             * (max-plus with linear decay is associative; wrapping adds
             * regroup exactly) */
            S f1 = L1, f2 = L2;
            if (wv > 0) {
                S C1 = sc_f1[0], C2 = sc_f2[0];
                for (int ww = 1; ww < wv; ++ww) {
                    C1 = smax(sc_f1[ww], (S)(C1 - (S)(WAVE * jb.e1)));
                    C2 = smax(sc_f2[ww], (S)(C2 - (S)(WAVE * jb.e2)));
                }
                f1 = smax(L1, (S)(C1 - (S)((lane + 1) * jb.e1)));
                f2 = smax(L2, (S)(C2 - (S)((lane + 1) * jb.e2)));
            }

            S hf = smax(hpre, smax(f1, f2));
            if (local_mode) hf = smax(hf, (S)0);
            S e1n = smax((S)(e1v - e1), (S)(hf - oe1));
            S e2n = smax((S)(e2v - e2), (S)(hf - oe2));
            if (local_mode) { e1n = smax(e1n, (S)0); e2n = smax(e2n, (S)0); }
            if (act) {
                H[j - beg] = hf; E1r[j - beg] = e1n; E2r[j - beg] = e2n;
                if (cache_fits) {
                    cw[j - beg] = hf;
                    cw[BMW + j - beg] = e1n;
                    cw[2 * BMW + j - beg] = e2n;
                }
                if (hf > lmax) { lmax = hf; lleft = j; lright = j; }
                else if (hf == lmax) { lright = j; }
            }
            KPROF_T(mt2);
            const bool more = ss + MWT <= end;
            if (more && j == ss + MWT - 1) {
                sc_carry[0] = hpre; sc_carry[1] = f1; sc_carry[2] = f2;
            }
            if (!more && (jb.banded || local_mode || extend_mode)) {
                /* last superchunk: fold the per-wave argmax publish into this
                 * barrier instead of paying a fourth barrier (and its store
                 * drain) after the loop */
                int mvw = wave_red_max_i32((int)lmax);
                int llw = ((int)lmax == mvw && lleft >= 0) ? lleft : 0x7fffffff;
                int rrw = ((int)lmax == mvw && lright >= 0) ? lright : -1;
                llw = wave_red_min_i32(llw);
                rrw = wave_red_max_i32(rrw);
                if (lane == 0) {
                    sc_red[wv] = mvw;
                    sc_red[MWAVES + wv] = llw;
                    sc_red[2 * MWAVES + wv] = rrw;
                }
            }
            __syncthreads(); /* also orders cw writes before the next row's reads */
            if (more) { carry_h = sc_carry[0]; f1c = sc_carry[1]; f2c = sc_carry[2]; }
#ifdef ABAMD_KPROF
            {
                unsigned long long mt3 = __builtin_readcyclecounter();
                mw_a += mt1 - mt0;   /* band + prefetch + gather + hpre */
                mw_b += mt2 - mt1;   /* B1 + scans + B2 + carries + fold + stores */
                mw_c += mt3 - mt2;   /* carry publish + B3 */
            }
#endif
        }

        if (cache_fits) { prev_ok = 1; prev_row = r; prev_beg = beg; prev_end = end; }
        else prev_ok = 0;
        buf_cur ^= 1;

        if (jb.banded || local_mode || extend_mode) {
            /* per-wave argmax published under the loop's last barrier; every
             * thread repeats the 8-way combine from LDS (uniform result) */
            int mv = sc_red[0], ll = sc_red[MWAVES], rr = sc_red[2 * MWAVES];
            #pragma unroll
            for (int ww = 1; ww < MWAVES; ++ww) {
                int m2 = sc_red[ww];
                if (m2 > mv) { mv = m2; ll = sc_red[MWAVES + ww]; rr = sc_red[2 * MWAVES + ww]; }
                else if (m2 == mv) {
                    if (sc_red[MWAVES + ww] < ll) ll = sc_red[MWAVES + ww];
                    if (sc_red[2 * MWAVES + ww] > rr) rr = sc_red[2 * MWAVES + ww];
                }
            }
            if (local_mode) {
                if (mv > run_best) { run_best = mv; run_best_i = r; run_best_j = ll; }
            } else if (extend_mode) {
                if (mv > run_best) {
                    run_best = mv; run_best_i = r; run_best_j = rr;
                    run_best_remain = row_remain;
                } else if (jb.zdrop > 0) {
                    int delta = run_best_remain - row_remain;
                    int dd = delta - (rr - run_best_j); if (dd < 0) dd = -dd;
                    if (run_best - mv > jb.zdrop + jb.e1 * dd) zdropped = 1;
                }
            }
            push_ml = 0x7fffffff; push_mr = -0x7fffffff;
            if (!zdropped && jb.banded) {
                for (int k = oo0; k < oo1; ++k) {
                    int o = jb.out_idx[k];
                    if (o == r + 1) {
                        if (rr + 1 > push_mr) push_mr = rr + 1;
                        if (ll + 1 < push_ml) push_ml = ll + 1;
                    } else {
                        /* every thread performs the same idempotent update so
                         * its own later prefetch of row o is self-ordered */
                        if (rr + 1 > jb.max_right[o]) jb.max_right[o] = rr + 1;
                        if (ll + 1 < jb.max_left[o]) jb.max_left[o] = ll + 1;
                    }
                }
            }
            if (zdropped) break;
        } else {
            push_ml = 0x7fffffff; push_mr = -0x7fffffff;
        }
#ifdef ABAMD_KPROF
        {
            unsigned long long mt4 = __builtin_readcyclecounter();
            mw_d += mt4 - mt0; mw_rows += 1; /* mt0 re-read below start */
        }
#endif
    }
#ifdef ABAMD_KPROF
    if (tid == 0) {
        atomicAdd(&abamd_kprof_acc[11], mw_a);
        atomicAdd(&abamd_kprof_acc[12], mw_b);
        atomicAdd(&abamd_kprof_acc[13], mw_c);
        atomicAdd(&abamd_kprof_acc[14], mw_d);
        atomicAdd(&abamd_kprof_acc[15], mw_rows);
    }
#endif

    __syncthreads();
    if (tid == 0) res->cells = used;

    /* ---- final best (computed redundantly by every thread: all inputs are
     * uniform, and every thread must stay alive for the cooperative
     * F-recompute below) ---- */
    int32_t best_score = run_best;
    int best_i = run_best_i, best_j = run_best_j;
    if (jb.align_mode == 0) {
        best_score = jb.inf_min; best_i = 0; best_j = 0;
        for (int k = jb.pre_off[n_rows - 1]; k < jb.pre_off[n_rows]; ++k) {
            const int p = jb.pre_idx[k];
            const abamd_row_meta_t pm = meta[p];
            int e = pm.end < qlen ? pm.end : qlen;
            const S *pH = arena + pm.off * 3;
            int32_t sc = (e >= pm.beg) ? (int32_t)pH[e - pm.beg] : jb.inf_min;
            if (sc > best_score) { best_score = sc; best_i = p; best_j = e; }
        }
    }
    if (tid == 0) {
        res->best_score = best_score;
        res->best_i = best_i; res->best_j = best_j;
    }
    if (!jb.ret_cigar) return;

    /* ---- backtrack (simd_abpoa_cg_backtrack transcription) ----
     * Thread 0 walks; the F planes are not stored, so when the insertion
     * branch needs F values the walk pauses and the whole block recomputes
     * the row's F window from the STORED (post-F) H plane:
     *   F'[rb] = inf_min - oe;  F'[j] = max(F'[j-1] - e, H[j-1] - oe)
     * — every backtrack decision is identical to stored-F (validated over
     * ~2.5M entries incl. entry-direction flips, tools/f_recompute_experiment.py,
     * and end-to-end by the parity suite). The window scan reuses the
     * per-wave local-scan + cross-wave-carry machinery of the forward pass
     * (same associativity/headroom argument). */
    __shared__ int bt_go, bt_bi_s, bt_bj_s;
    __shared__ S fbuf[2 * BMW];
    /* walk state: only thread 0's copies advance */
    int bi = best_i, bj = best_j, start_i = best_i, start_j = best_j;
    int cur_op = 0x1f, n_c = 0, status = ABAMD_JOB_OK;
    int look_end = jb.put_gap_at_end, put_right = jb.put_gap_on_right;
    int n_aln = 0, n_matched = 0;
    uint64_t *cig = jb.cigar;
    int id = jb.row_node_id[bi];
    int f_row = -1, f_wl = 0, f_hi = -1;
    if (tid == 0) {
        bt_go = 1;
        if (best_j < qlen) dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, qlen - best_j, -1, qlen - 1, &status);
    }
    for (;;) {
        __syncthreads();
        if (bt_go == 0) break;
        if (bt_go == 2) {
            /* recompute F for row bt_bi_s, window ending at bt_bj_s */
            const int rbi = bt_bi_s;
            const abamd_row_meta_t bm = meta[rbi];
            const int rb = bm.beg, re = bm.end;
            const int hi = bt_bj_s < re ? bt_bj_s : re;
            const int wl = (hi - (BMW - 2)) > rb ? hi - (BMW - 2) : rb;
            const S *Hrow = arena + bm.off * 3;
            S carry1 = inf_min, carry2 = inf_min; /* F' at position ss-1 */
            for (int ss = rb; ss <= hi; ss += MWT) {
                const int j = ss + tid;
                const bool act2 = j <= hi;
                S c1, c2;
                if (!act2) { c1 = inf_min; c2 = inf_min; }
                else if (j == rb) { c1 = (S)(inf_min - oe1); c2 = (S)(inf_min - oe2); }
                else {
                    const S hp = Hrow[j - 1 - rb];
                    c1 = (S)(hp - oe1);
                    c2 = (S)(hp - oe2);
                }
                S L1 = scan_maxplus(c1, jb.e1, inf_min, lane);
                S L2 = scan_maxplus(c2, jb.e2, inf_min, lane);
                if (lane == WAVE - 1) { sc_f1[wv] = L1; sc_f2[wv] = L2; }
                __syncthreads();
                /* F' entering this wave; no carry exists before the band
                 * start (mirrors the forward pass: wave 0 of the first
                 * superchunk takes its local scan verbatim) */
                S C1 = carry1, C2 = carry2;
                bool havec = ss != rb;
                for (int ww = 0; ww < wv; ++ww) {
                    if (havec) {
                        C1 = smax(sc_f1[ww], (S)(C1 - (S)(WAVE * jb.e1)));
                        C2 = smax(sc_f2[ww], (S)(C2 - (S)(WAVE * jb.e2)));
                    } else {
                        C1 = sc_f1[ww]; C2 = sc_f2[ww];
                        havec = true;
                    }
                }
                S f1 = L1, f2 = L2;
                if (havec) {
                    f1 = smax(L1, (S)(C1 - (S)((lane + 1) * jb.e1)));
                    f2 = smax(L2, (S)(C2 - (S)((lane + 1) * jb.e2)));
                }
                if (act2 && j >= wl) {
                    fbuf[j - wl] = f1;
                    fbuf[BMW + j - wl] = f2;
                }
                const bool more2 = ss + MWT <= hi;
                if (more2 && j == ss + MWT - 1) { sc_carry[0] = f1; sc_carry[1] = f2; }
                __syncthreads();
                if (more2) { carry1 = sc_carry[0]; carry2 = sc_carry[1]; }
            }
            f_row = rbi; f_wl = wl; f_hi = hi;
            __syncthreads();
        }
        if (tid == 0) {
            int pause = 0;
            while (bi > 0 && bj > 0 && status == ABAMD_JOB_OK) {
                const abamd_row_meta_t bm = meta[bi];
                const int rb = bm.beg, re = bm.end;
                const int64_t bw = re - rb + 1;
                const S *H = arena + bm.off * 3;
                const S *E1r = H + bw, *E2r = E1r + bw;
                const S Hj = (bj >= rb && bj <= re) ? H[bj - rb] : inf_min;
                const S Hjm1 = (bj - 1 >= rb && bj - 1 <= re) ? H[bj - 1 - rb] : inf_min;
                const S E1j = (bj >= rb && bj <= re) ? E1r[bj - rb] : inf_min;
                const S E2j = (bj >= rb && bj <= re) ? E2r[bj - rb] : inf_min;
                if (local_mode && Hj == 0) break;
                start_i = bi; start_j = bj;
                const int pq0 = jb.pre_off[bi], pq1 = jb.pre_off[bi + 1];
                const S s = (S)mat_lds[m * jb.row_base[bi] + query[bj - 1]];
                const int is_match = jb.row_base[bi] == query[bj - 1];
                int hit = 0;
                if (put_right == 0 && look_end == 0 && (cur_op & 0x1)) {
                    for (int k = pq0; k < pq1; ++k) {
                        const int p = jb.pre_idx[k];
                        const S ps = (S)jb.pre_ps[k];
                        const abamd_row_meta_t pm = meta[p];
                        if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                        const S *pH = arena + pm.off * 3;
                        if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                            dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                            bi = p; --bj; id = jb.row_node_id[bi]; hit = 1;
                            cur_op = 0x1f; ++n_aln; n_matched += is_match;
                            break;
                        }
                    }
                }
                if (!hit && (cur_op & 0x6)) { /* deletion */
                    for (int k = pq0; k < pq1; ++k) {
                        const int p = jb.pre_idx[k];
                        const S ps = (S)jb.pre_ps[k];
                        const abamd_row_meta_t pm = meta[p];
                        if (bj < pm.beg || bj > pm.end) continue;
                        const int poffc = bj - pm.beg;
                        const int64_t pbw = pm.end - pm.beg + 1;
                        const S *pH = arena + pm.off * 3;
                        const S *pE1 = pH + pbw, *pE2 = pE1 + pbw;
                        if (cur_op & 0x2) {
                            if (cur_op & 0x1) {
                                if (Hj == (S)(pE1[poffc] + ps)) {
                                    cur_op = ((S)(pH[poffc] - oe1) == pE1[poffc]) ? (0x1 | 0x18) : 0x2;
                                    hit = 1; dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                                    bi = p; id = jb.row_node_id[bi];
                                    if (look_end) look_end = 0;
                                    break;
                                }
                            } else {
                                if (E1j == (S)(pE1[poffc] - e1 + ps)) {
                                    cur_op = ((S)(pH[poffc] - oe1) == pE1[poffc]) ? (0x1 | 0x18) : 0x2;
                                    hit = 1; dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                                    bi = p; id = jb.row_node_id[bi];
                                    if (look_end) look_end = 0;
                                    break;
                                }
                            }
                        }
                        if (cur_op & 0x4) {
                            if (cur_op & 0x1) {
                                if (Hj == (S)(pE2[poffc] + ps)) {
                                    cur_op = ((S)(pH[poffc] - oe2) == pE2[poffc]) ? (0x1 | 0x18) : 0x4;
                                    hit = 1; dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                                    bi = p; id = jb.row_node_id[bi];
                                    if (look_end) look_end = 0;
                                    break;
                                }
                            } else {
                                if (E2j == (S)(pE2[poffc] - e2 + ps)) {
                                    cur_op = ((S)(pH[poffc] - oe2) == pE2[poffc]) ? (0x1 | 0x18) : 0x4;
                                    hit = 1; dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                                    bi = p; id = jb.row_node_id[bi];
                                    if (look_end) look_end = 0;
                                    break;
                                }
                            }
                        }
                    }
                }
                if (!hit && (cur_op & 0x18)) { /* insertion: needs recomputed F */
                    const bool needj = (bj >= rb && bj <= re);
                    const bool needj1 = (bj - 1 >= rb && bj - 1 <= re);
                    const bool cover = (f_row == bi)
                        && (!needj || (bj >= f_wl && bj <= f_hi))
                        && (!needj1 || (bj - 1 >= f_wl && bj - 1 <= f_hi));
                    if ((needj || needj1) && !cover) {
                        bt_bi_s = bi; bt_bj_s = bj; bt_go = 2;
                        pause = 1;
                        break; /* re-enter this step after the block recomputes */
                    }
                    const S F1j = needj ? fbuf[bj - f_wl] : inf_min;
                    const S F2j = needj ? fbuf[BMW + bj - f_wl] : inf_min;
                    const S F1jm1 = needj1 ? fbuf[bj - 1 - f_wl] : inf_min;
                    const S F2jm1 = needj1 ? fbuf[BMW + bj - 1 - f_wl] : inf_min;
                    if (cur_op & 0x8) {
                        if (cur_op & 0x1) {
                            if (Hj == F1j) {
                                if ((S)(Hjm1 - oe1) == F1j) { cur_op = 0x1 | 0x6; hit = 1; }
                                else if ((S)(F1jm1 - e1) == F1j) { cur_op = 0x8; hit = 1; }
                            }
                        } else {
                            if ((S)(Hjm1 - oe1) == F1j) { cur_op = 0x1 | 0x6; hit = 1; }
                            else if ((S)(F1jm1 - e1) == F1j) { cur_op = 0x8; hit = 1; }
                        }
                    }
                    if (!hit && (cur_op & 0x10)) {
                        if (cur_op & 0x1) {
                            if (Hj == F2j) {
                                if ((S)(Hjm1 - oe2) == F2j) { cur_op = 0x1 | 0x6; hit = 1; }
                                else if ((S)(F2jm1 - e2) == F2j) { cur_op = 0x10; hit = 1; }
                            }
                        } else {
                            if ((S)(Hjm1 - oe2) == F2j) { cur_op = 0x1 | 0x6; hit = 1; }
                            else if ((S)(F2jm1 - e2) == F2j) { cur_op = 0x10; hit = 1; }
                        }
                    }
                    if (hit) {
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, 1, id, bj - 1, &status);
                        --bj;
                        if (look_end) look_end = 0;
                        ++n_aln;
                    }
                }
                if (!hit && (cur_op & 0x1)) {
                    for (int k = pq0; k < pq1; ++k) {
                        const int p = jb.pre_idx[k];
                        const S ps = (S)jb.pre_ps[k];
                        const abamd_row_meta_t pm = meta[p];
                        if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                        const S *pH = arena + pm.off * 3;
                        if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                            dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                            bi = p; --bj; id = jb.row_node_id[bi]; hit = 1;
                            cur_op = 0x1f; ++n_aln; n_matched += is_match;
                            look_end = 0;
                            break;
                        }
                    }
                }
                if (!hit) { status = ABAMD_JOB_BT_DEAD_END; break; }
            }
            if (!pause) {
                if (status == ABAMD_JOB_OK && bj > 0)
                    dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, bj, -1, bj - 1, &status);
                res->status = status;
                res->n_cigar = n_c;
                res->n_aln_bases = n_aln;
                res->n_matched_bases = n_matched;
                res->node_e = jb.row_node_id[best_i]; res->query_e = best_j - 1;
                res->node_s = jb.row_node_id[start_i]; res->query_s = start_j - 1;
                bt_go = 0;
            }
        }
    }
}

extern "C" void abamd_launch_cg_i16(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                                    int n_jobs, void *stream) {
    hipLaunchKernelGGL((cg_global_mw_kernel<int16_t>), dim3(n_jobs), dim3(MWT), 0,
                       (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
}
extern "C" void abamd_launch_cg_i32(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                                    int n_jobs, void *stream) {
    hipLaunchKernelGGL((cg_global_mw_kernel<int32_t>), dim3(n_jobs), dim3(MWT), 0,
                       (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
}

/* ------------------------------------------------------------------ */
/* Affine-gap global kernel: 3 planes (H,E1,F1).                       */
/* Reference core simd_abpoa_ag_dp (abpoa_align_simd.c:817-933):       */
/* the F chain feeds from H BEFORE the E fold, and the stored E is     */
/* inf_min when the insertion won the cell (SIMDSetIfEqual, :930).     */
/* Backtrack: simd_abpoa_ag_backtrack (:196-307).                      */
/* ------------------------------------------------------------------ */
template <typename S>
__global__ __launch_bounds__(WAVE * JOBS_PER_BLOCK, 7)
void ag_global_kernel(const abamd_gpu_job_t *__restrict__ jobs,
                      abamd_gpu_res_t *__restrict__ results, int n_jobs) {
    const int wid = threadIdx.x / WAVE;
    const int jid = blockIdx.x * JOBS_PER_BLOCK + wid;
    const int lane = threadIdx.x % WAVE;

    __shared__ int mat_lds[27 * 27];
    constexpr int BMAX = BmaxOf<S>::v;
    /* cache H and E1 of the previous row */
    __shared__ S prev_lds[JOBS_PER_BLOCK][2][2 * BMAX];
    {
        const int m0 = jobs[0].m;
        const int *mat0 = jobs[0].mat;
        for (int i = threadIdx.x; i < m0 * m0; i += WAVE * JOBS_PER_BLOCK)
            mat_lds[i] = mat0[i];
    }
    __syncthreads();
    if (jid >= n_jobs) return;
    /* by VALUE: every field lives in registers (uniform loads become
       s_loads into SGPRs) — a reference would re-load fields from global
       memory inside the row loop because arena/steering stores could alias
       the jobs array in the compiler's view */
    const abamd_gpu_job_t jb = jobs[jid];
    abamd_gpu_res_t *res = &results[jid];
    abamd_row_meta_t *__restrict__ meta = (abamd_row_meta_t*)jb.row_meta;

    const int qlen = jb.qlen, n_rows = jb.n_rows, w = jb.w, m = jb.m;
    const S inf_min = (S)jb.inf_min;
    const S e1 = (S)jb.e1, oe1 = (S)jb.oe1;
    const int local_mode = jb.align_mode == 1, extend_mode = jb.align_mode == 2;
    int32_t run_best = jb.inf_min;
    int run_best_i = 0, run_best_j = 0, run_best_remain = jb.max_remain[0];
    int zdropped = 0;
    const int end_remain = jb.max_remain[n_rows - 1];
    S *arena = (S*)jb.arena;
    const uint8_t *__restrict__ query = jb.query;

    if (lane == 0) { res->status = ABAMD_JOB_OK; res->n_cigar = 0; }

    for (int i = lane; i < n_rows; i += WAVE) {
        jb.max_left[i] = jb.node_n_init;
        jb.max_right[i] = 0;
    }
    if (lane == 0) { jb.max_left[0] = 0; jb.max_right[0] = 0; }
    for (int k = jb.out_off[0] + lane; k < jb.out_off[1]; k += WAVE) {
        int o = jb.out_idx[k];
        jb.max_left[o] = 1; jb.max_right[o] = 1;
    }

    int buf_cur = 0;
    int prev_ok = 0, prev_row = -1, prev_beg = 0, prev_end = -1;

    /* first row (simd_abpoa_ag_first_dp, abpoa_align_simd.c:651-667) */
    int64_t used;
    {
        int mr = jb.max_remain[0] - end_remain - 1;
        int end0;
        if (jb.banded) {
            int t = jb.max_right[0] > qlen - mr ? jb.max_right[0] : qlen - mr;
            end0 = (qlen < t + w) ? qlen : t + w;
        } else end0 = qlen;
        if (lane == 0) { meta[0].beg = 0; meta[0].end = end0; meta[0].off = 0; }
        int64_t bw = end0 + 1;
        used = bw;
        S *H = arena, *E1 = arena + bw, *F1 = arena + 2 * bw;
        S *c = &prev_lds[wid][buf_cur][0];
        const int fits = end0 + 1 <= BMAX;
        for (int j = lane; j <= end0; j += WAVE) {
            S hv, e1v2;
            if (local_mode) {
                hv = 0; e1v2 = 0; F1[j] = 0;
            } else if (j == 0) {
                hv = 0; e1v2 = (S)(0 - oe1);
                F1[0] = inf_min;
            } else {
                S f1 = (S)(-(jb.o1 + jb.e1 * j));
                F1[j] = f1;
                hv = f1; e1v2 = inf_min;
            }
            H[j] = hv; E1[j] = e1v2;
            if (fits) { c[j] = hv; c[BMAX + j] = e1v2; }
        }
        if (fits) { prev_ok = 1; prev_row = 0; prev_beg = 0; prev_end = end0; }
        buf_cur ^= 1;
    }

    for (int r = 1; r < n_rows - 1; ++r) {
        const int pk0 = jb.pre_off[r], pk1 = jb.pre_off[r + 1];
        int beg, end;
        {
            int mr = jb.max_remain[r] - end_remain - 1;
            if (jb.banded) {
                int ml = jb.max_left[r], mrr = jb.max_right[r];
                int lo = ml < qlen - mr ? ml : qlen - mr;
                beg = lo - w; if (beg < 0) beg = 0;
                int hi = mrr > qlen - mr ? mrr : qlen - mr;
                end = hi + w; if (end > qlen) end = qlen;
                int min_pre_beg = 0x7fffffff;
                for (int k = pk0; k < pk1; ++k) {
                    const int pidx = jb.pre_idx[k];
                    int pb = (prev_ok && pidx == prev_row) ? prev_beg : meta[pidx].beg;
                    if (pb < min_pre_beg) min_pre_beg = pb;
                }
                if (beg < min_pre_beg) beg = min_pre_beg;
            } else { beg = 0; end = qlen; }
        }
        const int64_t bw = end - beg + 1;
        if (used + bw > jb.arena_cap) { if (lane == 0) res->status = ABAMD_JOB_ARENA_OVERFLOW; return; }
        const int64_t off = used;
        if (lane == 0) { meta[r].beg = beg; meta[r].end = end; meta[r].off = off; }
        used += bw;
        S *H = arena + off * 3, *E1r = H + bw, *F1r = E1r + bw;
        const uint8_t base = jb.row_base[r];
        const int *mrow = &mat_lds[base * m];
        const int cache_fits = bw <= BMAX;
        S *cw = &prev_lds[wid][buf_cur][0];
        const S *cr = &prev_lds[wid][buf_cur ^ 1][0];

        S carry_hm = inf_min, f1c = inf_min;
        S lmax = inf_min; int lleft = -1, lright = -1;

        for (int cs = beg; cs <= end; cs += WAVE) {
            const int j = cs + lane;
            const bool act = j <= end;
            S h = inf_min, e1v = inf_min;
            for (int k = pk0; k < pk1; ++k) {
                const int p = jb.pre_idx[k];
                const S ps = (S)jb.pre_ps[k];
                if (prev_ok && p == prev_row) {
                    if (act) {
                        if (local_mode && j == 0) { if (ps > h) h = ps; }
                        if (j - 1 >= prev_beg && j - 1 <= prev_end) {
                            S v = (S)(cr[j - 1 - prev_beg] + ps);
                            if (v > h) h = v;
                        }
                        if (j >= prev_beg && j <= prev_end) {
                            S v1 = (S)(cr[BMAX + j - prev_beg] + ps);
                            if (v1 > e1v) e1v = v1;
                        }
                    }
                    continue;
                }
                const abamd_row_meta_t pm = meta[p];
                const int64_t pbw = pm.end - pm.beg + 1;
                const S *__restrict__ pH = arena + pm.off * 3;
                const S *__restrict__ pE1 = pH + pbw;
                if (act) {
                    if (local_mode && j == 0) { if (ps > h) h = ps; }
                    if (j - 1 >= pm.beg && j - 1 <= pm.end) {
                        S v = (S)(pH[j - 1 - pm.beg] + ps);
                        if (v > h) h = v;
                    }
                    if (j >= pm.beg && j <= pm.end) {
                        S v1 = (S)(pE1[j - pm.beg] + ps);
                        if (v1 > e1v) e1v = v1;
                    }
                }
            }
            const S q = (S)((j == 0 || !act) ? 0 : mrow[query[j - 1]]);
            S hm = (S)(h + q);   /* M+q: the F chain feeds from this, pre-E */
            if (!act) hm = inf_min;

            S hmshift = (S)__shfl_up((int)hm, 1);
            S c1;
            if (lane == 0) {
                if (cs == beg) c1 = (S)(inf_min - oe1);
                else c1 = smax((S)(carry_hm - oe1), (S)(f1c - e1));
            } else c1 = (S)(hmshift - oe1);
            S f1 = scan_maxplus(c1, jb.e1, inf_min, lane);
            carry_hm = (S)__builtin_amdgcn_readlane((int)hm, WAVE - 1);
            f1c = (S)__builtin_amdgcn_readlane((int)f1, WAVE - 1);

            S tmp = smax(hm, e1v);
            S hf = smax(tmp, f1);
            if (local_mode) hf = smax(hf, (S)0);
            S e1n = (hf == tmp) ? smax((S)(e1v - e1), (S)(hf - oe1))
                                : (local_mode ? (S)0 : inf_min);
            if (act) {
                H[j - beg] = hf; E1r[j - beg] = e1n; F1r[j - beg] = f1;
                if (cache_fits) { cw[j - beg] = hf; cw[BMAX + j - beg] = e1n; }
                if (hf > lmax) { lmax = hf; lleft = j; lright = j; }
                else if (hf == lmax) { lright = j; }
            }
        }

        if (cache_fits) { prev_ok = 1; prev_row = r; prev_beg = beg; prev_end = end; }
        else prev_ok = 0;
        buf_cur ^= 1;

        if (jb.banded || local_mode || extend_mode) {
            int mv = wave_red_max_i32((int)lmax);
            int ll = ((int)lmax == mv && lleft >= 0) ? lleft : 0x7fffffff;
            int rr = ((int)lmax == mv && lright >= 0) ? lright : -1;
            ll = wave_red_min_i32(ll);
            rr = wave_red_max_i32(rr);
            if (local_mode) {
                if (mv > run_best) { run_best = mv; run_best_i = r; run_best_j = ll; }
            } else if (extend_mode) {
                if (mv > run_best) {
                    run_best = mv; run_best_i = r; run_best_j = rr;
                    run_best_remain = jb.max_remain[r];
                } else if (jb.zdrop > 0) {
                    int delta = run_best_remain - jb.max_remain[r];
                    int dd = delta - (rr - run_best_j); if (dd < 0) dd = -dd;
                    if (run_best - mv > jb.zdrop + jb.e1 * dd) zdropped = 1;
                }
            }
            if (!zdropped && jb.banded) {
                for (int k = jb.out_off[r] + lane; k < jb.out_off[r + 1]; k += WAVE) {
                    int o = jb.out_idx[k];
                    if (rr + 1 > jb.max_right[o]) jb.max_right[o] = rr + 1;
                    if (ll + 1 < jb.max_left[o]) jb.max_left[o] = ll + 1;
                }
            }
            if (zdropped) break;
        }
    }

    if (lane == 0) res->cells = used;
    if (lane != 0) return;

    int32_t best_score = run_best;
    int best_i = run_best_i, best_j = run_best_j;
    if (jb.align_mode == 0) {
        best_score = jb.inf_min; best_i = 0; best_j = 0;
        for (int k = jb.pre_off[n_rows - 1]; k < jb.pre_off[n_rows]; ++k) {
            const int p = jb.pre_idx[k];
            const abamd_row_meta_t pm = meta[p];
            int e = pm.end < qlen ? pm.end : qlen;
            const S *pH = arena + pm.off * 3;
            int32_t sc = (e >= pm.beg) ? (int32_t)pH[e - pm.beg] : jb.inf_min;
            if (sc > best_score) { best_score = sc; best_i = p; best_j = e; }
        }
    }
    res->best_score = best_score;
    res->best_i = best_i; res->best_j = best_j;
    if (!jb.ret_cigar) return;

    { /* simd_abpoa_ag_backtrack (:196-307) */
        int bi = best_i, bj = best_j, start_i = best_i, start_j = best_j;
        int cur_op = 0x1f, n_c = 0, status = ABAMD_JOB_OK;
        int look_end = jb.put_gap_at_end, put_right = jb.put_gap_on_right;
        int n_aln = 0, n_matched = 0;
        uint64_t *cig = jb.cigar;
        int id = jb.row_node_id[bi];
        if (best_j < qlen) dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, qlen - best_j, -1, qlen - 1, &status);
        while (bi > 0 && bj > 0 && status == ABAMD_JOB_OK) {
            const abamd_row_meta_t bm = meta[bi];
            const int rb = bm.beg, re = bm.end;
            const int64_t bw = re - rb + 1;
            const S *H = arena + bm.off * 3;
            const S *E1r = H + bw, *F1r = E1r + bw;
            const S Hj = (bj >= rb && bj <= re) ? H[bj - rb] : inf_min;
            const S Hjm1 = (bj - 1 >= rb && bj - 1 <= re) ? H[bj - 1 - rb] : inf_min;
            const S E1j = (bj >= rb && bj <= re) ? E1r[bj - rb] : inf_min;
            const S F1j = (bj >= rb && bj <= re) ? F1r[bj - rb] : inf_min;
            const S F1jm1 = (bj - 1 >= rb && bj - 1 <= re) ? F1r[bj - 1 - rb] : inf_min;
            if (local_mode && Hj == 0) break;
            start_i = bi; start_j = bj;
            const int pq0 = jb.pre_off[bi], pq1 = jb.pre_off[bi + 1];
            const S s = (S)mat_lds[m * jb.row_base[bi] + query[bj - 1]];
            const int is_match = jb.row_base[bi] == query[bj - 1];
            int hit = 0;
            if (put_right == 0 && look_end == 0 && (cur_op & 0x1)) {
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                    const S *pH = arena + pm.off * 3;
                    if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                        cur_op = 0x1f; hit = 1;
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                        bi = p; --bj; id = jb.row_node_id[bi];
                        ++n_aln; n_matched += is_match;
                        break;
                    }
                }
            }
            if (!hit && (cur_op & 0x2)) { /* deletion */
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj < pm.beg || bj > pm.end) continue;
                    const int poffc = bj - pm.beg;
                    const int64_t pbw = pm.end - pm.beg + 1;
                    const S *pH = arena + pm.off * 3;
                    const S *pE1 = pH + pbw;
                    if (cur_op & 0x1) {
                        if (Hj == (S)(pE1[poffc] + ps)) {
                            cur_op = ((S)(pH[poffc] - oe1) == pE1[poffc]) ? (0x1 | 0x18) : 0x2;
                            hit = 1; dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                            bi = p; id = jb.row_node_id[bi];
                            if (look_end) look_end = 0;
                            break;
                        }
                    } else {
                        if (E1j == (S)(pE1[poffc] - e1 + ps)) {
                            cur_op = ((S)(pH[poffc] - oe1) == pE1[poffc]) ? (0x1 | 0x18) : 0x2;
                            hit = 1; dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                            bi = p; id = jb.row_node_id[bi];
                            if (look_end) look_end = 0;
                            break;
                        }
                    }
                }
            }
            if (!hit && (cur_op & 0x18)) { /* insertion */
                if (cur_op & 0x1) {
                    if (Hj == F1j) {
                        if ((S)(Hjm1 - oe1) == F1j) { cur_op = 0x1 | 0x6; hit = 1; }
                        else if ((S)(F1jm1 - e1) == F1j) { cur_op = 0x8; hit = 1; }
                    }
                } else {
                    if ((S)(Hjm1 - oe1) == F1j) { cur_op = 0x1 | 0x6; hit = 1; }
                    else if ((S)(F1jm1 - e1) == F1j) { cur_op = 0x8; hit = 1; }
                }
                if (hit) {
                    dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, 1, id, bj - 1, &status);
                    --bj;
                    if (look_end) look_end = 0;
                    ++n_aln;
                }
            }
            if (!hit && (cur_op & 0x1)) {
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                    const S *pH = arena + pm.off * 3;
                    if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                        cur_op = 0x1f; hit = 1;
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                        bi = p; --bj; id = jb.row_node_id[bi];
                        ++n_aln; n_matched += is_match;
                        look_end = 0;
                        break;
                    }
                }
            }
            if (!hit) { status = ABAMD_JOB_BT_DEAD_END; break; }
        }
        if (status == ABAMD_JOB_OK && bj > 0)
            dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, bj, -1, bj - 1, &status);
        res->status = status;
        res->n_cigar = n_c;
        res->n_aln_bases = n_aln;
        res->n_matched_bases = n_matched;
        res->node_e = jb.row_node_id[best_i]; res->query_e = best_j - 1;
        res->node_s = jb.row_node_id[start_i]; res->query_s = start_j - 1;
    }
}

/* ------------------------------------------------------------------ */
/* Linear-gap global kernel: one H plane.                              */
/* Reference core simd_abpoa_lg_dp (abpoa_align_simd.c:727-815):       */
/* H = max over preds of (preH[j-1]+q, preH[j]-e1), then an in-row     */
/* max-plus insertion scan directly on H.                              */
/* Backtrack: simd_abpoa_lg_backtrack (:116-194).                      */
/* ------------------------------------------------------------------ */
template <typename S>
__global__ __launch_bounds__(WAVE * JOBS_PER_BLOCK, 7)
void lg_global_kernel(const abamd_gpu_job_t *__restrict__ jobs,
                      abamd_gpu_res_t *__restrict__ results, int n_jobs) {
    const int wid = threadIdx.x / WAVE;
    const int jid = blockIdx.x * JOBS_PER_BLOCK + wid;
    const int lane = threadIdx.x % WAVE;

    __shared__ int mat_lds[27 * 27];
    constexpr int BMAX = BmaxOf<S>::v;
    __shared__ S prev_lds[JOBS_PER_BLOCK][2][BMAX];
    {
        const int m0 = jobs[0].m;
        const int *mat0 = jobs[0].mat;
        for (int i = threadIdx.x; i < m0 * m0; i += WAVE * JOBS_PER_BLOCK)
            mat_lds[i] = mat0[i];
    }
    __syncthreads();
    if (jid >= n_jobs) return;
    /* by VALUE: every field lives in registers (uniform loads become
       s_loads into SGPRs) — a reference would re-load fields from global
       memory inside the row loop because arena/steering stores could alias
       the jobs array in the compiler's view */
    const abamd_gpu_job_t jb = jobs[jid];
    abamd_gpu_res_t *res = &results[jid];
    abamd_row_meta_t *__restrict__ meta = (abamd_row_meta_t*)jb.row_meta;

    const int qlen = jb.qlen, n_rows = jb.n_rows, w = jb.w, m = jb.m;
    const S inf_min = (S)jb.inf_min;
    const S e1 = (S)jb.e1;
    const int local_mode = jb.align_mode == 1, extend_mode = jb.align_mode == 2;
    int32_t run_best = jb.inf_min;
    int run_best_i = 0, run_best_j = 0, run_best_remain = jb.max_remain[0];
    int zdropped = 0;
    const int end_remain = jb.max_remain[n_rows - 1];
    S *arena = (S*)jb.arena;
    const uint8_t *__restrict__ query = jb.query;

    if (lane == 0) { res->status = ABAMD_JOB_OK; res->n_cigar = 0; }

    for (int i = lane; i < n_rows; i += WAVE) {
        jb.max_left[i] = jb.node_n_init;
        jb.max_right[i] = 0;
    }
    if (lane == 0) { jb.max_left[0] = 0; jb.max_right[0] = 0; }
    for (int k = jb.out_off[0] + lane; k < jb.out_off[1]; k += WAVE) {
        int o = jb.out_idx[k];
        jb.max_left[o] = 1; jb.max_right[o] = 1;
    }

    int buf_cur = 0;
    int prev_ok = 0, prev_row = -1, prev_beg = 0, prev_end = -1;

    /* first row (simd_abpoa_lg_first_dp, abpoa_align_simd.c:635-649) */
    int64_t used;
    {
        int mr = jb.max_remain[0] - end_remain - 1;
        int end0;
        if (jb.banded) {
            int t = jb.max_right[0] > qlen - mr ? jb.max_right[0] : qlen - mr;
            end0 = (qlen < t + w) ? qlen : t + w;
        } else end0 = qlen;
        if (lane == 0) { meta[0].beg = 0; meta[0].end = end0; meta[0].off = 0; }
        used = end0 + 1;
        S *H = arena;
        S *c = &prev_lds[wid][buf_cur][0];
        const int fits = end0 + 1 <= BMAX;
        for (int j = lane; j <= end0; j += WAVE) {
            S hv = local_mode ? (S)0 : (S)(-jb.e1 * j);
            H[j] = hv;
            if (fits) c[j] = hv;
        }
        if (fits) { prev_ok = 1; prev_row = 0; prev_beg = 0; prev_end = end0; }
        buf_cur ^= 1;
    }

    for (int r = 1; r < n_rows - 1; ++r) {
        const int pk0 = jb.pre_off[r], pk1 = jb.pre_off[r + 1];
        int beg, end;
        {
            int mr = jb.max_remain[r] - end_remain - 1;
            if (jb.banded) {
                int ml = jb.max_left[r], mrr = jb.max_right[r];
                int lo = ml < qlen - mr ? ml : qlen - mr;
                beg = lo - w; if (beg < 0) beg = 0;
                int hi = mrr > qlen - mr ? mrr : qlen - mr;
                end = hi + w; if (end > qlen) end = qlen;
                int min_pre_beg = 0x7fffffff;
                for (int k = pk0; k < pk1; ++k) {
                    const int pidx = jb.pre_idx[k];
                    int pb = (prev_ok && pidx == prev_row) ? prev_beg : meta[pidx].beg;
                    if (pb < min_pre_beg) min_pre_beg = pb;
                }
                if (beg < min_pre_beg) beg = min_pre_beg;
            } else { beg = 0; end = qlen; }
        }
        const int64_t bw = end - beg + 1;
        if (used + bw > jb.arena_cap) { if (lane == 0) res->status = ABAMD_JOB_ARENA_OVERFLOW; return; }
        const int64_t off = used;
        if (lane == 0) { meta[r].beg = beg; meta[r].end = end; meta[r].off = off; }
        used += bw;
        S *H = arena + off;
        const uint8_t base = jb.row_base[r];
        const int *mrow = &mat_lds[base * m];
        const int cache_fits = bw <= BMAX;
        S *cw = &prev_lds[wid][buf_cur][0];
        const S *cr = &prev_lds[wid][buf_cur ^ 1][0];

        S carry_h = inf_min;
        S lmax = inf_min; int lleft = -1, lright = -1;

        for (int cs = beg; cs <= end; cs += WAVE) {
            const int j = cs + lane;
            const bool act = j <= end;
            const S q = (S)((j == 0 || !act) ? 0 : mrow[query[j - 1]]);
            S h = inf_min;
            for (int k = pk0; k < pk1; ++k) {
                const int p = jb.pre_idx[k];
                const S ps = (S)jb.pre_ps[k];
                if (prev_ok && p == prev_row) {
                    if (act) {
                        if (local_mode && j == 0) { S v = (S)(ps + q); if (v > h) h = v; }
                        if (j - 1 >= prev_beg && j - 1 <= prev_end) {
                            S v = (S)(cr[j - 1 - prev_beg] + ps + q);
                            if (v > h) h = v;
                        }
                        if (j >= prev_beg && j <= prev_end) {
                            S v = (S)(cr[j - prev_beg] + ps - e1);
                            if (v > h) h = v;
                        }
                    }
                    continue;
                }
                const abamd_row_meta_t pm = meta[p];
                const S *__restrict__ pH = arena + pm.off;
                if (act) {
                    if (local_mode && j == 0) { S v = (S)(ps + q); if (v > h) h = v; }
                    if (j - 1 >= pm.beg && j - 1 <= pm.end) {
                        S v = (S)(pH[j - 1 - pm.beg] + ps + q);
                        if (v > h) h = v;
                    }
                    if (j >= pm.beg && j <= pm.end) {
                        S v = (S)(pH[j - pm.beg] + ps - e1);
                        if (v > h) h = v;
                    }
                }
            }
            if (!act) h = inf_min;
            /* in-row insertion scan on H with inter-chunk carry */
            S hs = (S)__shfl_up((int)h, 1);
            if (lane == 0) {
                if (cs != beg) h = smax(h, (S)(carry_h - e1));
            } else h = smax(h, (S)(hs - e1));
            h = scan_maxplus(h, jb.e1, inf_min, lane);
            carry_h = (S)__builtin_amdgcn_readlane((int)h, WAVE - 1);
            if (local_mode) h = smax(h, (S)0); /* clamp AFTER the scan+carry */
            if (act) {
                H[j - beg] = h;
                if (cache_fits) cw[j - beg] = h;
                if (h > lmax) { lmax = h; lleft = j; lright = j; }
                else if (h == lmax) { lright = j; }
            }
        }

        if (cache_fits) { prev_ok = 1; prev_row = r; prev_beg = beg; prev_end = end; }
        else prev_ok = 0;
        buf_cur ^= 1;

        if (jb.banded || local_mode || extend_mode) {
            int mv = wave_red_max_i32((int)lmax);
            int ll = ((int)lmax == mv && lleft >= 0) ? lleft : 0x7fffffff;
            int rr = ((int)lmax == mv && lright >= 0) ? lright : -1;
            ll = wave_red_min_i32(ll);
            rr = wave_red_max_i32(rr);
            if (local_mode) {
                if (mv > run_best) { run_best = mv; run_best_i = r; run_best_j = ll; }
            } else if (extend_mode) {
                if (mv > run_best) {
                    run_best = mv; run_best_i = r; run_best_j = rr;
                    run_best_remain = jb.max_remain[r];
                } else if (jb.zdrop > 0) {
                    int delta = run_best_remain - jb.max_remain[r];
                    int dd = delta - (rr - run_best_j); if (dd < 0) dd = -dd;
                    if (run_best - mv > jb.zdrop + jb.e1 * dd) zdropped = 1;
                }
            }
            if (!zdropped && jb.banded) {
                for (int k = jb.out_off[r] + lane; k < jb.out_off[r + 1]; k += WAVE) {
                    int o = jb.out_idx[k];
                    if (rr + 1 > jb.max_right[o]) jb.max_right[o] = rr + 1;
                    if (ll + 1 < jb.max_left[o]) jb.max_left[o] = ll + 1;
                }
            }
            if (zdropped) break;
        }
    }

    if (lane == 0) res->cells = used;
    if (lane != 0) return;

    int32_t best_score = run_best;
    int best_i = run_best_i, best_j = run_best_j;
    if (jb.align_mode == 0) {
        best_score = jb.inf_min; best_i = 0; best_j = 0;
        for (int k = jb.pre_off[n_rows - 1]; k < jb.pre_off[n_rows]; ++k) {
            const int p = jb.pre_idx[k];
            const abamd_row_meta_t pm = meta[p];
            int e = pm.end < qlen ? pm.end : qlen;
            const S *pH = arena + pm.off;
            int32_t sc = (e >= pm.beg) ? (int32_t)pH[e - pm.beg] : jb.inf_min;
            if (sc > best_score) { best_score = sc; best_i = p; best_j = e; }
        }
    }
    res->best_score = best_score;
    res->best_i = best_i; res->best_j = best_j;
    if (!jb.ret_cigar) return;

    { /* simd_abpoa_lg_backtrack (:116-194) */
        int bi = best_i, bj = best_j, start_i = best_i, start_j = best_j;
        int n_c = 0, status = ABAMD_JOB_OK;
        int look_end = jb.put_gap_at_end, put_right = jb.put_gap_on_right;
        int n_aln = 0, n_matched = 0;
        uint64_t *cig = jb.cigar;
        int id = jb.row_node_id[bi];
        if (best_j < qlen) dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, qlen - best_j, -1, qlen - 1, &status);
        while (bi > 0 && bj > 0 && status == ABAMD_JOB_OK) {
            const abamd_row_meta_t bm = meta[bi];
            const int rb = bm.beg, re = bm.end;
            const S *H = arena + bm.off;
            const S Hj = (bj >= rb && bj <= re) ? H[bj - rb] : inf_min;
            const S Hjm1 = (bj - 1 >= rb && bj - 1 <= re) ? H[bj - 1 - rb] : inf_min;
            if (local_mode && Hj == 0) break;
            start_i = bi; start_j = bj;
            const int pq0 = jb.pre_off[bi], pq1 = jb.pre_off[bi + 1];
            const S s = (S)mat_lds[m * jb.row_base[bi] + query[bj - 1]];
            const int is_match = jb.row_base[bi] == query[bj - 1];
            int hit = 0;
            if (put_right == 0 && look_end == 0) {
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                    const S *pH = arena + pm.off;
                    if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                        bi = p; --bj; id = jb.row_node_id[bi]; hit = 1;
                        ++n_aln; n_matched += is_match;
                        break;
                    }
                }
            }
            if (!hit) { /* deletion */
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj < pm.beg || bj > pm.end) continue;
                    const S *pH = arena + pm.off;
                    if ((S)(pH[bj - pm.beg] - e1 + ps) == Hj) {
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                        bi = p; id = jb.row_node_id[bi]; hit = 1;
                        if (look_end) look_end = 0;
                        break;
                    }
                }
            }
            if (!hit) { /* insertion */
                if ((S)(Hjm1 - e1) == Hj) {
                    dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, 1, id, bj - 1, &status);
                    --bj;
                    if (look_end) look_end = 0;
                    hit = 1; ++n_aln;
                }
            }
            if (!hit) {
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                    const S *pH = arena + pm.off;
                    if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                        bi = p; --bj; id = jb.row_node_id[bi]; hit = 1;
                        ++n_aln; n_matched += is_match;
                        look_end = 0;
                        break;
                    }
                }
            }
            if (!hit) { status = ABAMD_JOB_BT_DEAD_END; break; }
        }
        if (status == ABAMD_JOB_OK && bj > 0)
            dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, bj, -1, bj - 1, &status);
        res->status = status;
        res->n_cigar = n_c;
        res->n_aln_bases = n_aln;
        res->n_matched_bases = n_matched;
        res->node_e = jb.row_node_id[best_i]; res->query_e = best_j - 1;
        res->node_s = jb.row_node_id[start_i]; res->query_s = start_j - 1;
    }
}

/* ------------------------------------------------------------------ */
/* Multi-wave affine kernel (3 planes H,E1,F1): same 8-waves-per-job   */
/* design as the convex kernel; the F chain feeds from H BEFORE the E  */
/* fold (simd_abpoa_ag_dp, abpoa_align_simd.c:817-933) and the stored  */
/* E is inf_min when the insertion won the cell (SIMDSetIfEqual).      */
/* ------------------------------------------------------------------ */
template <typename S>
__global__ __launch_bounds__(MWT, 1)
void ag_global_mw_kernel(const abamd_gpu_job_t *__restrict__ jobs,
                         abamd_gpu_res_t *__restrict__ results, int n_jobs) {
    const int jid = blockIdx.x;
    if (jid >= n_jobs) return;
    const int tid = threadIdx.x;
    const int wv = tid / WAVE, lane = tid % WAVE;

    __shared__ int mat_lds[27 * 27];
    __shared__ S prev_lds[2][2 * BMW];
    __shared__ S sc_h[MWAVES], sc_f1[MWAVES];
    __shared__ int sc_red[3 * MWAVES];
    __shared__ S sc_carry[2];

    const abamd_gpu_job_t jb = jobs[jid];
    abamd_gpu_res_t *res = &results[jid];
    abamd_row_meta_t *__restrict__ meta = (abamd_row_meta_t*)jb.row_meta;

    const int qlen = jb.qlen, n_rows = jb.n_rows, w = jb.w, m = jb.m;
    const S inf_min = (S)jb.inf_min;
    const S e1 = (S)jb.e1, oe1 = (S)jb.oe1;
    const int local_mode = jb.align_mode == 1, extend_mode = jb.align_mode == 2;
    int32_t run_best = jb.inf_min;
    int run_best_i = 0, run_best_j = 0, run_best_remain = jb.max_remain[0];
    int zdropped = 0;
    const int end_remain = jb.max_remain[n_rows - 1];
    S *arena = (S*)jb.arena;
    const uint8_t *__restrict__ query = jb.query;

    for (int i = tid; i < m * m; i += MWT) mat_lds[i] = jb.mat[i];
    if (tid == 0) { res->status = ABAMD_JOB_OK; res->n_cigar = 0; }

    for (int i = tid; i < n_rows; i += MWT) {
        jb.max_left[i] = jb.node_n_init;
        jb.max_right[i] = 0;
    }
    if (tid == 0) { jb.max_left[0] = 0; jb.max_right[0] = 0; }
    for (int k = jb.out_off[0] + tid; k < jb.out_off[1]; k += MWT) {
        int o = jb.out_idx[k];
        jb.max_left[o] = 1; jb.max_right[o] = 1;
    }
    __syncthreads();

    int buf_cur = 0;
    int prev_ok = 0, prev_row = -1, prev_beg = 0, prev_end = -1;

    /* first row (simd_abpoa_ag_first_dp, abpoa_align_simd.c:651-667) */
    int64_t used;
    {
        int mr = jb.max_remain[0] - end_remain - 1;
        int end0;
        if (jb.banded) {
            int t = jb.max_right[0] > qlen - mr ? jb.max_right[0] : qlen - mr;
            end0 = (qlen < t + w) ? qlen : t + w;
        } else end0 = qlen;
        if (tid == 0) { meta[0].beg = 0; meta[0].end = end0; meta[0].off = 0; }
        int64_t bw = end0 + 1;
        used = bw;
        S *H = arena, *E1 = arena + bw, *F1 = arena + 2 * bw;
        S *c = &prev_lds[buf_cur][0];
        const int fits = end0 + 1 <= BMW;
        for (int j = tid; j <= end0; j += MWT) {
            S hv, e1v2;
            if (local_mode) {
                hv = 0; e1v2 = 0; F1[j] = 0;
            } else if (j == 0) {
                hv = 0; e1v2 = (S)(0 - oe1);
                F1[0] = inf_min;
            } else {
                S f1 = (S)(-(jb.o1 + jb.e1 * j));
                F1[j] = f1;
                hv = f1; e1v2 = inf_min;
            }
            H[j] = hv; E1[j] = e1v2;
            if (fits) { c[j] = hv; c[BMW + j] = e1v2; }
        }
        if (fits) { prev_ok = 1; prev_row = 0; prev_beg = 0; prev_end = end0; }
        buf_cur ^= 1;
        __syncthreads();
    }

    int cur_pk0 = jb.pre_off[1];
    int cur_pk1 = jb.pre_off[2 <= n_rows ? 2 : n_rows];
    int cur_oo0 = jb.out_off[1];
    int cur_oo1 = jb.out_off[2 <= n_rows ? 2 : n_rows];
    int cur_remain = jb.max_remain[1];
    int cur_ml_mem = jb.max_left[1];
    int cur_mr_mem = jb.max_right[1];
    int cur_pidx0 = jb.pre_idx[cur_pk0];
    int cur_ps0 = jb.pre_ps[cur_pk0];
    int push_ml = 0x7fffffff, push_mr = -0x7fffffff;
    for (int r = 1; r < n_rows - 1; ++r) {
        const int pk0 = cur_pk0, pk1 = cur_pk1;
        const int oo0 = cur_oo0, oo1 = cur_oo1;
        const int row_remain = cur_remain;
        const int pidx0 = cur_pidx0;
        const S ps0 = (S)cur_ps0;
        const int ml_eff = cur_ml_mem < push_ml ? cur_ml_mem : push_ml;
        const int mr_eff = cur_mr_mem > push_mr ? cur_mr_mem : push_mr;
        {
            const int nr = r + 1;
            cur_pk0 = pk1;
            cur_pk1 = jb.pre_off[nr + 1];
            cur_oo0 = oo1;
            cur_oo1 = jb.out_off[nr + 1];
            cur_remain = jb.max_remain[nr];
            cur_ml_mem = jb.max_left[nr];
            cur_mr_mem = jb.max_right[nr];
            cur_pidx0 = jb.pre_idx[cur_pk0];
            cur_ps0 = jb.pre_ps[cur_pk0];
        }
        int beg, end;
        {
            int mr = row_remain - end_remain - 1;
            if (jb.banded) {
                int lo = ml_eff < qlen - mr ? ml_eff : qlen - mr;
                beg = lo - w; if (beg < 0) beg = 0;
                int hi = mr_eff > qlen - mr ? mr_eff : qlen - mr;
                end = hi + w; if (end > qlen) end = qlen;
                int min_pre_beg;
                if (pk1 - pk0 == 1) {
                    min_pre_beg = (prev_ok && pidx0 == prev_row) ? prev_beg : meta[pidx0].beg;
                } else {
                    min_pre_beg = 0x7fffffff;
                    for (int k = pk0; k < pk1; ++k) {
                        const int pidx = jb.pre_idx[k];
                        int pb = (prev_ok && pidx == prev_row) ? prev_beg : meta[pidx].beg;
                        if (pb < min_pre_beg) min_pre_beg = pb;
                    }
                }
                if (beg < min_pre_beg) beg = min_pre_beg;
            } else { beg = 0; end = qlen; }
        }
        const int64_t bw = end - beg + 1;
        if (used + bw > jb.arena_cap) { if (tid == 0) res->status = ABAMD_JOB_ARENA_OVERFLOW; return; }
        const int64_t off = used;
        if (tid == 0) { meta[r].beg = beg; meta[r].end = end; meta[r].off = off; }
        used += bw;
        S *H = arena + off * 3, *E1r = H + bw, *F1r = E1r + bw;
        const uint8_t base = jb.row_base[r];
        const int *mrow = &mat_lds[base * m];
        const int cache_fits = bw <= BMW;
        S *cw = &prev_lds[buf_cur][0];
        const S *cr = &prev_lds[buf_cur ^ 1][0];
        const bool fast1 = (pk1 - pk0 == 1) && prev_ok && (pidx0 == prev_row);

        S carry_hm = inf_min, f1c = inf_min;
        S lmax = inf_min; int lleft = -1, lright = -1;

        for (int ss = beg; ss <= end; ss += MWT) {
            const int j = ss + tid;
            const bool act = j <= end;
            S h = inf_min, e1v = inf_min;
            if (fast1) {
                if (act) {
                    if (local_mode && j == 0) { if (ps0 > h) h = ps0; }
                    if (j - 1 >= prev_beg && j - 1 <= prev_end) {
                        S v = (S)(cr[j - 1 - prev_beg] + ps0);
                        if (v > h) h = v;
                    }
                    if (j >= prev_beg && j <= prev_end) {
                        S v1 = (S)(cr[BMW + j - prev_beg] + ps0);
                        if (v1 > e1v) e1v = v1;
                    }
                }
            } else for (int k = pk0; k < pk1; ++k) {
                const int p = jb.pre_idx[k];
                const S ps = (S)jb.pre_ps[k];
                if (prev_ok && p == prev_row) {
                    if (act) {
                        if (local_mode && j == 0) { if (ps > h) h = ps; }
                        if (j - 1 >= prev_beg && j - 1 <= prev_end) {
                            S v = (S)(cr[j - 1 - prev_beg] + ps);
                            if (v > h) h = v;
                        }
                        if (j >= prev_beg && j <= prev_end) {
                            S v1 = (S)(cr[BMW + j - prev_beg] + ps);
                            if (v1 > e1v) e1v = v1;
                        }
                    }
                    continue;
                }
                const abamd_row_meta_t pm = meta[p];
                const int64_t pbw = pm.end - pm.beg + 1;
                const S *__restrict__ pH = arena + pm.off * 3;
                const S *__restrict__ pE1 = pH + pbw;
                if (act) {
                    if (local_mode && j == 0) { if (ps > h) h = ps; }
                    if (j - 1 >= pm.beg && j - 1 <= pm.end) {
                        S v = (S)(pH[j - 1 - pm.beg] + ps);
                        if (v > h) h = v;
                    }
                    if (j >= pm.beg && j <= pm.end) {
                        S v1 = (S)(pE1[j - pm.beg] + ps);
                        if (v1 > e1v) e1v = v1;
                    }
                }
            }
            const S q = (S)((j == 0 || !act) ? 0 : mrow[query[j - 1]]);
            S hm = (S)(h + q);   /* M+q: the F chain feeds from this, pre-E */
            if (!act) hm = inf_min;

            if (lane == WAVE - 1) sc_h[wv] = hm;
            __syncthreads();
            S hmshift = (S)__shfl_up((int)hm, 1);
            S c1;
            if (lane == 0) {
                if (wv == 0) {
                    if (ss == beg) c1 = (S)(inf_min - oe1);
                    else c1 = smax((S)(carry_hm - oe1), (S)(f1c - e1));
                } else {
                    c1 = (S)(sc_h[wv - 1] - oe1);
                }
            } else c1 = (S)(hmshift - oe1);
            S L1 = scan_maxplus(c1, jb.e1, inf_min, lane);
            if (lane == WAVE - 1) sc_f1[wv] = L1;
            __syncthreads();
            S f1 = L1;
            if (wv > 0) {
                S C1 = sc_f1[0];
                for (int ww = 1; ww < wv; ++ww)
                    C1 = smax(sc_f1[ww], (S)(C1 - (S)(WAVE * jb.e1)));
                f1 = smax(L1, (S)(C1 - (S)((lane + 1) * jb.e1)));
            }

            S tmp = smax(hm, e1v);
            S hf = smax(tmp, f1);
            if (local_mode) hf = smax(hf, (S)0);
            S e1n = (hf == tmp) ? smax((S)(e1v - e1), (S)(hf - oe1))
                                : (local_mode ? (S)0 : inf_min);
            if (act) {
                H[j - beg] = hf; E1r[j - beg] = e1n; F1r[j - beg] = f1;
                if (cache_fits) { cw[j - beg] = hf; cw[BMW + j - beg] = e1n; }
                if (hf > lmax) { lmax = hf; lleft = j; lright = j; }
                else if (hf == lmax) { lright = j; }
            }
            const bool more = ss + MWT <= end;
            if (more && j == ss + MWT - 1) {
                sc_carry[0] = hm; sc_carry[1] = f1;
            }
            if (!more && (jb.banded || local_mode || extend_mode)) {
                int mvw = wave_red_max_i32((int)lmax);
                int llw = ((int)lmax == mvw && lleft >= 0) ? lleft : 0x7fffffff;
                int rrw = ((int)lmax == mvw && lright >= 0) ? lright : -1;
                llw = wave_red_min_i32(llw);
                rrw = wave_red_max_i32(rrw);
                if (lane == 0) {
                    sc_red[wv] = mvw;
                    sc_red[MWAVES + wv] = llw;
                    sc_red[2 * MWAVES + wv] = rrw;
                }
            }
            __syncthreads(); /* also orders cw writes before the next row's reads */
            if (more) { carry_hm = sc_carry[0]; f1c = sc_carry[1]; }
        }

        if (cache_fits) { prev_ok = 1; prev_row = r; prev_beg = beg; prev_end = end; }
        else prev_ok = 0;
        buf_cur ^= 1;

        if (jb.banded || local_mode || extend_mode) {
            int mv = sc_red[0], ll = sc_red[MWAVES], rr = sc_red[2 * MWAVES];
            #pragma unroll
            for (int ww = 1; ww < MWAVES; ++ww) {
                int m2 = sc_red[ww];
                if (m2 > mv) { mv = m2; ll = sc_red[MWAVES + ww]; rr = sc_red[2 * MWAVES + ww]; }
                else if (m2 == mv) {
                    if (sc_red[MWAVES + ww] < ll) ll = sc_red[MWAVES + ww];
                    if (sc_red[2 * MWAVES + ww] > rr) rr = sc_red[2 * MWAVES + ww];
                }
            }
            if (local_mode) {
                if (mv > run_best) { run_best = mv; run_best_i = r; run_best_j = ll; }
            } else if (extend_mode) {
                if (mv > run_best) {
                    run_best = mv; run_best_i = r; run_best_j = rr;
                    run_best_remain = row_remain;
                } else if (jb.zdrop > 0) {
                    int delta = run_best_remain - row_remain;
                    int dd = delta - (rr - run_best_j); if (dd < 0) dd = -dd;
                    if (run_best - mv > jb.zdrop + jb.e1 * dd) zdropped = 1;
                }
            }
            push_ml = 0x7fffffff; push_mr = -0x7fffffff;
            if (!zdropped && jb.banded) {
                for (int k = oo0; k < oo1; ++k) {
                    int o = jb.out_idx[k];
                    if (o == r + 1) {
                        if (rr + 1 > push_mr) push_mr = rr + 1;
                        if (ll + 1 < push_ml) push_ml = ll + 1;
                    } else {
                        if (rr + 1 > jb.max_right[o]) jb.max_right[o] = rr + 1;
                        if (ll + 1 < jb.max_left[o]) jb.max_left[o] = ll + 1;
                    }
                }
            }
            if (zdropped) break;
        } else {
            push_ml = 0x7fffffff; push_mr = -0x7fffffff;
        }
    }

    __syncthreads();
    if (tid == 0) res->cells = used;
    if (tid != 0) return;

    int32_t best_score = run_best;
    int best_i = run_best_i, best_j = run_best_j;
    if (jb.align_mode == 0) {
        best_score = jb.inf_min; best_i = 0; best_j = 0;
        for (int k = jb.pre_off[n_rows - 1]; k < jb.pre_off[n_rows]; ++k) {
            const int p = jb.pre_idx[k];
            const abamd_row_meta_t pm = meta[p];
            int e = pm.end < qlen ? pm.end : qlen;
            const S *pH = arena + pm.off * 3;
            int32_t sc = (e >= pm.beg) ? (int32_t)pH[e - pm.beg] : jb.inf_min;
            if (sc > best_score) { best_score = sc; best_i = p; best_j = e; }
        }
    }
    res->best_score = best_score;
    res->best_i = best_i; res->best_j = best_j;
    if (!jb.ret_cigar) return;
    { /* simd_abpoa_ag_backtrack (:196-307) */
        int bi = best_i, bj = best_j, start_i = best_i, start_j = best_j;
        int cur_op = 0x1f, n_c = 0, status = ABAMD_JOB_OK;
        int look_end = jb.put_gap_at_end, put_right = jb.put_gap_on_right;
        int n_aln = 0, n_matched = 0;
        uint64_t *cig = jb.cigar;
        int id = jb.row_node_id[bi];
        if (best_j < qlen) dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, qlen - best_j, -1, qlen - 1, &status);
        while (bi > 0 && bj > 0 && status == ABAMD_JOB_OK) {
            const abamd_row_meta_t bm = meta[bi];
            const int rb = bm.beg, re = bm.end;
            const int64_t bw = re - rb + 1;
            const S *H = arena + bm.off * 3;
            const S *E1r = H + bw, *F1r = E1r + bw;
            const S Hj = (bj >= rb && bj <= re) ? H[bj - rb] : inf_min;
            const S Hjm1 = (bj - 1 >= rb && bj - 1 <= re) ? H[bj - 1 - rb] : inf_min;
            const S E1j = (bj >= rb && bj <= re) ? E1r[bj - rb] : inf_min;
            const S F1j = (bj >= rb && bj <= re) ? F1r[bj - rb] : inf_min;
            const S F1jm1 = (bj - 1 >= rb && bj - 1 <= re) ? F1r[bj - 1 - rb] : inf_min;
            if (local_mode && Hj == 0) break;
            start_i = bi; start_j = bj;
            const int pq0 = jb.pre_off[bi], pq1 = jb.pre_off[bi + 1];
            const S s = (S)mat_lds[m * jb.row_base[bi] + query[bj - 1]];
            const int is_match = jb.row_base[bi] == query[bj - 1];
            int hit = 0;
            if (put_right == 0 && look_end == 0 && (cur_op & 0x1)) {
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                    const S *pH = arena + pm.off * 3;
                    if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                        cur_op = 0x1f; hit = 1;
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                        bi = p; --bj; id = jb.row_node_id[bi];
                        ++n_aln; n_matched += is_match;
                        break;
                    }
                }
            }
            if (!hit && (cur_op & 0x2)) { /* deletion */
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj < pm.beg || bj > pm.end) continue;
                    const int poffc = bj - pm.beg;
                    const int64_t pbw = pm.end - pm.beg + 1;
                    const S *pH = arena + pm.off * 3;
                    const S *pE1 = pH + pbw;
                    if (cur_op & 0x1) {
                        if (Hj == (S)(pE1[poffc] + ps)) {
                            cur_op = ((S)(pH[poffc] - oe1) == pE1[poffc]) ? (0x1 | 0x18) : 0x2;
                            hit = 1; dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                            bi = p; id = jb.row_node_id[bi];
                            if (look_end) look_end = 0;
                            break;
                        }
                    } else {
                        if (E1j == (S)(pE1[poffc] - e1 + ps)) {
                            cur_op = ((S)(pH[poffc] - oe1) == pE1[poffc]) ? (0x1 | 0x18) : 0x2;
                            hit = 1; dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                            bi = p; id = jb.row_node_id[bi];
                            if (look_end) look_end = 0;
                            break;
                        }
                    }
                }
            }
            if (!hit && (cur_op & 0x18)) { /* insertion */
                if (cur_op & 0x1) {
                    if (Hj == F1j) {
                        if ((S)(Hjm1 - oe1) == F1j) { cur_op = 0x1 | 0x6; hit = 1; }
                        else if ((S)(F1jm1 - e1) == F1j) { cur_op = 0x8; hit = 1; }
                    }
                } else {
                    if ((S)(Hjm1 - oe1) == F1j) { cur_op = 0x1 | 0x6; hit = 1; }
                    else if ((S)(F1jm1 - e1) == F1j) { cur_op = 0x8; hit = 1; }
                }
                if (hit) {
                    dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, 1, id, bj - 1, &status);
                    --bj;
                    if (look_end) look_end = 0;
                    ++n_aln;
                }
            }
            if (!hit && (cur_op & 0x1)) {
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                    const S *pH = arena + pm.off * 3;
                    if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                        cur_op = 0x1f; hit = 1;
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                        bi = p; --bj; id = jb.row_node_id[bi];
                        ++n_aln; n_matched += is_match;
                        look_end = 0;
                        break;
                    }
                }
            }
            if (!hit) { status = ABAMD_JOB_BT_DEAD_END; break; }
        }
        if (status == ABAMD_JOB_OK && bj > 0)
            dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, bj, -1, bj - 1, &status);
        res->status = status;
        res->n_cigar = n_c;
        res->n_aln_bases = n_aln;
        res->n_matched_bases = n_matched;
        res->node_e = jb.row_node_id[best_i]; res->query_e = best_j - 1;
        res->node_s = jb.row_node_id[start_i]; res->query_s = start_j - 1;
    }
}


/* ------------------------------------------------------------------ */
/* Multi-wave linear kernel (1 plane): candidates carry the query score */
/* and the deletion term, then the insertion max-plus scan runs on H    */
/* itself (simd_abpoa_lg_dp, abpoa_align_simd.c:727-815); the local     */
/* clamp applies AFTER the scan and the cross-wave carries are          */
/* PRE-clamp, exactly like the one-wave kernel's readlane carry.        */
/* ------------------------------------------------------------------ */
template <typename S>
__global__ __launch_bounds__(MWT, 1)
void lg_global_mw_kernel(const abamd_gpu_job_t *__restrict__ jobs,
                         abamd_gpu_res_t *__restrict__ results, int n_jobs) {
    const int jid = blockIdx.x;
    if (jid >= n_jobs) return;
    const int tid = threadIdx.x;
    const int wv = tid / WAVE, lane = tid % WAVE;

    __shared__ int mat_lds[27 * 27];
    __shared__ S prev_lds[2][BMW];
    __shared__ S sc_f1[MWAVES];
    __shared__ int sc_red[3 * MWAVES];
    __shared__ S sc_carry[1];

    const abamd_gpu_job_t jb = jobs[jid];
    abamd_gpu_res_t *res = &results[jid];
    abamd_row_meta_t *__restrict__ meta = (abamd_row_meta_t*)jb.row_meta;

    const int qlen = jb.qlen, n_rows = jb.n_rows, w = jb.w, m = jb.m;
    const S inf_min = (S)jb.inf_min;
    const S e1 = (S)jb.e1;
    const int local_mode = jb.align_mode == 1, extend_mode = jb.align_mode == 2;
    int32_t run_best = jb.inf_min;
    int run_best_i = 0, run_best_j = 0, run_best_remain = jb.max_remain[0];
    int zdropped = 0;
    const int end_remain = jb.max_remain[n_rows - 1];
    S *arena = (S*)jb.arena;
    const uint8_t *__restrict__ query = jb.query;

    for (int i = tid; i < m * m; i += MWT) mat_lds[i] = jb.mat[i];
    if (tid == 0) { res->status = ABAMD_JOB_OK; res->n_cigar = 0; }

    for (int i = tid; i < n_rows; i += MWT) {
        jb.max_left[i] = jb.node_n_init;
        jb.max_right[i] = 0;
    }
    if (tid == 0) { jb.max_left[0] = 0; jb.max_right[0] = 0; }
    for (int k = jb.out_off[0] + tid; k < jb.out_off[1]; k += MWT) {
        int o = jb.out_idx[k];
        jb.max_left[o] = 1; jb.max_right[o] = 1;
    }
    __syncthreads();

    int buf_cur = 0;
    int prev_ok = 0, prev_row = -1, prev_beg = 0, prev_end = -1;

    /* first row (simd_abpoa_lg_first_dp, abpoa_align_simd.c:635-649) */
    int64_t used;
    {
        int mr = jb.max_remain[0] - end_remain - 1;
        int end0;
        if (jb.banded) {
            int t = jb.max_right[0] > qlen - mr ? jb.max_right[0] : qlen - mr;
            end0 = (qlen < t + w) ? qlen : t + w;
        } else end0 = qlen;
        if (tid == 0) { meta[0].beg = 0; meta[0].end = end0; meta[0].off = 0; }
        used = end0 + 1;
        S *H = arena;
        S *c = &prev_lds[buf_cur][0];
        const int fits = end0 + 1 <= BMW;
        for (int j = tid; j <= end0; j += MWT) {
            S hv = local_mode ? (S)0 : (S)(-jb.e1 * j);
            H[j] = hv;
            if (fits) c[j] = hv;
        }
        if (fits) { prev_ok = 1; prev_row = 0; prev_beg = 0; prev_end = end0; }
        buf_cur ^= 1;
        __syncthreads();
    }

    int cur_pk0 = jb.pre_off[1];
    int cur_pk1 = jb.pre_off[2 <= n_rows ? 2 : n_rows];
    int cur_oo0 = jb.out_off[1];
    int cur_oo1 = jb.out_off[2 <= n_rows ? 2 : n_rows];
    int cur_remain = jb.max_remain[1];
    int cur_ml_mem = jb.max_left[1];
    int cur_mr_mem = jb.max_right[1];
    int cur_pidx0 = jb.pre_idx[cur_pk0];
    int cur_ps0 = jb.pre_ps[cur_pk0];
    int push_ml = 0x7fffffff, push_mr = -0x7fffffff;
    for (int r = 1; r < n_rows - 1; ++r) {
        const int pk0 = cur_pk0, pk1 = cur_pk1;
        const int oo0 = cur_oo0, oo1 = cur_oo1;
        const int row_remain = cur_remain;
        const int pidx0 = cur_pidx0;
        const S ps0 = (S)cur_ps0;
        const int ml_eff = cur_ml_mem < push_ml ? cur_ml_mem : push_ml;
        const int mr_eff = cur_mr_mem > push_mr ? cur_mr_mem : push_mr;
        {
            const int nr = r + 1;
            cur_pk0 = pk1;
            cur_pk1 = jb.pre_off[nr + 1];
            cur_oo0 = oo1;
            cur_oo1 = jb.out_off[nr + 1];
            cur_remain = jb.max_remain[nr];
            cur_ml_mem = jb.max_left[nr];
            cur_mr_mem = jb.max_right[nr];
            cur_pidx0 = jb.pre_idx[cur_pk0];
            cur_ps0 = jb.pre_ps[cur_pk0];
        }
        int beg, end;
        {
            int mr = row_remain - end_remain - 1;
            if (jb.banded) {
                int lo = ml_eff < qlen - mr ? ml_eff : qlen - mr;
                beg = lo - w; if (beg < 0) beg = 0;
                int hi = mr_eff > qlen - mr ? mr_eff : qlen - mr;
                end = hi + w; if (end > qlen) end = qlen;
                int min_pre_beg;
                if (pk1 - pk0 == 1) {
                    min_pre_beg = (prev_ok && pidx0 == prev_row) ? prev_beg : meta[pidx0].beg;
                } else {
                    min_pre_beg = 0x7fffffff;
                    for (int k = pk0; k < pk1; ++k) {
                        const int pidx = jb.pre_idx[k];
                        int pb = (prev_ok && pidx == prev_row) ? prev_beg : meta[pidx].beg;
                        if (pb < min_pre_beg) min_pre_beg = pb;
                    }
                }
                if (beg < min_pre_beg) beg = min_pre_beg;
            } else { beg = 0; end = qlen; }
        }
        const int64_t bw = end - beg + 1;
        if (used + bw > jb.arena_cap) { if (tid == 0) res->status = ABAMD_JOB_ARENA_OVERFLOW; return; }
        const int64_t off = used;
        if (tid == 0) { meta[r].beg = beg; meta[r].end = end; meta[r].off = off; }
        used += bw;
        S *H = arena + off;
        const uint8_t base = jb.row_base[r];
        const int *mrow = &mat_lds[base * m];
        const int cache_fits = bw <= BMW;
        S *cw = &prev_lds[buf_cur][0];
        const S *cr = &prev_lds[buf_cur ^ 1][0];
        const bool fast1 = (pk1 - pk0 == 1) && prev_ok && (pidx0 == prev_row);

        S carry_h = inf_min;                 /* PRE-clamp scan value at ss-1 */
        S lmax = inf_min; int lleft = -1, lright = -1;

        for (int ss = beg; ss <= end; ss += MWT) {
            const int j = ss + tid;
            const bool act = j <= end;
            const S q = (S)((j == 0 || !act) ? 0 : mrow[query[j - 1]]);
            S h = inf_min;
            if (fast1) {
                if (act) {
                    if (local_mode && j == 0) { S v = (S)(ps0 + q); if (v > h) h = v; }
                    if (j - 1 >= prev_beg && j - 1 <= prev_end) {
                        S v = (S)(cr[j - 1 - prev_beg] + ps0 + q);
                        if (v > h) h = v;
                    }
                    if (j >= prev_beg && j <= prev_end) {
                        S v = (S)(cr[j - prev_beg] + ps0 - e1);
                        if (v > h) h = v;
                    }
                }
            } else for (int k = pk0; k < pk1; ++k) {
                const int p = jb.pre_idx[k];
                const S ps = (S)jb.pre_ps[k];
                if (prev_ok && p == prev_row) {
                    if (act) {
                        if (local_mode && j == 0) { S v = (S)(ps + q); if (v > h) h = v; }
                        if (j - 1 >= prev_beg && j - 1 <= prev_end) {
                            S v = (S)(cr[j - 1 - prev_beg] + ps + q);
                            if (v > h) h = v;
                        }
                        if (j >= prev_beg && j <= prev_end) {
                            S v = (S)(cr[j - prev_beg] + ps - e1);
                            if (v > h) h = v;
                        }
                    }
                    continue;
                }
                const abamd_row_meta_t pm = meta[p];
                const S *__restrict__ pH = arena + pm.off;
                if (act) {
                    if (local_mode && j == 0) { S v = (S)(ps + q); if (v > h) h = v; }
                    if (j - 1 >= pm.beg && j - 1 <= pm.end) {
                        S v = (S)(pH[j - 1 - pm.beg] + ps + q);
                        if (v > h) h = v;
                    }
                    if (j >= pm.beg && j <= pm.end) {
                        S v = (S)(pH[j - pm.beg] + ps - e1);
                        if (v > h) h = v;
                    }
                }
            }
            if (!act) h = inf_min;

            /* insertion scan ON H: per-wave local scan + cross-wave carry
             * (the merged-candidate + scan form of the one-wave kernel
             * regroups to exactly this under wrapping arithmetic) */
            S L1 = scan_maxplus(h, jb.e1, inf_min, lane);
            if (lane == WAVE - 1) sc_f1[wv] = L1;
            __syncthreads();
            S hsc = L1;
            {
                S C1 = carry_h;
                bool havec = ss != beg;
                for (int ww = 0; ww < wv; ++ww) {
                    if (havec) C1 = smax(sc_f1[ww], (S)(C1 - (S)(WAVE * jb.e1)));
                    else { C1 = sc_f1[ww]; havec = true; }
                }
                if (havec) hsc = smax(L1, (S)(C1 - (S)((lane + 1) * jb.e1)));
            }
            S hfin = local_mode ? smax(hsc, (S)0) : hsc; /* clamp AFTER scan+carry */
            if (act) {
                H[j - beg] = hfin;
                if (cache_fits) cw[j - beg] = hfin;
                if (hfin > lmax) { lmax = hfin; lleft = j; lright = j; }
                else if (hfin == lmax) { lright = j; }
            }
            const bool more = ss + MWT <= end;
            if (more && j == ss + MWT - 1) sc_carry[0] = hsc; /* PRE-clamp */
            if (!more && (jb.banded || local_mode || extend_mode)) {
                int mvw = wave_red_max_i32((int)lmax);
                int llw = ((int)lmax == mvw && lleft >= 0) ? lleft : 0x7fffffff;
                int rrw = ((int)lmax == mvw && lright >= 0) ? lright : -1;
                llw = wave_red_min_i32(llw);
                rrw = wave_red_max_i32(rrw);
                if (lane == 0) {
                    sc_red[wv] = mvw;
                    sc_red[MWAVES + wv] = llw;
                    sc_red[2 * MWAVES + wv] = rrw;
                }
            }
            __syncthreads(); /* also orders cw writes before the next row's reads */
            if (more) carry_h = sc_carry[0];
        }

        if (cache_fits) { prev_ok = 1; prev_row = r; prev_beg = beg; prev_end = end; }
        else prev_ok = 0;
        buf_cur ^= 1;

        if (jb.banded || local_mode || extend_mode) {
            int mv = sc_red[0], ll = sc_red[MWAVES], rr = sc_red[2 * MWAVES];
            #pragma unroll
            for (int ww = 1; ww < MWAVES; ++ww) {
                int m2 = sc_red[ww];
                if (m2 > mv) { mv = m2; ll = sc_red[MWAVES + ww]; rr = sc_red[2 * MWAVES + ww]; }
                else if (m2 == mv) {
                    if (sc_red[MWAVES + ww] < ll) ll = sc_red[MWAVES + ww];
                    if (sc_red[2 * MWAVES + ww] > rr) rr = sc_red[2 * MWAVES + ww];
                }
            }
            if (local_mode) {
                if (mv > run_best) { run_best = mv; run_best_i = r; run_best_j = ll; }
            } else if (extend_mode) {
                if (mv > run_best) {
                    run_best = mv; run_best_i = r; run_best_j = rr;
                    run_best_remain = row_remain;
                } else if (jb.zdrop > 0) {
                    int delta = run_best_remain - row_remain;
                    int dd = delta - (rr - run_best_j); if (dd < 0) dd = -dd;
                    if (run_best - mv > jb.zdrop + jb.e1 * dd) zdropped = 1;
                }
            }
            push_ml = 0x7fffffff; push_mr = -0x7fffffff;
            if (!zdropped && jb.banded) {
                for (int k = oo0; k < oo1; ++k) {
                    int o = jb.out_idx[k];
                    if (o == r + 1) {
                        if (rr + 1 > push_mr) push_mr = rr + 1;
                        if (ll + 1 < push_ml) push_ml = ll + 1;
                    } else {
                        if (rr + 1 > jb.max_right[o]) jb.max_right[o] = rr + 1;
                        if (ll + 1 < jb.max_left[o]) jb.max_left[o] = ll + 1;
                    }
                }
            }
            if (zdropped) break;
        } else {
            push_ml = 0x7fffffff; push_mr = -0x7fffffff;
        }
    }

    __syncthreads();
    if (tid == 0) res->cells = used;
    if (tid != 0) return;

    int32_t best_score = run_best;
    int best_i = run_best_i, best_j = run_best_j;
    if (jb.align_mode == 0) {
        best_score = jb.inf_min; best_i = 0; best_j = 0;
        for (int k = jb.pre_off[n_rows - 1]; k < jb.pre_off[n_rows]; ++k) {
            const int p = jb.pre_idx[k];
            const abamd_row_meta_t pm = meta[p];
            int e = pm.end < qlen ? pm.end : qlen;
            const S *pH = arena + pm.off;
            int32_t sc = (e >= pm.beg) ? (int32_t)pH[e - pm.beg] : jb.inf_min;
            if (sc > best_score) { best_score = sc; best_i = p; best_j = e; }
        }
    }
    res->best_score = best_score;
    res->best_i = best_i; res->best_j = best_j;
    if (!jb.ret_cigar) return;

    { /* simd_abpoa_lg_backtrack (:116-194) */
        int bi = best_i, bj = best_j, start_i = best_i, start_j = best_j;
        int n_c = 0, status = ABAMD_JOB_OK;
        int look_end = jb.put_gap_at_end, put_right = jb.put_gap_on_right;
        int n_aln = 0, n_matched = 0;
        uint64_t *cig = jb.cigar;
        int id = jb.row_node_id[bi];
        if (best_j < qlen) dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, qlen - best_j, -1, qlen - 1, &status);
        while (bi > 0 && bj > 0 && status == ABAMD_JOB_OK) {
            const abamd_row_meta_t bm = meta[bi];
            const int rb = bm.beg, re = bm.end;
            const S *H = arena + bm.off;
            const S Hj = (bj >= rb && bj <= re) ? H[bj - rb] : inf_min;
            const S Hjm1 = (bj - 1 >= rb && bj - 1 <= re) ? H[bj - 1 - rb] : inf_min;
            if (local_mode && Hj == 0) break;
            start_i = bi; start_j = bj;
            const int pq0 = jb.pre_off[bi], pq1 = jb.pre_off[bi + 1];
            const S s = (S)mat_lds[m * jb.row_base[bi] + query[bj - 1]];
            const int is_match = jb.row_base[bi] == query[bj - 1];
            int hit = 0;
            if (put_right == 0 && look_end == 0) {
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                    const S *pH = arena + pm.off;
                    if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                        bi = p; --bj; id = jb.row_node_id[bi]; hit = 1;
                        ++n_aln; n_matched += is_match;
                        break;
                    }
                }
            }
            if (!hit) { /* deletion */
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj < pm.beg || bj > pm.end) continue;
                    const S *pH = arena + pm.off;
                    if ((S)(pH[bj - pm.beg] - e1 + ps) == Hj) {
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 2, 1, id, bj - 1, &status);
                        bi = p; id = jb.row_node_id[bi]; hit = 1;
                        if (look_end) look_end = 0;
                        break;
                    }
                }
            }
            if (!hit) { /* insertion */
                if ((S)(Hjm1 - e1) == Hj) {
                    dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, 1, id, bj - 1, &status);
                    --bj;
                    if (look_end) look_end = 0;
                    hit = 1; ++n_aln;
                }
            }
            if (!hit) {
                for (int k = pq0; k < pq1; ++k) {
                    const int p = jb.pre_idx[k];
                    const S ps = (S)jb.pre_ps[k];
                    const abamd_row_meta_t pm = meta[p];
                    if (bj - 1 < pm.beg || bj - 1 > pm.end) continue;
                    const S *pH = arena + pm.off;
                    if ((S)(pH[bj - 1 - pm.beg] + s + ps) == Hj) {
                        dev_push_cigar(cig, &n_c, jb.cigar_cap, 0, 1, id, bj - 1, &status);
                        bi = p; --bj; id = jb.row_node_id[bi]; hit = 1;
                        ++n_aln; n_matched += is_match;
                        look_end = 0;
                        break;
                    }
                }
            }
            if (!hit) { status = ABAMD_JOB_BT_DEAD_END; break; }
        }
        if (status == ABAMD_JOB_OK && bj > 0)
            dev_push_cigar(cig, &n_c, jb.cigar_cap, 1, bj, -1, bj - 1, &status);
        res->status = status;
        res->n_cigar = n_c;
        res->n_aln_bases = n_aln;
        res->n_matched_bases = n_matched;
        res->node_e = jb.row_node_id[best_i]; res->query_e = best_j - 1;
        res->node_s = jb.row_node_id[start_i]; res->query_s = start_j - 1;
    }
}

/* one-wave ag/lg kernels kept as an A/B fallback (ABPOA_AMD_SW_AGLG=1) */
static int use_sw_aglg(void) {
    static int v = -1;
    if (v < 0) v = getenv("ABPOA_AMD_SW_AGLG") != nullptr;
    return v;
}

extern "C" void abamd_launch_ag_i16(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                                    int n_jobs, void *stream) {
    if (use_sw_aglg()) {
        int blocks = (n_jobs + JOBS_PER_BLOCK - 1) / JOBS_PER_BLOCK;
        hipLaunchKernelGGL((ag_global_kernel<int16_t>), dim3(blocks), dim3(WAVE * JOBS_PER_BLOCK), 0,
                           (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
        return;
    }
    hipLaunchKernelGGL((ag_global_mw_kernel<int16_t>), dim3(n_jobs), dim3(MWT), 0,
                       (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
}
extern "C" void abamd_launch_ag_i32(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                                    int n_jobs, void *stream) {
    if (use_sw_aglg()) {
        int blocks = (n_jobs + JOBS_PER_BLOCK - 1) / JOBS_PER_BLOCK;
        hipLaunchKernelGGL((ag_global_kernel<int32_t>), dim3(blocks), dim3(WAVE * JOBS_PER_BLOCK), 0,
                           (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
        return;
    }
    hipLaunchKernelGGL((ag_global_mw_kernel<int32_t>), dim3(n_jobs), dim3(MWT), 0,
                       (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
}
extern "C" void abamd_launch_lg_i16(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                                    int n_jobs, void *stream) {
    if (use_sw_aglg()) {
        int blocks = (n_jobs + JOBS_PER_BLOCK - 1) / JOBS_PER_BLOCK;
        hipLaunchKernelGGL((lg_global_kernel<int16_t>), dim3(blocks), dim3(WAVE * JOBS_PER_BLOCK), 0,
                           (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
        return;
    }
    hipLaunchKernelGGL((lg_global_mw_kernel<int16_t>), dim3(n_jobs), dim3(MWT), 0,
                       (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
}
extern "C" void abamd_launch_lg_i32(const abamd_gpu_job_t *dev_jobs, abamd_gpu_res_t *dev_res,
                                    int n_jobs, void *stream) {
    if (use_sw_aglg()) {
        int blocks = (n_jobs + JOBS_PER_BLOCK - 1) / JOBS_PER_BLOCK;
        hipLaunchKernelGGL((lg_global_kernel<int32_t>), dim3(blocks), dim3(WAVE * JOBS_PER_BLOCK), 0,
                           (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
        return;
    }
    hipLaunchKernelGGL((lg_global_mw_kernel<int32_t>), dim3(n_jobs), dim3(MWT), 0,
                       (hipStream_t)stream, dev_jobs, dev_res, n_jobs);
}
