/* ============================================================================
 * ORACLE — TEST INFRASTRUCTURE ONLY. NOT THE PRODUCT PATH.
 *
 * Scalar CPU restatement of abPOA's adaptive-banded sequence-to-graph DP
 * (the hot path named by BASELINE.json's north_star), used exclusively as the
 * bit-exactness checker for the HIP/CDNA4 core:
 *   - only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg
 *     may load and call this library (via abpoa_amd_set_test_aligner or the
 *     ABPOA_AMD_TEST_ALIGNER_SO env hook, both of which print loud notices);
 *   - the product library never links it and aborts rather than fall back.
 *
 * The restatement follows, cell by cell, the reference implementation:
 *   wrapper/width pick/inf_min:  abpoa_align_simd.c:1250-1332
 *   convex DP core:              abpoa_align_simd.c:935-1074 (simd_abpoa_cg_dp)
 *   affine DP core:              abpoa_align_simd.c:817-933
 *   linear DP core:              abpoa_align_simd.c:727-815
 *   first row:                   abpoa_align_simd.c:617-688
 *   row max / band update:       abpoa_align_simd.c:1076-1130
 *   backtrack:                   abpoa_align_simd.c:116-458
 *   band formulas:               abpoa_align.h:34-35 (GET_AD_DP_BEGIN/END)
 *
 * Vector-width independence: the reference computes rows in SIMD segments of
 * pn lanes. Its in-band cell values are pn-independent (its own CI asserts
 * golden equality across SSE2/SSE4.1/AVX2/AVX512BW, pn = 8/16/32), so this
 * restatement uses the clean cell-granularity limit of the recurrence:
 *   - out-of-band predecessor cells read as inf_min,
 *   - the band start is clamped to min(pred dp_beg) at cell granularity
 *     (the reference clamps at vector granularity; the cells in between are
 *     provably inf-valued in both formulations),
 *   - the F (insertion) chain is seeded with F[beg] = inf_min - gap_oe,
 *     matching the value the reference's masked log-scan leaves there.
 * int16 arithmetic wraps (the reference uses non-saturating _mm*_add_epi16),
 * reproduced here with int16_t casts after every operation.
 * Parity pinned against the reference binary itself (oracle/_ref/abpoa, built
 * unmodified from /root/reference) on the committed goldens and on synthetic
 * read sets: see tests/.
 * ==========================================================================*/
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <stdint.h>
#include <limits.h>
#include "abpoa_amd.h"

#define OMIN2(a,b) ((a)<(b)?(a):(b))
#define OMAX2(a,b) ((a)>(b)?(a):(b))
#define OMIN3(a,b,c) OMIN2(OMIN2(a,b),(c))
#define OMAX3(a,b,c) OMAX2(OMAX2(a,b),(c))

/* backtrack op-state bits (abpoa_align.h:20-27) */
#define OP_M   0x1
#define OP_E1  0x2
#define OP_E2  0x4
#define OP_E   0x6
#define OP_F1  0x8
#define OP_F2  0x10
#define OP_F   0x18
#define OP_ALL 0x1f

static void *oxmalloc(size_t n) {
    void *p = malloc(n ? n : 1);
    if (!p) { fprintf(stderr, "[oracle] out of memory (%zu)\n", n); exit(1); }
    return p;
}
static void *oxcalloc(size_t n, size_t s) {
    void *p = calloc(n ? n : 1, s);
    if (!p) { fprintf(stderr, "[oracle] out of memory\n"); exit(1); }
    return p;
}

/* packed-cigar push (abpoa_align.h:54-73) */
static abpoa_cigar_t *push_cigar(int *n_c, int *m_c, abpoa_cigar_t *cigar, int op, int len, int32_t node_id, int32_t query_id) {
    abpoa_cigar_t l = (abpoa_cigar_t)len;
    if (*n_c == 0 || (op != ABPOA_CINS && op != ABPOA_CSOFT_CLIP && op != ABPOA_CHARD_CLIP) || op != (int)(cigar[(*n_c)-1] & 0xf)) {
        if (*n_c == *m_c) {
            *m_c = *m_c ? (*m_c) << 1 : 4;
            cigar = (abpoa_cigar_t*)realloc(cigar, (size_t)(*m_c) * sizeof(abpoa_cigar_t));
        }
        abpoa_cigar_t n_id = (abpoa_cigar_t)node_id, q_id = (abpoa_cigar_t)query_id;
        if (op == ABPOA_CMATCH || op == ABPOA_CDIFF) cigar[(*n_c)++] = n_id << 34 | q_id << 4 | (abpoa_cigar_t)op;
        else if (op == ABPOA_CINS || op == ABPOA_CSOFT_CLIP || op == ABPOA_CHARD_CLIP) cigar[(*n_c)++] = q_id << 34 | l << 4 | (abpoa_cigar_t)op;
        else if (op == ABPOA_CDEL) cigar[(*n_c)++] = n_id << 34 | l << 4 | (abpoa_cigar_t)op;
        else { fprintf(stderr, "[oracle] unknown cigar op %d\n", op); exit(1); }
    } else cigar[(*n_c)-1] += l << 4;
    return cigar;
}
static abpoa_cigar_t *reverse_cigar(int n, abpoa_cigar_t *c) {
    int i; abpoa_cigar_t t;
    for (i = 0; i < n >> 1; ++i) { t = c[i]; c[i] = c[n-1-i]; c[n-1-i] = t; }
    return c;
}

static int incre_path_score(abpoa_graph_t *g, int node_id, int k) {
    /* abpoa_graph.c:429-437 */
    int pre = g->node[node_id].in_id[k], i, node_w = 0;
    for (i = 0; i < g->node[pre].out_edge_n; ++i) node_w += g->node[pre].out_edge_weight[i];
    int edge_w = g->node[node_id].in_edge_weight[k];
    if (node_w == 0 || edge_w == 0) return 0;
    double r = (double)edge_w / (double)node_w;
    /* round(log(r)) as in the reference (math.h round/log) */
    extern double log(double); extern double round(double);
    int score = (int)round(log(r));
    return OMAX2(score, -20);
}

/* Banded row store: one growing arena per plane set; per-row beg/end/offset. */
typedef struct {
    int64_t *row_off;     /* offset of row's band in the arena */
    int *dp_beg, *dp_end;
    void *arena;          /* score_t arena: planes interleaved per row */
    int64_t cap, used;    /* in cells (per plane) */
    int n_planes;
} band_store_t;

#define ROWP(T, bs, r, plane) ((T*)(bs)->arena + ((bs)->row_off[r] * (bs)->n_planes + (int64_t)(plane) * ((bs)->dp_end[r] - (bs)->dp_beg[r] + 1)))

/* The DP proper, templated over score width via macro expansion. */
#define DEFINE_ORACLE_CG(SCORE_T, SUFFIX)                                                                     \
static int oracle_cg_##SUFFIX(abpoa_t *ab, abpoa_para_t *abpt, int beg_node_id, int beg_index,                \
        int end_node_id, int end_index, uint8_t *index_map, uint8_t *query, int qlen,                         \
        int32_t inf_min32, abpoa_res_t *res) {                                                                \
    abpoa_graph_t *g = ab->abg;                                                                               \
    const SCORE_T inf_min = (SCORE_T)inf_min32;                                                               \
    int64_t n_rows = end_index - beg_index + 1;                                                               \
    int i, j, k, dp_i, index_i;                                                                               \
    SCORE_T gap_o1 = (SCORE_T)abpt->gap_open1, gap_o2 = (SCORE_T)abpt->gap_open2;                             \
    SCORE_T gap_e1 = (SCORE_T)abpt->gap_ext1, gap_e2 = (SCORE_T)abpt->gap_ext2;                               \
    SCORE_T gap_oe1 = (SCORE_T)(abpt->gap_open1 + abpt->gap_ext1), gap_oe2 = (SCORE_T)(abpt->gap_open2 + abpt->gap_ext2); \
    int w = abpt->wb < 0 ? qlen : abpt->wb + (int)(abpt->wf * qlen);                                          \
    int *mat = abpt->mat, m = abpt->m;                                                                        \
    int local_mode = abpt->align_mode == ABPOA_LOCAL_MODE;                                                    \
    /* predecessor lists in adjacency (weight-sorted) order, filtered by reachability */                      \
    int **pre_index = (int**)oxcalloc(n_rows, sizeof(int*));                                                  \
    int *pre_n = (int*)oxcalloc(n_rows, sizeof(int));                                                         \
    for (index_i = beg_index+1, dp_i = 1; index_i <= end_index; ++index_i, ++dp_i) {                          \
        int node_id = g->index_to_node_id[index_i], pn = g->node[node_id].in_edge_n, c = 0;                   \
        pre_index[dp_i] = (int*)oxmalloc((size_t)(pn>0?pn:1) * sizeof(int));                                  \
        for (j = 0; j < pn; ++j) {                                                                            \
            int pidx = g->node_id_to_index[g->node[node_id].in_id[j]];                                        \
            if (index_map[pidx]) pre_index[dp_i][c++] = pidx - beg_index;                                     \
        }                                                                                                     \
        pre_n[dp_i] = c;                                                                                      \
    }                                                                                                         \
    /* banded plane store: H,E1,E2,F1,F2 */                                                                   \
    band_store_t bs;                                                                                          \
    bs.n_planes = 5;                                                                                          \
    bs.row_off = (int64_t*)oxmalloc((size_t)n_rows * sizeof(int64_t));                                        \
    bs.dp_beg = (int*)oxmalloc((size_t)n_rows * sizeof(int));                                                 \
    bs.dp_end = (int*)oxmalloc((size_t)n_rows * sizeof(int));                                                 \
    bs.cap = 4 * (int64_t)(qlen + 1024); bs.used = 0;                                                         \
    bs.arena = oxmalloc((size_t)bs.cap * 5 * sizeof(SCORE_T));                                                \
    int *dp_beg = bs.dp_beg, *dp_end = bs.dp_end;                                                             \
    /* first row (simd_abpoa_cg_first_dp, abpoa_align_simd.c:617-688) */                                      \
    if (abpt->wb >= 0) {                                                                                      \
        g->node_id_to_max_pos_left[beg_node_id] = g->node_id_to_max_pos_right[beg_node_id] = 0;               \
        for (i = 0; i < g->node[beg_node_id].out_edge_n; ++i) {                                               \
            int out_id = g->node[beg_node_id].out_id[i];                                                      \
            if (index_map[g->node_id_to_index[out_id]])                                                       \
                g->node_id_to_max_pos_left[out_id] = g->node_id_to_max_pos_right[out_id] = 1;                 \
        }                                                                                                     \
        dp_beg[0] = 0;                                                                                        \
        int mr = g->node_id_to_max_remain[beg_node_id] - g->node_id_to_max_remain[end_node_id] - 1;           \
        dp_end[0] = OMIN2(qlen, OMAX2(g->node_id_to_max_pos_right[beg_node_id], qlen - mr) + w);              \
    } else { dp_beg[0] = 0; dp_end[0] = qlen; }                                                               \
    bs.row_off[0] = 0; bs.used = dp_end[0] - dp_beg[0] + 1;                                                   \
    {                                                                                                         \
        SCORE_T *H = ROWP(SCORE_T, &bs, 0, 0), *E1 = ROWP(SCORE_T, &bs, 0, 1), *E2 = ROWP(SCORE_T, &bs, 0, 2);\
        SCORE_T *F1 = ROWP(SCORE_T, &bs, 0, 3), *F2 = ROWP(SCORE_T, &bs, 0, 4);                               \
        if (local_mode) {                                                                                     \
            for (i = 0; i <= dp_end[0]; ++i) H[i] = E1[i] = E2[i] = F1[i] = F2[i] = 0;                        \
        } else {                                                                                              \
            H[0] = 0; E1[0] = (SCORE_T)(0 - gap_oe1); E2[0] = (SCORE_T)(0 - gap_oe2);                         \
            F1[0] = inf_min; F2[0] = inf_min;                                                                 \
            for (i = 1; i <= dp_end[0]; ++i) {                                                                \
                F1[i] = (SCORE_T)(-(abpt->gap_open1 + abpt->gap_ext1 * i));                                   \
                F2[i] = (SCORE_T)(-(abpt->gap_open2 + abpt->gap_ext2 * i));                                   \
                H[i] = OMAX2(F1[i], F2[i]);                                                                   \
                E1[i] = inf_min; E2[i] = inf_min;                                                             \
            }                                                                                                 \
        }                                                                                                     \
    }                                                                                                         \
    int32_t best_score = inf_min32; int best_i = 0, best_j = 0, best_id = 0, zdropped = 0;                    \
    /* per-row loop over topo-sorted reachable nodes */                                                       \
    for (index_i = beg_index+1, dp_i = 1; index_i < end_index; ++index_i, ++dp_i) {                           \
        if (index_map[index_i] == 0) continue;                                                                \
        int node_id = g->index_to_node_id[index_i];                                                           \
        uint8_t base = g->node[node_id].base;                                                                 \
        int beg, end;                                                                                         \
        if (abpt->wb < 0) { beg = 0; end = qlen; }                                                            \
        else {                                                                                                \
            int mr = g->node_id_to_max_remain[node_id] - g->node_id_to_max_remain[end_node_id] - 1;           \
            beg = OMAX2(0, OMIN2(g->node_id_to_max_pos_left[node_id], qlen - mr) - w);                        \
            end = OMIN2(qlen, OMAX2(g->node_id_to_max_pos_right[node_id], qlen - mr) + w);                    \
            int min_pre_beg = INT_MAX;                                                                        \
            for (i = 0; i < pre_n[dp_i]; ++i) min_pre_beg = OMIN2(min_pre_beg, dp_beg[pre_index[dp_i][i]]);   \
            if (beg < min_pre_beg) beg = min_pre_beg; /* cell-granularity clamp (see header) */               \
        }                                                                                                     \
        dp_beg[dp_i] = beg; dp_end[dp_i] = end;                                                               \
        int64_t bw = end - beg + 1;                                                                           \
        if (bs.used + bw > bs.cap) {                                                                          \
            while (bs.used + bw > bs.cap) bs.cap <<= 1;                                                       \
            bs.arena = realloc(bs.arena, (size_t)bs.cap * 5 * sizeof(SCORE_T));                               \
            if (!bs.arena) { fprintf(stderr, "[oracle] OOM arena\n"); exit(1); }                              \
        }                                                                                                     \
        bs.row_off[dp_i] = bs.used; bs.used += bw;                                                            \
        SCORE_T *H = ROWP(SCORE_T, &bs, dp_i, 0), *E1 = ROWP(SCORE_T, &bs, dp_i, 1), *E2 = ROWP(SCORE_T, &bs, dp_i, 2); \
        SCORE_T *F1 = ROWP(SCORE_T, &bs, dp_i, 3), *F2 = ROWP(SCORE_T, &bs, dp_i, 4);                         \
        /* M/E gather over predecessors, in pre_index order */                                                \
        for (j = beg; j <= end; ++j) { H[j-beg] = inf_min; E1[j-beg] = inf_min; E2[j-beg] = inf_min; }        \
        for (k = 0; k < pre_n[dp_i]; ++k) {                                                                   \
            int pre_i = pre_index[dp_i][k];                                                                   \
            SCORE_T ps = 0;                                                                                   \
            if (abpt->inc_path_score) ps = (SCORE_T)incre_path_score(g, node_id, k);                          \
            int pbeg = dp_beg[pre_i], pend = dp_end[pre_i];                                                   \
            SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0), *pE1 = ROWP(SCORE_T, &bs, pre_i, 1), *pE2 = ROWP(SCORE_T, &bs, pre_i, 2); \
            int lo, hi;                                                                                       \
            /* M from (pre, j-1) */                                                                           \
            if (local_mode) {                                                                                 \
                /* local: unbanded rows; the virtual column left of 0 reads 0  */                             \
                for (j = beg; j <= end; ++j) {                                                                \
                    SCORE_T v = (SCORE_T)((j == 0 ? (SCORE_T)0 : pH[j-1-pbeg]) + ps);                         \
                    if (v > H[j-beg]) H[j-beg] = v;                                                           \
                }                                                                                             \
            } else {                                                                                          \
                lo = OMAX2(beg, pbeg + 1); hi = OMIN2(end, pend + 1);                                         \
                for (j = lo; j <= hi; ++j) {                                                                  \
                    SCORE_T v = (SCORE_T)(pH[j-1-pbeg] + ps);                                                 \
                    if (v > H[j-beg]) H[j-beg] = v;                                                           \
                }                                                                                             \
            }                                                                                                 \
            /* E from (pre, j) */                                                                             \
            lo = local_mode ? beg : OMAX2(beg, pbeg); hi = OMIN2(end, pend);                                  \
            for (j = lo; j <= hi; ++j) {                                                                      \
                SCORE_T v1 = (SCORE_T)(pE1[j-pbeg] + ps), v2 = (SCORE_T)(pE2[j-pbeg] + ps);                   \
                if (v1 > E1[j-beg]) E1[j-beg] = v1;                                                           \
                if (v2 > E2[j-beg]) E2[j-beg] = v2;                                                           \
            }                                                                                                 \
        }                                                                                                     \
        /* add query profile; then E/F folds with the in-row F chain */                                       \
        SCORE_T f1 = (SCORE_T)(inf_min), f2 = (SCORE_T)(inf_min);                                             \
        SCORE_T hprev = inf_min; /* Hpre[j-1] (post-E, pre-F) */                                              \
        for (j = beg; j <= end; ++j) {                                                                        \
            SCORE_T q = (SCORE_T)(j == 0 ? 0 : mat[m * base + query[j-1]]);                                   \
            SCORE_T h = (SCORE_T)(H[j-beg] + q);                                                              \
            h = OMAX3(h, E1[j-beg], E2[j-beg]);                                                               \
            /* F recurrence: F[j] = max(Hpre[j-1], F[j-1]+o) - oe  (abpoa_align_simd.c:1052-1058) */          \
            if (j == beg) { f1 = (SCORE_T)(inf_min - gap_oe1); f2 = (SCORE_T)(inf_min - gap_oe2); }           \
            else {                                                                                            \
                SCORE_T c1 = OMAX2(hprev, (SCORE_T)(f1 + gap_o1));                                            \
                SCORE_T c2 = OMAX2(hprev, (SCORE_T)(f2 + gap_o2));                                            \
                f1 = (SCORE_T)(c1 - gap_oe1); f2 = (SCORE_T)(c2 - gap_oe2);                                   \
            }                                                                                                 \
            F1[j-beg] = f1; F2[j-beg] = f2;                                                                   \
            hprev = h;                                                                                        \
            SCORE_T hf = OMAX3(h, f1, f2);                                                                    \
            if (local_mode) hf = OMAX2(hf, (SCORE_T)0);                                                       \
            H[j-beg] = hf;                                                                                    \
            SCORE_T e1n = OMAX2((SCORE_T)(E1[j-beg] - gap_e1), (SCORE_T)(hf - gap_oe1));                      \
            SCORE_T e2n = OMAX2((SCORE_T)(E2[j-beg] - gap_e2), (SCORE_T)(hf - gap_oe2));                      \
            if (local_mode) { e1n = OMAX2(e1n, (SCORE_T)0); e2n = OMAX2(e2n, (SCORE_T)0); }                   \
            E1[j-beg] = e1n; E2[j-beg] = e2n;                                                                 \
        }                                                                                                     \
        /* row max + adaptive band push (abpoa_align_simd.c:1107-1130) */                                     \
        if (local_mode || abpt->align_mode == ABPOA_EXTEND_MODE || abpt->wb >= 0) {                           \
            int32_t mx = inf_min32; int left = -1, right = -1;                                                \
            for (j = beg; j <= end; ++j) {                                                                    \
                if ((int32_t)H[j-beg] > mx) { mx = (int32_t)H[j-beg]; left = right = j; }                     \
                else if ((int32_t)H[j-beg] == mx) right = j;                                                  \
            }                                                                                                 \
            if (local_mode) {                                                                                 \
                if (mx > best_score) { best_score = mx; best_i = dp_i; best_j = left; }                       \
            } else if (abpt->align_mode == ABPOA_EXTEND_MODE) {                                               \
                if (mx > best_score) { best_score = mx; best_i = dp_i; best_j = right; best_id = node_id; }   \
                else if (abpt->zdrop > 0) {                                                                   \
                    int delta = g->node_id_to_max_remain[best_id] - g->node_id_to_max_remain[node_id];        \
                    int dd = delta - (right - best_j); if (dd < 0) dd = -dd;                                  \
                    if (best_score - mx > abpt->zdrop + abpt->gap_ext1 * dd) { zdropped = 1; }                \
                }                                                                                             \
            }                                                                                                 \
            if (!zdropped && abpt->wb >= 0) {                                                                 \
                if (abpt->align_mode == ABPOA_GLOBAL_MODE) { /* recompute is the same scan */ }               \
                for (i = 0; i < g->node[node_id].out_edge_n; ++i) {                                           \
                    int out_id = g->node[node_id].out_id[i];                                                  \
                    if (right + 1 > g->node_id_to_max_pos_right[out_id]) g->node_id_to_max_pos_right[out_id] = right + 1; \
                    if (left + 1 < g->node_id_to_max_pos_left[out_id]) g->node_id_to_max_pos_left[out_id] = left + 1;     \
                }                                                                                             \
            }                                                                                                 \
            if (zdropped) break;                                                                              \
        }                                                                                                     \
    }                                                                                                         \
    /* final best for global mode (abpoa_align_simd.c:1092-1105) */                                           \
    if (abpt->align_mode == ABPOA_GLOBAL_MODE) {                                                              \
        for (i = 0; i < g->node[end_node_id].in_edge_n; ++i) {                                                \
            int in_id = g->node[end_node_id].in_id[i];                                                        \
            int in_index = g->node_id_to_index[in_id];                                                        \
            if (index_map[in_index] == 0) continue;                                                           \
            int in_dp_i = in_index - beg_index;                                                               \
            int e = OMIN2(qlen, dp_end[in_dp_i]);                                                             \
            SCORE_T *pH = ROWP(SCORE_T, &bs, in_dp_i, 0);                                                     \
            int32_t sc = (e >= dp_beg[in_dp_i]) ? (int32_t)pH[e - dp_beg[in_dp_i]] : inf_min32;               \
            if (sc > best_score) { best_score = sc; best_i = in_dp_i; best_j = e; }                           \
        }                                                                                                     \
    }                                                                                                         \
    res->best_score = best_score;                                                                             \
    /* backtrack (simd_abpoa_cg_backtrack, abpoa_align_simd.c:309-458) */                                     \
    if (abpt->ret_cigar) {                                                                                    \
        int bi = best_i, bj = best_j, _start_i = best_i, _start_j = best_j;                                   \
        int n_c = 0, m_c = 0, cur_op = OP_ALL, hit, id, s, is_match, path_score = 0;                          \
        abpoa_cigar_t *cigar = 0;                                                                             \
        id = g->index_to_node_id[bi + beg_index];                                                             \
        if (best_j < qlen) cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, qlen - best_j, -1, qlen - 1);    \
        int look_end = abpt->put_gap_at_end, put_right = abpt->put_gap_on_right;                              \
        /* banded accessors: out-of-band reads return inf_min (the reference  */                              \
        /* stores inf_min in those vector-padded cells)                       */                              \
        while (bi > 0 && bj > 0) {                                                                            \
            SCORE_T *H = ROWP(SCORE_T, &bs, bi, 0), *E1r = ROWP(SCORE_T, &bs, bi, 1), *E2r = ROWP(SCORE_T, &bs, bi, 2); \
            SCORE_T *F1r = ROWP(SCORE_T, &bs, bi, 3), *F2r = ROWP(SCORE_T, &bs, bi, 4);                       \
            int rb = dp_beg[bi], re = dp_end[bi];                                                             \
            /* banded reads; out-of-band cells are inf_min (the reference      */                             \
            /* stores inf_min in its vector-padded out-of-band cells)          */                             \
            int32_t Hj   = (bj   >= rb && bj   <= re) ? (int32_t)H[bj-rb]    : inf_min32;                     \
            int32_t Hjm1 = (bj-1 >= rb && bj-1 <= re) ? (int32_t)H[bj-1-rb]  : inf_min32;                     \
            int32_t E1j  = (bj   >= rb && bj   <= re) ? (int32_t)E1r[bj-rb]  : inf_min32;                     \
            int32_t E2j  = (bj   >= rb && bj   <= re) ? (int32_t)E2r[bj-rb]  : inf_min32;                     \
            int32_t F1j  = (bj   >= rb && bj   <= re) ? (int32_t)F1r[bj-rb]  : inf_min32;                     \
            int32_t F2j  = (bj   >= rb && bj   <= re) ? (int32_t)F2r[bj-rb]  : inf_min32;                     \
            int32_t F1jm1= (bj-1 >= rb && bj-1 <= re) ? (int32_t)F1r[bj-1-rb]: inf_min32;                     \
            int32_t F2jm1= (bj-1 >= rb && bj-1 <= re) ? (int32_t)F2r[bj-1-rb]: inf_min32;                     \
            if (local_mode && Hj == 0) break;                                                                 \
            _start_i = bi; _start_j = bj;                                                                     \
            int *pre_index_i = pre_index[bi];                                                                 \
            s = mat[m * g->node[id].base + query[bj-1]]; hit = 0;                                             \
            is_match = g->node[id].base == query[bj-1];                                                       \
            if (put_right == 0 && look_end == 0 && (cur_op & OP_M)) {                                         \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj-1 < dp_beg[pre_i] || bj-1 > dp_end[pre_i]) continue;                               \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                               \
                    if ((SCORE_T)(pH[bj-1-dp_beg[pre_i]] + (SCORE_T)s + (SCORE_T)path_score) == (SCORE_T)Hj) {\
                        cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CMATCH, 1, id, bj-1);                     \
                        bi = pre_i; --bj; id = g->index_to_node_id[bi + beg_index]; hit = 1;                  \
                        cur_op = OP_ALL;                                                                      \
                        ++res->n_aln_bases; res->n_matched_bases += is_match ? 1 : 0;                         \
                        break;                                                                                \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0 && (cur_op & OP_E)) { /* deletion */                                                 \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj < dp_beg[pre_i] || bj > dp_end[pre_i]) continue;                                   \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                               \
                    SCORE_T *pE1 = ROWP(SCORE_T, &bs, pre_i, 1), *pE2 = ROWP(SCORE_T, &bs, pre_i, 2);         \
                    int off = bj - dp_beg[pre_i];                                                             \
                    if (cur_op & OP_E1) {                                                                     \
                        if (cur_op & OP_M) {                                                                  \
                            if ((SCORE_T)Hj == (SCORE_T)(pE1[off] + (SCORE_T)path_score)) {                   \
                                cur_op = ((SCORE_T)(pH[off] - gap_oe1) == pE1[off]) ? (OP_M|OP_F) : OP_E1;    \
                                hit = 1; cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CDEL, 1, id, bj-1);      \
                                bi = pre_i; id = g->index_to_node_id[bi + beg_index];                         \
                                if (look_end) look_end = 0;                                                   \
                                break;                                                                        \
                            }                                                                                 \
                        } else {                                                                              \
                            if ((SCORE_T)E1j == (SCORE_T)(pE1[off] - gap_ext_1_st + (SCORE_T)path_score)) {   \
                                cur_op = ((SCORE_T)(pH[off] - gap_oe1) == pE1[off]) ? (OP_M|OP_F) : OP_E1;    \
                                hit = 1; cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CDEL, 1, id, bj-1);      \
                                bi = pre_i; id = g->index_to_node_id[bi + beg_index];                         \
                                if (look_end) look_end = 0;                                                   \
                                break;                                                                        \
                            }                                                                                 \
                        }                                                                                     \
                    }                                                                                         \
                    if (cur_op & OP_E2) {                                                                     \
                        if (cur_op & OP_M) {                                                                  \
                            if ((SCORE_T)Hj == (SCORE_T)(pE2[off] + (SCORE_T)path_score)) {                   \
                                cur_op = ((SCORE_T)(pH[off] - gap_oe2) == pE2[off]) ? (OP_M|OP_F) : OP_E2;    \
                                hit = 1; cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CDEL, 1, id, bj-1);      \
                                bi = pre_i; id = g->index_to_node_id[bi + beg_index];                         \
                                if (look_end) look_end = 0;                                                   \
                                break;                                                                        \
                            }                                                                                 \
                        } else {                                                                              \
                            if ((SCORE_T)E2j == (SCORE_T)(pE2[off] - gap_ext_2_st + (SCORE_T)path_score)) {   \
                                cur_op = ((SCORE_T)(pH[off] - gap_oe2) == pE2[off]) ? (OP_M|OP_F) : OP_E2;    \
                                hit = 1; cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CDEL, 1, id, bj-1);      \
                                bi = pre_i; id = g->index_to_node_id[bi + beg_index];                         \
                                if (look_end) look_end = 0;                                                   \
                                break;                                                                        \
                            }                                                                                 \
                        }                                                                                     \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0 && (cur_op & OP_F)) { /* insertion */                                                \
                if (cur_op & OP_F1) {                                                                         \
                    if (cur_op & OP_M) {                                                                      \
                        if (Hj == F1j) {                                                                      \
                            if ((SCORE_T)(Hjm1 - gap_oe1) == (SCORE_T)F1j) cur_op = OP_M|OP_E, hit = 1;       \
                            else if ((SCORE_T)(F1jm1 - gap_ext_1_st) == (SCORE_T)F1j) cur_op = OP_F1, hit = 1;\
                        }                                                                                     \
                    } else {                                                                                  \
                        if ((SCORE_T)(Hjm1 - gap_oe1) == (SCORE_T)F1j) cur_op = OP_M|OP_E, hit = 1;           \
                        else if ((SCORE_T)(F1jm1 - gap_ext_1_st) == (SCORE_T)F1j) cur_op = OP_F1, hit = 1;    \
                    }                                                                                         \
                }                                                                                             \
                if (hit == 0 && (cur_op & OP_F2)) {                                                           \
                    if (cur_op & OP_M) {                                                                      \
                        if (Hj == F2j) {                                                                      \
                            if ((SCORE_T)(Hjm1 - gap_oe2) == (SCORE_T)F2j) cur_op = OP_M|OP_E, hit = 1;       \
                            else if ((SCORE_T)(F2jm1 - gap_ext_2_st) == (SCORE_T)F2j) cur_op = OP_F2, hit = 1;\
                        }                                                                                     \
                    } else {                                                                                  \
                        if ((SCORE_T)(Hjm1 - gap_oe2) == (SCORE_T)F2j) cur_op = OP_M|OP_E, hit = 1;           \
                        else if ((SCORE_T)(F2jm1 - gap_ext_2_st) == (SCORE_T)F2j) cur_op = OP_F2, hit = 1;    \
                    }                                                                                         \
                }                                                                                             \
                if (hit == 1) {                                                                               \
                    cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, 1, id, bj-1); --bj;                     \
                    if (look_end) look_end = 0;                                                               \
                    ++res->n_aln_bases;                                                                       \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0 && (cur_op & OP_M)) {                                                                \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj-1 < dp_beg[pre_i] || bj-1 > dp_end[pre_i]) continue;                               \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                               \
                    if ((SCORE_T)(pH[bj-1-dp_beg[pre_i]] + (SCORE_T)s + (SCORE_T)path_score) == (SCORE_T)Hj) {\
                        cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CMATCH, 1, id, bj-1);                     \
                        bi = pre_i; --bj; id = g->index_to_node_id[bi + beg_index]; hit = 1;                  \
                        cur_op = OP_ALL;                                                                      \
                        ++res->n_aln_bases; res->n_matched_bases += is_match ? 1 : 0;                         \
                        look_end = 0;                                                                         \
                        break;                                                                                \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0) { fprintf(stderr, "[oracle] backtrack dead end at row %d col %d\n", bi, bj); exit(1); } \
        }                                                                                                     \
        if (bj > 0) cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, bj, -1, bj-1);                          \
        res->graph_cigar = abpt->rev_cigar ? cigar : reverse_cigar(n_c, cigar);                               \
        res->n_cigar = n_c; res->m_cigar = m_c;                                                               \
        res->node_e = g->index_to_node_id[best_i + beg_index]; res->query_e = best_j - 1;                     \
        res->node_s = g->index_to_node_id[_start_i + beg_index]; res->query_s = _start_j - 1;                 \
    }                                                                                                         \
    for (i = 0; i < n_rows; ++i) free(pre_index[i]);                                                          \
    free(pre_index); free(pre_n);                                                                             \
    free(bs.row_off); free(bs.dp_beg); free(bs.dp_end); free(bs.arena);                                       \
    return best_score;                                                                                        \
}

/* gap_ext as SCORE_T inside the macro */
#define gap_ext_1_st gap_e1
#define gap_ext_2_st gap_e2

DEFINE_ORACLE_CG(int16_t, i16)
DEFINE_ORACLE_CG(int32_t, i32)

/* ---------------- affine gap (simd_abpoa_ag_dp, abpoa_align_simd.c:817-933;
 * backtrack :196-307). Note: F candidates use H BEFORE the E fold, and the
 * stored E is inf_min (0 local) when the insertion won the cell (:916-931). */
#define DEFINE_ORACLE_AG(SCORE_T, SUFFIX)                                                                     \
static int oracle_ag_##SUFFIX(abpoa_t *ab, abpoa_para_t *abpt, int beg_node_id, int beg_index,                \
        int end_node_id, int end_index, uint8_t *index_map, uint8_t *query, int qlen,                         \
        int32_t inf_min32, abpoa_res_t *res) {                                                                \
    abpoa_graph_t *g = ab->abg;                                                                               \
    const SCORE_T inf_min = (SCORE_T)inf_min32;                                                               \
    int64_t n_rows = end_index - beg_index + 1;                                                               \
    int i, j, k, dp_i, index_i;                                                                               \
    SCORE_T gap_e1 = (SCORE_T)abpt->gap_ext1;                                                                 \
    SCORE_T gap_oe1 = (SCORE_T)(abpt->gap_open1 + abpt->gap_ext1);                                            \
    int w = abpt->wb < 0 ? qlen : abpt->wb + (int)(abpt->wf * qlen);                                          \
    int *mat = abpt->mat, m = abpt->m;                                                                        \
    int local_mode = abpt->align_mode == ABPOA_LOCAL_MODE;                                                    \
    int **pre_index = (int**)oxcalloc(n_rows, sizeof(int*));                                                  \
    int *pre_n = (int*)oxcalloc(n_rows, sizeof(int));                                                         \
    for (index_i = beg_index+1, dp_i = 1; index_i <= end_index; ++index_i, ++dp_i) {                          \
        int node_id = g->index_to_node_id[index_i], pn = g->node[node_id].in_edge_n, c = 0;                   \
        pre_index[dp_i] = (int*)oxmalloc((size_t)(pn>0?pn:1) * sizeof(int));                                  \
        for (j = 0; j < pn; ++j) {                                                                            \
            int pidx = g->node_id_to_index[g->node[node_id].in_id[j]];                                        \
            if (index_map[pidx]) pre_index[dp_i][c++] = pidx - beg_index;                                     \
        }                                                                                                     \
        pre_n[dp_i] = c;                                                                                      \
    }                                                                                                         \
    band_store_t bs;                                                                                          \
    bs.n_planes = 3;                                                                                          \
    bs.row_off = (int64_t*)oxmalloc((size_t)n_rows * sizeof(int64_t));                                        \
    bs.dp_beg = (int*)oxmalloc((size_t)n_rows * sizeof(int));                                                 \
    bs.dp_end = (int*)oxmalloc((size_t)n_rows * sizeof(int));                                                 \
    bs.cap = 4 * (int64_t)(qlen + 1024); bs.used = 0;                                                         \
    bs.arena = oxmalloc((size_t)bs.cap * 3 * sizeof(SCORE_T));                                                \
    int *dp_beg = bs.dp_beg, *dp_end = bs.dp_end;                                                             \
    if (abpt->wb >= 0) {                                                                                      \
        g->node_id_to_max_pos_left[beg_node_id] = g->node_id_to_max_pos_right[beg_node_id] = 0;               \
        for (i = 0; i < g->node[beg_node_id].out_edge_n; ++i) {                                               \
            int out_id = g->node[beg_node_id].out_id[i];                                                      \
            if (index_map[g->node_id_to_index[out_id]])                                                       \
                g->node_id_to_max_pos_left[out_id] = g->node_id_to_max_pos_right[out_id] = 1;                 \
        }                                                                                                     \
        dp_beg[0] = 0;                                                                                        \
        int mr = g->node_id_to_max_remain[beg_node_id] - g->node_id_to_max_remain[end_node_id] - 1;           \
        dp_end[0] = OMIN2(qlen, OMAX2(g->node_id_to_max_pos_right[beg_node_id], qlen - mr) + w);              \
    } else { dp_beg[0] = 0; dp_end[0] = qlen; }                                                               \
    bs.row_off[0] = 0; bs.used = dp_end[0] - dp_beg[0] + 1;                                                   \
    {                                                                                                         \
        SCORE_T *H = ROWP(SCORE_T, &bs, 0, 0), *E1 = ROWP(SCORE_T, &bs, 0, 1), *F1 = ROWP(SCORE_T, &bs, 0, 2);\
        if (local_mode) { for (i = 0; i <= dp_end[0]; ++i) H[i] = E1[i] = F1[i] = 0; }                        \
        else {                                                                                                \
            H[0] = 0; E1[0] = (SCORE_T)(0 - gap_oe1); F1[0] = inf_min;                                        \
            for (i = 1; i <= dp_end[0]; ++i) {                                                                \
                F1[i] = (SCORE_T)(-(abpt->gap_open1 + abpt->gap_ext1 * i));                                   \
                H[i] = F1[i]; E1[i] = inf_min;                                                                \
            }                                                                                                 \
        }                                                                                                     \
    }                                                                                                         \
    int32_t best_score = inf_min32; int best_i = 0, best_j = 0, best_id = 0, zdropped = 0;                    \
    for (index_i = beg_index+1, dp_i = 1; index_i < end_index; ++index_i, ++dp_i) {                           \
        if (index_map[index_i] == 0) continue;                                                                \
        int node_id = g->index_to_node_id[index_i];                                                           \
        uint8_t base = g->node[node_id].base;                                                                 \
        int beg, end;                                                                                         \
        if (abpt->wb < 0) { beg = 0; end = qlen; }                                                            \
        else {                                                                                                \
            int mr = g->node_id_to_max_remain[node_id] - g->node_id_to_max_remain[end_node_id] - 1;           \
            beg = OMAX2(0, OMIN2(g->node_id_to_max_pos_left[node_id], qlen - mr) - w);                        \
            end = OMIN2(qlen, OMAX2(g->node_id_to_max_pos_right[node_id], qlen - mr) + w);                    \
            int min_pre_beg = INT_MAX;                                                                        \
            for (i = 0; i < pre_n[dp_i]; ++i) min_pre_beg = OMIN2(min_pre_beg, dp_beg[pre_index[dp_i][i]]);   \
            if (beg < min_pre_beg) beg = min_pre_beg;                                                         \
        }                                                                                                     \
        dp_beg[dp_i] = beg; dp_end[dp_i] = end;                                                               \
        int64_t bwid = end - beg + 1;                                                                         \
        if (bs.used + bwid > bs.cap) {                                                                        \
            while (bs.used + bwid > bs.cap) bs.cap <<= 1;                                                     \
            bs.arena = realloc(bs.arena, (size_t)bs.cap * 3 * sizeof(SCORE_T));                               \
            if (!bs.arena) { fprintf(stderr, "[oracle] OOM arena\n"); exit(1); }                              \
        }                                                                                                     \
        bs.row_off[dp_i] = bs.used; bs.used += bwid;                                                          \
        SCORE_T *H = ROWP(SCORE_T, &bs, dp_i, 0), *E1 = ROWP(SCORE_T, &bs, dp_i, 1), *F1 = ROWP(SCORE_T, &bs, dp_i, 2); \
        for (j = beg; j <= end; ++j) { H[j-beg] = inf_min; E1[j-beg] = inf_min; }                             \
        for (k = 0; k < pre_n[dp_i]; ++k) {                                                                   \
            int pre_i = pre_index[dp_i][k];                                                                   \
            SCORE_T ps = 0;                                                                                   \
            if (abpt->inc_path_score) ps = (SCORE_T)incre_path_score(g, node_id, k);                          \
            int pbeg = dp_beg[pre_i], pend = dp_end[pre_i];                                                   \
            SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0), *pE1 = ROWP(SCORE_T, &bs, pre_i, 1);                  \
            int lo, hi;                                                                                       \
            if (local_mode) {                                                                                 \
                for (j = beg; j <= end; ++j) {                                                                \
                    SCORE_T v = (SCORE_T)((j == 0 ? (SCORE_T)0 : pH[j-1-pbeg]) + ps);                         \
                    if (v > H[j-beg]) H[j-beg] = v;                                                           \
                }                                                                                             \
            } else {                                                                                          \
                lo = OMAX2(beg, pbeg + 1); hi = OMIN2(end, pend + 1);                                         \
                for (j = lo; j <= hi; ++j) {                                                                  \
                    SCORE_T v = (SCORE_T)(pH[j-1-pbeg] + ps);                                                 \
                    if (v > H[j-beg]) H[j-beg] = v;                                                           \
                }                                                                                             \
            }                                                                                                 \
            lo = local_mode ? beg : OMAX2(beg, pbeg); hi = OMIN2(end, pend);                                  \
            for (j = lo; j <= hi; ++j) {                                                                      \
                SCORE_T v1 = (SCORE_T)(pE1[j-pbeg] + ps);                                                     \
                if (v1 > E1[j-beg]) E1[j-beg] = v1;                                                           \
            }                                                                                                 \
        }                                                                                                     \
        SCORE_T f1 = inf_min, hm_prev = inf_min;                                                              \
        for (j = beg; j <= end; ++j) {                                                                        \
            SCORE_T q = (SCORE_T)(j == 0 ? 0 : mat[m * base + query[j-1]]);                                   \
            SCORE_T hm = (SCORE_T)(H[j-beg] + q); /* M+q, E not folded yet */                                 \
            if (j == beg) f1 = (SCORE_T)(inf_min - gap_oe1);                                                  \
            else f1 = OMAX2((SCORE_T)(hm_prev - gap_oe1), (SCORE_T)(f1 - gap_e1));                            \
            F1[j-beg] = f1;                                                                                   \
            hm_prev = hm;                                                                                     \
            SCORE_T tmp = OMAX2(hm, E1[j-beg]);                                                               \
            SCORE_T hf = OMAX2(tmp, f1);                                                                      \
            if (local_mode) hf = OMAX2(hf, (SCORE_T)0);                                                       \
            H[j-beg] = hf;                                                                                    \
            if (hf == tmp) E1[j-beg] = OMAX2((SCORE_T)(E1[j-beg] - gap_e1), (SCORE_T)(hf - gap_oe1));         \
            else E1[j-beg] = local_mode ? (SCORE_T)0 : inf_min;                                               \
        }                                                                                                     \
        if (local_mode || abpt->align_mode == ABPOA_EXTEND_MODE || abpt->wb >= 0) {                           \
            int32_t mx = inf_min32; int left = -1, right = -1;                                                \
            for (j = beg; j <= end; ++j) {                                                                    \
                if ((int32_t)H[j-beg] > mx) { mx = (int32_t)H[j-beg]; left = right = j; }                     \
                else if ((int32_t)H[j-beg] == mx) right = j;                                                  \
            }                                                                                                 \
            if (local_mode) {                                                                                 \
                if (mx > best_score) { best_score = mx; best_i = dp_i; best_j = left; }                       \
            } else if (abpt->align_mode == ABPOA_EXTEND_MODE) {                                               \
                if (mx > best_score) { best_score = mx; best_i = dp_i; best_j = right; best_id = node_id; }   \
                else if (abpt->zdrop > 0) {                                                                   \
                    int delta = g->node_id_to_max_remain[best_id] - g->node_id_to_max_remain[node_id];        \
                    int dd = delta - (right - best_j); if (dd < 0) dd = -dd;                                  \
                    if (best_score - mx > abpt->zdrop + abpt->gap_ext1 * dd) { zdropped = 1; }                \
                }                                                                                             \
            }                                                                                                 \
            if (!zdropped && abpt->wb >= 0) {                                                                 \
                for (i = 0; i < g->node[node_id].out_edge_n; ++i) {                                           \
                    int out_id = g->node[node_id].out_id[i];                                                  \
                    if (right + 1 > g->node_id_to_max_pos_right[out_id]) g->node_id_to_max_pos_right[out_id] = right + 1; \
                    if (left + 1 < g->node_id_to_max_pos_left[out_id]) g->node_id_to_max_pos_left[out_id] = left + 1;     \
                }                                                                                             \
            }                                                                                                 \
            if (zdropped) break;                                                                              \
        }                                                                                                     \
    }                                                                                                         \
    if (abpt->align_mode == ABPOA_GLOBAL_MODE) {                                                              \
        for (i = 0; i < g->node[end_node_id].in_edge_n; ++i) {                                                \
            int in_id = g->node[end_node_id].in_id[i];                                                        \
            int in_index = g->node_id_to_index[in_id];                                                        \
            if (index_map[in_index] == 0) continue;                                                           \
            int in_dp_i = in_index - beg_index;                                                               \
            int e = OMIN2(qlen, dp_end[in_dp_i]);                                                             \
            SCORE_T *pH = ROWP(SCORE_T, &bs, in_dp_i, 0);                                                     \
            int32_t sc = (e >= dp_beg[in_dp_i]) ? (int32_t)pH[e - dp_beg[in_dp_i]] : inf_min32;               \
            if (sc > best_score) { best_score = sc; best_i = in_dp_i; best_j = e; }                           \
        }                                                                                                     \
    }                                                                                                         \
    res->best_score = best_score;                                                                             \
    if (abpt->ret_cigar) { /* simd_abpoa_ag_backtrack (:196-307) */                                           \
        int bi = best_i, bj = best_j, _start_i = best_i, _start_j = best_j;                                   \
        int n_c = 0, m_c = 0, cur_op = OP_ALL, hit, id, s, is_match, path_score = 0;                          \
        abpoa_cigar_t *cigar = 0;                                                                             \
        id = g->index_to_node_id[bi + beg_index];                                                             \
        if (best_j < qlen) cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, qlen - best_j, -1, qlen - 1);    \
        int look_end = abpt->put_gap_at_end, put_right = abpt->put_gap_on_right;                              \
        while (bi > 0 && bj > 0) {                                                                            \
            SCORE_T *H = ROWP(SCORE_T, &bs, bi, 0), *E1r = ROWP(SCORE_T, &bs, bi, 1), *F1r = ROWP(SCORE_T, &bs, bi, 2); \
            int rb = dp_beg[bi], re = dp_end[bi];                                                             \
            int32_t Hj   = (bj   >= rb && bj   <= re) ? (int32_t)H[bj-rb]    : inf_min32;                     \
            int32_t Hjm1 = (bj-1 >= rb && bj-1 <= re) ? (int32_t)H[bj-1-rb]  : inf_min32;                     \
            int32_t E1j  = (bj   >= rb && bj   <= re) ? (int32_t)E1r[bj-rb]  : inf_min32;                     \
            int32_t F1j  = (bj   >= rb && bj   <= re) ? (int32_t)F1r[bj-rb]  : inf_min32;                     \
            int32_t F1jm1= (bj-1 >= rb && bj-1 <= re) ? (int32_t)F1r[bj-1-rb]: inf_min32;                     \
            if (local_mode && Hj == 0) break;                                                                 \
            _start_i = bi; _start_j = bj;                                                                     \
            int *pre_index_i = pre_index[bi];                                                                 \
            s = mat[m * g->node[id].base + query[bj-1]]; hit = 0;                                             \
            is_match = g->node[id].base == query[bj-1];                                                       \
            if (put_right == 0 && look_end == 0 && (cur_op & OP_M)) {                                         \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj-1 < dp_beg[pre_i] || bj-1 > dp_end[pre_i]) continue;                               \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                               \
                    if ((SCORE_T)(pH[bj-1-dp_beg[pre_i]] + (SCORE_T)s + (SCORE_T)path_score) == (SCORE_T)Hj) {\
                        cur_op = OP_ALL; hit = 1;                                                             \
                        cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CMATCH, 1, id, bj-1);                     \
                        bi = pre_i; --bj; id = g->index_to_node_id[bi + beg_index];                           \
                        ++res->n_aln_bases; res->n_matched_bases += is_match ? 1 : 0;                         \
                        break;                                                                                \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0 && (cur_op & OP_E1)) {                                                               \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj < dp_beg[pre_i] || bj > dp_end[pre_i]) continue;                                   \
                    int off = bj - dp_beg[pre_i];                                                             \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0), *pE1 = ROWP(SCORE_T, &bs, pre_i, 1);          \
                    if (cur_op & OP_M) {                                                                      \
                        if ((SCORE_T)Hj == (SCORE_T)(pE1[off] + (SCORE_T)path_score)) {                       \
                            cur_op = ((SCORE_T)(pH[off] - gap_oe1) == pE1[off]) ? (OP_M|OP_F) : OP_E1;        \
                            hit = 1; cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CDEL, 1, id, bj-1);          \
                            bi = pre_i; id = g->index_to_node_id[bi + beg_index];                             \
                            if (look_end) look_end = 0;                                                       \
                            break;                                                                            \
                        }                                                                                     \
                    } else {                                                                                  \
                        if ((SCORE_T)E1j == (SCORE_T)(pE1[off] - gap_e1 + (SCORE_T)path_score)) {             \
                            cur_op = ((SCORE_T)(pH[off] - gap_oe1) == pE1[off]) ? (OP_M|OP_F) : OP_E1;        \
                            hit = 1; cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CDEL, 1, id, bj-1);          \
                            bi = pre_i; id = g->index_to_node_id[bi + beg_index];                             \
                            if (look_end) look_end = 0;                                                       \
                            break;                                                                            \
                        }                                                                                     \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0 && (cur_op & OP_F)) {                                                                \
                if (cur_op & OP_M) {                                                                          \
                    if (Hj == F1j) {                                                                          \
                        if ((SCORE_T)(Hjm1 - gap_oe1) == (SCORE_T)F1j) cur_op = OP_M|OP_E, hit = 1;           \
                        else if ((SCORE_T)(F1jm1 - gap_e1) == (SCORE_T)F1j) cur_op = OP_F1, hit = 1;          \
                    }                                                                                         \
                } else {                                                                                      \
                    if ((SCORE_T)(Hjm1 - gap_oe1) == (SCORE_T)F1j) cur_op = OP_M|OP_E, hit = 1;               \
                    else if ((SCORE_T)(F1jm1 - gap_e1) == (SCORE_T)F1j) cur_op = OP_F1, hit = 1;              \
                }                                                                                             \
                if (hit == 1) {                                                                               \
                    cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, 1, id, bj-1); --bj;                     \
                    if (look_end) look_end = 0;                                                               \
                    ++res->n_aln_bases;                                                                       \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0 && (cur_op & OP_M)) {                                                                \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj-1 < dp_beg[pre_i] || bj-1 > dp_end[pre_i]) continue;                               \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                               \
                    if ((SCORE_T)(pH[bj-1-dp_beg[pre_i]] + (SCORE_T)s + (SCORE_T)path_score) == (SCORE_T)Hj) {\
                        cur_op = OP_ALL; hit = 1;                                                             \
                        cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CMATCH, 1, id, bj-1);                     \
                        bi = pre_i; --bj; id = g->index_to_node_id[bi + beg_index];                           \
                        ++res->n_aln_bases; res->n_matched_bases += is_match ? 1 : 0;                         \
                        look_end = 0;                                                                         \
                        break;                                                                                \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0) { fprintf(stderr, "[oracle] ag backtrack dead end at %d,%d\n", bi, bj); exit(1); }  \
        }                                                                                                     \
        if (bj > 0) cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, bj, -1, bj-1);                          \
        res->graph_cigar = abpt->rev_cigar ? cigar : reverse_cigar(n_c, cigar);                               \
        res->n_cigar = n_c; res->m_cigar = m_c;                                                               \
        res->node_e = g->index_to_node_id[best_i + beg_index]; res->query_e = best_j - 1;                     \
        res->node_s = g->index_to_node_id[_start_i + beg_index]; res->query_s = _start_j - 1;                 \
    }                                                                                                         \
    for (i = 0; i < n_rows; ++i) free(pre_index[i]);                                                          \
    free(pre_index); free(pre_n);                                                                             \
    free(bs.row_off); free(bs.dp_beg); free(bs.dp_end); free(bs.arena);                                       \
    return best_score;                                                                                        \
}

DEFINE_ORACLE_AG(int16_t, i16)
DEFINE_ORACLE_AG(int32_t, i32)

/* ---------------- linear gap (simd_abpoa_lg_dp, abpoa_align_simd.c:727-815;
 * backtrack :116-194). One H plane; deletions preH[j]-e1, insertions via an
 * in-row max-plus scan directly on H. */
#define DEFINE_ORACLE_LG(SCORE_T, SUFFIX)                                                                     \
static int oracle_lg_##SUFFIX(abpoa_t *ab, abpoa_para_t *abpt, int beg_node_id, int beg_index,                \
        int end_node_id, int end_index, uint8_t *index_map, uint8_t *query, int qlen,                         \
        int32_t inf_min32, abpoa_res_t *res) {                                                                \
    abpoa_graph_t *g = ab->abg;                                                                               \
    const SCORE_T inf_min = (SCORE_T)inf_min32;                                                               \
    int64_t n_rows = end_index - beg_index + 1;                                                               \
    int i, j, k, dp_i, index_i;                                                                               \
    SCORE_T gap_e1 = (SCORE_T)abpt->gap_ext1;                                                                 \
    int w = abpt->wb < 0 ? qlen : abpt->wb + (int)(abpt->wf * qlen);                                          \
    int *mat = abpt->mat, m = abpt->m;                                                                        \
    int local_mode = abpt->align_mode == ABPOA_LOCAL_MODE;                                                    \
    int **pre_index = (int**)oxcalloc(n_rows, sizeof(int*));                                                  \
    int *pre_n = (int*)oxcalloc(n_rows, sizeof(int));                                                         \
    for (index_i = beg_index+1, dp_i = 1; index_i <= end_index; ++index_i, ++dp_i) {                          \
        int node_id = g->index_to_node_id[index_i], pn = g->node[node_id].in_edge_n, c = 0;                   \
        pre_index[dp_i] = (int*)oxmalloc((size_t)(pn>0?pn:1) * sizeof(int));                                  \
        for (j = 0; j < pn; ++j) {                                                                            \
            int pidx = g->node_id_to_index[g->node[node_id].in_id[j]];                                        \
            if (index_map[pidx]) pre_index[dp_i][c++] = pidx - beg_index;                                     \
        }                                                                                                     \
        pre_n[dp_i] = c;                                                                                      \
    }                                                                                                         \
    band_store_t bs;                                                                                          \
    bs.n_planes = 1;                                                                                          \
    bs.row_off = (int64_t*)oxmalloc((size_t)n_rows * sizeof(int64_t));                                        \
    bs.dp_beg = (int*)oxmalloc((size_t)n_rows * sizeof(int));                                                 \
    bs.dp_end = (int*)oxmalloc((size_t)n_rows * sizeof(int));                                                 \
    bs.cap = 4 * (int64_t)(qlen + 1024); bs.used = 0;                                                         \
    bs.arena = oxmalloc((size_t)bs.cap * sizeof(SCORE_T));                                                    \
    int *dp_beg = bs.dp_beg, *dp_end = bs.dp_end;                                                             \
    if (abpt->wb >= 0) {                                                                                      \
        g->node_id_to_max_pos_left[beg_node_id] = g->node_id_to_max_pos_right[beg_node_id] = 0;               \
        for (i = 0; i < g->node[beg_node_id].out_edge_n; ++i) {                                               \
            int out_id = g->node[beg_node_id].out_id[i];                                                      \
            if (index_map[g->node_id_to_index[out_id]])                                                       \
                g->node_id_to_max_pos_left[out_id] = g->node_id_to_max_pos_right[out_id] = 1;                 \
        }                                                                                                     \
        dp_beg[0] = 0;                                                                                        \
        int mr = g->node_id_to_max_remain[beg_node_id] - g->node_id_to_max_remain[end_node_id] - 1;           \
        dp_end[0] = OMIN2(qlen, OMAX2(g->node_id_to_max_pos_right[beg_node_id], qlen - mr) + w);              \
    } else { dp_beg[0] = 0; dp_end[0] = qlen; }                                                               \
    bs.row_off[0] = 0; bs.used = dp_end[0] - dp_beg[0] + 1;                                                   \
    {                                                                                                         \
        SCORE_T *H = ROWP(SCORE_T, &bs, 0, 0);                                                                \
        if (local_mode) { for (i = 0; i <= dp_end[0]; ++i) H[i] = 0; }                                        \
        else for (i = 0; i <= dp_end[0]; ++i) H[i] = (SCORE_T)(-abpt->gap_ext1 * i);                          \
    }                                                                                                         \
    int32_t best_score = inf_min32; int best_i = 0, best_j = 0, best_id = 0, zdropped = 0;                    \
    for (index_i = beg_index+1, dp_i = 1; index_i < end_index; ++index_i, ++dp_i) {                           \
        if (index_map[index_i] == 0) continue;                                                                \
        int node_id = g->index_to_node_id[index_i];                                                           \
        uint8_t base = g->node[node_id].base;                                                                 \
        int beg, end;                                                                                         \
        if (abpt->wb < 0) { beg = 0; end = qlen; }                                                            \
        else {                                                                                                \
            int mr = g->node_id_to_max_remain[node_id] - g->node_id_to_max_remain[end_node_id] - 1;           \
            beg = OMAX2(0, OMIN2(g->node_id_to_max_pos_left[node_id], qlen - mr) - w);                        \
            end = OMIN2(qlen, OMAX2(g->node_id_to_max_pos_right[node_id], qlen - mr) + w);                    \
            int min_pre_beg = INT_MAX;                                                                        \
            for (i = 0; i < pre_n[dp_i]; ++i) min_pre_beg = OMIN2(min_pre_beg, dp_beg[pre_index[dp_i][i]]);   \
            if (beg < min_pre_beg) beg = min_pre_beg;                                                         \
        }                                                                                                     \
        dp_beg[dp_i] = beg; dp_end[dp_i] = end;                                                               \
        int64_t bwid = end - beg + 1;                                                                         \
        if (bs.used + bwid > bs.cap) {                                                                        \
            while (bs.used + bwid > bs.cap) bs.cap <<= 1;                                                     \
            bs.arena = realloc(bs.arena, (size_t)bs.cap * sizeof(SCORE_T));                                   \
            if (!bs.arena) { fprintf(stderr, "[oracle] OOM arena\n"); exit(1); }                              \
        }                                                                                                     \
        bs.row_off[dp_i] = bs.used; bs.used += bwid;                                                          \
        SCORE_T *H = ROWP(SCORE_T, &bs, dp_i, 0);                                                             \
        for (j = beg; j <= end; ++j) H[j-beg] = inf_min;                                                      \
        for (k = 0; k < pre_n[dp_i]; ++k) {                                                                   \
            int pre_i = pre_index[dp_i][k];                                                                   \
            SCORE_T ps = 0;                                                                                   \
            if (abpt->inc_path_score) ps = (SCORE_T)incre_path_score(g, node_id, k);                          \
            int pbeg = dp_beg[pre_i], pend = dp_end[pre_i];                                                   \
            SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                                       \
            for (j = beg; j <= end; ++j) {                                                                    \
                SCORE_T q = (SCORE_T)(j == 0 ? 0 : mat[m * base + query[j-1]]);                               \
                SCORE_T v = inf_min;                                                                          \
                if (local_mode) {                                                                             \
                    SCORE_T mm = (SCORE_T)((j == 0 ? (SCORE_T)0 : pH[j-1-pbeg]) + ps + q);                    \
                    SCORE_T de = (SCORE_T)(pH[j-pbeg] + ps - gap_e1);                                         \
                    v = OMAX2(mm, de);                                                                        \
                } else {                                                                                      \
                    if (j-1 >= pbeg && j-1 <= pend) v = (SCORE_T)(pH[j-1-pbeg] + ps + q);                     \
                    if (j >= pbeg && j <= pend) {                                                             \
                        SCORE_T de = (SCORE_T)(pH[j-pbeg] + ps - gap_e1);                                     \
                        if (de > v) v = de;                                                                   \
                    }                                                                                         \
                }                                                                                             \
                if (v > H[j-beg]) H[j-beg] = v;                                                               \
            }                                                                                                 \
        }                                                                                                     \
        for (j = beg + 1; j <= end; ++j) { /* in-row insertion scan */                                        \
            SCORE_T v = (SCORE_T)(H[j-1-beg] - gap_e1);                                                       \
            if (v > H[j-beg]) H[j-beg] = v;                                                                   \
        }                                                                                                     \
        if (local_mode) for (j = beg; j <= end; ++j) H[j-beg] = OMAX2(H[j-beg], (SCORE_T)0);                  \
        if (local_mode || abpt->align_mode == ABPOA_EXTEND_MODE || abpt->wb >= 0) {                           \
            int32_t mx = inf_min32; int left = -1, right = -1;                                                \
            for (j = beg; j <= end; ++j) {                                                                    \
                if ((int32_t)H[j-beg] > mx) { mx = (int32_t)H[j-beg]; left = right = j; }                     \
                else if ((int32_t)H[j-beg] == mx) right = j;                                                  \
            }                                                                                                 \
            if (local_mode) {                                                                                 \
                if (mx > best_score) { best_score = mx; best_i = dp_i; best_j = left; }                       \
            } else if (abpt->align_mode == ABPOA_EXTEND_MODE) {                                               \
                if (mx > best_score) { best_score = mx; best_i = dp_i; best_j = right; best_id = node_id; }   \
                else if (abpt->zdrop > 0) {                                                                   \
                    int delta = g->node_id_to_max_remain[best_id] - g->node_id_to_max_remain[node_id];        \
                    int dd = delta - (right - best_j); if (dd < 0) dd = -dd;                                  \
                    if (best_score - mx > abpt->zdrop + abpt->gap_ext1 * dd) { zdropped = 1; }                \
                }                                                                                             \
            }                                                                                                 \
            if (!zdropped && abpt->wb >= 0) {                                                                 \
                for (i = 0; i < g->node[node_id].out_edge_n; ++i) {                                           \
                    int out_id = g->node[node_id].out_id[i];                                                  \
                    if (right + 1 > g->node_id_to_max_pos_right[out_id]) g->node_id_to_max_pos_right[out_id] = right + 1; \
                    if (left + 1 < g->node_id_to_max_pos_left[out_id]) g->node_id_to_max_pos_left[out_id] = left + 1;     \
                }                                                                                             \
            }                                                                                                 \
            if (zdropped) break;                                                                              \
        }                                                                                                     \
    }                                                                                                         \
    if (abpt->align_mode == ABPOA_GLOBAL_MODE) {                                                              \
        for (i = 0; i < g->node[end_node_id].in_edge_n; ++i) {                                                \
            int in_id = g->node[end_node_id].in_id[i];                                                        \
            int in_index = g->node_id_to_index[in_id];                                                        \
            if (index_map[in_index] == 0) continue;                                                           \
            int in_dp_i = in_index - beg_index;                                                               \
            int e = OMIN2(qlen, dp_end[in_dp_i]);                                                             \
            SCORE_T *pH = ROWP(SCORE_T, &bs, in_dp_i, 0);                                                     \
            int32_t sc = (e >= dp_beg[in_dp_i]) ? (int32_t)pH[e - dp_beg[in_dp_i]] : inf_min32;               \
            if (sc > best_score) { best_score = sc; best_i = in_dp_i; best_j = e; }                           \
        }                                                                                                     \
    }                                                                                                         \
    res->best_score = best_score;                                                                             \
    if (abpt->ret_cigar) { /* simd_abpoa_lg_backtrack (:116-194) */                                           \
        int bi = best_i, bj = best_j, _start_i = best_i, _start_j = best_j;                                   \
        int n_c = 0, m_c = 0, hit, id, s, is_match, path_score = 0;                                           \
        abpoa_cigar_t *cigar = 0;                                                                             \
        id = g->index_to_node_id[bi + beg_index];                                                             \
        if (best_j < qlen) cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, qlen - best_j, -1, qlen - 1);    \
        int look_end = abpt->put_gap_at_end, put_right = abpt->put_gap_on_right;                              \
        while (bi > 0 && bj > 0) {                                                                            \
            SCORE_T *H = ROWP(SCORE_T, &bs, bi, 0);                                                           \
            int rb = dp_beg[bi], re = dp_end[bi];                                                             \
            int32_t Hj   = (bj   >= rb && bj   <= re) ? (int32_t)H[bj-rb]   : inf_min32;                      \
            int32_t Hjm1 = (bj-1 >= rb && bj-1 <= re) ? (int32_t)H[bj-1-rb] : inf_min32;                      \
            if (local_mode && Hj == 0) break;                                                                 \
            _start_i = bi; _start_j = bj;                                                                     \
            int *pre_index_i = pre_index[bi];                                                                 \
            s = mat[m * g->node[id].base + query[bj-1]]; hit = 0;                                             \
            is_match = g->node[id].base == query[bj-1];                                                       \
            if (put_right == 0 && look_end == 0) {                                                            \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj-1 < dp_beg[pre_i] || bj-1 > dp_end[pre_i]) continue;                               \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                               \
                    if ((SCORE_T)(pH[bj-1-dp_beg[pre_i]] + (SCORE_T)s + (SCORE_T)path_score) == (SCORE_T)Hj) {\
                        cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CMATCH, 1, id, bj-1);                     \
                        bi = pre_i; --bj; id = g->index_to_node_id[bi + beg_index]; hit = 1;                  \
                        ++res->n_aln_bases; res->n_matched_bases += is_match ? 1 : 0;                         \
                        break;                                                                                \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0) { /* deletion */                                                                    \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj < dp_beg[pre_i] || bj > dp_end[pre_i]) continue;                                   \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                               \
                    if ((SCORE_T)(pH[bj-dp_beg[pre_i]] - gap_e1 + (SCORE_T)path_score) == (SCORE_T)Hj) {      \
                        cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CDEL, 1, id, bj-1);                       \
                        bi = pre_i; id = g->index_to_node_id[bi + beg_index]; hit = 1;                        \
                        if (look_end) look_end = 0;                                                           \
                        break;                                                                                \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0) { /* insertion */                                                                   \
                if ((SCORE_T)(Hjm1 - gap_e1) == (SCORE_T)Hj) {                                                \
                    cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, 1, id, bj-1); bj--;                     \
                    if (look_end) look_end = 0;                                                               \
                    hit = 1; ++res->n_aln_bases;                                                              \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0) { /* match again (gap-flag variants) */                                             \
                for (k = 0; k < pre_n[bi]; ++k) {                                                             \
                    int pre_i = pre_index_i[k];                                                               \
                    if (abpt->inc_path_score) path_score = incre_path_score(g, id, k);                        \
                    if (bj-1 < dp_beg[pre_i] || bj-1 > dp_end[pre_i]) continue;                               \
                    SCORE_T *pH = ROWP(SCORE_T, &bs, pre_i, 0);                                               \
                    if ((SCORE_T)(pH[bj-1-dp_beg[pre_i]] + (SCORE_T)s + (SCORE_T)path_score) == (SCORE_T)Hj) {\
                        cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CMATCH, 1, id, bj-1);                     \
                        bi = pre_i; --bj; id = g->index_to_node_id[bi + beg_index]; hit = 1;                  \
                        ++res->n_aln_bases; res->n_matched_bases += is_match ? 1 : 0;                         \
                        look_end = 0;                                                                         \
                        break;                                                                                \
                    }                                                                                         \
                }                                                                                             \
            }                                                                                                 \
            if (hit == 0) { fprintf(stderr, "[oracle] lg backtrack dead end at %d,%d\n", bi, bj); exit(1); }  \
        }                                                                                                     \
        if (bj > 0) cigar = push_cigar(&n_c, &m_c, cigar, ABPOA_CINS, bj, -1, bj-1);                          \
        res->graph_cigar = abpt->rev_cigar ? cigar : reverse_cigar(n_c, cigar);                               \
        res->n_cigar = n_c; res->m_cigar = m_c;                                                               \
        res->node_e = g->index_to_node_id[best_i + beg_index]; res->query_e = best_j - 1;                     \
        res->node_s = g->index_to_node_id[_start_i + beg_index]; res->query_s = _start_j - 1;                 \
    }                                                                                                         \
    for (i = 0; i < n_rows; ++i) free(pre_index[i]);                                                          \
    free(pre_index); free(pre_n);                                                                             \
    free(bs.row_off); free(bs.dp_beg); free(bs.dp_end); free(bs.arena);                                       \
    return best_score;                                                                                        \
}

DEFINE_ORACLE_LG(int16_t, i16)
DEFINE_ORACLE_LG(int32_t, i32)

/* seam entry, mirroring simd_abpoa_align_sequence_to_subgraph
 * (abpoa_align_simd.c:1250-1332) */
int oracle_align_sequence_to_subgraph(abpoa_t *ab, abpoa_para_t *abpt,
        int beg_node_id, int end_node_id, uint8_t *query, int qlen, abpoa_res_t *res) {
    abpoa_graph_t *g = ab->abg;
    int i, j;
    int beg_index = g->node_id_to_index[beg_node_id], end_index = g->node_id_to_index[end_node_id];
    int gn = end_index - beg_index + 1;
    uint8_t *index_map = (uint8_t*)oxcalloc(g->node_n, 1);
    index_map[beg_index] = index_map[end_index] = 1;
    for (i = beg_index; i < end_index - 1; ++i) {
        if (index_map[i] == 0) continue;
        int node_id = g->index_to_node_id[i];
        for (j = 0; j < g->node[node_id].out_edge_n; ++j)
            index_map[g->node_id_to_index[g->node[node_id].out_id[j]]] = 1;
    }
    int32_t gap_ext1 = abpt->gap_ext1, gap_ext2 = abpt->gap_ext2;
    int32_t gap_oe1 = abpt->gap_open1 + gap_ext1, gap_oe2 = abpt->gap_open2 + gap_ext2;
    int len = qlen > gn ? qlen : gn;
    int32_t max_score = (int32_t)qlen * abpt->max_mat;
    {
        int32_t alt = (int32_t)len * abpt->gap_ext1 + abpt->gap_open1;
        if (alt > max_score) max_score = alt;
    }
    if (max_score <= INT16_MAX - abpt->min_mis - gap_oe1 - gap_oe2) {
        int32_t inf_min = INT16_MIN + abpt->min_mis;
        if (INT16_MIN + gap_oe1 > inf_min) inf_min = INT16_MIN + gap_oe1;
        if (INT16_MIN + gap_oe2 > inf_min) inf_min = INT16_MIN + gap_oe2;
        inf_min += 512 * (gap_ext1 > gap_ext2 ? gap_ext1 : gap_ext2);
        if (abpt->gap_mode == ABPOA_CONVEX_GAP)
            oracle_cg_i16(ab, abpt, beg_node_id, beg_index, end_node_id, end_index, index_map, query, qlen, inf_min, res);
        else if (abpt->gap_mode == ABPOA_AFFINE_GAP)
            oracle_ag_i16(ab, abpt, beg_node_id, beg_index, end_node_id, end_index, index_map, query, qlen, inf_min, res);
        else
            oracle_lg_i16(ab, abpt, beg_node_id, beg_index, end_node_id, end_index, index_map, query, qlen, inf_min, res);
    } else {
        int32_t inf_min = INT32_MIN + abpt->min_mis;
        if (INT32_MIN + gap_oe1 > inf_min) inf_min = INT32_MIN + gap_oe1;
        if (INT32_MIN + gap_oe2 > inf_min) inf_min = INT32_MIN + gap_oe2;
        inf_min += 512 * (gap_ext1 > gap_ext2 ? gap_ext1 : gap_ext2);
        if (abpt->gap_mode == ABPOA_CONVEX_GAP)
            oracle_cg_i32(ab, abpt, beg_node_id, beg_index, end_node_id, end_index, index_map, query, qlen, inf_min, res);
        else if (abpt->gap_mode == ABPOA_AFFINE_GAP)
            oracle_ag_i32(ab, abpt, beg_node_id, beg_index, end_node_id, end_index, index_map, query, qlen, inf_min, res);
        else
            oracle_lg_i32(ab, abpt, beg_node_id, beg_index, end_node_id, end_index, index_map, query, qlen, inf_min, res);
    }
    free(index_map);
    return 0;
}
