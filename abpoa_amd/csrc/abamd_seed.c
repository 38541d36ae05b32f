/* Minimizer seeding, progressive guide tree and anchor chaining (-S / -p).
 *
 * Restates abpoa_seed.c:1-756 exactly:
 *   - (w,k)-minimizer sketch (the minimap2 sketch algorithm, both-strand for
 *     nt with symmetric-k-mer skip, forward-only 5-bit k-mers for aa);
 *   - guide tree: per-pair minimizer-hit counts -> jaccard similarity ->
 *     greedy insertion order (abpoa_build_guide_tree, :244-337);
 *   - per-adjacent-pair anchors (cartesian products of equal-hash minimizer
 *     runs) and two-level DP chaining (local chains then chain-of-chains,
 *     abpoa_dp_chaining, :497-591);
 *   - anchor-windowed POA: each read aligns per anchor window through the
 *     subgraph seam, anchor k-mers become exact-match cigar runs, and the
 *     previous read's qpos->node map becomes the target map
 *     (abpoa_anchor_poa, abpoa_align.c:209-310).
 *
 * Sort-order fidelity: the reference sorts with klib's MSD radix sort
 * (8-bit digits from the top byte, in-place bucket cycling, insertion sort
 * at <= 64 elements), which is NOT stable — tie order is a function of the
 * algorithm. rs_sort_* below restate that algorithm exactly so runs of
 * equal keys land in the same order. Two reference quirks are preserved
 * deliberately: the chain-collection loop reads the strand bit from
 * anchors[i] where i indexes local_chains (abpoa_seed.c:566), and the
 * chain-score gap penalty is computed in double and truncated on the int
 * subtraction, with ilog2(0) = -1 (:478-492). */
#include <assert.h>
#include <math.h>
#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

typedef struct { uint64_t x, y; } ab_u128_t;
typedef struct { size_t n, m; ab_u128_t *a; } u128v_t;
typedef abamd_u64v_t u64v_t;

static void u128v_push(u128v_t *v, ab_u128_t e) {
    if (v->n == v->m) {
        v->m = v->m ? v->m << 1 : 16;
        v->a = (ab_u128_t*)abamd_realloc(v->a, v->m * sizeof(ab_u128_t));
    }
    v->a[v->n++] = e;
}
static void u64v_push(u64v_t *v, uint64_t e) {
    if (v->n == v->m) {
        v->m = v->m ? v->m << 1 : 16;
        v->a = (uint64_t*)abamd_realloc(v->a, v->m * sizeof(uint64_t));
    }
    v->a[v->n++] = e;
}

/* ---- klib MSD radix sort, restated (ksort.h KRADIX_SORT_INIT) ---- */
#define RS_MIN_SIZE 64

#define DEFINE_RADIX_SORT(name, type_t, KEY)                                   \
static void rs_insertsort_##name(type_t *beg, type_t *end) {                   \
    type_t *i;                                                                 \
    for (i = beg + 1; i < end; ++i)                                            \
        if (KEY(*i) < KEY(*(i - 1))) {                                         \
            type_t *j, tmp = *i;                                               \
            for (j = i; j > beg && KEY(tmp) < KEY(*(j - 1)); --j) *j = *(j-1); \
            *j = tmp;                                                          \
        }                                                                      \
}                                                                              \
static void rs_sort_##name(type_t *beg, type_t *end, int n_bits, int s) {      \
    type_t *i;                                                                 \
    int size = 1 << n_bits, m = size - 1;                                      \
    struct { type_t *b, *e; } b[256], *k, *be = b + size;                      \
    for (k = b; k != be; ++k) k->b = k->e = beg;                               \
    for (i = beg; i != end; ++i) ++b[KEY(*i) >> s & m].e;                      \
    for (k = b + 1; k != be; ++k) k->e += (k - 1)->e - beg, k->b = (k - 1)->e; \
    for (k = b; k != be;) {                                                    \
        if (k->b != k->e) {                                                    \
            __typeof__(k) l;                                                   \
            if ((l = b + (KEY(*k->b) >> s & m)) != k) {                        \
                type_t tmp = *k->b, swap;                                      \
                do {                                                           \
                    swap = tmp; tmp = *l->b; *l->b++ = swap;                   \
                    l = b + (KEY(tmp) >> s & m);                               \
                } while (l != k);                                              \
                *k->b++ = tmp;                                                 \
            } else ++k->b;                                                     \
        } else ++k;                                                            \
    }                                                                          \
    for (b->b = beg, k = b + 1; k != be; ++k) k->b = (k - 1)->e;               \
    if (s) {                                                                   \
        s = s > n_bits ? s - n_bits : 0;                                       \
        for (k = b; k != be; ++k)                                              \
            if (k->e - k->b > RS_MIN_SIZE) rs_sort_##name(k->b, k->e, n_bits, s); \
            else if (k->e - k->b > 1) rs_insertsort_##name(k->b, k->e);        \
    }                                                                          \
}                                                                              \
static void radix_sort_##name(type_t *beg, type_t *end) {                      \
    if (end - beg <= RS_MIN_SIZE) rs_insertsort_##name(beg, end);              \
    else rs_sort_##name(beg, end, 8, 56);                                      \
}

#define KEY_128X(a) ((a).x)
#define KEY_64(a) (a)
DEFINE_RADIX_SORT(u128x, ab_u128_t, KEY_128X)
DEFINE_RADIX_SORT(u64, uint64_t, KEY_64)

/* ---- minimizer sketch (minimap2 algorithm; abpoa_seed.c:48-240) ---- */

static inline uint64_t mm_hash64(uint64_t key, uint64_t mask) {
    key = (~key + (key << 21)) & mask;
    key = key ^ key >> 24;
    key = ((key + (key << 3)) + (key << 8)) & mask;
    key = key ^ key >> 14;
    key = ((key + (key << 2)) + (key << 4)) & mask;
    key = key ^ key >> 28;
    key = (key + (key << 31)) & mask;
    return key;
}

static const signed char seed_log_table256[256] = {
#define ABAMD_LT(n) n, n, n, n, n, n, n, n, n, n, n, n, n, n, n, n
    -1, 0, 1, 1, 2, 2, 2, 2, 3, 3, 3, 3, 3, 3, 3, 3,
    ABAMD_LT(4), ABAMD_LT(5), ABAMD_LT(5), ABAMD_LT(6), ABAMD_LT(6), ABAMD_LT(6), ABAMD_LT(6),
    ABAMD_LT(7), ABAMD_LT(7), ABAMD_LT(7), ABAMD_LT(7), ABAMD_LT(7), ABAMD_LT(7), ABAMD_LT(7), ABAMD_LT(7)
};
static inline int seed_ilog2_32(uint32_t v) {
    uint32_t t, tt;
    if ((tt = v >> 16)) return (t = tt >> 8) ? 24 + seed_log_table256[t] : 16 + seed_log_table256[tt];
    return (t = v >> 8) ? 8 + seed_log_table256[t] : seed_log_table256[v];
}

/* sliding-window min over the last w k-mers; bits = 2 (nt) or 5 (aa) */
static void mm_sketch(const uint8_t *str, int len, int w, int k, uint32_t rid,
                      int both_strand, int bits, int max_code, u128v_t *p) {
    uint64_t shift1 = (uint64_t)bits * (k - 1), mask = (bits == 2 ? (1ULL << 2 * k) : (1ULL << 5 * k)) - 1;
    uint64_t kmer[2] = {0, 0};
    int i, j, l, buf_pos, min_pos, kmer_span = 0;
    ab_u128_t buf[256], min = { UINT64_MAX, UINT64_MAX };

    if (!(len > 0 && w > 0 && w < 256 && k > 0 && (bits == 2 ? k <= 28 : k <= 11)))
        abamd_fatal("mm_sketch", "bad sketch parameters (len %d, w %d, k %d)", len, w, k);
    memset(buf, 0xff, (size_t)w * 16);

    for (i = l = buf_pos = min_pos = 0; i < len; ++i) {
        int c = str[i];
        ab_u128_t info = { UINT64_MAX, UINT64_MAX };
        if (c < max_code) {
            uint32_t z;
            kmer_span = l + 1 < k ? l + 1 : k;
            if (both_strand) {
                kmer[0] = (kmer[0] << 2 | (uint64_t)c) & mask;
                kmer[1] = (kmer[1] >> 2) | ((3ULL ^ c) << shift1);
                if (kmer[0] == kmer[1]) continue; /* symmetric k-mer: strand unknown */
                z = kmer[0] < kmer[1] ? 0 : 1;
            } else {
                kmer[0] = (kmer[0] << bits | (uint64_t)c) & mask;
                z = 0;
            }
            ++l;
            if (l >= k && kmer_span < 256) {
                info.x = mm_hash64(kmer[z], mask) << 8 | (uint64_t)kmer_span;
                info.y = (uint64_t)rid << 32 | (uint32_t)i << 1 | z;
            }
        } else l = 0, kmer_span = 0;
        buf[buf_pos] = info;
        if (l == w + k - 1 && min.x != UINT64_MAX) {
            for (j = buf_pos + 1; j < w; ++j)
                if (min.x == buf[j].x && buf[j].y != min.y) u128v_push(p, buf[j]);
            for (j = 0; j < buf_pos; ++j)
                if (min.x == buf[j].x && buf[j].y != min.y) u128v_push(p, buf[j]);
        }
        if (info.x <= min.x) {
            if (l >= w + k && min.x != UINT64_MAX) u128v_push(p, min);
            min = info, min_pos = buf_pos;
        } else if (buf_pos == min_pos) {
            if (l >= w + k - 1 && min.x != UINT64_MAX) u128v_push(p, min);
            for (j = buf_pos + 1, min.x = UINT64_MAX; j < w; ++j)
                if (min.x >= buf[j].x) min = buf[j], min_pos = j;
            for (j = 0; j <= buf_pos; ++j)
                if (min.x >= buf[j].x) min = buf[j], min_pos = j;
            if (l >= w + k - 1 && min.x != UINT64_MAX) {
                for (j = buf_pos + 1; j < w; ++j)
                    if (min.x == buf[j].x && min.y != buf[j].y) u128v_push(p, buf[j]);
                for (j = 0; j <= buf_pos; ++j)
                    if (min.x == buf[j].x && min.y != buf[j].y) u128v_push(p, buf[j]);
            }
        }
        if (++buf_pos == w) buf_pos = 0;
    }
    if (min.x != UINT64_MAX) u128v_push(p, min);
}

static void collect_mm(uint8_t **seqs, int *seq_lens, int n_seq, abpoa_para_t *abpt,
                       u128v_t *mm, int *mm_c) {
    int i;
    mm_c[0] = 0;
    for (i = 0; i < n_seq; ++i) {
        if (abpt->m > 5) mm_sketch(seqs[i], seq_lens[i], abpt->w, abpt->k, (uint32_t)i, 0, 5, 26, mm);
        else mm_sketch(seqs[i], seq_lens[i], abpt->w, abpt->k, (uint32_t)i, abpt->amb_strand, 2, 4, mm);
        mm_c[i + 1] = (int)mm->n;
    }
}

/* ---- progressive guide tree (abpoa_seed.c:244-337) ---- */
static void build_guide_tree(abpoa_para_t *abpt, int n_seq, u128v_t *mm, int *tree_id_map) {
    (void)abpt;
    if (mm->n == 0) return;
    size_t i, _i, j;
    int rid1, rid2;
    int *mm_hit_n = (int*)abamd_calloc((size_t)(n_seq * (n_seq + 1)) >> 1, sizeof(int));
    radix_sort_u128x(mm->a, mm->a + mm->n);
    uint64_t last_x = mm->a[0].x;
    int *mm_cnt = (int*)abamd_malloc((size_t)n_seq * sizeof(int));
    for (_i = 0, i = 1; i < mm->n; ++i) {
        if (mm->a[i].x != last_x) {
            memset(mm_cnt, 0, (size_t)n_seq * sizeof(int));
            for (j = _i; j < i; ++j) {
                rid1 = (int)(mm->a[j].y >> 32);
                ++mm_cnt[rid1];
                ++mm_hit_n[((rid1 * (rid1 + 1)) >> 1) + rid1];
            }
            for (rid1 = 0; rid1 < n_seq - 1; ++rid1)
                for (rid2 = rid1 + 1; rid2 < n_seq; ++rid2)
                    mm_hit_n[((rid2 * (rid2 + 1)) >> 1) + rid1] += AB_MIN2(mm_cnt[rid1], mm_cnt[rid2]);
            last_x = mm->a[i].x, _i = i;
        }
    }
    memset(mm_cnt, 0, (size_t)n_seq * sizeof(int));
    for (j = _i; j < i; ++j) {
        rid1 = (int)(mm->a[j].y >> 32);
        ++mm_cnt[rid1];
        ++mm_hit_n[((rid1 * (rid1 + 1)) >> 1) + rid1];
    }
    for (rid1 = 0; rid1 < n_seq - 1; ++rid1)
        for (rid2 = rid1 + 1; rid2 < n_seq; ++rid2)
            mm_hit_n[((rid2 * (rid2 + 1)) >> 1) + rid1] += AB_MIN2(mm_cnt[rid1], mm_cnt[rid2]);
    free(mm_cnt);

    double *jac_sim = (double*)abamd_calloc((size_t)(n_seq * (n_seq - 1)) >> 1, sizeof(double));
    double max_jac = -1.0, jac;
    int max_i = -1, max_j = -1;
    for (i = 1; i < (size_t)n_seq; ++i) {
        for (j = 0; j < i; ++j) {
            int tot_n = mm_hit_n[((i * (i + 1)) >> 1) + i] + mm_hit_n[((j * (j + 1)) >> 1) + j]
                      - mm_hit_n[((i * (i + 1)) >> 1) + j];
            if (tot_n == 0) jac = 0;
            else if (tot_n < 0) abamd_fatal("build_guide_tree", "negative minimizer totals");
            else jac = (0.0 + mm_hit_n[((i * (i + 1)) >> 1) + j]) / tot_n;
            jac_sim[((i * (i - 1)) >> 1) + j] = jac;
            if (jac > max_jac) { max_jac = jac; max_i = (int)i; max_j = (int)j; }
        }
    }
    int n_in_map = 2;
    tree_id_map[0] = max_j, tree_id_map[1] = max_i;
    while (n_in_map < n_seq) {
        max_jac = -1.0, max_i = n_seq;
        for (rid1 = 0; rid1 < n_seq; ++rid1) {
            jac = 0.0;
            for (i = 0; i < (size_t)n_in_map; ++i) {
                rid2 = tree_id_map[i];
                if (rid1 == rid2) { jac = -1.0; break; }
                else if (rid1 > rid2) jac += jac_sim[(((size_t)rid1 * (rid1 - 1)) >> 1) + rid2];
                else jac += jac_sim[(((size_t)rid2 * (rid2 - 1)) >> 1) + rid1];
            }
            if (jac > max_jac) { max_jac = jac; max_i = rid1; }
        }
        if (max_i == n_seq) abamd_fatal("build_guide_tree", "no next sequence found");
        tree_id_map[n_in_map++] = max_i;
    }
    free(mm_hit_n); free(jac_sim);
}

/* ---- anchors between adjacent guide-tree reads (abpoa_seed.c:344-377) ---- */
static int collect_anchors1(u64v_t *anchors, u128v_t mm, int *mm_c, int tid, int qid,
                            int qlen, int k) {
    int i, j, _i, _j;
    uint64_t xi, xj, _xi, _xj, _yi, _yj, a;
    i = mm_c[tid], j = mm_c[qid];
    radix_sort_u128x(mm.a + j, mm.a + mm_c[qid + 1]);
    while (i < mm_c[tid + 1] && j < mm_c[qid + 1]) {
        xi = mm.a[i].x, xj = mm.a[j].x;
        if (xi == xj) {
            for (_i = i; _i < mm_c[tid + 1]; ++_i) {
                _xi = mm.a[_i].x;
                if (_xi != xi) break;
                _yi = mm.a[_i].y;
                for (_j = j; _j < mm_c[qid + 1]; ++_j) {
                    _xj = mm.a[_j].x;
                    if (_xj != xj) break;
                    _yj = mm.a[_j].y;
                    if ((_yi & 1) == (_yj & 1)) {
                        a = (uint64_t)((uint32_t)_yi >> 1) << 32 | ((uint32_t)_yj >> 1);
                    } else {
                        a = 1ULL << 63 | (uint64_t)((uint32_t)_yi >> 1) << 32
                          | (uint64_t)(qlen - (((uint32_t)_yj >> 1) + 1 - k) - 1);
                    }
                    u64v_push(anchors, a);
                }
            }
            i = _i, j = _j;
        } else if (xi < xj) ++i;
        else ++j;
    }
    radix_sort_u64(anchors->a, anchors->a + anchors->n);
    return (int)anchors->n;
}

/* ---- two-level DP chaining (abpoa_seed.c:380-591) ---- */

static int get_local_chain_score(int j_end_tpos, int j_end_qpos, int i_end_anchor_i,
                                 u64v_t *anchors, int *pre_id, int *score) {
    int i = i_end_anchor_i, chain_score;
    int i_tpos, i_qpos;
    do {
        i_tpos = (int)((anchors->a[i] >> 32) & 0x7fffffff), i_qpos = (int32_t)anchors->a[i];
        if (i_tpos <= j_end_tpos && i_qpos <= j_end_qpos) break;
        i = pre_id[i];
    } while (i != -1);
    if (i == -1) chain_score = score[i_end_anchor_i];
    else chain_score = score[i_end_anchor_i] - score[i];
    return chain_score;
}

static void dp_chaining_of_local_chains(ab_u128_t *local_chains, int n_local_chains,
                                        u64v_t *anchors, int *score, int *pre_id,
                                        u64v_t *par_anchors, int min_w, int tlen, int qlen) {
    int i, j, st, score1, global_max_score = INT32_MIN, global_max_i = -1;
    int *chain_score = (int*)abamd_malloc((size_t)n_local_chains * 4);
    int *pre_chain_id = (int*)abamd_malloc((size_t)n_local_chains * 4);
    size_t _n = par_anchors->n;

    for (i = st = 0; i < n_local_chains; ++i) {
        uint64_t ix = local_chains[i].x, iy = local_chains[i].y;
        int istrand = (int)(ix >> 63), i_end_qpos = (int32_t)ix;
        int i_end_anchor_i = (int)(iy >> 32), i_start_anchor_i = (int32_t)iy;
        int i_start_tpos = (int)((anchors->a[i_start_anchor_i] >> 32) & 0x7fffffff);
        int i_start_qpos = (int32_t)anchors->a[i_start_anchor_i];
        int max_j = -1, max_score = score[i_end_anchor_i];
        while (st < i) {
            if ((int)((local_chains[st].x) >> 63) != istrand) ++st;
            else break;
        }
        for (j = i - 1; j >= st; --j) {
            uint64_t jx = local_chains[j].x;
            int j_end_tpos = (int)((jx >> 32) & 0x7fffffff), j_end_qpos = (int32_t)jx;
            if (j_end_qpos >= i_end_qpos) continue;
            if (i_start_tpos > j_end_tpos && i_start_qpos > j_end_qpos)
                score1 = chain_score[j] + score[i_end_anchor_i];
            else
                score1 = chain_score[j] + get_local_chain_score(j_end_tpos, j_end_qpos, i_end_anchor_i, anchors, pre_id, score);
            if (score1 > max_score) { max_score = score1; max_j = j; }
        }
        chain_score[i] = max_score; pre_chain_id[i] = max_j;
        if (max_score > global_max_score) { global_max_score = max_score; global_max_i = i; }
    }
    if (global_max_i < 0) { free(chain_score); free(pre_chain_id); return; }
    int cur_i = global_max_i, pre_i = pre_chain_id[global_max_i];
    uint64_t cur_y = local_chains[cur_i].y, pre_x, pre_y;
    int last_tpos = tlen, last_qpos = qlen;
    while (pre_i != -1) {
        pre_x = local_chains[pre_i].x, pre_y = local_chains[pre_i].y;
        int pre_end_tpos = (int)((pre_x >> 32) & 0x7fffffff), pre_end_qpos = (int32_t)pre_x;
        i = (int)(cur_y >> 32);
        while (i != -1) {
            int cur_tpos = (int)((anchors->a[i] >> 32) & 0x7fffffff), cur_qpos = (int32_t)anchors->a[i];
            if (cur_tpos > pre_end_tpos && cur_qpos > pre_end_qpos) {
                if (last_tpos - cur_tpos >= min_w && last_qpos - cur_qpos >= min_w) {
                    u64v_push(par_anchors, anchors->a[i]);
                    last_tpos = cur_tpos, last_qpos = cur_qpos;
                }
            } else break;
            i = pre_id[i];
        }
        cur_i = pre_i, pre_i = pre_chain_id[pre_i], cur_y = pre_y;
    }
    i = (int)(cur_y >> 32);
    while (i != -1) {
        int cur_tpos = (int)((anchors->a[i] >> 32) & 0x7fffffff), cur_qpos = (int32_t)anchors->a[i];
        if (last_tpos - cur_tpos >= min_w && last_qpos - cur_qpos >= min_w) {
            u64v_push(par_anchors, anchors->a[i]);
            last_tpos = cur_tpos, last_qpos = cur_qpos;
        }
        i = pre_id[i];
    }
    for (i = 0; i < (int)(par_anchors->n - _n) >> 1; ++i) {
        uint64_t tmp = par_anchors->a[_n + i];
        par_anchors->a[_n + i] = par_anchors->a[par_anchors->n - i - 1];
        par_anchors->a[par_anchors->n - i - 1] = tmp;
    }
    free(chain_score); free(pre_chain_id);
}

/* gap-cost term computed in double, truncated by the int subtraction; note
 * seed_ilog2_32(0) == -1 (both exactly as the reference) */
static int get_chain_score(int max_bw, int *score, int i_qpos, int i_tpos,
                           int j_qpos, int j_tpos, int k) {
    int delta_q, delta_t, delta_tq, min_d;
    delta_q = i_qpos - j_qpos; delta_t = i_tpos - j_tpos;
    min_d = AB_MIN3(delta_q, delta_t, k);
    *score = min_d;
    if (delta_q >= delta_t) {
        if ((delta_tq = delta_q - delta_t) > max_bw) return 0;
    } else {
        if ((delta_tq = delta_t - delta_q) > max_bw) return 0;
    }
    *score -= ((seed_ilog2_32((uint32_t)delta_tq) >> 1) + delta_tq * 0.01 * k);
    return 1;
}

static void dp_chaining(u64v_t *anchors, u64v_t *par_anchors, abpoa_para_t *abpt,
                        int tlen, int qlen) {
    int i, j, st, n_a = (int)anchors->n;
    if (n_a == 0) return;
    int *score = (int*)abamd_malloc((size_t)n_a * 4);
    int *pre_id = (int*)abamd_malloc((size_t)n_a * 4);
    int *end_pos = (int*)abamd_malloc((size_t)n_a * 4);
    memset(end_pos, 0, (size_t)n_a * 4);

    int max_bw = 100, max_dis = 100, max_skip_anchors = 25, max_non_best_anchors = 50, min_local_chain_score = 100;
    int min_w = abpt->min_w + abpt->k;
    int i_qpos, i_tpos, i_tstrand, j_qpos, j_tpos;
    for (i = st = 0; i < n_a; ++i) {
        uint64_t ia = anchors->a[i];
        i_qpos = (int32_t)ia, i_tpos = (int)((ia >> 32) & 0x7fffffff), i_tstrand = (int)(ia >> 63);
        int max_j = -1, n_skip = 0, non_best_iter_n = 0, max_score = abpt->k, _score;
        while (st < i) {
            uint64_t st_a = anchors->a[st];
            if ((int)(st_a >> 63) != i_tstrand || (int)((st_a >> 32) & 0x7fffffff) + max_dis < i_tpos) ++st;
            else break;
        }
        for (j = i - 1; j >= st; --j) {
            uint64_t ja = anchors->a[j];
            j_qpos = (int)(uint32_t)ja; j_tpos = (int)((ja >> 32) & 0x7fffffff);
            if (j_qpos >= i_qpos || j_qpos + max_dis < i_qpos) continue;
            if (!get_chain_score(max_bw, &_score, i_qpos, i_tpos, j_qpos, j_tpos, abpt->k)) continue;
            _score += score[j];
            if (_score > max_score) {
                max_score = _score; max_j = j;
                non_best_iter_n = 0;
                if (n_skip > 0) --n_skip;
            } else if (end_pos[j] == i) {
                if (++n_skip > max_skip_anchors) break;
            } else if (++non_best_iter_n > max_non_best_anchors) break;
            if (pre_id[j] >= 0) end_pos[pre_id[j]] = i;
        }
        score[i] = max_score, pre_id[i] = max_j;
    }

    memset(end_pos, 0, (size_t)n_a * 4);
    int n_local_chains = 0;
    for (i = n_a - 1; i >= 0; --i) {
        if (pre_id[i] >= 0) end_pos[pre_id[i]] = 1;
        if (end_pos[i] == 0 && score[i] >= min_local_chain_score) {
            end_pos[i] = 2;
            ++n_local_chains;
        }
    }
    ab_u128_t *local_chains = (ab_u128_t*)abamd_malloc((size_t)(n_local_chains ? n_local_chains : 1) * sizeof(ab_u128_t));
    for (i = n_local_chains = 0; i < n_a; ++i) {
        if (end_pos[i] == 2) {
            local_chains[n_local_chains].x = (uint64_t)score[i];
            local_chains[n_local_chains++].y = (uint64_t)i;
        }
    }
    radix_sort_u128x(local_chains, local_chains + n_local_chains);

    int32_t *anchor_map = end_pos; memset(anchor_map, 0, (size_t)n_a * 4);
    int start_id = 0, end_id, tot_chain_i;
    uint64_t strand, tpos, qpos;
    for (i = tot_chain_i = n_local_chains - 1; i >= 0; --i) {
        j = (int)local_chains[i].y; end_id = j;
        /* reference reads the strand from anchors[i] (the CHAIN index) here */
        strand = anchors->a[i] >> 63;
        tpos = (anchors->a[j] >> 32) & 0x7fffffff, qpos = (uint64_t)(uint32_t)(int32_t)anchors->a[j];
        do {
            start_id = j;
            anchor_map[j] = 1;
            j = pre_id[j];
        } while (j >= 0 && anchor_map[j] == 0);
        if (j < 0) {
            local_chains[tot_chain_i].x = strand << 63 | tpos << 32 | qpos;
            local_chains[tot_chain_i--].y = (uint64_t)end_id << 32 | (uint32_t)start_id;
        }
    }
    radix_sort_u128x(local_chains + tot_chain_i + 1, local_chains + n_local_chains);
    dp_chaining_of_local_chains(local_chains + tot_chain_i + 1, n_local_chains - 1 - tot_chain_i,
                                anchors, score, pre_id, par_anchors, min_w, tlen, qlen);
    free(score); free(pre_id); free(end_pos); free(local_chains);
}

/* ---- driver: guide tree + per-pair partition (abpoa_seed.c:716-756) ---- */
int abamd_build_guide_tree_partition(uint8_t **seqs, int *seq_lens, int n_seq,
                                     abpoa_para_t *abpt, int *read_id_map,
                                     u64v_t *par_anchors, int *par_c) {
    int i;
    for (i = 0; i < n_seq; ++i) read_id_map[i] = i;
    u128v_t mm1; memset(&mm1, 0, sizeof(mm1));
    int *mm_c = (int*)abamd_malloc((size_t)(n_seq + 1) * sizeof(int));
    collect_mm(seqs, seq_lens, n_seq, abpt, &mm1, mm_c);

    if (abpt->progressive_poa && n_seq > 2) {
        u128v_t mm2; memset(&mm2, 0, sizeof(mm2));
        for (i = 0; i < (int)mm1.n; ++i) u128v_push(&mm2, mm1.a[i]);
        build_guide_tree(abpt, n_seq, &mm2, read_id_map);
        free(mm2.a);
    }
    if (abpt->disable_seeding || n_seq < 2) {
        free(mm1.a); free(mm_c);
        return 0;
    }
    int qid, tid;
    tid = read_id_map[0];
    radix_sort_u128x(mm1.a + mm_c[tid], mm1.a + mm_c[tid + 1]);
    par_c[0] = 0;
    for (i = 1; i < n_seq; ++i) {
        tid = read_id_map[i - 1]; qid = read_id_map[i];
        u64v_t anchors; memset(&anchors, 0, sizeof(anchors));
        collect_anchors1(&anchors, mm1, mm_c, tid, qid, seq_lens[qid], abpt->k);
        dp_chaining(&anchors, par_anchors, abpt, seq_lens[tid], seq_lens[qid]);
        par_c[i] = (int)par_anchors->n;
        free(anchors.a);
    }
    free(mm1.a); free(mm_c);
    return 0;
}

/* ---- anchor-windowed POA (abpoa_anchor_poa, abpoa_align.c:209-310) ---- */

static abpoa_cigar_t *seed_push_cigar(int *n_cigar, int *m_cigar, abpoa_cigar_t *cigar,
                                      int op, int len, int32_t node_id, int32_t query_id) {
    abpoa_cigar_t l = (abpoa_cigar_t)len;
    if (*n_cigar == 0 || (op != ABPOA_CINS && op != ABPOA_CSOFT_CLIP && op != ABPOA_CHARD_CLIP)
        || op != (int)(cigar[(*n_cigar) - 1] & 0xf)) {
        if (*n_cigar == *m_cigar) {
            *m_cigar = *m_cigar ? (*m_cigar) << 1 : 4;
            cigar = (abpoa_cigar_t*)abamd_realloc(cigar, (size_t)(*m_cigar) * sizeof(abpoa_cigar_t));
        }
        abpoa_cigar_t n_id = (abpoa_cigar_t)node_id, q_id = (abpoa_cigar_t)query_id;
        if (op == ABPOA_CMATCH || op == ABPOA_CDIFF)
            cigar[(*n_cigar)++] = n_id << 34 | q_id << 4 | op;
        else if (op == ABPOA_CINS || op == ABPOA_CSOFT_CLIP || op == ABPOA_CHARD_CLIP)
            cigar[(*n_cigar)++] = q_id << 34 | l << 4 | op;
        else if (op == ABPOA_CDEL)
            cigar[(*n_cigar)++] = n_id << 34 | l << 4 | op;
        else abamd_fatal("seed_push_cigar", "unknown cigar op %d", op);
    } else cigar[(*n_cigar) - 1] += l << 4;
    return cigar;
}

static void push_whole_cigar(int *dn, int *dm, abpoa_cigar_t **dc, int sn, abpoa_cigar_t *sc) {
    int i, dest_n = *dn;
    *dn += sn;
    if (*dn > *dm) {
        *dm = AB_MAX2((*dm) << 1, *dn);
        *dc = (abpoa_cigar_t*)abamd_realloc(*dc, (size_t)(*dm) * sizeof(abpoa_cigar_t));
    }
    for (i = 0; i < sn; ++i) (*dc)[dest_n + i] = sc[i];
}

int abamd_anchor_poa(abpoa_t *ab, abpoa_para_t *abpt, uint8_t **seqs, int **weights,
                     int *seq_lens, u64v_t par_anchors, int *par_c,
                     int *tpos_to_node_id, int *qpos_to_node_id, int *read_id_map,
                     int exist_n_seq, int n_seq) {
    abpoa_res_t res;
    int read_id, last_read_id = -1, m_c = 0, k = abpt->k, qlen;
    abpoa_seq_t *abs = ab->abs;
    int *tmp;
    int i, _i, ai, j, tot_n_seq = exist_n_seq + n_seq;
    uint8_t *qseq; int *weight; abpoa_res_t whole_res;
    for (_i = 0; _i < n_seq; ++_i) {
        i = read_id_map[_i]; read_id = exist_n_seq + i; qlen = seq_lens[i];
        whole_res.n_cigar = 0, whole_res.m_cigar = 0, whole_res.graph_cigar = 0;
        if (_i == 0) ai = 0; else ai = par_c[_i - 1];

        int beg_id = ABPOA_SRC_NODE_ID, beg_qpos = 0, end_id = -1, end_tpos = -1, end_qpos = -1;
        if (ai < par_c[_i]) {
            abs->is_rc[read_id] = (uint8_t)(abs->is_rc[last_read_id] ^ (par_anchors.a[ai] >> 63));
            if (abs->is_rc[read_id]) {
                qseq = (uint8_t*)abamd_malloc((size_t)qlen);
                weight = (int*)abamd_malloc((size_t)qlen * sizeof(int));
                for (j = 0; j < qlen; ++j) {
                    if (seqs[i][qlen - j - 1] < 4) qseq[j] = (uint8_t)(3 - seqs[i][qlen - j - 1]);
                    else qseq[j] = 4;
                    weight[j] = weights[i][qlen - j - 1];
                }
            } else {
                qseq = seqs[i];
                weight = weights[i];
            }
            if (abs->is_rc[last_read_id]) { /* rewrite anchors into the flipped frame */
                int last_qlen = seq_lens[read_id_map[_i - 1]];
                for (j = ai; j < par_c[_i]; ++j) {
                    end_tpos = (int)((par_anchors.a[j] >> 32) & 0x7fffffff); end_qpos = (int32_t)par_anchors.a[j];
                    par_anchors.a[j] = (par_anchors.a[j] >> 63) << 63
                        | (uint64_t)(last_qlen - end_tpos + k) << 32 | (uint64_t)(uint32_t)(qlen - end_qpos + k);
                }
                for (j = 0; j < (par_c[_i] - ai) / 2; ++j) {
                    uint64_t t = par_anchors.a[ai + j];
                    par_anchors.a[ai + j] = par_anchors.a[par_c[_i] - 1 - j];
                    par_anchors.a[par_c[_i] - 1 - j] = t;
                }
            }
        } else {
            abs->is_rc[read_id] = 0, qseq = seqs[i]; weight = weights[i];
        }

        for (; ai < par_c[_i]; ++ai) {
            end_tpos = (int)((par_anchors.a[ai] >> 32) & 0x7fffffff) - k + 1; end_id = tpos_to_node_id[end_tpos];
            end_qpos = (int32_t)par_anchors.a[ai] - k + 1;

            res.graph_cigar = 0; res.n_cigar = 0;
            abpoa_align_sequence_to_subgraph(ab, abpt, beg_id, end_id, qseq + beg_qpos, end_qpos - beg_qpos, &res);
            push_whole_cigar(&whole_res.n_cigar, &whole_res.m_cigar, &whole_res.graph_cigar, res.n_cigar, res.graph_cigar);
            if (res.n_cigar) free(res.graph_cigar);

            /* the anchor k-mer becomes an exact match run */
            res.graph_cigar = (abpoa_cigar_t*)abamd_malloc((size_t)k * sizeof(abpoa_cigar_t)); res.n_cigar = 0; m_c = k;
            for (j = 0; j < k; ++j)
                res.graph_cigar = seed_push_cigar(&res.n_cigar, &m_c, res.graph_cigar, ABPOA_CMATCH, 1, tpos_to_node_id[end_tpos + j], j);
            push_whole_cigar(&whole_res.n_cigar, &whole_res.m_cigar, &whole_res.graph_cigar, res.n_cigar, res.graph_cigar);
            if (res.n_cigar) free(res.graph_cigar);

            beg_id = tpos_to_node_id[end_tpos + k - 1]; beg_qpos = end_qpos + k;
        }
        end_id = ABPOA_SINK_NODE_ID; end_qpos = seq_lens[i];

        res.graph_cigar = 0; res.n_cigar = 0;
        abpoa_align_sequence_to_subgraph(ab, abpt, beg_id, end_id, qseq + beg_qpos, end_qpos - beg_qpos, &res);
        push_whole_cigar(&whole_res.n_cigar, &whole_res.m_cigar, &whole_res.graph_cigar, res.n_cigar, res.graph_cigar);
        if (res.n_cigar) free(res.graph_cigar);

        abpoa_add_subgraph_alignment(ab, abpt, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID, qseq, weight, qlen,
                                     qpos_to_node_id, whole_res, read_id, tot_n_seq, 1);
        if (abs->is_rc[read_id]) { free(qseq); free(weight); }
        if (whole_res.n_cigar) free(whole_res.graph_cigar);

        tmp = qpos_to_node_id; qpos_to_node_id = tpos_to_node_id; tpos_to_node_id = tmp;
        last_read_id = read_id;
    }
    return 0;
}
