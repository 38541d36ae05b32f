/* Host side of the CDNA4 aligner core.
 *
 * Packs one alignment job (compact reachable rows of the topo-sorted graph +
 * query + parameters) into a device slab, launches the DP+backtrack kernel
 * (gpu_kernels.hip), and unpacks abpoa_res_t. Fails loudly when no AMD GPU is
 * usable — there is no CPU fallback on this path.
 *
 * Width selection and inf_min mirror the reference dispatch
 * (abpoa_align_simd.c:1284-1302): int16 unless the score bound overflows,
 * then int32. Arena overflow (adaptive band wider than the reserved slab)
 * retries with a doubled arena.
 *
 * Batching: abamd_gpu_align_batch() packs many independent jobs into one
 * launch so the 256-CU chip sees thousands of wavefronts; the single-job seam
 * entry is a batch of one.
 */
#include <hip/hip_runtime.h>
#include <atomic>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>
#include <cmath>
#include <ctime>
#include <thread>
#include "abpoa_amd.h"
#include "gpu_core.h"
#include "abamd_util.h"

#define HIP_CHECK(x) do { hipError_t _e = (x); if (_e != hipSuccess) { \
    fprintf(stderr, "[abpoa_amd] HIP error %s at %s:%d: %s\n", hipGetErrorName(_e), __FILE__, __LINE__, hipGetErrorString(_e)); \
    exit(EXIT_FAILURE); } } while (0)

static std::atomic<uint64_t> g_dp_cells{0}, g_kernel_ns{0}, g_launches{0}, g_alg_bytes{0};
static std::atomic<uint64_t> g_pack_ns{0}, g_stage_ns{0}, g_gpu_ns{0}, g_unpack_ns{0};
static std::atomic<uint64_t> g_retry_jobs{0};
static std::atomic<uint64_t> g_dev_held{0};   /* bytes held by our DevBufs */
/* per-slot arena growth clamp, set by the batch driver to its per-launch
 * memory budget so two resident arenas can never overshoot free HBM */
static std::atomic<uint64_t> g_arena_cap{0};
extern "C" void abamd_gpu_set_arena_cap(uint64_t bytes) { g_arena_cap = bytes; }
static inline uint64_t now_ns() {
    struct timespec ts; clock_gettime(CLOCK_MONOTONIC, &ts);
    return (uint64_t)ts.tv_sec * 1000000000ull + ts.tv_nsec;
}
extern "C" void abamd_timing_report(const char *tag) {
    if (!getenv("ABPOA_AMD_TIMING")) return;
    fprintf(stderr, "[abamd timing %s] pack %.2fs stage %.2fs gpu(upload->sync) %.2fs unpack %.2fs kernel %.2fs launches %llu retry_jobs %llu\n",
            tag, g_pack_ns/1e9, g_stage_ns/1e9, g_gpu_ns/1e9, g_unpack_ns/1e9, g_kernel_ns/1e9,
            (unsigned long long)g_launches.load(), (unsigned long long)g_retry_jobs.load());
}

extern "C" void abpoa_amd_get_stats(uint64_t *dp_cells, uint64_t *kernel_ns, uint64_t *n_launches) {
    if (dp_cells) *dp_cells = g_dp_cells.load();
    if (kernel_ns) *kernel_ns = g_kernel_ns.load();
    if (n_launches) *n_launches = g_launches.load();
}
extern "C" void abpoa_amd_get_stats2(uint64_t *alg_bytes) {
    if (alg_bytes) *alg_bytes = g_alg_bytes.load();
}
extern "C" void abpoa_amd_reset_stats(void) {
    g_dp_cells = 0; g_kernel_ns = 0; g_launches = 0; g_alg_bytes = 0;
    g_pack_ns = 0; g_stage_ns = 0; g_gpu_ns = 0; g_unpack_ns = 0;
}

/* shared-counter hooks for the device-resident batch driver
 * (gpu_batch_resident.cpp), so abpoa_amd_get_stats covers both paths */
extern "C" void abamd_stats_add_cells(uint64_t cells, uint64_t alg_bytes) {
    g_dp_cells += cells; g_alg_bytes += alg_bytes;
}
extern "C" void abamd_stats_add_kernel(uint64_t ns) {
    g_kernel_ns += ns; g_launches += 1;
}

/* ---------------- device context (per thread) ---------------- */

namespace {

struct DevBuf {
    void *p = nullptr;
    size_t cap = 0;
    /* grow with headroom, never shrink: each regrow is a hipFree+hipMalloc
     * (hipFree synchronizes the WHOLE device, stalling the other slot's
     * in-flight kernel) and mapping tens of GB costs ~1 s, so regrows must
     * stay rare. The arena's demand grows monotonically with the graphs
     * (~3x first-to-last round), so x2 headroom caps regrows at ~2-3 per
     * slot; power-of-2 rounding on a ~100 GB arena would overshoot the
     * device, exact x2 of demand does not. */
    void ensure(size_t n, int big = 0, size_t limit = 0) {
        if (n <= cap) return;
        size_t want = big ? n * 2 : n + n / 2;
        if (big && limit) {
            /* budget-capped arena: overshoot aggressively toward the cap so
             * the slot allocates once and never regrows mid-pipeline */
            want = n * 8;
            if (want > limit) want = limit;
            if (want < n) want = n;
        } else if (limit && want > limit) {
            want = n > limit ? n : limit;
        }
        if (want < 4096) want = 4096;
        if (p) HIP_CHECK(hipFree(p));
        HIP_CHECK(hipMalloc(&p, want));
        g_dev_held += want - cap;
        cap = want;
    }
};

struct HostBuf {
    std::vector<uint8_t> v;
    size_t used = 0;
    void reset() { used = 0; }
    /* reserve n bytes aligned to 16 and return the offset */
    size_t alloc(size_t n) {
        size_t off = (used + 15) & ~(size_t)15;
        used = off + n;
        if (used > v.size()) v.resize(used * 2 + 4096);
        return off;
    }
    uint8_t *at(size_t off) { return v.data() + off; }
};

struct PinnedBuf {
    uint8_t *p = nullptr;
    size_t cap = 0;
    void ensure(size_t n) {
        if (n <= cap) return;
        size_t want = cap ? cap : 1 << 20;
        while (want < n) want <<= 1;
        if (p) HIP_CHECK(hipHostFree(p));
        HIP_CHECK(hipHostMalloc((void**)&p, want));
        cap = want;
    }
};

/* Per-slot device context: slots 0/1 are the two pipeline groups, slot 2 the
 * overflow-retry sub-batch. Each slot owns its stream and device buffers so
 * the two groups' kernels can be IN FLIGHT CONCURRENTLY (the per-row latency
 * chain leaves most CUs idle at ~500 waves/launch; two overlapped launches
 * roughly double chip fill). Streams are non-blocking so the null-stream
 * hipMemcpy in unpack never serializes against the other slot's kernel. */
struct SlotDev {
    bool init = false;
    hipStream_t stream;
    hipEvent_t ev0, ev1, ev_h2d;
    DevBuf slab;        /* all per-job inputs + metadata */
    DevBuf arena;       /* DP plane arena */
    DevBuf jobs, results, cigars;
    void ensure_init() {
        if (init) return;
        HIP_CHECK(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
        HIP_CHECK(hipEventCreate(&ev0));
        HIP_CHECK(hipEventCreate(&ev1));
        HIP_CHECK(hipEventCreate(&ev_h2d));
        init = true;
    }
};

struct GpuCtx {
    bool init = false;
    SlotDev dev[8];
    PinnedBuf stage2[8];              /* per-slot pinned H2D staging (slot 7 = retry/big) */
    std::vector<HostBuf> jb_bufs2[8]; /* per-slot per-job pack buffers */
    HostBuf hb;
    std::vector<uint64_t> hcig;
    void ensure_init() {
        if (init) return;
        int n = 0;
        hipError_t e = hipGetDeviceCount(&n);
        if (e != hipSuccess || n == 0) {
            fprintf(stderr, "[abpoa_amd] FATAL: no usable AMD GPU (hipGetDeviceCount: %s). "
                            "The abpoa_amd aligner is GPU-only; there is no CPU fallback.\n",
                    hipGetErrorString(e));
            exit(EXIT_FAILURE);
        }
        init = true;
    }
};

thread_local GpuCtx g_ctx;

/* host-side packed job (offsets into the slab) */
struct JobPack {
    abamd_gpu_job_t jb;     /* device pointers filled after upload */
    size_t o_query, o_base, o_nodeid, o_preoff, o_preidx, o_preps, o_outoff, o_outidx, o_remain;
    size_t o_ml, o_mr, o_meta;
    int64_t arena_off;      /* cells */
    size_t slab_base;       /* this job's base offset in the device slab */
    int64_t cigar_off;      /* entries */
    int n_rows, qlen;
    int n_pre, n_out;
};

/* Build the compact-row job into its own host buffer; returns per-job arena
 * demand (cells). Thread-safe: touches only hb and P. */
static int64_t pack_job(HostBuf &hb, JobPack &P, abpoa_t *ab, abpoa_para_t *abpt,
                        int beg_node_id, int end_node_id, uint8_t *query, int qlen) {
    abpoa_graph_t *g = ab->abg;
    int beg_index = g->node_id_to_index[beg_node_id];
    int end_index = g->node_id_to_index[end_node_id];
    int span = end_index - beg_index + 1;

    /* reachability map (abpoa_align_simd.c:1259-1269); thread-local scratch
     * (pack runs on a 64-thread pool: per-call allocation serializes on the
     * allocator) */
    static thread_local std::vector<uint8_t> imap_s;
    static thread_local std::vector<int> idx2row_s;
    imap_s.assign(g->node_n, 0);
    std::vector<uint8_t> &imap = imap_s;
    imap[beg_index] = imap[end_index] = 1;
    for (int i = beg_index; i < end_index - 1; ++i) {
        if (!imap[i]) continue;
        int nid = g->index_to_node_id[i];
        for (int j = 0; j < g->node[nid].out_edge_n; ++j)
            imap[g->node_id_to_index[g->node[nid].out_id[j]]] = 1;
    }
    /* compact row numbering */
    idx2row_s.assign(span, -1);
    std::vector<int> &idx2row = idx2row_s;
    int n_rows = 0;
    for (int i = 0; i < span; ++i)
        if (imap[beg_index + i]) idx2row[i] = n_rows++;

    P.n_rows = n_rows; P.qlen = qlen;

    size_t o_query = hb.alloc(qlen);
    size_t o_base = hb.alloc(n_rows);
    size_t o_nodeid = hb.alloc((size_t)n_rows * 4);
    size_t o_preoff = hb.alloc((size_t)(n_rows + 1) * 4);
    size_t o_outoff = hb.alloc((size_t)(n_rows + 1) * 4);
    size_t o_remain = hb.alloc((size_t)n_rows * 4);
    memcpy(hb.at(o_query), query, qlen);
    uint8_t *base = hb.at(o_base);
    int *nodeid = (int*)hb.at(o_nodeid);
    int *preoff = (int*)hb.at(o_preoff);
    int *outoff = (int*)hb.at(o_outoff);
    int *remain = (int*)hb.at(o_remain);

    int n_pre = 0, n_out = 0;
    for (int i = 0, r = 0; i < span; ++i) {
        if (idx2row[i] < 0) continue;
        int nid = g->index_to_node_id[beg_index + i];
        base[r] = g->node[nid].base;
        nodeid[r] = nid;
        remain[r] = (abpt->wb >= 0 || abpt->zdrop > 0) ? g->node_id_to_max_remain[nid] : 0;
        preoff[r] = n_pre; outoff[r] = n_out;
        if (r > 0) {
            for (int j = 0; j < g->node[nid].in_edge_n; ++j) {
                int pidx = g->node_id_to_index[g->node[nid].in_id[j]];
                if (pidx >= beg_index && pidx <= end_index && imap[pidx] && idx2row[pidx - beg_index] >= 0)
                    ++n_pre;
            }
        }
        for (int j = 0; j < g->node[nid].out_edge_n; ++j) {
            int oidx = g->node_id_to_index[g->node[nid].out_id[j]];
            if (oidx >= beg_index && oidx <= end_index && imap[oidx] && idx2row[oidx - beg_index] >= 0)
                ++n_out;
        }
        ++r;
    }
    preoff[n_rows] = n_pre; outoff[n_rows] = n_out;

    size_t o_preidx = hb.alloc((size_t)(n_pre > 0 ? n_pre : 1) * 4);
    size_t o_preps = hb.alloc((size_t)(n_pre > 0 ? n_pre : 1) * 4);
    size_t o_outidx = hb.alloc((size_t)(n_out > 0 ? n_out : 1) * 4);
    int *preidx = (int*)hb.at(o_preidx);
    int *preps = (int*)hb.at(o_preps);
    int *outidx = (int*)hb.at(o_outidx);
    /* re-read offsets (hb.alloc may have resized the vector) */
    base = hb.at(o_base); nodeid = (int*)hb.at(o_nodeid);
    preoff = (int*)hb.at(o_preoff); outoff = (int*)hb.at(o_outoff);
    remain = (int*)hb.at(o_remain);

    n_pre = 0; n_out = 0;
    for (int i = 0, r = 0; i < span; ++i) {
        if (idx2row[i] < 0) continue;
        int nid = g->index_to_node_id[beg_index + i];
        if (r > 0) {
            for (int j = 0; j < g->node[nid].in_edge_n; ++j) {
                int pidx = g->node_id_to_index[g->node[nid].in_id[j]];
                if (pidx >= beg_index && pidx <= end_index && imap[pidx] && idx2row[pidx - beg_index] >= 0) {
                    preidx[n_pre] = idx2row[pidx - beg_index];
                    preps[n_pre] = 0; /* inc_path_score handled below */
                    if (abpt->inc_path_score) {
                        /* abpoa_get_incre_path_score (abpoa_graph.c:429-437) */
                        int pre_id = g->node[nid].in_id[j], node_w = 0;
                        for (int t = 0; t < g->node[pre_id].out_edge_n; ++t) node_w += g->node[pre_id].out_edge_weight[t];
                        int edge_w = g->node[nid].in_edge_weight[j];
                        int sc = 0;
                        if (node_w != 0 && edge_w != 0) {
                            double r2 = (double)edge_w / (double)node_w;
                            sc = (int)lround(log(r2));
                            if (sc < -20) sc = -20;
                        }
                        preps[n_pre] = sc;
                    }
                    ++n_pre;
                }
            }
        }
        for (int j = 0; j < g->node[nid].out_edge_n; ++j) {
            int oidx = g->node_id_to_index[g->node[nid].out_id[j]];
            if (oidx >= beg_index && oidx <= end_index && imap[oidx] && idx2row[oidx - beg_index] >= 0)
                outidx[n_out++] = idx2row[oidx - beg_index];
        }
        ++r;
    }

    /* working arrays (device-initialized) */
    size_t o_ml = hb.alloc((size_t)n_rows * 4);
    size_t o_mr = hb.alloc((size_t)n_rows * 4);
    size_t o_meta = hb.alloc((size_t)n_rows * sizeof(abamd_row_meta_t));

    P.o_query = o_query; P.o_base = o_base; P.o_nodeid = o_nodeid;
    P.o_preoff = o_preoff; P.o_preidx = o_preidx; P.o_preps = o_preps;
    P.o_outoff = o_outoff; P.o_outidx = o_outidx; P.o_remain = o_remain;
    P.o_ml = o_ml; P.o_mr = o_mr; P.o_meta = o_meta;
    P.n_pre = n_pre; P.n_out = n_out;

    abamd_gpu_job_t &jb = P.jb;
    memset(&jb, 0, sizeof(jb));
    jb.n_rows = n_rows; jb.qlen = qlen; jb.m = abpt->m;
    jb.w = abpt->wb < 0 ? qlen : abpt->wb + (int)(abpt->wf * qlen);
    jb.banded = abpt->wb >= 0;
    jb.o1 = abpt->gap_open1; jb.e1 = abpt->gap_ext1;
    jb.o2 = abpt->gap_open2; jb.e2 = abpt->gap_ext2;
    jb.oe1 = abpt->gap_open1 + abpt->gap_ext1;
    jb.oe2 = abpt->gap_open2 + abpt->gap_ext2;
    jb.align_mode = abpt->align_mode;
    jb.put_gap_on_right = abpt->put_gap_on_right;
    jb.put_gap_at_end = abpt->put_gap_at_end;
    jb.zdrop = abpt->zdrop;
    jb.inc_path_score = abpt->inc_path_score;
    jb.node_n_init = g->node_n;
    jb.ret_cigar = abpt->ret_cigar;

    /* arena demand estimate: adaptive bands are ~2w + drift; reserve slack */
    int64_t est = (int64_t)n_rows * (2 * (int64_t)jb.w + 160) + qlen + 64;
    return est;
}

struct BatchJob {
    abpoa_t *ab;
    abpoa_para_t *abpt;
    int beg_node_id, end_node_id;
    uint8_t *query;
    int qlen;
    abpoa_res_t *res;
    int64_t est_cells_hint;  /* measured cells of this set's previous round
                                (0 = none): tightens the arena reservation */
    int64_t *cells_out;      /* optional: actual banded cells written back */
};

/* score-width pick (abpoa_align_simd.c:1284-1302); assumes uniform paras */
static void pick_width(abpoa_para_t *abpt, int qlen, int gn, int *bits, int *inf_min) {
    int32_t gap_oe1 = abpt->gap_open1 + abpt->gap_ext1;
    int32_t gap_oe2 = abpt->gap_open2 + abpt->gap_ext2;
    int len = qlen > gn ? qlen : gn;
    int32_t max_score = (int32_t)qlen * abpt->max_mat;
    int32_t alt = (int32_t)len * abpt->gap_ext1 + abpt->gap_open1;
    if (alt > max_score) max_score = alt;
    int32_t ext_max = abpt->gap_ext1 > abpt->gap_ext2 ? abpt->gap_ext1 : abpt->gap_ext2;
    if (max_score <= INT16_MAX - abpt->min_mis - gap_oe1 - gap_oe2) {
        int32_t im = INT16_MIN + abpt->min_mis;
        if (INT16_MIN + gap_oe1 > im) im = INT16_MIN + gap_oe1;
        if (INT16_MIN + gap_oe2 > im) im = INT16_MIN + gap_oe2;
        *inf_min = im + 512 * ext_max; *bits = 16;
    } else {
        int32_t im = INT32_MIN + abpt->min_mis;
        if (INT32_MIN + gap_oe1 > im) im = INT32_MIN + gap_oe1;
        if (INT32_MIN + gap_oe2 > im) im = INT32_MIN + gap_oe2;
        *inf_min = im + 512 * ext_max; *bits = 32;
    }
}

} // namespace

/* score-width pick, shared with the device-resident batch driver */
extern "C" void abamd_pick_width(abpoa_para_t *abpt, int qlen, int gn, int *bits, int *inf_min) {
    pick_width(abpt, qlen, gn, bits, inf_min);
}

/* Pipelined batch interface: begin() packs, uploads and launches without
 * waiting; finish() synchronizes and unpacks. The caller overlaps host-side
 * graph folds of one set group with the kernel of the other. begin/finish
 * pairs must not nest. abamd_gpu_align_batch() = begin + finish. */
extern "C" int abamd_gpu_align_batch(BatchJob *batch, int n_jobs);
extern "C" int abamd_gpu_batch_prepare(BatchJob *batch, int n_jobs, int slot);
extern "C" int abamd_gpu_batch_launch(int slot);
extern "C" int abamd_gpu_batch_finish_slot(int slot);

namespace {
struct PendingBatch {
    BatchJob *batch = nullptr;
    int slot = 0;
    int n_jobs = 0;
    std::vector<JobPack> packs;
    std::vector<int64_t> arena_est;
    std::vector<abamd_gpu_job_t> hjobs;   /* per-slot: an async D2H/H2D may be
    in flight for one slot while the other slot repacks */
    std::vector<abamd_gpu_res_t> hres;
    int bits = 16;
    int planes = 5;
    size_t total = 0;
    bool active = false;
};
thread_local PendingBatch g_slots[8];
}

static int batch_launch(GpuCtx &C, PendingBatch &P); /* fwd */

extern "C" int abamd_gpu_align_batch_slot(BatchJob *batch, int n_jobs, int slot) {
    int r = abamd_gpu_batch_prepare(batch, n_jobs, slot);
    if (r) return r;
    r = abamd_gpu_batch_launch(slot);
    if (r) return r;
    return abamd_gpu_batch_finish_slot(slot);
}

extern "C" int abamd_gpu_align_batch(BatchJob *batch, int n_jobs) {
    return abamd_gpu_align_batch_slot(batch, n_jobs, 0);
}

static int prepare_internal(BatchJob *batch, int n_jobs, int slot, const int64_t *min_est);

extern "C" int abamd_gpu_batch_prepare(BatchJob *batch, int n_jobs, int slot) {
    return prepare_internal(batch, n_jobs, slot, nullptr);
}

/* min_est: per-job arena floor (cells) used by the overflow retry, which must
 * REPACK: the shared pinned stage may already hold the other slot's batch */
static int prepare_internal(BatchJob *batch, int n_jobs, int slot, const int64_t *min_est) {
    if (n_jobs <= 0) { g_slots[slot].active = false; g_slots[slot].n_jobs = 0; return 0; }
    GpuCtx &C = g_ctx;
    C.ensure_init();
    abpoa_para_t *abpt = batch[0].abpt;
    /* convex stores 3 planes (H,E1,E2): F recomputed at backtrack */
    const int planes = abpt->gap_mode == ABPOA_CONVEX_GAP ? 3
                     : abpt->gap_mode == ABPOA_AFFINE_GAP ? 3 : 1;

    uint64_t t_pack0 = now_ns();
    std::vector<HostBuf> &jbufs = C.jb_bufs2[slot];
    if ((int)jbufs.size() < n_jobs) jbufs.resize(n_jobs);
    std::vector<JobPack> packs(n_jobs);
    std::vector<int64_t> arena_est(n_jobs);
    std::vector<int> bits_v(n_jobs, 16);

    /* parallel pack on the persistent pool: each job into its own (reused)
     * host buffer; pool threads keep their thread-local scratch warm */
    {
        struct PackCtx {
            std::atomic<int> next{0};
            BatchJob *batch; int n_jobs;
            std::vector<HostBuf> *jbufs;
            std::vector<JobPack> *packs;
            std::vector<int64_t> *arena_est;
            std::vector<int> *bits_v;
            const int64_t *min_est;
            bool no_hint;
        } ctx;
        ctx.batch = batch; ctx.n_jobs = n_jobs; ctx.jbufs = &jbufs;
        ctx.packs = &packs; ctx.arena_est = &arena_est; ctx.bits_v = &bits_v;
        ctx.min_est = min_est; ctx.no_hint = getenv("ABPOA_AMD_NO_HINT") != nullptr;
        auto worker = [](void *p, int, int) {
            PackCtx &c = *(PackCtx*)p;
            for (;;) {
                int i = c.next.fetch_add(1);
                if (i >= c.n_jobs) break;
                BatchJob &B = c.batch[i];
                (*c.jbufs)[i].reset();
                (*c.arena_est)[i] = pack_job((*c.jbufs)[i], (*c.packs)[i], B.ab, B.abpt,
                                             B.beg_node_id, B.end_node_id, B.query, B.qlen);
                if (B.est_cells_hint > 0 && !c.no_hint) {
                    /* bands drift a few % per round; 50% headroom + overflow retry */
                    int64_t tight = B.est_cells_hint + B.est_cells_hint / 2 + B.qlen;
                    if (tight < (*c.arena_est)[i]) (*c.arena_est)[i] = tight;
                }
                if (c.min_est && c.min_est[i] > (*c.arena_est)[i]) (*c.arena_est)[i] = c.min_est[i];
                int inf_min;
                int span = B.ab->abg->node_id_to_index[B.end_node_id] - B.ab->abg->node_id_to_index[B.beg_node_id] + 1;
                pick_width(B.abpt, B.qlen, span, &(*c.bits_v)[i], &inf_min);
                (*c.packs)[i].jb.inf_min = inf_min;
                (*c.packs)[i].cigar_off = 0;
            }
        };
        int nthr = abamd_pool_size();
        if (nthr > 64) nthr = 64; /* measured: pack scales poorly past ~64 */
        if (nthr > n_jobs) nthr = n_jobs;
        abamd_pool_run(worker, &ctx, nthr);
    }
    int bits_max = 16;
    for (int i = 0; i < n_jobs; ++i) if (bits_v[i] > bits_max) bits_max = bits_v[i];
    g_pack_ns += now_ns() - t_pack0;
    uint64_t t_stage0 = now_ns();

    /* assemble the pinned staging slab: [mat][job 0][job 1]... */
    const size_t mat_bytes = (size_t)abpt->m * abpt->m * 4;
    size_t o_mat = 0;
    size_t total = (mat_bytes + 255) & ~(size_t)255;
    for (int i = 0; i < n_jobs; ++i) {
        packs[i].slab_base = total;
        total += (jbufs[i].used + 255) & ~(size_t)255;
    }
    PinnedBuf &stage = C.stage2[slot];
    stage.ensure(total);
    memcpy(stage.p + o_mat, abpt->mat, mat_bytes);
    {
        struct StageCtx {
            std::atomic<int> next{0};
            PinnedBuf *stage; int n_jobs;
            std::vector<HostBuf> *jbufs;
            std::vector<JobPack> *packs;
        } ctx;
        ctx.stage = &stage; ctx.n_jobs = n_jobs; ctx.jbufs = &jbufs; ctx.packs = &packs;
        auto worker = [](void *p, int, int) {
            StageCtx &c = *(StageCtx*)p;
            for (;;) {
                int i = c.next.fetch_add(1);
                if (i >= c.n_jobs) break;
                memcpy(c.stage->p + (*c.packs)[i].slab_base, (*c.jbufs)[i].v.data(), (*c.jbufs)[i].used);
            }
        };
        int nthr = 64; /* memcpy saturates well below core count */
        if (nthr > n_jobs) nthr = n_jobs;
        abamd_pool_run(worker, &ctx, nthr);
    }
    /* a mixed batch runs at the widest type; widths are identical across jobs
     * of one workload in practice (same scoring paras, similar qlen) */
    int bits = bits_max;
    if (bits == 32) {
        /* recompute inf_min at 32-bit for every job */
        for (int i = 0; i < n_jobs; ++i) {
            int b2, im;
            BatchJob &B = batch[i];
            int span = B.ab->abg->node_id_to_index[B.end_node_id] - B.ab->abg->node_id_to_index[B.beg_node_id] + 1;
            (void)span; (void)b2;
            int32_t gap_oe1 = abpt->gap_open1 + abpt->gap_ext1, gap_oe2 = abpt->gap_open2 + abpt->gap_ext2;
            int32_t ext_max = abpt->gap_ext1 > abpt->gap_ext2 ? abpt->gap_ext1 : abpt->gap_ext2;
            im = INT32_MIN + abpt->min_mis;
            if (INT32_MIN + gap_oe1 > im) im = INT32_MIN + gap_oe1;
            if (INT32_MIN + gap_oe2 > im) im = INT32_MIN + gap_oe2;
            packs[i].jb.inf_min = im + 512 * ext_max;
        }
    }
    PendingBatch &PB = g_slots[slot];
    PB.batch = batch;
    PB.slot = slot;
    PB.n_jobs = n_jobs;
    PB.packs = std::move(packs);
    PB.arena_est = std::move(arena_est);
    PB.bits = bits;
    PB.planes = planes;
    PB.total = total;
    PB.hjobs.resize(n_jobs);
    PB.hres.resize(n_jobs);
    PB.active = true;
    g_stage_ns += now_ns() - t_stage0;
    return 0;
}

extern "C" int abamd_gpu_batch_launch(int slot) {
    GpuCtx &C = g_ctx;
    PendingBatch &PB = g_slots[slot];
    if (!PB.active) return 0;
    uint64_t t_gpu0 = now_ns();
    int rr = batch_launch(C, PB);
    /* wait for the H2D copies only: the caller may repack the pinned staging
     * buffer for the other slot while this kernel runs */
    HIP_CHECK(hipEventSynchronize(C.dev[slot].ev_h2d));
    g_gpu_ns += now_ns() - t_gpu0;
    return rr;
}

/* pack+upload happen in begin(); this issues the H2D + kernel (async) */
static int batch_launch(GpuCtx &C, PendingBatch &PB) {
    const int n_jobs = PB.n_jobs;
    std::vector<JobPack> &packs = PB.packs;
    std::vector<int64_t> &arena_est = PB.arena_est;
    abpoa_para_t *abpt = PB.batch[0].abpt;
    const int bits = PB.bits;
    const int planes = PB.planes;
    const size_t total = PB.total;
    const size_t ssz = bits == 16 ? 2 : 4;
    SlotDev &D = C.dev[PB.slot];
    D.ensure_init();
    (void)abpt;
    {
        /* arena layout */
        int64_t arena_cells = 0;
        for (int i = 0; i < n_jobs; ++i) {
            packs[i].arena_off = arena_cells;
            arena_cells += arena_est[i];
        }
        D.arena.ensure((size_t)arena_cells * planes * ssz, 1, g_arena_cap.load());

        /* cigar buffers. Deterministic bound on backtrack entries: every
         * M/I step consumes one query base (<= qlen total), every D step
         * strictly decreases the topo row (<= n_rows total), plus the
         * initial/terminal soft pushes — so qlen + n_rows + margin can
         * never overflow (graphs with far more rows than qlen included,
         * e.g. incremental restore + short reads). */
        int64_t cig_total = 0;
        for (int i = 0; i < n_jobs; ++i) {
            packs[i].cigar_off = cig_total;
            cig_total += (int64_t)packs[i].qlen + packs[i].n_rows + 64;
        }
        D.cigars.ensure((size_t)cig_total * 8);

        /* upload slab + jobs */
        D.slab.ensure(total);
        HIP_CHECK(hipMemcpyAsync(D.slab.p, C.stage2[PB.slot].p, total, hipMemcpyHostToDevice, D.stream));
        uint8_t *S0 = (uint8_t*)D.slab.p;
        for (int i = 0; i < n_jobs; ++i) {
            JobPack &P = packs[i];
            abamd_gpu_job_t &jb = P.jb;
            uint8_t *S = S0 + P.slab_base;
            jb.query = S + P.o_query;
            jb.row_base = S + P.o_base;
            jb.row_node_id = (int*)(S + P.o_nodeid);
            jb.pre_off = (int*)(S + P.o_preoff);
            jb.pre_idx = (int*)(S + P.o_preidx);
            jb.pre_ps = (int*)(S + P.o_preps);
            jb.out_off = (int*)(S + P.o_outoff);
            jb.out_idx = (int*)(S + P.o_outidx);
            jb.max_remain = (int*)(S + P.o_remain);
            jb.max_left = (int*)(S + P.o_ml);
            jb.max_right = (int*)(S + P.o_mr);
            jb.row_meta = S + P.o_meta;
            jb.mat = (int*)(S0 + 0); /* matrix is at slab offset 0 */
            jb.arena = (uint8_t*)D.arena.p + (size_t)P.arena_off * planes * ssz;
            jb.arena_cap = arena_est[i];
            jb.cigar = (uint64_t*)D.cigars.p + P.cigar_off;
            jb.cigar_cap = P.qlen + P.n_rows + 64;
            PB.hjobs[i] = jb;
        }
        D.jobs.ensure((size_t)n_jobs * sizeof(abamd_gpu_job_t));
        D.results.ensure((size_t)n_jobs * sizeof(abamd_gpu_res_t));
        HIP_CHECK(hipMemcpyAsync(D.jobs.p, PB.hjobs.data(), (size_t)n_jobs * sizeof(abamd_gpu_job_t), hipMemcpyHostToDevice, D.stream));
        HIP_CHECK(hipEventRecord(D.ev_h2d, D.stream));

        HIP_CHECK(hipEventRecord(D.ev0, D.stream));
        {
            abamd_gpu_job_t *J = (abamd_gpu_job_t*)D.jobs.p;
            abamd_gpu_res_t *R = (abamd_gpu_res_t*)D.results.p;
            if (abpt->gap_mode == ABPOA_CONVEX_GAP)
                bits == 16 ? abamd_launch_cg_i16(J, R, n_jobs, D.stream) : abamd_launch_cg_i32(J, R, n_jobs, D.stream);
            else if (abpt->gap_mode == ABPOA_AFFINE_GAP)
                bits == 16 ? abamd_launch_ag_i16(J, R, n_jobs, D.stream) : abamd_launch_ag_i32(J, R, n_jobs, D.stream);
            else
                bits == 16 ? abamd_launch_lg_i16(J, R, n_jobs, D.stream) : abamd_launch_lg_i32(J, R, n_jobs, D.stream);
        }
        HIP_CHECK(hipGetLastError());
        HIP_CHECK(hipEventRecord(D.ev1, D.stream));
        /* results D2H happens in finish(): an async copy into pageable host
         * memory would silently synchronize and stall the pipeline */
    }
    return 0;
}

extern "C" int abamd_gpu_batch_finish_slot(int slot) {
    GpuCtx &C = g_ctx;
    PendingBatch &PB = g_slots[slot];
    if (!PB.active) return 0;
    const int n_jobs = PB.n_jobs;
    BatchJob *batch = PB.batch;
    std::vector<JobPack> &packs = PB.packs;
    const size_t ssz = PB.bits == 16 ? 2 : 4;
    const int planes = PB.planes;
    SlotDev &D = C.dev[slot];
    uint64_t t_gpu0 = now_ns();
    HIP_CHECK(hipStreamSynchronize(D.stream));
    HIP_CHECK(hipMemcpy(PB.hres.data(), D.results.p, (size_t)n_jobs * sizeof(abamd_gpu_res_t), hipMemcpyDeviceToHost));
    {
        float ms = 0.f;
        HIP_CHECK(hipEventElapsedTime(&ms, D.ev0, D.ev1));
        g_kernel_ns += (uint64_t)(ms * 1e6);
        g_launches += 1;
    }
    g_gpu_ns += now_ns() - t_gpu0;
    if (slot == 7) {
        /* retry slot itself overflowed: escalate reservations in place */
        for (int attempt = 0;; ++attempt) {
            bool overflow = false;
            std::vector<int64_t> floor_est(n_jobs, 0);
            for (int i = 0; i < n_jobs; ++i) {
                floor_est[i] = PB.arena_est[i];
                if (PB.hres[i].status == ABAMD_JOB_ARENA_OVERFLOW) { floor_est[i] *= 2; overflow = true; }
            }
            if (!overflow) break;
            if (attempt > 8) {
                fprintf(stderr, "[abpoa_amd] arena overflow persists after %d retries\n", attempt);
                exit(EXIT_FAILURE);
            }
            prepare_internal(batch, n_jobs, 7, floor_est.data());
            batch_launch(C, PB);
            HIP_CHECK(hipStreamSynchronize(D.stream));
            HIP_CHECK(hipMemcpy(PB.hres.data(), D.results.p, (size_t)n_jobs * sizeof(abamd_gpu_res_t), hipMemcpyDeviceToHost));
            float ms = 0.f;
            HIP_CHECK(hipEventElapsedTime(&ms, D.ev0, D.ev1));
            g_kernel_ns += (uint64_t)(ms * 1e6);
            g_launches += 1;
        }
    }
    uint64_t t_unpack0 = now_ns();
    /* unpack results */
    C.hcig.clear();
    for (int i = 0; i < n_jobs; ++i) {
        abamd_gpu_res_t &R = PB.hres[i];
        if (slot != 7 && R.status == ABAMD_JOB_ARENA_OVERFLOW)
            continue; /* re-run and unpacked by the retry sub-batch */
        if (R.status != ABAMD_JOB_OK) {
            fprintf(stderr, "[abpoa_amd] GPU job %d failed with status %d\n", i, R.status);
            exit(EXIT_FAILURE);
        }
        g_dp_cells += (uint64_t)R.cells;
        g_alg_bytes += (uint64_t)R.cells * planes * ssz;
        if (batch[i].cells_out) *batch[i].cells_out = R.cells;
        abpoa_res_t *res = batch[i].res;
        res->best_score = R.best_score;
        if (batch[i].abpt->ret_cigar && R.n_cigar > 0) {
            uint64_t *tmp = (uint64_t*)malloc((size_t)R.n_cigar * 8);
            HIP_CHECK(hipMemcpy(tmp, (uint64_t*)D.cigars.p + packs[i].cigar_off, (size_t)R.n_cigar * 8, hipMemcpyDeviceToHost));
            if (!batch[i].abpt->rev_cigar) { /* reverse to front-to-back order */
                for (int a = 0, b = R.n_cigar - 1; a < b; ++a, --b) {
                    uint64_t t = tmp[a]; tmp[a] = tmp[b]; tmp[b] = t;
                }
            }
            res->graph_cigar = tmp;
            res->n_cigar = R.n_cigar; res->m_cigar = R.n_cigar;
        } else { res->n_cigar = 0; res->graph_cigar = nullptr; }
        res->n_aln_bases += R.n_aln_bases;
        res->n_matched_bases += R.n_matched_bases;
        res->node_s = R.node_s; res->node_e = R.node_e;
        res->query_s = R.query_s; res->query_e = R.query_e;
    }
    g_unpack_ns += now_ns() - t_unpack0;

    /* rare: a job's adaptive band outgrew its reservation. Re-run ONLY the
     * overflowed jobs as a fresh sub-batch on the dedicated retry slot
     * (slot 2) with doubled reservations — a whole-batch relaunch costs a
     * full latency-bound kernel pass. The sub-batch repacks from the graphs
     * (the pinned stage may already hold another slot's data). */
    if (slot != 7) {
        std::vector<BatchJob> failed;
        std::vector<int64_t> floors;
        for (int i = 0; i < n_jobs; ++i) {
            if (PB.hres[i].status == ABAMD_JOB_ARENA_OVERFLOW) {
                BatchJob fj = batch[i];
                fj.est_cells_hint = 0; /* use the formula floor below instead */
                failed.push_back(fj);
                floors.push_back(PB.arena_est[i] * 2);
            }
        }
        if (!failed.empty()) {
            g_retry_jobs += failed.size();
            prepare_internal(failed.data(), (int)failed.size(), 7, floors.data());
            int r2 = abamd_gpu_batch_launch(7);
            if (!r2) abamd_gpu_batch_finish_slot(7);
        }
    }
    PB.active = false;
    return 0;
}

/* free HBM PLUS what our own never-shrink buffers already hold: the batch
 * driver's per-launch budget must not ratchet down as the arenas it sized on
 * a previous call stay resident (they are reused, not re-allocated) */
extern "C" int64_t abamd_gpu_free_mem(void) {
    size_t free_b = 0, total_b = 0;
    if (hipMemGetInfo(&free_b, &total_b) != hipSuccess) return 0;
    return (int64_t)(free_b + g_dev_held.load());
}

extern "C" int abamd_gpu_align_sequence_to_subgraph(abpoa_t *ab, abpoa_para_t *abpt,
        int beg_node_id, int end_node_id, uint8_t *query, int qlen, abpoa_res_t *res) {
    BatchJob b;
    b.ab = ab; b.abpt = abpt;
    b.beg_node_id = beg_node_id; b.end_node_id = end_node_id;
    b.query = query; b.qlen = qlen; b.res = res;
    b.est_cells_hint = 0; b.cells_out = nullptr;
    return abamd_gpu_align_batch(&b, 1);
}
