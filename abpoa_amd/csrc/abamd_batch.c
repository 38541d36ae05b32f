/* Batched multi-set POA driver — HOST-FOLD FALLBACK + dispatcher.
 *
 * The reference processes one read set at a time (abpoa_poa,
 * abpoa_align.c:313-353); a single alignment cannot fill a 256-CU GPU, so
 * the MI355X-native driver advances MANY independent read sets in lockstep
 * "rounds". The PRODUCT path is the device-resident driver
 * (gpu_batch_resident.cpp: graphs live in HBM, the fold kernel consumes
 * CIGARs in place); abpoa_amd_msa_batch dispatches to it whenever the
 * configuration allows. THIS file keeps the round-1 host-fold pipeline as
 * the fallback (configs the flat pools don't carry: inc_path_score,
 * per-read quality weights; ABPOA_AMD_HOST_FOLD=1 forces it for A/B) and
 * as the CPU-test route (gpu_stub builds always land here). Per-set
 * results are identical on every path: sets are fully independent
 * (SURVEY.md §5/§8e).
 */
#include <pthread.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

void abamd_timing_report(const char *tag);
static double g_fold_s, g_cons_s, g_build_s;
static long g_big_items;
static _Atomic unsigned long long g_fold_work_ns; /* summed across workers */

/* from gpu_align.cpp */
typedef struct {
    abpoa_t *ab;
    abpoa_para_t *abpt;
    int beg_node_id, end_node_id;
    uint8_t *query;
    int qlen;
    abpoa_res_t *res;
    int64_t est_cells_hint;
    int64_t *cells_out;
} abamd_batch_job_t;
int abamd_gpu_align_batch(abamd_batch_job_t *batch, int n_jobs);
int abamd_gpu_align_batch_slot(abamd_batch_job_t *batch, int n_jobs, int slot);
int64_t abamd_gpu_free_mem(void);
int abamd_gpu_batch_prepare(abamd_batch_job_t *batch, int n_jobs, int slot);
int abamd_gpu_batch_launch(int slot);
int abamd_gpu_batch_finish_slot(int slot);
void abamd_gpu_set_arena_cap(uint64_t bytes);

typedef struct {
    abpoa_t *ab;
    const int *seq_lens;
    const uint8_t *const *seqs;
    int n_seqs;
    abpoa_res_t res;
    int *weight_buf; int weight_cap;
    int active; /* has a job this round */
    int64_t last_cells; /* banded cells of this set's previous alignment */
} set_state_t;

typedef struct {
    set_state_t *sets;
    int n_sets;
    abpoa_para_t *abpt;
    int round;
    int next;             /* work-stealing cursor */
    pthread_mutex_t mu;
} fold_work_t;

static int *ones_weight(set_state_t *st, int len) {
    if (len > st->weight_cap) {
        st->weight_buf = (int*)abamd_realloc(st->weight_buf, (size_t)len * sizeof(int));
        for (int i = st->weight_cap; i < len; ++i) st->weight_buf[i] = 1;
        for (int i = 0; i < len; ++i) st->weight_buf[i] = 1;
        st->weight_cap = len;
    }
    return st->weight_buf;
}

static void fold_one(fold_work_t *w, int si) {
    set_state_t *st = &w->sets[si];
    int r = w->round;
    if (r >= st->n_seqs) return;
    int qlen = st->seq_lens[r];
    uint8_t *q = (uint8_t*)st->seqs[r];
    double t0 = abamd_realtime();
    abpoa_add_graph_alignment(st->ab, w->abpt, q, ones_weight(st, qlen), qlen, NULL,
                              st->res, r, st->n_seqs, 1);
    g_fold_work_ns += (unsigned long long)((abamd_realtime() - t0) * 1e9);
    if (st->res.n_cigar) { free(st->res.graph_cigar); st->res.graph_cigar = NULL; st->res.n_cigar = 0; }
}

static void fold_worker(void *arg, int tid, int nthr) {
    fold_work_t *w = (fold_work_t*)arg;
    (void)tid; (void)nthr;
    for (;;) {
        pthread_mutex_lock(&w->mu);
        int si = w->next++;
        pthread_mutex_unlock(&w->mu);
        if (si >= w->n_sets) break;
        if (w->sets[si].active) fold_one(w, si);
    }
}

typedef struct {
    set_state_t *sets;
    int n_sets, next;
    abpoa_para_t *abpt;
    abpoa_amd_cons_cb cb; void *user;
    pthread_mutex_t mu;
} cons_work_t;

static void cons_worker(void *arg, int tid, int nthr) {
    cons_work_t *w = (cons_work_t*)arg;
    (void)tid; (void)nthr;
    for (;;) {
        pthread_mutex_lock(&w->mu);
        int si = w->next++;
        pthread_mutex_unlock(&w->mu);
        if (si >= w->n_sets) break;
        w->sets[si].ab->abs->n_seq = w->sets[si].n_seqs; /* consensus reads n_seq */
        abpoa_generate_consensus(w->sets[si].ab, w->abpt);
    }
}

/* device-resident driver (gpu_batch_resident.cpp; stubbed out in the
 * CPU-only test build, where this host-fold driver always runs) */
int abamd_batch_resident_supported(const abpoa_para_t *abpt);
int abpoa_amd_msa_batch_resident(abpoa_para_t *abpt, int n_sets, const int *n_seqs,
                                 const int *const *seq_lens, const uint8_t *const *const *seqs,
                                 abpoa_amd_cons_cb cb, void *user, int n_host_threads);

int abpoa_amd_msa_batch(abpoa_para_t *abpt, int n_sets, const int *n_seqs,
                        const int *const *seq_lens, const uint8_t *const *const *seqs,
                        abpoa_amd_cons_cb cb, void *user, int n_host_threads) {
    if (n_sets <= 0) return 0;
    if (abamd_batch_resident_supported(abpt)) {
        /* device-resident graphs need every read non-empty (the reference
         * fold indexes weight[len-1]); scan is trivial vs one alignment */
        int ok = 1, s, i;
        for (s = 0; s < n_sets && ok; ++s)
            for (i = 0; i < n_seqs[s]; ++i)
                if (seq_lens[s][i] < 1) { ok = 0; break; }
        if (ok)
            return abpoa_amd_msa_batch_resident(abpt, n_sets, n_seqs, seq_lens, seqs,
                                                cb, user, n_host_threads);
    }
    {
        /* fold thread-count override: at ~250 threads the concurrent folds'
         * ~10 MB graph working sets thrash the LLC (measured ~20x per-call
         * inflation); knob for tuning without a rebuild */
        const char *ft = getenv("ABPOA_AMD_FOLD_THREADS");
        if (ft && *ft) n_host_threads = atoi(ft);
    }
    if (n_host_threads < 1) n_host_threads = 1;
    int i, max_reads = 0;
    set_state_t *sets = (set_state_t*)abamd_calloc(n_sets, sizeof(set_state_t));
    for (i = 0; i < n_sets; ++i) {
        sets[i].ab = abpoa_init();
        abpoa_reset(sets[i].ab, abpt, 1024);
        sets[i].seq_lens = seq_lens[i];
        sets[i].seqs = seqs[i];
        sets[i].n_seqs = n_seqs[i];
        if (n_seqs[i] > max_reads) max_reads = n_seqs[i];
    }

    /* GPU memory budget per launch (arena is the dominant term); default
     * mirrors the shim's 70%-of-free-HBM reservation */
    double mem_gb = 0.0;
    {
        const char *s = getenv("ABPOA_AMD_MEM_GB");
        if (s && *s) mem_gb = atof(s);
    }
    if (mem_gb <= 0.0) {
        /* up to three pipeline groups keep three arenas resident
         * concurrently, so each launch gets just under a third of the
         * usable HBM (abamd_gpu_free_mem counts our own held buffers) */
        int64_t free_b = abamd_gpu_free_mem();
        int np = n_sets >= 3 ? 3 : (n_sets >= 1 ? n_sets : 1);
        {
            const char *gs = getenv("ABPOA_AMD_GROUPS");
            if (gs && *gs) { int g = atoi(gs); if (g >= 1 && g <= 6) np = g; }
        }
        mem_gb = free_b > 0 ? (double)free_b * 0.84 / np / 1e9 : 48.0;
    }
    const double budget_bytes = mem_gb * 1e9;
    abamd_gpu_set_arena_cap((uint64_t)budget_bytes);

    /* ---- round 0: first reads thread straight into their graphs ---- */
    {
        double tf0 = abamd_realtime();
        for (i = 0; i < n_sets; ++i) {
            sets[i].active = sets[i].n_seqs > 0;
            memset(&sets[i].res, 0, sizeof(abpoa_res_t));
        }
        fold_work_t fw = { sets, n_sets, abpt, 0, 0, PTHREAD_MUTEX_INITIALIZER };
        abamd_pool_run(fold_worker, &fw, n_host_threads);
        g_fold_s += abamd_realtime() - tf0;
    }

    /* ---- rounds 1..max: software pipeline over set groups -------------
     * Group count is sized so one group-round fits the device memory
     * budget at the LAST (largest) round; with >= 2 groups, the host fold
     * and the next pack run while the other group's kernel executes. */
    /* three groups, lookahead two: while item `it` is finished, items
     * it+1 and it+2 (disjoint set groups) are already in flight on their
     * own streams — the DP kernel is latency-bound at ~1 wave per job, so
     * two overlapped launches nearly double chip fill. A group-round whose
     * arena demand exceeds the budget falls back to sequential sub-chunks
     * on the dedicated slot 3. */
    int n_groups = n_sets >= 3 ? 3 : n_sets;
    {
        /* up to 6 pipeline groups (device slots 0..5; 7 = retry/big). 3 is
         * the measured default; deeper overlap is a round-2 tuning knob */
        const char *gs = getenv("ABPOA_AMD_GROUPS");
        if (gs && *gs) {
            int g = atoi(gs);
            if (g >= 1 && g <= 6 && g <= n_sets) n_groups = g;
        }
    }
    int *grp_of = (int*)abamd_malloc((size_t)n_sets * sizeof(int));
    for (i = 0; i < n_sets; ++i) grp_of[i] = i % n_groups;

    /* item = (round r >= 1, group g); g-major within a round */
    long n_items = (long)(max_reads - 1) * n_groups;
    abamd_batch_job_t *slot_jobs[6];
    for (i = 0; i < 6; ++i)
        slot_jobs[i] = (abamd_batch_job_t*)abamd_malloc((size_t)n_sets * sizeof(abamd_batch_job_t));
    int slot_nj[6] = {0};

    #define ITEM_R(it) (1 + (int)((it) / n_groups))
    #define ITEM_G(it) ((int)((it) % n_groups))

    int slot_big[6] = {0};

    /* arena bytes one job will demand (matches gpu_align.cpp's reservation,
     * including the measured-cells tightening) */
    #define JOB_EST(J) ({ \
        int _w = abpt->wb < 0 ? (J)->qlen : abpt->wb + (int)(abpt->wf * (J)->qlen); \
        int _gn = (J)->ab->abg->node_n, _q = (J)->qlen; \
        long _ms = (long)_q * abpt->max_mat; \
        long _alt = (long)(_q > _gn ? _q : _gn) * abpt->gap_ext1 + abpt->gap_open1; \
        if (_alt > _ms) _ms = _alt; \
        int _oe1 = abpt->gap_open1 + abpt->gap_ext1, _oe2 = abpt->gap_open2 + abpt->gap_ext2; \
        int _ssz = (_ms <= 32767 - abpt->min_mis - _oe1 - _oe2) ? 2 : 4; \
        int _pl = abpt->gap_mode == ABPOA_CONVEX_GAP ? 3 : abpt->gap_mode == ABPOA_AFFINE_GAP ? 3 : 1; \
        double _cells = (double)_gn * (2.0 * _w + 160.0); \
        if ((J)->est_cells_hint > 0) { \
            double _t = (double)(J)->est_cells_hint * 1.5 + _q; \
            if (_t < _cells) _cells = _t; \
        } \
        _cells * _pl * _ssz; })

    /* build the job list for one (round, group) item into a slot */
    #define BUILD_ITEM(it, slot) do { \
        double _tb0 = abamd_realtime(); \
        int _r = ITEM_R(it), _g = ITEM_G(it), _nj = 0; \
        double _est = 0; \
        for (i = 0; i < n_sets; ++i) { \
            if (grp_of[i] != _g || _r >= sets[i].n_seqs) continue; \
            memset(&sets[i].res, 0, sizeof(abpoa_res_t)); \
            if (sets[i].ab->abg->node_n <= 2) continue; \
            if (sets[i].ab->abg->is_topological_sorted == 0) \
                abpoa_topological_sort(sets[i].ab->abg, abpt); \
            abamd_batch_job_t *J = &slot_jobs[slot][_nj]; \
            J->ab = sets[i].ab; J->abpt = abpt; \
            J->beg_node_id = ABPOA_SRC_NODE_ID; J->end_node_id = ABPOA_SINK_NODE_ID; \
            J->query = (uint8_t*)sets[i].seqs[_r]; \
            J->qlen = sets[i].seq_lens[_r]; \
            J->res = &sets[i].res; \
            J->est_cells_hint = sets[i].last_cells; \
            J->cells_out = &sets[i].last_cells; \
            _est += JOB_EST(J); \
            ++_nj; \
        } \
        slot_nj[slot] = _nj; slot_big[slot] = _est > budget_bytes; \
        if (slot_big[slot]) ++g_big_items; \
        g_build_s += abamd_realtime() - _tb0; \
    } while (0)

    /* oversized item: memory-bounded sequential chunks (not pipelined),
     * on device slot 3 so an in-flight pipeline slot is never clobbered */
    #define RUN_BIG_ITEM(slot) do { \
        int _done = 0; \
        while (_done < slot_nj[slot]) { \
            double _acc = 0; int _take = 0; \
            while (_done + _take < slot_nj[slot]) { \
                double _e = JOB_EST(&slot_jobs[slot][_done + _take]); \
                if (_take > 0 && _acc + _e > budget_bytes) break; \
                _acc += _e; ++_take; \
            } \
            abamd_gpu_align_batch_slot(&slot_jobs[slot][_done], _take, 7); \
            _done += _take; \
        } \
    } while (0)

    /* fold one (round, group) item on the host thread pool */
    #define FOLD_ITEM(it) do { \
        double _tf0 = abamd_realtime(); \
        int _r = ITEM_R(it), _g = ITEM_G(it); \
        for (i = 0; i < n_sets; ++i) \
            sets[i].active = (grp_of[i] == _g && _r < sets[i].n_seqs); \
        fold_work_t fw = { sets, n_sets, abpt, _r, 0, PTHREAD_MUTEX_INITIALIZER }; \
        abamd_pool_run(fold_worker, &fw, n_host_threads); \
        g_fold_s += abamd_realtime() - _tf0; \
    } while (0)

    if (n_items > 0) {
        /* lookahead = n_groups-1 items launched ahead of the one being
         * finished: item nx = it+LA is the same GROUP as item it-1, whose
         * fold completes just before BUILD(nx), so every launched item's
         * graphs are complete up to its round. LA=0 (single set) degrades
         * to strict build->launch->finish->fold per round. */
        const int LA = n_groups - 1;
        const int n_pipe = LA > 0 ? LA + 1 : 1;
        for (long k = 0; k < LA && k < n_items; ++k) {
            int s = (int)(k % n_pipe);
            BUILD_ITEM(k, s);
            if (slot_big[s]) RUN_BIG_ITEM(s);
            else {
                abamd_gpu_batch_prepare(slot_jobs[s], slot_nj[s], s);
                abamd_gpu_batch_launch(s);
            }
        }
        for (long it = 0; it < n_items; ++it) {
            int s = (int)(it % n_pipe);
            if (it >= 1) FOLD_ITEM(it - 1);
            long nx = it + LA;
            int ns = (int)(nx % n_pipe);
            if (LA > 0 && nx < n_items) {
                BUILD_ITEM(nx, ns);
                if (!slot_big[ns]) {
                    abamd_gpu_batch_prepare(slot_jobs[ns], slot_nj[ns], ns);
                    abamd_gpu_batch_launch(ns);
                }
            } else if (LA == 0) {
                BUILD_ITEM(it, 0);
                if (!slot_big[0]) {
                    abamd_gpu_batch_prepare(slot_jobs[0], slot_nj[0], 0);
                    abamd_gpu_batch_launch(0);
                }
            }
            if (!slot_big[s]) abamd_gpu_batch_finish_slot(s);
            if (LA > 0 && nx < n_items && slot_big[ns]) RUN_BIG_ITEM(ns);
            else if (LA == 0 && slot_big[0]) RUN_BIG_ITEM(0);
        }
        FOLD_ITEM(n_items - 1);
    }

    free(grp_of);
    for (i = 0; i < 6; ++i) free(slot_jobs[i]);

    /* consensus on host threads, then emit callbacks in order */
    double tc0 = abamd_realtime();
    cons_work_t cw = { sets, n_sets, 0, abpt, cb, user, PTHREAD_MUTEX_INITIALIZER };
    abamd_pool_run(cons_worker, &cw, n_host_threads);
    g_cons_s += abamd_realtime() - tc0;
    if (cb) for (i = 0; i < n_sets; ++i) cb(i, sets[i].ab->abc, user);

    if (getenv("ABPOA_AMD_TIMING")) {
        fprintf(stderr, "[abamd timing batch] fold %.2fs (work %.2f cpu-s) build+toposort %.2fs cons %.2fs big_items %ld (threads %d)\n",
                g_fold_s, (double)g_fold_work_ns / 1e9, g_build_s, g_cons_s, g_big_items, n_host_threads);
        abamd_timing_report("batch");
    }
    for (i = 0; i < n_sets; ++i) {
        free(sets[i].weight_buf);
        abpoa_free(sets[i].ab);
    }
    free(sets);
    return 0;
}
