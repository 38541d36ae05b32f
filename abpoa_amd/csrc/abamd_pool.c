/* Persistent host thread pool.
 *
 * The pipeline runs a fold pass and a pack pass per (round, group) item —
 * ~150 items x ~250 threads per bench step. Creating and joining pthreads
 * per pass costs 20-30 ms/item and leaves every worker's thread-local
 * scratch cold; this pool creates the workers once and broadcasts work via
 * a generation counter. fn(arg, tid, nthr) is invoked on workers
 * tid = 0..nthr-1; abamd_pool_run returns when all have finished. Calls
 * must not nest (all call sites run on the driver's main thread). */
#include <pthread.h>
#include <stdlib.h>
#include <unistd.h>
#include "abamd_util.h"

typedef void (*abamd_pool_fn)(void *arg, int tid, int nthr);

/* Spawn-per-call is the DEFAULT: a same-box A/B (profiles/r01_pool_ab.txt)
 * showed the persistent broadcast pool tripling per-fold CPU time on the
 * 254-thread GPU-box host (fold work 5255 vs 1717 cpu-s, pack 16.1 vs
 * 12.8 s wall) — staggered thread starts behave better for these short
 * memory-bound passes. The pool stays opt-in via ABPOA_AMD_POOL=1. */
typedef struct { abamd_pool_fn fn; void *arg; int tid, nthr; } spawn_t;
static void *spawn_tramp(void *p) {
    spawn_t *s = (spawn_t*)p;
    s->fn(s->arg, s->tid, s->nthr);
    return NULL;
}
static int pool_disabled(void) {
    static int v = -1;
    if (v < 0) { const char *e = getenv("ABPOA_AMD_POOL"); v = (e && *e && *e != '0') ? 0 : 1; }
    return v;
}

static pthread_mutex_t mu = PTHREAD_MUTEX_INITIALIZER;
static pthread_cond_t cv_work = PTHREAD_COND_INITIALIZER;
static pthread_cond_t cv_done = PTHREAD_COND_INITIALIZER;
static int pool_n = 0;
static int started = 0;
static unsigned long long gen = 0;
static int n_active = 0;
static int n_running = 0;
static abamd_pool_fn cur_fn;
static void *cur_arg;

static void *pool_worker(void *p) {
    long tid = (long)p;
    unsigned long long seen = 0;
    pthread_mutex_lock(&mu);
    for (;;) {
        while (gen == seen) pthread_cond_wait(&cv_work, &mu);
        seen = gen;
        abamd_pool_fn fn = cur_fn;
        void *arg = cur_arg;
        int na = n_active;
        pthread_mutex_unlock(&mu);
        if ((int)tid < na) fn(arg, (int)tid, na);
        pthread_mutex_lock(&mu);
        if (--n_running == 0) pthread_cond_signal(&cv_done);
    }
    return NULL;
}

int abamd_pool_size(void) {
    long n = sysconf(_SC_NPROCESSORS_ONLN);
    if (n < 1) n = 1;
    if (n > 256) n = 256;
    return (int)n;
}

void abamd_pool_run(abamd_pool_fn fn, void *arg, int nthr) {
    int N = abamd_pool_size();
    if (nthr > N) nthr = N;
    if (nthr <= 1) { fn(arg, 0, 1); return; }
    if (pool_disabled()) {
        spawn_t *ss = (spawn_t*)abamd_malloc((size_t)nthr * sizeof(spawn_t));
        pthread_t *ts = (pthread_t*)abamd_malloc((size_t)nthr * sizeof(pthread_t));
        for (int t = 1; t < nthr; ++t) {
            ss[t].fn = fn; ss[t].arg = arg; ss[t].tid = t; ss[t].nthr = nthr;
            pthread_create(&ts[t], NULL, spawn_tramp, &ss[t]);
        }
        fn(arg, 0, nthr);
        for (int t = 1; t < nthr; ++t) pthread_join(ts[t], NULL);
        free(ss); free(ts);
        return;
    }
    pthread_mutex_lock(&mu);
    if (!started) {
        pool_n = N;
        for (long t = 0; t < pool_n; ++t) {
            pthread_t th;
            if (pthread_create(&th, NULL, pool_worker, (void*)t) != 0) {
                pool_n = (int)t;
                break;
            }
            pthread_detach(th);
        }
        started = 1;
    }
    if (pool_n < 1) { pthread_mutex_unlock(&mu); fn(arg, 0, 1); return; }
    if (nthr > pool_n) nthr = pool_n;
    cur_fn = fn;
    cur_arg = arg;
    n_active = nthr;
    n_running = pool_n; /* every worker wakes; only tid < nthr runs fn */
    ++gen;
    pthread_cond_broadcast(&cv_work);
    while (n_running) pthread_cond_wait(&cv_done, &mu);
    pthread_mutex_unlock(&mu);
}
