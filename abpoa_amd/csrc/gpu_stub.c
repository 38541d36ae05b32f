/* Link stub used ONLY by the CPU-side test build (abpoa_amd_cputest): the
 * product GPU core is not linked, so any attempt to use the product aligner
 * without an explicitly injected test oracle aborts loudly. */
#include "abpoa_amd.h"
#include "abamd_util.h"

int abamd_gpu_align_sequence_to_subgraph(abpoa_t *ab, abpoa_para_t *abpt,
        int beg_node_id, int end_node_id, uint8_t *query, int qlen, abpoa_res_t *res) {
    (void)ab; (void)abpt; (void)beg_node_id; (void)end_node_id; (void)query; (void)qlen; (void)res;
    abamd_fatal("abpoa_amd", "this is the CPU-only TEST build: no GPU aligner is linked. "
                "Inject the oracle via ABPOA_AMD_TEST_ALIGNER_SO, or use the real abpoa_amd build.");
    return -1;
}

void abpoa_amd_get_stats(uint64_t *dp_cells, uint64_t *kernel_ns, uint64_t *n_launches) {
    if (dp_cells) *dp_cells = 0;
    if (kernel_ns) *kernel_ns = 0;
    if (n_launches) *n_launches = 0;
}
void abpoa_amd_reset_stats(void) {}

typedef struct {
    abpoa_t *ab; abpoa_para_t *abpt;
    int beg_node_id, end_node_id;
    uint8_t *query; int qlen; abpoa_res_t *res;
    int64_t est_cells_hint; int64_t *cells_out;
} abamd_batch_job_t;

int abamd_gpu_align_batch(abamd_batch_job_t *batch, int n_jobs) {
    /* CPU test build: route each job through the dispatching seam so an
     * injected test oracle serves batched runs too. */
    int i;
    for (i = 0; i < n_jobs; ++i)
        simd_abpoa_align_sequence_to_subgraph(batch[i].ab, batch[i].abpt,
            batch[i].beg_node_id, batch[i].end_node_id, batch[i].query, batch[i].qlen, batch[i].res);
    return 0;
}

void abpoa_amd_get_stats2(uint64_t *alg_bytes) { if (alg_bytes) *alg_bytes = 0; }
void abamd_timing_report(const char *tag) { (void)tag; }

static abamd_batch_job_t *stub_slot_batch[8];
static int stub_slot_n[8];
int abamd_gpu_batch_prepare(abamd_batch_job_t *batch, int n_jobs, int slot) {
    stub_slot_batch[slot] = batch; stub_slot_n[slot] = n_jobs; return 0;
}
int abamd_gpu_batch_launch(int slot) {
    return abamd_gpu_align_batch(stub_slot_batch[slot], stub_slot_n[slot]);
}
int abamd_gpu_batch_finish_slot(int slot) { (void)slot; return 0; }

int abamd_gpu_align_batch_slot(abamd_batch_job_t *batch, int n_jobs, int slot) {
    (void)slot; return abamd_gpu_align_batch(batch, n_jobs);
}

int64_t abamd_gpu_free_mem(void) { return 0; }
void abamd_gpu_set_arena_cap(uint64_t bytes) { (void)bytes; }

/* device-resident batch driver: GPU build only — the CPU test build always
 * takes the host-fold driver */
int abamd_batch_resident_supported(const abpoa_para_t *abpt) { (void)abpt; return 0; }
int abpoa_amd_msa_batch_resident(abpoa_para_t *abpt, int n_sets, const int *n_seqs,
                                 const int *const *seq_lens, const uint8_t *const *const *seqs,
                                 abpoa_amd_cons_cb cb, void *user, int n_host_threads) {
    (void)abpt; (void)n_sets; (void)n_seqs; (void)seq_lens; (void)seqs;
    (void)cb; (void)user; (void)n_host_threads;
    abamd_fatal("abpoa_amd", "device-resident batch driver is not in the CPU test build");
    return -1;
}
