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
