/* Aligner seam + dispatch.
 *
 * Exports the exact symbols the reference host code calls
 * (abpoa_align_simd.h:11-12 / dispatch precedent abpoa_dispatch_simd.c:59-82)
 * and routes them to the HIP/CDNA4 core. There is NO silent CPU fallback:
 * with no usable GPU the call aborts. Tests may explicitly inject the CPU
 * oracle (oracle/liboracle.so) via abpoa_amd_set_test_aligner() or the
 * ABPOA_AMD_TEST_ALIGNER_SO environment variable; doing so prints a loud
 * notice on stderr.
 */
#include <dlfcn.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

/* product GPU implementation (gpu_align.cpp) */
int abamd_gpu_align_sequence_to_subgraph(abpoa_t *ab, abpoa_para_t *abpt,
        int beg_node_id, int end_node_id, uint8_t *query, int qlen, abpoa_res_t *res);

static abpoa_amd_aligner_fn g_test_aligner = NULL;
static int g_env_checked = 0;

void abpoa_amd_set_test_aligner(abpoa_amd_aligner_fn fn) {
    g_test_aligner = fn;
    g_env_checked = 1; /* explicit registration overrides the env hook */
    if (fn) fprintf(stderr, "[abpoa_amd] NOTICE: TEST ALIGNER INJECTED (CPU oracle) — this is not the product GPU path.\n");
}

static void check_env_aligner(void) {
    if (g_env_checked) return;
    g_env_checked = 1;
    const char *so = getenv("ABPOA_AMD_TEST_ALIGNER_SO");
    if (!so || !*so) return;
    void *h = dlopen(so, RTLD_NOW | RTLD_LOCAL);
    if (!h) abamd_fatal("abpoa_amd", "ABPOA_AMD_TEST_ALIGNER_SO: dlopen(%s) failed: %s", so, dlerror());
    abpoa_amd_aligner_fn fn = (abpoa_amd_aligner_fn)(size_t)dlsym(h, "oracle_align_sequence_to_subgraph");
    if (!fn) abamd_fatal("abpoa_amd", "ABPOA_AMD_TEST_ALIGNER_SO: symbol oracle_align_sequence_to_subgraph not found in %s", so);
    g_test_aligner = fn;
    fprintf(stderr, "[abpoa_amd] NOTICE: TEST ALIGNER INJECTED from %s (CPU oracle) — this is not the product GPU path.\n", so);
}

int simd_abpoa_align_sequence_to_subgraph(abpoa_t *ab, abpoa_para_t *abpt,
        int beg_node_id, int end_node_id, uint8_t *query, int qlen, abpoa_res_t *res) {
    check_env_aligner();
    if (g_test_aligner)
        return g_test_aligner(ab, abpt, beg_node_id, end_node_id, query, qlen, res);
    return abamd_gpu_align_sequence_to_subgraph(ab, abpt, beg_node_id, end_node_id, query, qlen, res);
}

int simd_abpoa_align_sequence_to_graph(abpoa_t *ab, abpoa_para_t *abpt, uint8_t *query, int qlen, abpoa_res_t *res) {
    return simd_abpoa_align_sequence_to_subgraph(ab, abpt, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID, query, qlen, res);
}

/* public wrappers (abpoa_align.c:195-207). Excluded from the seam-only shim
 * build (ABAMD_SHIM_ONLY), where the reference host objects define them. */
#ifndef ABAMD_SHIM_ONLY
int abpoa_align_sequence_to_subgraph(abpoa_t *ab, abpoa_para_t *abpt, int beg_node_id, int end_node_id,
                                     uint8_t *query, int qlen, abpoa_res_t *res) {
    if (ab->abg->node_n <= 2) return -1;
    if (ab->abg->is_topological_sorted == 0) abpoa_topological_sort(ab->abg, abpt);
    simd_abpoa_align_sequence_to_subgraph(ab, abpt, beg_node_id, end_node_id, query, qlen, res);
    return 0;
}

int abpoa_align_sequence_to_graph(abpoa_t *ab, abpoa_para_t *abpt, uint8_t *query, int qlen, abpoa_res_t *res) {
    if (ab->abg->node_n <= 2) return -1;
    if (ab->abg->is_topological_sorted == 0) abpoa_topological_sort(ab->abg, abpt);
    simd_abpoa_align_sequence_to_graph(ab, abpt, query, qlen, res);
    return 0;
}
#endif /* !ABAMD_SHIM_ONLY */

/* Opaque matrix handle: the DP arena lives on the device inside the GPU
 * shim's per-thread context, so this is a placeholder kept only for layout
 * compatibility of abpoa_t (abpoa.h:136). */
struct abpoa_simd_matrix_t { int unused; };

abpoa_simd_matrix_t *abamd_matrix_new(void) {
    return (abpoa_simd_matrix_t*)abamd_calloc(1, sizeof(struct abpoa_simd_matrix_t));
}
void abamd_matrix_destroy(abpoa_simd_matrix_t *m) { free(m); }
