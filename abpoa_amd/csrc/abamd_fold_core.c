/* Flat-array graph fold core — round-2 groundwork for the on-device fold.
 *
 * Re-expresses the order-sensitive graph mutation of
 * abpoa_add_subgraph_alignment (abamd_graph.c:492-562, itself restating
 * abpoa_graph.c:689-774) over allocation-free flat arrays: append-only edge
 * pools with per-node head/tail links, a word pool for per-out-edge read-id
 * bitsets, and a linked aligned-group pool. This is the representation a
 * CDNA4 fold kernel can own per set (one block per set, the order-sensitive
 * walk on lane 0); compiled for the host today so bit-exactness against the
 * pointer-based graph is provable on CPU (tests/test_fold_twin via
 * abamd_fold_twin_test.c), before any GPU port.
 *
 * Scope: node/edge/aligned-group/weight/read-bitset mutation — everything
 * whose ORDER determines downstream tie-breaks. The derived passes (topo
 * index, remain, msa rank) consume the materialized adjacency and are
 * mechanical; they stay in abamd_graph.c until the device port.
 */
#include <string.h>
#include "abpoa_amd.h"
#include "abamd_util.h"
#include <math.h>
#include "abamd_fold_core.h"

void abamd_flat_init(flat_graph_t *fg, int node_cap, int edge_cap, int aln_cap, int rid_n) {
    memset(fg, 0, sizeof(*fg));
    fg->node_cap = node_cap; fg->edge_cap = edge_cap; fg->aln_cap = aln_cap; fg->rid_n = rid_n;
    fg->base = (uint8_t*)abamd_calloc(node_cap, 1);
    fg->n_read = (int*)abamd_calloc(node_cap, sizeof(int));
    fg->n_span_read = (int*)abamd_calloc(node_cap, sizeof(int));
    fg->in_head = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    fg->in_tail = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    fg->out_head = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    fg->out_tail = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    fg->aln_head = (int*)abamd_malloc((size_t)node_cap * sizeof(int));
    for (int i = 0; i < node_cap; ++i)
        fg->in_head[i] = fg->in_tail[i] = fg->out_head[i] = fg->out_tail[i] = fg->aln_head[i] = -1;
    fg->in_to = (int*)abamd_malloc((size_t)edge_cap * sizeof(int));
    fg->in_w = (int*)abamd_malloc((size_t)edge_cap * sizeof(int));
    fg->in_next = (int*)abamd_malloc((size_t)edge_cap * sizeof(int));
    fg->out_to = (int*)abamd_malloc((size_t)edge_cap * sizeof(int));
    fg->out_w = (int*)abamd_malloc((size_t)edge_cap * sizeof(int));
    fg->out_next = (int*)abamd_malloc((size_t)edge_cap * sizeof(int));
    fg->rid_pool = rid_n > 0 ? (uint64_t*)abamd_calloc((size_t)edge_cap * rid_n, sizeof(uint64_t)) : NULL;
    fg->aln_id = (int*)abamd_malloc((size_t)aln_cap * sizeof(int));
    fg->aln_next = (int*)abamd_malloc((size_t)aln_cap * sizeof(int));
    /* nodes 0/1 = SRC/SINK, as the pointer graph after abpoa_reset */
    fg->node_n = 2;
}

void abamd_flat_free(flat_graph_t *fg) {
    free(fg->base); free(fg->n_read); free(fg->n_span_read);
    free(fg->in_head); free(fg->in_tail); free(fg->out_head); free(fg->out_tail);
    free(fg->aln_head);
    free(fg->in_to); free(fg->in_w); free(fg->in_next);
    free(fg->out_to); free(fg->out_w); free(fg->out_next);
    free(fg->rid_pool); free(fg->aln_id); free(fg->aln_next);
}


#include "abamd_fold_core.inc"
#include "abamd_cons_core.inc"
