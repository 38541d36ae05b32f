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

static int flat_add_node(flat_graph_t *fg, uint8_t base) {
    if (fg->node_n >= fg->node_cap)
        abamd_fatal("abamd_flat", "node pool exhausted (%d)", fg->node_cap);
    int id = fg->node_n++;
    fg->base[id] = base;
    return id;
}

/* append-order edge add with the exact existing-edge and read-id semantics
 * of abpoa_add_graph_edge (abamd_graph.c:150-212) */
static void flat_add_edge(flat_graph_t *fg, int from, int to, int check_edge, int w,
                          int add_read_id, int read_id) {
    int e, out_e = -1, exists = 0;
    if (check_edge) {
        for (e = fg->in_head[to]; e != -1; e = fg->in_next[e])
            if (fg->in_to[e] == from) { fg->in_w[e] += w; break; }
        for (e = fg->out_head[from]; e != -1; e = fg->out_next[e])
            if (fg->out_to[e] == to) { fg->out_w[e] += w; exists = 1; out_e = e; break; }
    }
    if (!exists) {
        if (fg->edge_n_in >= fg->edge_cap || fg->edge_n_out >= fg->edge_cap)
            abamd_fatal("abamd_flat", "edge pool exhausted (%d)", fg->edge_cap);
        e = fg->edge_n_in++;
        fg->in_to[e] = from; fg->in_w[e] = w; fg->in_next[e] = -1;
        if (fg->in_tail[to] == -1) fg->in_head[to] = e;
        else fg->in_next[fg->in_tail[to]] = e;
        fg->in_tail[to] = e;

        e = fg->edge_n_out++;
        fg->out_to[e] = to; fg->out_w[e] = w; fg->out_next[e] = -1;
        if (fg->out_tail[from] == -1) fg->out_head[from] = e;
        else fg->out_next[fg->out_tail[from]] = e;
        fg->out_tail[from] = e;
        out_e = e;
    }
    if (add_read_id && fg->rid_n > 0)
        fg->rid_pool[(size_t)out_e * fg->rid_n + (read_id >> 6)] |= 1ull << (read_id & 0x3f);
    fg->n_read[from] += 1;
}

static int flat_find_aligned(const flat_graph_t *fg, int node_id, uint8_t base) {
    for (int a = fg->aln_head[node_id]; a != -1; a = fg->aln_next[a])
        if (fg->base[fg->aln_id[a]] == base) return fg->aln_id[a];
    return -1;
}

static void flat_push_aligned(flat_graph_t *fg, int node_id, int aligned_id) {
    if (fg->aln_n >= fg->aln_cap)
        abamd_fatal("abamd_flat", "aligned pool exhausted (%d)", fg->aln_cap);
    int a = fg->aln_n++;
    fg->aln_id[a] = aligned_id;
    /* append at tail to preserve the pointer graph's list order */
    int *slot = &fg->aln_head[node_id];
    while (*slot != -1) slot = &fg->aln_next[*slot];
    fg->aln_next[a] = -1;
    *slot = a;
}

/* mutual group registration (add_aligned_pair, abamd_graph.c:220-229) */
static void flat_add_aligned_pair(flat_graph_t *fg, int node_id, int new_id) {
    int a;
    for (a = fg->aln_head[node_id]; a != -1; a = fg->aln_next[a]) {
        int other = fg->aln_id[a];
        flat_push_aligned(fg, other, new_id);
        flat_push_aligned(fg, new_id, other);
        /* NOTE: new_id's list grows while we iterate node_id's list only —
         * matches the pointer version's iteration over node_id's snapshot */
    }
    flat_push_aligned(fg, node_id, new_id);
    flat_push_aligned(fg, new_id, node_id);
}

/* weight-descending adjacency sort with the pointer version's exact
 * pairwise-swap pattern (sort_adjacency, abamd_graph.c:352-373), applied by
 * permuting pool indices and relinking the per-node chains; runs after every
 * fold exactly where abpoa_topological_sort runs it */
void abamd_flat_sort_adjacency(flat_graph_t *fg) {
    int i, j, k, e;
    int scratch[1024];
    for (i = 0; i < fg->node_n; ++i) {
        /* in edges */
        int n = 0;
        for (e = fg->in_head[i]; e != -1; e = fg->in_next[e]) {
            if (n >= (int)(sizeof(scratch) / sizeof(int)))
                abamd_fatal("abamd_flat", "in-degree > scratch at node %d", i);
            scratch[n++] = e;
        }
        for (j = 0; j < n - 1; ++j)
            for (k = j + 1; k < n; ++k)
                if (fg->in_w[scratch[j]] < fg->in_w[scratch[k]]) {
                    int t = scratch[j]; scratch[j] = scratch[k]; scratch[k] = t;
                }
        fg->in_head[i] = n ? scratch[0] : -1;
        for (j = 0; j + 1 < n; ++j) fg->in_next[scratch[j]] = scratch[j + 1];
        if (n) { fg->in_next[scratch[n - 1]] = -1; fg->in_tail[i] = scratch[n - 1]; }
        /* out edges (read-id bitsets travel with the edge index) */
        n = 0;
        for (e = fg->out_head[i]; e != -1; e = fg->out_next[e]) {
            if (n >= (int)(sizeof(scratch) / sizeof(int)))
                abamd_fatal("abamd_flat", "out-degree > scratch at node %d", i);
            scratch[n++] = e;
        }
        for (j = 0; j < n - 1; ++j)
            for (k = j + 1; k < n; ++k)
                if (fg->out_w[scratch[j]] < fg->out_w[scratch[k]]) {
                    int t = scratch[j]; scratch[j] = scratch[k]; scratch[k] = t;
                }
        fg->out_head[i] = n ? scratch[0] : -1;
        for (j = 0; j + 1 < n; ++j) fg->out_next[scratch[j]] = scratch[j + 1];
        if (n) { fg->out_next[scratch[n - 1]] = -1; fg->out_tail[i] = scratch[n - 1]; }
    }
}

/* Derived passes over the flat adjacency, matching abamd_graph.c's
 * abpoa_BFS_set_node_index (:292-316) and abpoa_BFS_set_node_remain
 * (:320-348) exactly: Kahn BFS with whole-aligned-group queue entry, then
 * the reverse max-weight-out-edge remain BFS. Outputs into caller arrays
 * sized node_n; scratch = 2*node_n ints (degree + queue). */
void abamd_flat_topo_index(const flat_graph_t *fg, int *index_to_node_id,
                           int *node_id_to_index, int *scratch) {
    int i, e, a, cur, index = 0;
    int *in_deg = scratch, *q = scratch + fg->node_n;
    int qh = 0, qt = 0;
    for (i = 0; i < fg->node_n; ++i) {
        int d = 0;
        for (e = fg->in_head[i]; e != -1; e = fg->in_next[e]) ++d;
        in_deg[i] = d;
    }
    q[qt++] = 0; /* SRC */
    while (qh < qt) {
        cur = q[qh++];
        index_to_node_id[index] = cur;
        node_id_to_index[cur] = index++;
        if (cur == 1 /* SINK */) return;
        for (e = fg->out_head[cur]; e != -1; e = fg->out_next[e]) {
            int out = fg->out_to[e];
            if (--in_deg[out] == 0) {
                int ready = 1;
                for (a = fg->aln_head[out]; a != -1; a = fg->aln_next[a])
                    if (in_deg[fg->aln_id[a]] != 0) { ready = 0; break; }
                if (!ready) continue;
                q[qt++] = out;
                for (a = fg->aln_head[out]; a != -1; a = fg->aln_next[a])
                    q[qt++] = fg->aln_id[a];
            }
        }
    }
    abamd_fatal("abamd_flat_topo_index", "graph is not a connected DAG");
}

void abamd_flat_remain(const flat_graph_t *fg, int *max_remain, int *scratch) {
    int i, e, cur;
    int *out_deg = scratch, *q = scratch + fg->node_n;
    int qh = 0, qt = 0;
    for (i = 0; i < fg->node_n; ++i) {
        int d = 0;
        for (e = fg->out_head[i]; e != -1; e = fg->out_next[e]) ++d;
        out_deg[i] = d;
        max_remain[i] = 0;
    }
    q[qt++] = 1; /* SINK */
    max_remain[1] = -1;
    while (qh < qt) {
        cur = q[qh++];
        if (cur != 1) {
            int max_w = -1, max_id = -1;
            for (e = fg->out_head[cur]; e != -1; e = fg->out_next[e])
                if (fg->out_w[e] > max_w) { max_w = fg->out_w[e]; max_id = fg->out_to[e]; }
            max_remain[cur] = max_remain[max_id] + 1;
        }
        if (cur == 0 /* SRC */) return;
        for (e = fg->in_head[cur]; e != -1; e = fg->in_next[e]) {
            int in = fg->in_to[e];
            if (--out_deg[in] == 0) q[qt++] = in;
        }
    }
    abamd_fatal("abamd_flat_remain", "graph is not a connected DAG");
}

/* same walk as abpoa_add_subgraph_alignment minus n_span/topo (those are
 * derived passes). qpos_to_node_id is filled identically. */
void abamd_flat_apply_alignment(flat_graph_t *fg, int beg_node_id, int end_node_id,
                                const uint8_t *seq, const int *weight, int seq_l,
                                int *qpos_to_node_id, int n_cigar, const abpoa_cigar_t *cig,
                                int read_id, int add_read_id, int inc_both_ends) {
    int i, j;
    if (fg->node_n == 2) { /* first read: plain chain */
        int last = 0 /* SRC */;
        for (i = 0; i < seq_l; ++i) {
            int cur = flat_add_node(fg, seq[i]);
            if (qpos_to_node_id) qpos_to_node_id[i] = cur;
            flat_add_edge(fg, last, cur, 0, weight[i], add_read_id, read_id);
            fg->n_span_read[cur] = fg->n_span_read[last];
            last = cur;
        }
        flat_add_edge(fg, last, 1 /* SINK */, 0, weight[seq_l - 1], add_read_id, read_id);
        return;
    }
    if (n_cigar == 0) return;
    int op, len, node_id, query_id = -1, last_new = 0, last_id = beg_node_id, new_id, aligned_id, add;
    for (i = 0; i < n_cigar; ++i) {
        op = (int)(cig[i] & 0xf);
        if (op == ABPOA_CMATCH) {
            node_id = (int)((cig[i] >> 34) & 0x3fffffff);
            query_id++;
            if (fg->base[node_id] != seq[query_id]) {
                if ((aligned_id = flat_find_aligned(fg, node_id, seq[query_id])) != -1) {
                    add = (last_id != beg_node_id || inc_both_ends) ? 1 : 0;
                    flat_add_edge(fg, last_id, aligned_id, 1 - last_new, weight[query_id], add_read_id & add, read_id);
                    if (add == 0) fg->n_read[last_id]--;
                    last_id = aligned_id; last_new = 0;
                } else {
                    new_id = flat_add_node(fg, seq[query_id]);
                    add = (last_id != beg_node_id || inc_both_ends) ? 1 : 0;
                    flat_add_edge(fg, last_id, new_id, 0, weight[query_id], add_read_id & add, read_id);
                    fg->n_span_read[new_id] = fg->n_span_read[last_id];
                    if (add == 0) fg->n_read[last_id]--;
                    last_id = new_id; last_new = 1;
                    flat_add_aligned_pair(fg, node_id, new_id);
                }
            } else {
                add = (last_id != beg_node_id || inc_both_ends) ? 1 : 0;
                flat_add_edge(fg, last_id, node_id, 1 - last_new, weight[query_id], add_read_id & add, read_id);
                if (add == 0) fg->n_read[last_id]--;
                last_id = node_id; last_new = 0;
            }
            if (qpos_to_node_id) qpos_to_node_id[query_id] = last_id;
        } else if (op == ABPOA_CINS || op == ABPOA_CSOFT_CLIP || op == ABPOA_CHARD_CLIP) {
            len = (int)((cig[i] >> 4) & 0x3fffffff);
            query_id += len;
            for (j = len - 1; j >= 0; --j) {
                new_id = flat_add_node(fg, seq[query_id - j]);
                add = (last_id != beg_node_id || inc_both_ends) ? 1 : 0;
                flat_add_edge(fg, last_id, new_id, 0, weight[query_id - j], add_read_id & add, read_id);
                fg->n_span_read[new_id] = fg->n_span_read[last_id];
                if (add == 0) fg->n_read[last_id]--;
                last_id = new_id; last_new = 1;
                if (qpos_to_node_id) qpos_to_node_id[query_id - j] = last_id;
            }
        } /* ABPOA_CDEL consumes nothing on the query side */
    }
    flat_add_edge(fg, last_id, end_node_id, 1 - last_new, weight[seq_l - 1], add_read_id, read_id);
}
