/* Partial-order graph: DAG store, topological index, band metadata, and the
 * fold of a read's graph-CIGAR back into the DAG.
 *
 * Behaviour mirrors the reference implementation exactly where results are
 * observable (edge/adjacency insertion order, tie-breaking of the
 * weight-descending adjacency sort, BFS queue order) because the aligner's
 * backtrack and the consensus walk both depend on those orders:
 *   - topo index + band metadata: abpoa_graph.c:221-309
 *   - adjacency sort:             abpoa_graph.c:192-219
 *   - CIGAR fold:                 abpoa_graph.c:689-778
 * All code here is written from scratch.
 */
#include "abpoa_amd.h"
#include "abamd_util.h"

static void init_node(abpoa_node_t *v, int id) {
    memset(v, 0, sizeof(*v));
    v->node_id = id;
}

/* ---------------- allocation ---------------- */

static abpoa_graph_t *graph_new(void) {
    abpoa_graph_t *g = (abpoa_graph_t*)abamd_calloc(1, sizeof(abpoa_graph_t));
    g->node_m = 2; g->node_n = 2;
    g->node = (abpoa_node_t*)abamd_calloc(2, sizeof(abpoa_node_t));
    init_node(&g->node[0], ABPOA_SRC_NODE_ID);
    init_node(&g->node[1], ABPOA_SINK_NODE_ID);
    return g;
}

static void free_node_arrays(abpoa_node_t *v) {
    int j;
    if (v->in_edge_m > 0) { free(v->in_id); free(v->in_edge_weight); }
    if (v->out_edge_m > 0) {
        free(v->out_id); free(v->out_edge_weight);
        if (v->read_ids_n > 0) {
            for (j = 0; j < v->out_edge_m; ++j) free(v->read_ids[j]);
            free(v->read_ids);
        } else if (v->read_ids) free(v->read_ids);
    }
    if (v->m_read > 0) free(v->read_weight);
    if (v->aligned_node_m > 0) free(v->aligned_node_id);
}

static void graph_destroy(abpoa_graph_t *g) {
    int i;
    for (i = 0; i < g->node_m; ++i) free_node_arrays(&g->node[i]);
    free(g->node);
    free(g->index_to_node_id); free(g->node_id_to_index);
    free(g->node_id_to_msa_rank);
    free(g->node_id_to_max_pos_left); free(g->node_id_to_max_pos_right);
    free(g->node_id_to_max_remain);
    free(g);
}

static abpoa_cons_t *cons_new(void) {
    return (abpoa_cons_t*)abamd_calloc(1, sizeof(abpoa_cons_t));
}

static void cons_clear(abpoa_cons_t *c) {
    int i;
    if (c->n_cons > 0) {
        free(c->clu_n_seq); free(c->cons_len);
        if (c->cons_node_ids) { for (i = 0; i < c->n_cons; ++i) free(c->cons_node_ids[i]); free(c->cons_node_ids); }
        if (c->cons_base)     { for (i = 0; i < c->n_cons; ++i) free(c->cons_base[i]);     free(c->cons_base); }
        if (c->cons_cov)      { for (i = 0; i < c->n_cons; ++i) free(c->cons_cov[i]);      free(c->cons_cov); }
        if (c->clu_read_ids)  { for (i = 0; i < c->n_cons; ++i) free(c->clu_read_ids[i]);  free(c->clu_read_ids); }
        if (c->cons_phred_score) { for (i = 0; i < c->n_cons; ++i) free(c->cons_phred_score[i]); free(c->cons_phred_score); }
    }
    if (c->msa_len > 0 && c->msa_base) {
        for (i = 0; i < c->n_seq + c->n_cons; ++i) free(c->msa_base[i]);
        free(c->msa_base);
    }
    memset(c, 0, sizeof(*c));
}

/* declared in abamd_seq.c */
abpoa_seq_t *abamd_seq_new(void);
void abamd_seq_destroy(abpoa_seq_t *abs);
/* declared in the aligner shim */
abpoa_simd_matrix_t *abamd_matrix_new(void);
void abamd_matrix_destroy(abpoa_simd_matrix_t *m);

abpoa_t *abpoa_init(void) {
    abpoa_t *ab = (abpoa_t*)abamd_malloc(sizeof(abpoa_t));
    ab->abg = graph_new();
    ab->abs = abamd_seq_new();
    ab->abm = abamd_matrix_new();
    ab->abc = cons_new();
    return ab;
}

void abpoa_free(abpoa_t *ab) {
    graph_destroy(ab->abg);
    abamd_seq_destroy(ab->abs);
    abamd_matrix_destroy(ab->abm);
    cons_clear(ab->abc); free(ab->abc);
    free(ab);
}

void abpoa_clean_msa_cons(abpoa_t *ab) { cons_clear(ab->abc); }

/* ---------------- node / edge mutation ---------------- */

int abpoa_add_graph_node(abpoa_graph_t *g, uint8_t base) {
    int id = g->node_n;
    if (g->node_n == g->node_m) {
        int i, old = g->node_m;
        g->node_m = old ? old << 1 : 2;
        g->node = (abpoa_node_t*)abamd_realloc(g->node, (size_t)g->node_m * sizeof(abpoa_node_t));
        for (i = old; i < g->node_m; ++i) init_node(&g->node[i], i);
    }
    g->node[id].node_id = id;
    g->node[id].base = base;
    ++g->node_n;
    return id;
}

static void set_read_id_bit(uint64_t *bits, int read_id) {
    bits[read_id >> 6] |= ((uint64_t)1) << (read_id & 63);
}

/* make room for one more out-edge; allocate/extend read_ids slabs when the
 * node tracks per-edge read bitsets (abpoa_graph.c:49-85) */
static void grow_out_edges(abpoa_node_t *v, int want_read_ids) {
    int i;
    if (v->out_edge_m == 0) {
        v->out_edge_m = AB_MAX2(v->out_edge_n, 1);
        v->out_id = (int*)abamd_malloc((size_t)v->out_edge_m * sizeof(int));
        v->out_edge_weight = (int*)abamd_malloc((size_t)v->out_edge_m * sizeof(int));
        if (want_read_ids || v->read_ids_n > 0) {
            v->read_ids = (uint64_t**)abamd_malloc((size_t)v->out_edge_m * sizeof(uint64_t*));
            for (i = 0; i < v->out_edge_m; ++i)
                v->read_ids[i] = v->read_ids_n > 0 ? (uint64_t*)abamd_calloc(v->read_ids_n, sizeof(uint64_t)) : NULL;
            if (v->read_ids_n == 0) for (i = 0; i < v->out_edge_m; ++i) v->read_ids[i] = NULL;
        }
    } else if (v->out_edge_n >= v->out_edge_m) {
        int old = v->out_edge_m;
        v->out_edge_m = ab_round_up_pow2_32(v->out_edge_n + 1);
        v->out_id = (int*)abamd_realloc(v->out_id, (size_t)v->out_edge_m * sizeof(int));
        v->out_edge_weight = (int*)abamd_realloc(v->out_edge_weight, (size_t)v->out_edge_m * sizeof(int));
        if (want_read_ids || v->read_ids_n > 0) {
            v->read_ids = (uint64_t**)abamd_realloc(v->read_ids, (size_t)v->out_edge_m * sizeof(uint64_t*));
            for (i = old; i < v->out_edge_m; ++i)
                v->read_ids[i] = v->read_ids_n > 0 ? (uint64_t*)abamd_calloc(v->read_ids_n, sizeof(uint64_t)) : NULL;
        }
    }
}

int abpoa_add_graph_edge(abpoa_graph_t *g, int from_id, int to_id, int check_edge, int w,
                         uint8_t add_read_id, uint8_t add_read_weight, int read_id, int read_ids_n, int tot_read_n) {
    if (from_id < 0 || from_id >= g->node_n || to_id < 0 || to_id >= g->node_n)
        abamd_fatal("abpoa_add_graph_edge", "bad edge %d->%d (node_n %d)", from_id, to_id, g->node_n);
    abpoa_node_t *from = &g->node[from_id], *to = &g->node[to_id];
    int i, exists = 0, out_i = -1;

    if (check_edge) {
        for (i = 0; i < to->in_edge_n; ++i)
            if (to->in_id[i] == from_id) { to->in_edge_weight[i] += w; break; }
        for (i = 0; i < from->out_edge_n; ++i)
            if (from->out_id[i] == to_id) { from->out_edge_weight[i] += w; exists = 1; out_i = i; break; }
    }
    if (!exists) {
        AB_GROW(int, to->in_id, to->in_edge_n, to->in_edge_m);
        /* in_edge_weight shares the same capacity variable; grow it in lockstep */
        to->in_edge_weight = (int*)abamd_realloc(to->in_edge_weight, (size_t)to->in_edge_m * sizeof(int));
        to->in_id[to->in_edge_n] = from_id;
        to->in_edge_weight[to->in_edge_n] = w;
        ++to->in_edge_n;

        grow_out_edges(from, add_read_id);
        from->out_id[from->out_edge_n] = to_id;
        from->out_edge_weight[from->out_edge_n] = w;
        out_i = from->out_edge_n;
        ++from->out_edge_n;
    }

    if (add_read_id) {
        if (out_i < 0) abamd_fatal("abpoa_add_graph_edge", "edge bookkeeping failure");
        if (read_ids_n <= 0) abamd_fatal("abpoa_add_graph_edge", "read_ids_n %d", read_ids_n);
        if (from->read_ids_n == 0) {
            if (!from->read_ids) {
                from->read_ids = (uint64_t**)abamd_malloc((size_t)from->out_edge_m * sizeof(uint64_t*));
                for (i = 0; i < from->out_edge_m; ++i) from->read_ids[i] = NULL;
            }
            for (i = 0; i < from->out_edge_m; ++i) {
                free(from->read_ids[i]);
                from->read_ids[i] = (uint64_t*)abamd_calloc(read_ids_n, sizeof(uint64_t));
            }
            from->read_ids_n = read_ids_n;
        } else if (from->read_ids_n < read_ids_n) {
            int j;
            for (i = 0; i < from->out_edge_m; ++i) {
                from->read_ids[i] = (uint64_t*)abamd_realloc(from->read_ids[i], (size_t)read_ids_n * sizeof(uint64_t));
                for (j = from->read_ids_n; j < read_ids_n; ++j) from->read_ids[i][j] = 0;
            }
            from->read_ids_n = read_ids_n;
        }
        set_read_id_bit(from->read_ids[out_i], read_id);
    }
    from->n_read += 1;
    if (add_read_weight) {
        if (tot_read_n > from->m_read) {
            from->read_weight = (int*)abamd_realloc(from->read_weight, (size_t)tot_read_n * sizeof(int));
            for (i = from->m_read; i < tot_read_n; ++i) from->read_weight[i] = 0;
            from->m_read = tot_read_n;
        }
        from->read_weight[read_id] = w;
    }
    return 1;
}

/* mutual "aligned node" (mismatch rank group) registration
 * (abpoa_graph.c:450-463) */
static void push_aligned(abpoa_node_t *v, int aligned_id) {
    AB_GROW(int, v->aligned_node_id, v->aligned_node_n, v->aligned_node_m);
    v->aligned_node_id[v->aligned_node_n++] = aligned_id;
}
static void add_aligned_pair(abpoa_graph_t *g, int node_id, int new_id) {
    int i;
    for (i = 0; i < g->node[node_id].aligned_node_n; ++i) {
        int other = g->node[node_id].aligned_node_id[i];
        push_aligned(&g->node[other], new_id);
        push_aligned(&g->node[new_id], other);
    }
    push_aligned(&g->node[node_id], new_id);
    push_aligned(&g->node[new_id], node_id);
}

static int find_aligned_with_base(abpoa_graph_t *g, int node_id, uint8_t base) {
    int i;
    for (i = 0; i < g->node[node_id].aligned_node_n; ++i) {
        int a = g->node[node_id].aligned_node_id[i];
        if (g->node[a].base == base) return a;
    }
    return -1;
}

/* non-static entries for the graph-restore path (abamd_restore.c) */
int abamd_get_aligned_id(abpoa_graph_t *g, int node_id, uint8_t base) {
    return find_aligned_with_base(g, node_id, base);
}
void abamd_add_aligned_pair(abpoa_graph_t *g, int node_id, int new_id) {
    add_aligned_pair(g, node_id, new_id);
}

/* ---------------- topological machinery ---------------- */

/* simple int FIFO over a thread-local reusable buffer: the topological
 * machinery runs once per read fold, and per-call malloc traffic serializes
 * the 200+-thread fold pool on the allocator */
typedef struct { int *a; int head, tail, cap; } ab_queue_t;
static __thread int *tls_q_buf = NULL;
static __thread int tls_q_cap = 0;
static __thread int *tls_deg_buf = NULL;
static __thread int tls_deg_cap = 0;

static int *tls_degree(int n) {
    if (n > tls_deg_cap) {
        tls_deg_buf = (int*)abamd_realloc(tls_deg_buf, (size_t)(n * 2) * sizeof(int));
        tls_deg_cap = n * 2;
    }
    return tls_deg_buf;
}
static void q_init(ab_queue_t *q, int cap_hint) {
    int want = cap_hint > 16 ? cap_hint : 16;
    if (want > tls_q_cap) {
        tls_q_buf = (int*)abamd_realloc(tls_q_buf, (size_t)(want * 2) * sizeof(int));
        tls_q_cap = want * 2;
    }
    q->a = tls_q_buf; q->cap = tls_q_cap;
    q->head = q->tail = 0;
}
static void q_push(ab_queue_t *q, int v) {
    if (q->tail == q->cap) {
        q->cap <<= 1;
        q->a = (int*)abamd_realloc(q->a, (size_t)q->cap * sizeof(int));
        tls_q_buf = q->a; tls_q_cap = q->cap;
    }
    q->a[q->tail++] = v;
}
static int q_pop(ab_queue_t *q, int *v) {
    if (q->head == q->tail) return 0;
    *v = q->a[q->head++]; return 1;
}
static void q_free(ab_queue_t *q) { (void)q; /* thread-local, reused */ }

/* Kahn BFS from src: assign dense topo indices; an aligned-node group enters
 * the queue together once every member's in-degree is exhausted
 * (abpoa_graph.c:221-266). */
void abpoa_BFS_set_node_index(abpoa_graph_t *g, int src_id, int sink_id) {
    int i, j, cur, index = 0;
    int *in_deg = tls_degree(g->node_n);
    for (i = 0; i < g->node_n; ++i) in_deg[i] = g->node[i].in_edge_n;
    ab_queue_t q; q_init(&q, g->node_n);
    q_push(&q, src_id);
    while (q_pop(&q, &cur)) {
        g->index_to_node_id[index] = cur;
        g->node_id_to_index[cur] = index++;
        if (cur == sink_id) { q_free(&q); return; }
        for (i = 0; i < g->node[cur].out_edge_n; ++i) {
            int out = g->node[cur].out_id[i];
            if (--in_deg[out] == 0) {
                int ready = 1;
                for (j = 0; j < g->node[out].aligned_node_n; ++j)
                    if (in_deg[g->node[out].aligned_node_id[j]] != 0) { ready = 0; break; }
                if (!ready) continue;
                q_push(&q, out);
                for (j = 0; j < g->node[out].aligned_node_n; ++j)
                    q_push(&q, g->node[out].aligned_node_id[j]);
            }
        }
    }
    abamd_fatal("abpoa_BFS_set_node_index", "graph is not a connected DAG");
}

/* reverse BFS from sink: remaining path length along max-weight out-edges,
 * used by the adaptive band (abpoa_graph.c:268-309) */
void abpoa_BFS_set_node_remain(abpoa_graph_t *g, int src_id, int sink_id) {
    int i, cur;
    int *out_deg = tls_degree(g->node_n);
    for (i = 0; i < g->node_n; ++i) {
        out_deg[i] = g->node[i].out_edge_n;
        g->node_id_to_max_remain[i] = 0;
    }
    ab_queue_t q; q_init(&q, g->node_n);
    q_push(&q, sink_id);
    g->node_id_to_max_remain[sink_id] = -1;
    while (q_pop(&q, &cur)) {
        if (cur != sink_id) {
            /* INT32_MIN sentinel, not the reference's -1: all-negative edge
             * weights (possible only via malformed FASTQ qualities) would
             * leave max_id at -1 and index out of bounds */
            int max_w = INT32_MIN, max_id = -1;
            for (i = 0; i < g->node[cur].out_edge_n; ++i) {
                if (g->node[cur].out_edge_weight[i] > max_w) {
                    max_w = g->node[cur].out_edge_weight[i];
                    max_id = g->node[cur].out_id[i];
                }
            }
            g->node_id_to_max_remain[cur] = g->node_id_to_max_remain[max_id] + 1;
        }
        if (cur == src_id) { q_free(&q); return; }
        for (i = 0; i < g->node[cur].in_edge_n; ++i) {
            int in = g->node[cur].in_id[i];
            if (--out_deg[in] == 0) q_push(&q, in);
        }
    }
    abamd_fatal("abpoa_BFS_set_node_remain", "graph is not a connected DAG");
}

/* adjacency sort by weight, descending; ties keep insertion order. The exact
 * swap pattern matters for backtrack tie-breaking, so this replicates the
 * reference's pairwise-swap pass (abpoa_graph.c:192-219). */
static void sort_adjacency(abpoa_graph_t *g) {
    int i, j, k, tmp; uint64_t *tmp_ids;
    for (i = 0; i < g->node_n; ++i) {
        abpoa_node_t *v = &g->node[i];
        for (j = 0; j < v->in_edge_n - 1; ++j)
            for (k = j + 1; k < v->in_edge_n; ++k)
                if (v->in_edge_weight[j] < v->in_edge_weight[k]) {
                    tmp = v->in_id[j]; v->in_id[j] = v->in_id[k]; v->in_id[k] = tmp;
                    tmp = v->in_edge_weight[j]; v->in_edge_weight[j] = v->in_edge_weight[k]; v->in_edge_weight[k] = tmp;
                }
        for (j = 0; j < v->out_edge_n - 1; ++j)
            for (k = j + 1; k < v->out_edge_n; ++k)
                if (v->out_edge_weight[j] < v->out_edge_weight[k]) {
                    tmp = v->out_id[j]; v->out_id[j] = v->out_id[k]; v->out_id[k] = tmp;
                    tmp = v->out_edge_weight[j]; v->out_edge_weight[j] = v->out_edge_weight[k]; v->out_edge_weight[k] = tmp;
                    if (v->read_ids_n > 0) {
                        tmp_ids = v->read_ids[j]; v->read_ids[j] = v->read_ids[k]; v->read_ids[k] = tmp_ids;
                    }
                }
    }
}

void abpoa_topological_sort(abpoa_graph_t *g, abpoa_para_t *abpt) {
    if (g->node_n <= 0) return;
    int i, n = g->node_n;
    if (n > g->index_rank_m) {
        g->index_rank_m = ab_round_up_pow2_32(n);
        g->index_to_node_id = (int*)abamd_realloc(g->index_to_node_id, (size_t)g->index_rank_m * sizeof(int));
        g->node_id_to_index = (int*)abamd_realloc(g->node_id_to_index, (size_t)g->index_rank_m * sizeof(int));
        if (abpt->out_msa || abpt->max_n_cons > 1 || abpt->cons_algrm == ABPOA_MF)
            g->node_id_to_msa_rank = (int*)abamd_realloc(g->node_id_to_msa_rank, (size_t)g->index_rank_m * sizeof(int));
        if (abpt->wb >= 0) {
            g->node_id_to_max_pos_left = (int*)abamd_realloc(g->node_id_to_max_pos_left, (size_t)g->index_rank_m * sizeof(int));
            g->node_id_to_max_pos_right = (int*)abamd_realloc(g->node_id_to_max_pos_right, (size_t)g->index_rank_m * sizeof(int));
            g->node_id_to_max_remain = (int*)abamd_realloc(g->node_id_to_max_remain, (size_t)g->index_rank_m * sizeof(int));
        } else if (abpt->zdrop > 0) {
            g->node_id_to_max_remain = (int*)abamd_realloc(g->node_id_to_max_remain, (size_t)g->index_rank_m * sizeof(int));
        }
    }
    abpoa_BFS_set_node_index(g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID);
    sort_adjacency(g);
    if (abpt->wb >= 0) {
        for (i = 0; i < n; ++i) {
            g->node_id_to_max_pos_right[i] = 0;
            g->node_id_to_max_pos_left[i] = n;
        }
        abpoa_BFS_set_node_remain(g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID);
    } else if (abpt->zdrop > 0)
        abpoa_BFS_set_node_remain(g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID);
    g->is_topological_sorted = 1;
}

/* DFS-flavoured rank assignment for row-column MSA: LIFO traversal, aligned
 * groups share one rank (abpoa_graph.c:359-419). */
static void dfs_set_msa_rank(abpoa_graph_t *g, int src_id, int sink_id, int *in_deg) {
    if (g->node_n > g->index_rank_m) {
        int m = ab_round_up_pow2_32(g->node_n);
        g->node_id_to_msa_rank = (int*)abamd_realloc(g->node_id_to_msa_rank, (size_t)m * sizeof(int));
    }
    int i, j, cur, msa_rank = 0;
    /* LIFO stack (kdq_pop pops from the tail in the reference) */
    int sp = 0, cap = 16;
    int *stack = (int*)abamd_malloc((size_t)cap * sizeof(int));
    stack[sp++] = src_id;
    g->node_id_to_msa_rank[src_id] = -1;
    while (sp > 0) {
        cur = stack[--sp];
        if (g->node_id_to_msa_rank[cur] < 0) {
            g->node_id_to_msa_rank[cur] = msa_rank;
            for (i = 0; i < g->node[cur].aligned_node_n; ++i)
                g->node_id_to_msa_rank[g->node[cur].aligned_node_id[i]] = msa_rank;
            msa_rank++;
        }
        if (cur == sink_id) { free(stack); g->is_set_msa_rank = 1; return; }
        for (i = 0; i < g->node[cur].out_edge_n; ++i) {
            int out = g->node[cur].out_id[i];
            if (--in_deg[out] == 0) {
                int ready = 1;
                for (j = 0; j < g->node[out].aligned_node_n; ++j)
                    if (in_deg[g->node[out].aligned_node_id[j]] != 0) { ready = 0; break; }
                if (!ready) continue;
                if (sp + 1 + g->node[out].aligned_node_n > cap) {
                    while (sp + 1 + g->node[out].aligned_node_n > cap) cap <<= 1;
                    stack = (int*)abamd_realloc(stack, (size_t)cap * sizeof(int));
                }
                stack[sp++] = out;
                g->node_id_to_msa_rank[out] = -1;
                for (j = 0; j < g->node[out].aligned_node_n; ++j) {
                    int a = g->node[out].aligned_node_id[j];
                    stack[sp++] = a;
                    g->node_id_to_msa_rank[a] = -1;
                }
            }
        }
    }
    abamd_fatal("dfs_set_msa_rank", "graph is not a connected DAG");
}

void abamd_set_msa_rank(abpoa_graph_t *g, int src_id, int sink_id) {
    if (g->is_set_msa_rank == 0) {
        int i, *in_deg = (int*)abamd_malloc((size_t)g->node_n * sizeof(int));
        for (i = 0; i < g->node_n; ++i) in_deg[i] = g->node[i].in_edge_n;
        dfs_set_msa_rank(g, src_id, sink_id, in_deg);
        free(in_deg);
    }
}

/* ---------------- span-read counting & sequence/alignment fold ---------------- */

static void update_n_span(abpoa_graph_t *g, int src_id, int sink_id, int inc_both_ends) {
    int i;
    int s = g->node_id_to_index[src_id], e = g->node_id_to_index[sink_id];
    for (i = s + 1; i < e; ++i) g->node[g->index_to_node_id[i]].n_span_read += 1;
    if (inc_both_ends) {
        g->node[src_id].n_span_read += 1;
        g->node[sink_id].n_span_read += 1;
    }
}

/* first read: thread the whole sequence as a simple chain
 * (abpoa_graph.c:573-593) */
static void add_first_sequence(abpoa_graph_t *g, abpoa_para_t *abpt, const uint8_t *seq, const int *weight,
                               int len, int *qpos_to_node_id, uint8_t add_read_id, uint8_t add_read_weight,
                               int read_id, int read_ids_n, int tot_read_n) {
    int i, last = ABPOA_SRC_NODE_ID;
    for (i = 0; i < len; ++i) {
        int cur = abpoa_add_graph_node(g, seq[i]);
        if (qpos_to_node_id) qpos_to_node_id[i] = cur;
        abpoa_add_graph_edge(g, last, cur, 0, weight[i], add_read_id, add_read_weight, read_id, read_ids_n, tot_read_n);
        g->node[cur].n_span_read = g->node[last].n_span_read;
        last = cur;
    }
    /* len == 0 would read weight[-1] (the reference does); use 0 instead */
    abpoa_add_graph_edge(g, last, ABPOA_SINK_NODE_ID, 0, len > 0 ? weight[len-1] : 0, add_read_id, add_read_weight, read_id, read_ids_n, tot_read_n);
    g->is_called_cons = g->is_set_msa_rank = g->is_topological_sorted = 0;
    abpoa_topological_sort(g, abpt);
    update_n_span(g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID, 1);
}

int abpoa_add_subgraph_alignment(abpoa_t *ab, abpoa_para_t *abpt, int beg_node_id, int end_node_id,
                                 uint8_t *seq, int *_weight, int seq_l, int *qpos_to_node_id,
                                 abpoa_res_t res, int read_id, int tot_read_n, int inc_both_ends) {
    abpoa_graph_t *g = ab->abg;
    int n_cigar = res.n_cigar; abpoa_cigar_t *cig = res.graph_cigar;
    int read_ids_n = 1 + ((tot_read_n - 1) >> 6);
    uint8_t add_read_id = abpt->use_read_ids, add_read_weight = (uint8_t)(abpt->use_qv & (abpt->max_n_cons > 1)), add;
    int i, j, *weight;
    if (_weight == NULL) {
        weight = (int*)abamd_malloc((size_t)seq_l * sizeof(int));
        for (i = 0; i < seq_l; ++i) weight[i] = 1;
    } else weight = _weight;

    if (g->node_n == 2) {
        add_first_sequence(g, abpt, seq, weight, seq_l, qpos_to_node_id, add_read_id, add_read_weight, read_id, read_ids_n, tot_read_n);
        if (_weight == NULL) free(weight);
        return 0;
    }
    if (g->node_n < 2) abamd_fatal("abpoa_add_subgraph_alignment", "graph node_n %d", g->node_n);
    if (n_cigar == 0) { if (_weight == NULL) free(weight); return 0; }

    int op, len, node_id, query_id = -1, last_new = 0, last_id = beg_node_id, new_id, aligned_id;
    for (i = 0; i < n_cigar; ++i) {
        op = (int)(cig[i] & 0xf);
        if (op == ABPOA_CMATCH) {
            node_id = (int)((cig[i] >> 34) & 0x3fffffff);
            query_id++;
            if (g->node[node_id].base != seq[query_id]) { /* mismatch column */
                if ((aligned_id = find_aligned_with_base(g, node_id, seq[query_id])) != -1) {
                    add = (last_id != beg_node_id || inc_both_ends) ? 1 : 0;
                    abpoa_add_graph_edge(g, last_id, aligned_id, 1 - last_new, weight[query_id], add_read_id & add, add_read_weight, read_id, read_ids_n, tot_read_n);
                    if (add == 0) g->node[last_id].n_read--;
                    last_id = aligned_id; last_new = 0;
                } else {
                    new_id = abpoa_add_graph_node(g, seq[query_id]);
                    add = (last_id != beg_node_id || inc_both_ends) ? 1 : 0;
                    abpoa_add_graph_edge(g, last_id, new_id, 0, weight[query_id], add_read_id & add, add_read_weight, read_id, read_ids_n, tot_read_n);
                    g->node[new_id].n_span_read = g->node[last_id].n_span_read;
                    if (add == 0) g->node[last_id].n_read--;
                    last_id = new_id; last_new = 1;
                    add_aligned_pair(g, node_id, new_id);
                }
            } else { /* match */
                add = (last_id != beg_node_id || inc_both_ends) ? 1 : 0;
                abpoa_add_graph_edge(g, last_id, node_id, 1 - last_new, weight[query_id], add_read_id & add, add_read_weight, read_id, read_ids_n, tot_read_n);
                if (add == 0) g->node[last_id].n_read--;
                last_id = node_id; last_new = 0;
            }
            if (qpos_to_node_id) qpos_to_node_id[query_id] = last_id;
        } else if (op == ABPOA_CINS || op == ABPOA_CSOFT_CLIP || op == ABPOA_CHARD_CLIP) {
            len = (int)((cig[i] >> 4) & 0x3fffffff);
            query_id += len;
            for (j = len - 1; j >= 0; --j) {
                new_id = abpoa_add_graph_node(g, seq[query_id - j]);
                add = (last_id != beg_node_id || inc_both_ends) ? 1 : 0;
                abpoa_add_graph_edge(g, last_id, new_id, 0, weight[query_id - j], add_read_id & add, add_read_weight, read_id, read_ids_n, tot_read_n);
                g->node[new_id].n_span_read = g->node[last_id].n_span_read;
                if (add == 0) g->node[last_id].n_read--;
                last_id = new_id; last_new = 1;
                if (qpos_to_node_id) qpos_to_node_id[query_id - j] = last_id;
            }
        } /* ABPOA_CDEL: consumes nothing on the query side */
    }
    abpoa_add_graph_edge(g, last_id, end_node_id, 1 - last_new, seq_l > 0 ? weight[seq_l - 1] : 0, add_read_id, add_read_weight, read_id, read_ids_n, tot_read_n);
    g->is_called_cons = g->is_set_msa_rank = g->is_topological_sorted = 0;
    abpoa_topological_sort(g, abpt);
    update_n_span(g, beg_node_id, end_node_id, inc_both_ends);
    if (_weight == NULL) free(weight);
    return 0;
}

int abpoa_add_graph_alignment(abpoa_t *ab, abpoa_para_t *abpt, uint8_t *seq, int *weight, int seq_l,
                              int *qpos_to_node_id, abpoa_res_t res, int read_id, int tot_read_n, int inc_both_ends) {
    return abpoa_add_subgraph_alignment(ab, abpt, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID,
                                        seq, weight, seq_l, qpos_to_node_id, res, read_id, tot_read_n, inc_both_ends);
}

/* ---------------- reset ---------------- */

void abpoa_reset(abpoa_t *ab, abpoa_para_t *abpt, int qlen) {
    abpoa_graph_t *g = ab->abg;
    int i, j, k;
    g->is_topological_sorted = g->is_called_cons = 0;
    for (i = 0; i < g->node_n; ++i) {
        for (j = 0; j < g->node[i].out_edge_n; ++j)
            for (k = 0; k < g->node[i].read_ids_n; ++k) g->node[i].read_ids[j][k] = 0;
        g->node[i].in_edge_n = g->node[i].out_edge_n = g->node[i].aligned_node_n = 0;
        g->node[i].n_read = 0; g->node[i].n_span_read = 0;
    }
    g->node_n = 2;
    if (qlen + 2 > g->node_m) {
        int node_m = ab_round_up_pow2_32(qlen + 2);
        g->node = (abpoa_node_t*)abamd_realloc(g->node, (size_t)node_m * sizeof(abpoa_node_t));
        for (i = g->node_m; i < node_m; ++i) init_node(&g->node[i], i);
        g->node_m = g->index_rank_m = node_m;
        g->index_to_node_id = (int*)abamd_realloc(g->index_to_node_id, (size_t)node_m * sizeof(int));
        g->node_id_to_index = (int*)abamd_realloc(g->node_id_to_index, (size_t)node_m * sizeof(int));
        if (abpt->out_msa || abpt->max_n_cons > 1 || abpt->cons_algrm == ABPOA_MF)
            g->node_id_to_msa_rank = (int*)abamd_realloc(g->node_id_to_msa_rank, (size_t)node_m * sizeof(int));
        if (abpt->wb >= 0) {
            g->node_id_to_max_pos_left = (int*)abamd_realloc(g->node_id_to_max_pos_left, (size_t)node_m * sizeof(int));
            g->node_id_to_max_pos_right = (int*)abamd_realloc(g->node_id_to_max_pos_right, (size_t)node_m * sizeof(int));
            g->node_id_to_max_remain = (int*)abamd_realloc(g->node_id_to_max_remain, (size_t)node_m * sizeof(int));
        } else if (abpt->zdrop > 0)
            g->node_id_to_max_remain = (int*)abamd_realloc(g->node_id_to_max_remain, (size_t)node_m * sizeof(int));
    }
    ab->abs->n_seq = 0;
    cons_clear(ab->abc);
}

/* ---------------- subgraph window expansion (abpoa_graph.c:595-678) ---------------- */

static int full_upstream(abpoa_graph_t *g, int up_index, int down_index, int beg_index, int end_index) {
    int i, j;
    int min_index = AB_MIN2(up_index, beg_index);
    int max_index = AB_MAX2(down_index, end_index);
    for (i = up_index + 1; i <= down_index; ++i) {
        int id = g->index_to_node_id[i];
        for (j = 0; j < g->node[id].in_edge_n; ++j) {
            int ii = g->node_id_to_index[g->node[id].in_id[j]];
            if (ii < min_index || ii > max_index) return 0;
        }
    }
    return 1;
}

static int upstream_index(abpoa_graph_t *g, int beg_index, int end_index) {
    for (;;) {
        int min_index = beg_index, i, j;
        for (i = beg_index; i <= end_index; ++i) {
            int id = g->index_to_node_id[i];
            for (j = 0; j < g->node[id].in_edge_n; ++j)
                min_index = AB_MIN2(min_index, g->node_id_to_index[g->node[id].in_id[j]]);
        }
        if (full_upstream(g, min_index, beg_index, beg_index, end_index)) return min_index;
        end_index = beg_index; beg_index = min_index;
    }
}

static int downstream_index(abpoa_graph_t *g, int beg_index, int end_index) {
    for (;;) {
        int max_index = end_index, i, j;
        for (i = beg_index; i <= end_index; ++i) {
            int id = g->index_to_node_id[i];
            for (j = 0; j < g->node[id].out_edge_n; ++j)
                max_index = AB_MAX2(max_index, g->node_id_to_index[g->node[id].out_id[j]]);
        }
        if (full_upstream(g, end_index, max_index, beg_index, end_index)) return max_index;
        beg_index = end_index; end_index = max_index;
    }
}

void abpoa_subgraph_nodes(abpoa_t *ab, abpoa_para_t *abpt, int inc_beg, int inc_end, int *exc_beg, int *exc_end) {
    abpoa_graph_t *g = ab->abg;
    if (g->is_topological_sorted == 0) abpoa_topological_sort(g, abpt);
    int b = g->node_id_to_index[inc_beg], e = g->node_id_to_index[inc_end];
    int eb = upstream_index(g, b, e), ee = downstream_index(g, b, e);
    if (eb < 0 || ee >= g->node_n) abamd_fatal("abpoa_subgraph_nodes", "bad subgraph window");
    *exc_beg = g->index_to_node_id[eb];
    *exc_end = g->index_to_node_id[ee];
}

/* ---------------- flat-graph import (device-resident batch path) ----------------
 *
 * Rebuild the pointer graph from a host copy of a device-resident flat graph
 * (abamd_fold_core.h) after the GPU has folded every read of a set. The flat
 * fold is CPU-twin-proven bit-equal to the pointer fold (tests/test_fold_twin,
 * device-validated by the gpu fold test), so walking its chains in order
 * reproduces the exact adjacency order - including the weight-sort order the
 * device applied - and the final abpoa_topological_sort here re-derives the
 * index arrays with the identical algorithm, leaving the graph byte-equal to
 * what the host fold path would have produced.
 *
 * ALL node-level arrays are carved from ONE malloc per set (returned to the
 * caller): with ~50k nodes x 4 arrays x 1000 sets, per-array malloc would be
 * ~200M allocator calls and serializes a 200-thread consensus pool on the
 * allocator lock (measured: 12 s of a 34 s step). The caller must NOT
 * abpoa_free() the graph while the slab is live - it releases the slab with
 * abamd_graph_arena_release(), which restores a normally-allocated empty
 * graph first. `ab` must be fresh (abpoa_init, untouched). read_ids_n > 0
 * copies per-out-edge read bitsets (fg->rid_n must equal read_ids_n). */
#include "abamd_fold_core.h"

void *abamd_graph_from_flat(abpoa_t *ab, const flat_graph_t *fg, abpoa_para_t *abpt, int read_ids_n,
                            const int *i2n, const int *n2i, const int *remain) {
    abpoa_graph_t *g = ab->abg;
    int id, e;
    if (read_ids_n > 0 && fg->rid_n != read_ids_n)
        abamd_fatal("abamd_graph_from_flat", "rid_n %d != read_ids_n %d", fg->rid_n, read_ids_n);
    const size_t nn = (size_t)fg->node_n;
    const size_t n_in = (size_t)fg->edge_n_in, n_out = (size_t)fg->edge_n_out;
    const size_t n_aln = (size_t)fg->aln_n;
    size_t sz_nodes = nn * sizeof(abpoa_node_t);
    size_t sz_in = 2 * n_in * sizeof(int);
    size_t sz_out = 2 * n_out * sizeof(int);
    size_t sz_aln = n_aln * sizeof(int);
    size_t sz_ridp = read_ids_n > 0 ? n_out * sizeof(uint64_t*) : 0;
    size_t sz_rid = read_ids_n > 0 ? n_out * (size_t)read_ids_n * sizeof(uint64_t) : 0;
    size_t a = 0;
    #define AL16(x) (((x) + 15) & ~(size_t)15)
    size_t o_nodes = 0;                 a = AL16(sz_nodes);
    size_t o_in = a;                    a += AL16(sz_in);
    size_t o_out = a;                   a += AL16(sz_out);
    size_t o_aln = a;                   a += AL16(sz_aln);
    size_t o_ridp = a;                  a += AL16(sz_ridp);
    size_t o_rid = a;                   a += AL16(sz_rid);
    uint8_t *slab = (uint8_t*)abamd_malloc(a ? a : 16);
    abpoa_node_t *nodes = (abpoa_node_t*)(slab + o_nodes);
    int *in_pool = (int*)(slab + o_in);
    int *out_pool = (int*)(slab + o_out);
    int *aln_pool = (int*)(slab + o_aln);
    uint64_t **ridp_pool = (uint64_t**)(slab + o_ridp);
    uint64_t *rid_pool = (uint64_t*)(slab + o_rid);
    size_t c_in = 0, c_out = 0, c_aln = 0;
    for (id = 0; id < fg->node_n; ++id) {
        abpoa_node_t *v = &nodes[id];
        memset(v, 0, sizeof(*v));
        v->node_id = id;
        v->base = fg->base[id];
        v->n_read = fg->n_read[id];
        v->n_span_read = fg->n_span_read[id];
        /* in edges, chain order */
        v->in_id = in_pool + c_in * 2; /* ids then weights, contiguous per node */
        int cnt = 0;
        for (e = fg->in_head[id]; e != -1; e = fg->in_next[e]) v->in_id[cnt++] = fg->in_to[e];
        v->in_edge_weight = v->in_id + cnt;
        cnt = 0;
        for (e = fg->in_head[id]; e != -1; e = fg->in_next[e]) v->in_edge_weight[cnt++] = fg->in_w[e];
        v->in_edge_n = v->in_edge_m = cnt;
        c_in += (size_t)cnt;
        /* out edges (+ read-id bitsets travelling with each edge) */
        v->out_id = out_pool + c_out * 2;
        cnt = 0;
        for (e = fg->out_head[id]; e != -1; e = fg->out_next[e]) v->out_id[cnt++] = fg->out_to[e];
        v->out_edge_weight = v->out_id + cnt;
        cnt = 0;
        for (e = fg->out_head[id]; e != -1; e = fg->out_next[e]) v->out_edge_weight[cnt++] = fg->out_w[e];
        if (read_ids_n > 0 && cnt > 0) {
            v->read_ids = ridp_pool + c_out;
            int k = 0;
            for (e = fg->out_head[id]; e != -1; e = fg->out_next[e]) {
                v->read_ids[k] = rid_pool + (c_out + (size_t)k) * read_ids_n;
                memcpy(v->read_ids[k], fg->rid_pool + (size_t)e * fg->rid_n,
                       (size_t)read_ids_n * sizeof(uint64_t));
                ++k;
            }
            v->read_ids_n = read_ids_n;
        }
        v->out_edge_n = v->out_edge_m = cnt;
        c_out += (size_t)cnt;
        /* aligned (mismatch-column) group, chain order */
        v->aligned_node_id = aln_pool + c_aln;
        cnt = 0;
        for (e = fg->aln_head[id]; e != -1; e = fg->aln_next[e]) v->aligned_node_id[cnt++] = fg->aln_id[e];
        v->aligned_node_n = v->aligned_node_m = cnt;
        c_aln += (size_t)cnt;
    }
    /* swap the slab-carved node array in; the original (tiny) array is freed
     * here and restored by abamd_graph_arena_release */
    for (id = 0; id < g->node_m; ++id) free_node_arrays(&g->node[id]);
    free(g->node);
    g->node = nodes;
    g->node_n = fg->node_n;
    g->node_m = fg->node_n;
    g->is_topological_sorted = g->is_called_cons = g->is_set_msa_rank = 0;
    abpoa_topological_sort(g, abpt);
    if (i2n && n2i) {
        /* The reference's index arrays come from the BFS over the PRE-sort
         * adjacency of the LAST fold; the re-derivation above ran over the
         * already-sorted adjacency, which can order ready groups
         * differently. Restore the faithful arrays the device computed in
         * the live pass order (the sort itself is idempotent, so the
         * adjacency state is untouched). */
        memcpy(g->index_to_node_id, i2n, (size_t)fg->node_n * sizeof(int));
        memcpy(g->node_id_to_index, n2i, (size_t)fg->node_n * sizeof(int));
        if (remain && g->node_id_to_max_remain)
            memcpy(g->node_id_to_max_remain, remain, (size_t)fg->node_n * sizeof(int));
    }
    return slab;
}

/* undo abamd_graph_from_flat's slab carve so abpoa_free() is valid again */
void abamd_graph_arena_release(abpoa_t *ab, void *slab) {
    abpoa_graph_t *g = ab->abg;
    g->node = (abpoa_node_t*)abamd_calloc(2, sizeof(abpoa_node_t));
    init_node(&g->node[0], ABPOA_SRC_NODE_ID);
    init_node(&g->node[1], ABPOA_SINK_NODE_ID);
    g->node_n = g->node_m = 2;
    free(slab);
}
