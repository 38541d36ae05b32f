/* Fold-twin test driver: replays every read of a FASTA through BOTH the
 * pointer-based graph fold (abamd_graph.c) and the flat-array device-layout
 * core (abamd_fold_core.c), comparing the full structure after every read:
 * node bases, per-node in/out adjacency (order, ids, weights), per-out-edge
 * read-id bitsets, aligned-group lists, n_read and qpos->node maps.
 *
 * Built against gpu_stub.c so the aligner is the injected oracle
 * (ABPOA_AMD_TEST_ALIGNER_SO). Prints "twin OK (<reads> reads, <nodes>
 * nodes)" on success; aborts on the first structural difference. */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include "abpoa_amd.h"
#include "abamd_util.h"
#include "abamd_fold_core.h"

typedef struct abamd_fx_t abamd_fx_t;
abpoa_seq_t *abamd_seq_new(void);
void abamd_seq_destroy(abpoa_seq_t *abs);
abamd_fx_t *abamd_fx_open(const char *fn);
void abamd_fx_close(abamd_fx_t *x);
int abamd_read_seq(abpoa_seq_t *abs, abamd_fx_t *x);

static void die(const char *what, int read_i, int node) {
    fprintf(stderr, "TWIN MISMATCH after read %d at node %d: %s\n", read_i, node, what);
    exit(1);
}

static void compare(abpoa_graph_t *g, flat_graph_t *fg, int read_i, int rid_n) {
    int i, j, e, a;
    if (g->node_n != fg->node_n) die("node_n", read_i, -1);
    for (i = 0; i < g->node_n; ++i) {
        abpoa_node_t *v = &g->node[i];
        if (v->base != fg->base[i]) die("base", read_i, i);
        if (v->n_read != fg->n_read[i]) die("n_read", read_i, i);
        if (v->n_span_read != fg->n_span_read[i]) die("n_span_read", read_i, i);
        /* out edges in order */
        for (j = 0, e = fg->out_head[i]; j < v->out_edge_n; ++j, e = fg->out_next[e]) {
            if (e == -1) die("out edge count (flat short)", read_i, i);
            if (fg->out_to[e] != v->out_id[j]) die("out id order", read_i, i);
            if (fg->out_w[e] != v->out_edge_weight[j]) die("out weight", read_i, i);
            if (rid_n > 0 && v->read_ids_n > 0) {
                int k;
                for (k = 0; k < rid_n; ++k)
                    if (fg->rid_pool[(size_t)e * rid_n + k] != v->read_ids[j][k])
                        die("read-id bitset", read_i, i);
            }
        }
        if (e != -1) die("out edge count (flat long)", read_i, i);
        /* in edges in order */
        for (j = 0, e = fg->in_head[i]; j < v->in_edge_n; ++j, e = fg->in_next[e]) {
            if (e == -1) die("in edge count (flat short)", read_i, i);
            if (fg->in_to[e] != v->in_id[j]) die("in id order", read_i, i);
            if (fg->in_w[e] != v->in_edge_weight[j]) die("in weight", read_i, i);
        }
        if (e != -1) die("in edge count (flat long)", read_i, i);
        /* aligned groups in order */
        for (j = 0, a = fg->aln_head[i]; j < v->aligned_node_n; ++j, a = fg->aln_next[a]) {
            if (a == -1) die("aligned count (flat short)", read_i, i);
            if (fg->aln_id[a] != v->aligned_node_id[j]) die("aligned order", read_i, i);
        }
        if (a != -1) die("aligned count (flat long)", read_i, i);
    }
}

int main(int argc, char **argv) {
    if (argc < 2) { fprintf(stderr, "usage: %s reads.fa [-r1]\n", argv[0]); return 2; }
    abpoa_para_t *abpt = abpoa_init_para();
    if (argc > 2 && strcmp(argv[2], "-r1") == 0) abpt->out_msa = 1; /* turn read-id tracking on */
    abpoa_post_set_para(abpt);
    abpoa_t *ab = abpoa_init();
    abpoa_reset(ab, abpt, 1024);

    abpoa_seq_t *abs = abamd_seq_new();
    abamd_fx_t *fx = abamd_fx_open(argv[1]);
    int n_seq = abamd_read_seq(abs, fx);
    abamd_fx_close(fx);
    ab->abs->n_seq = n_seq;

    int total_len = 0, max_len = 0, i, j;
    for (i = 0; i < n_seq; ++i) {
        total_len += abs->seq[i].l;
        if (abs->seq[i].l > max_len) max_len = abs->seq[i].l;
    }
    int rid_n = abpt->use_read_ids ? 1 + ((n_seq - 1) >> 6) : 0;
    flat_graph_t fg;
    abamd_flat_init(&fg, total_len + 2, 4 * total_len + 64, 8 * total_len + 1024, rid_n);

    int *w = (int*)abamd_malloc((size_t)max_len * sizeof(int));
    int *qmap_live = (int*)abamd_malloc((size_t)max_len * sizeof(int));
    int *qmap_flat = (int*)abamd_malloc((size_t)max_len * sizeof(int));
    for (i = 0; i < max_len; ++i) w[i] = 1;

    uint8_t *codes = (uint8_t*)abamd_malloc((size_t)max_len);
    for (i = 0; i < n_seq; ++i) {
        int qlen = abs->seq[i].l;
        for (j = 0; j < qlen; ++j) codes[j] = (uint8_t)ab_amd_char26_table[(int)abs->seq[i].s[j]];
        abpoa_res_t res; memset(&res, 0, sizeof(res));
        abpoa_align_sequence_to_graph(ab, abpt, codes, qlen, &res);
        abamd_flat_apply_alignment(&fg, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID, codes, w, qlen,
                                   qmap_flat, res.n_cigar, res.graph_cigar, i,
                                   abpt->use_read_ids, 1);
        abpoa_add_graph_alignment(ab, abpt, codes, w, qlen, qmap_live, res, i, n_seq, 1);
        if (res.n_cigar) free(res.graph_cigar);
        for (j = 0; j < qlen; ++j)
            if (qmap_live[j] != qmap_flat[j]) die("qpos_to_node_id", i, j);
        {   /* derived passes in the live order (abpoa_topological_sort):
             * index BFS on the pre-sort adjacency, THEN the weight sort,
             * then remain BFS — only after that does the structural
             * compare see the same post-sort adjacency the live graph has */
            int n = fg.node_n;
            int *i2n = (int*)abamd_malloc((size_t)n * sizeof(int));
            int *n2i = (int*)abamd_malloc((size_t)n * sizeof(int));
            int *rem = (int*)abamd_malloc((size_t)n * sizeof(int));
            int *scr = (int*)abamd_malloc((size_t)2 * n * sizeof(int));
            abamd_flat_topo_index(&fg, i2n, n2i, scr);
            abamd_flat_sort_adjacency(&fg);
            abamd_flat_remain(&fg, rem, scr);
            abamd_flat_update_n_span(&fg, i2n, n2i, 1);
            compare(ab->abg, &fg, i, rid_n);
            if (abpt->out_msa) { /* msa rank over the post-sort adjacency;
                 * the live array only exists when MSA output is on */
                int *mr = (int*)abamd_malloc((size_t)n * sizeof(int));
                int k;
                abamd_flat_msa_rank(&fg, mr, scr);
                ab->abg->is_set_msa_rank = 0;
                abamd_set_msa_rank(ab->abg, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID);
                for (k = 0; k < n; ++k)
                    if (mr[k] != ab->abg->node_id_to_msa_rank[k]) die("msa rank", i, k);
                free(mr);
            }
            {   /* DP-row CSR: flat materializer vs the pointer graph,
                 * mirroring pack_job's whole-graph loops (gpu_align.cpp) */
                abpoa_graph_t *pg = ab->abg;
                int nr = pg->node_id_to_index[ABPOA_SINK_NODE_ID] + 1;
                int ecap = fg.edge_n_out + 4;
                uint8_t *fb = (uint8_t*)abamd_malloc((size_t)nr);
                int *fni = (int*)abamd_malloc((size_t)nr * sizeof(int));
                int *fpo = (int*)abamd_malloc((size_t)(nr + 1) * sizeof(int));
                int *foo = (int*)abamd_malloc((size_t)(nr + 1) * sizeof(int));
                int *frm = (int*)abamd_malloc((size_t)nr * sizeof(int));
                int *fpi = (int*)abamd_malloc((size_t)ecap * sizeof(int));
                int *foi = (int*)abamd_malloc((size_t)ecap * sizeof(int));
                int flat_nr = abamd_flat_build_rows(&fg, i2n, n2i, rem, 1,
                                                    fb, fni, fpo, foo, frm, fpi, foi);
                if (flat_nr != nr) die("csr n_rows", i, flat_nr);
                int r, e2, np = 0, no = 0;
                for (r = 0; r < nr; ++r) {
                    int nid = pg->index_to_node_id[r];
                    if (fb[r] != pg->node[nid].base) die("csr base", i, r);
                    if (fni[r] != nid) die("csr node id", i, r);
                    if (frm[r] != pg->node_id_to_max_remain[nid]) die("csr remain", i, r);
                    if (fpo[r] != np || foo[r] != no) die("csr offsets", i, r);
                    if (r > 0)
                        for (e2 = 0; e2 < pg->node[nid].in_edge_n; ++e2) {
                            if (fpi[np] != pg->node_id_to_index[pg->node[nid].in_id[e2]])
                                die("csr pre idx", i, r);
                            ++np;
                        }
                    for (e2 = 0; e2 < pg->node[nid].out_edge_n; ++e2) {
                        if (foi[no] != pg->node_id_to_index[pg->node[nid].out_id[e2]])
                            die("csr out idx", i, r);
                        ++no;
                    }
                }
                if (fpo[nr] != np || foo[nr] != no) die("csr totals", i, nr);
                free(fb); free(fni); free(fpo); free(foo); free(frm); free(fpi); free(foi);
            }
            abpoa_graph_t *g = ab->abg;
            /* the BFS returns when the SINK pops (reference quirk): nodes
             * still queued keep stale indices, and the DP only reads the
             * topo prefix 0..sink_index — compare exactly that */
            for (j = 0; j < n; ++j) {
                int id;
                if (i2n[j] != g->index_to_node_id[j]) die("topo index order", i, j);
                id = i2n[j];
                if (n2i[id] != g->node_id_to_index[id]) die("node->index", i, id);
                if (g->node_id_to_max_remain && rem[id] != g->node_id_to_max_remain[id])
                    die("max_remain", i, id);
                if (id == ABPOA_SINK_NODE_ID) break;
            }
            {   /* flat -> pointer import (the resident driver's consensus
                 * hand-off, abamd_graph_from_flat): must reproduce the live
                 * graph field-for-field, topo arrays included */
                abpoa_t *ab2 = abpoa_init();
                void *slab = abamd_graph_from_flat(ab2, &fg, abpt, rid_n, i2n, n2i, rem);
                abpoa_graph_t *ga = ab->abg, *gb = ab2->abg;
                int id2, k2, sink_idx = ga->node_id_to_index[ABPOA_SINK_NODE_ID];
                if (ga->node_n != gb->node_n) die("import node_n", i, -1);
                for (id2 = 0; id2 < ga->node_n; ++id2) {
                    abpoa_node_t *va = &ga->node[id2], *vb = &gb->node[id2];
                    if (va->base != vb->base || va->n_read != vb->n_read ||
                        va->n_span_read != vb->n_span_read) die("import node fields", i, id2);
                    if (va->in_edge_n != vb->in_edge_n || va->out_edge_n != vb->out_edge_n ||
                        va->aligned_node_n != vb->aligned_node_n) die("import counts", i, id2);
                    for (k2 = 0; k2 < va->in_edge_n; ++k2)
                        if (va->in_id[k2] != vb->in_id[k2] ||
                            va->in_edge_weight[k2] != vb->in_edge_weight[k2])
                            die("import in edges", i, id2);
                    for (k2 = 0; k2 < va->out_edge_n; ++k2) {
                        if (va->out_id[k2] != vb->out_id[k2] ||
                            va->out_edge_weight[k2] != vb->out_edge_weight[k2])
                            die("import out edges", i, id2);
                        if (rid_n > 0 && va->read_ids_n > 0) {
                            int k3;
                            if (vb->read_ids_n != rid_n) die("import read_ids_n", i, id2);
                            for (k3 = 0; k3 < rid_n; ++k3)
                                if (va->read_ids[k2][k3] != vb->read_ids[k2][k3])
                                    die("import read bitsets", i, id2);
                        }
                    }
                    for (k2 = 0; k2 < va->aligned_node_n; ++k2)
                        if (va->aligned_node_id[k2] != vb->aligned_node_id[k2])
                            die("import aligned", i, id2);
                }
                for (k2 = 0; k2 <= sink_idx; ++k2) {
                    if (ga->index_to_node_id[k2] != gb->index_to_node_id[k2])
                        die("import topo", i, k2);
                    id2 = ga->index_to_node_id[k2];
                    if (ga->node_id_to_index[id2] != gb->node_id_to_index[id2])
                        die("import n2i", i, id2);
                    if (ga->node_id_to_max_remain && gb->node_id_to_max_remain &&
                        ga->node_id_to_max_remain[id2] != gb->node_id_to_max_remain[id2])
                        die("import remain", i, id2);
                }
                abamd_graph_arena_release(ab2, slab);
                abpoa_free(ab2);
            }
            {   /* flat heaviest-bundle consensus (abamd_cons_core.inc, the
                 * round-3 device-consensus core) vs the live pointer-graph
                 * consensus on the same state */
                int *score = (int*)abamd_malloc((size_t)n * sizeof(int));
                int *mout = (int*)abamd_malloc((size_t)n * sizeof(int));
                int *cid = (int*)abamd_malloc((size_t)n * sizeof(int));
                uint8_t *cb = (uint8_t*)abamd_malloc((size_t)n);
                int *cc = (int*)abamd_malloc((size_t)n * sizeof(int));
                int *cp = (int*)abamd_malloc((size_t)n * sizeof(int));
                int clen = abamd_flat_hb_consensus(&fg, n_seq, scr, score, mout,
                                                   cid, cb, cc, cp);
                abpoa_clean_msa_cons(ab);
                ab->abg->is_called_cons = 0;
                abpoa_generate_consensus(ab, abpt);
                abpoa_cons_t *abc = ab->abc;
                int k4;
                if (abc->n_cons != 1 || abc->cons_len[0] != clen)
                    die("hb cons length", i, clen);
                for (k4 = 0; k4 < clen; ++k4) {
                    if (cid[k4] != abc->cons_node_ids[0][k4]) die("hb cons node", i, k4);
                    if (cb[k4] != abc->cons_base[0][k4]) die("hb cons base", i, k4);
                    if (cc[k4] != abc->cons_cov[0][k4]) die("hb cons cov", i, k4);
                    if (cp[k4] != abc->cons_phred_score[0][k4]) die("hb cons phred", i, k4);
                }
                abpoa_clean_msa_cons(ab);
                ab->abg->is_called_cons = 0;
                free(score); free(mout); free(cid); free(cb); free(cc); free(cp);
            }
            free(i2n); free(n2i); free(rem); free(scr);
        }
    }
    printf("twin OK (%d reads, %d nodes, %d out-edges, %d aligned entries)\n",
           n_seq, fg.node_n, fg.edge_n_out, fg.aln_n);
    abamd_flat_free(&fg);
    return 0;
}
