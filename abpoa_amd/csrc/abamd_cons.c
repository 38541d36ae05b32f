/* Consensus extraction and result emission.
 *   heaviest-bundle consensus:  abpoa_output.c:476-548 (tie-breaks preserved)
 *   majority-vote consensus:    abpoa_output.c:394-452, 550-587
 *   RC-MSA:                     abpoa_output.c:106-193
 *   GFA:                        abpoa_output.c:196-295
 *   FASTA/FASTQ emission:       abpoa_output.c:297-303, 589-628
 * Multi-consensus clustering (k-medoids, -d >= 2) is not in this build yet;
 * requesting it aborts with a clear message.
 */
#include <math.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

void abamd_set_msa_rank(abpoa_graph_t *g, int src_id, int sink_id);

#define ABAMD_NAT_E 2.718281828459045

static int cons_phred_score(int n_cov, int n_seq) {
    if (n_cov > n_seq) abamd_fatal("cons_phred_score", "n_cov %d > n_seq %d", n_cov, n_seq);
    double x = 13.8 * (1.25 * n_cov / n_seq - 0.25);
    double p = 1 - 1.0 / (1.0 + pow(ABAMD_NAT_E, -1 * x));
    return 33 + (int)(-10 * log10(p) + 0.499);
}

static abpoa_cons_t *allocate_cons(abpoa_cons_t *c, int n_node, int n_seq, int n_cons) {
    int i;
    c->n_cons = n_cons; c->n_seq = n_seq;
    c->clu_n_seq = (int*)abamd_calloc(n_cons, sizeof(int));
    c->cons_len = (int*)abamd_calloc(n_cons, sizeof(int));
    c->cons_node_ids = (int**)abamd_malloc((size_t)n_cons * sizeof(int*));
    c->cons_base = (uint8_t**)abamd_malloc((size_t)n_cons * sizeof(uint8_t*));
    c->cons_cov = (int**)abamd_malloc((size_t)n_cons * sizeof(int*));
    c->clu_read_ids = (int**)abamd_malloc((size_t)n_cons * sizeof(int*));
    c->cons_phred_score = (int**)abamd_malloc((size_t)n_cons * sizeof(int*));
    for (i = 0; i < n_cons; ++i) {
        c->cons_node_ids[i] = (int*)abamd_malloc((size_t)n_node * sizeof(int));
        c->cons_base[i] = (uint8_t*)abamd_malloc((size_t)n_node * sizeof(uint8_t));
        c->cons_cov[i] = (int*)abamd_malloc((size_t)n_node * sizeof(int));
        c->clu_read_ids[i] = (int*)abamd_malloc((size_t)(n_seq > 0 ? n_seq : 1) * sizeof(int));
        c->cons_phred_score[i] = (int*)abamd_malloc((size_t)n_node * sizeof(int));
    }
    return c;
}

/* single-cluster edge weight is just the stored weight (abpoa_output.c:330-336) */
static int edge_weight(const abpoa_node_t *v, int edge_i) { return v->out_edge_weight[edge_i]; }

/* Reverse-BFS heaviest-bundle walk. Per node choose the out-edge with max
 * weight; ties prefer the later-scanned edge when its downstream score is >=
 * (abpoa_output.c:521-535); the src node instead prefers strictly higher
 * weight with score as secondary (:507-517). */
static void heaviest_bundling(abpoa_graph_t *g, abpoa_para_t *abpt, int src_id, int sink_id,
                              const int *out_degree, abpoa_cons_t *abc) {
    (void)abpt;
    int i, cur;
    int n = g->node_n;
    int *deg = (int*)abamd_malloc((size_t)n * sizeof(int));
    int *score = (int*)abamd_malloc((size_t)n * sizeof(int));
    int *max_out_id = (int*)abamd_malloc((size_t)n * sizeof(int));
    abc->clu_n_seq[0] = abc->n_seq;
    for (i = 0; i < abc->n_seq; ++i) abc->clu_read_ids[0][i] = i;

    for (i = 0; i < n; ++i) deg[i] = out_degree[i];
    int cap = n, sp_head = 0, sp_tail = 0;
    int *q = (int*)abamd_malloc((size_t)cap * sizeof(int));
    q[sp_tail++] = sink_id;
    while (sp_head < sp_tail) {
        cur = q[sp_head++];
        if (cur == sink_id) {
            max_out_id[cur] = -1;
            score[cur] = 0;
        } else if (cur == src_id) {
            int max_id = -1, path_score = -1, path_max_w = -1;
            for (i = 0; i < g->node[cur].out_edge_n; ++i) {
                int out_id = g->node[cur].out_id[i];
                int w = edge_weight(&g->node[cur], i);
                if (w > path_max_w || (w == path_max_w && score[out_id] > path_score)) {
                    max_id = out_id; path_score = score[out_id]; path_max_w = w;
                }
            }
            max_out_id[cur] = max_id;
            break;
        } else {
            int max_id = -1, max_w = INT32_MIN;
            for (i = 0; i < g->node[cur].out_edge_n; ++i) {
                int out_id = g->node[cur].out_id[i];
                int w = edge_weight(&g->node[cur], i);
                if (max_w < w) { max_w = w; max_id = out_id; }
                else if (max_w == w && score[max_id] <= score[out_id]) max_id = out_id;
            }
            score[cur] = max_w + score[max_id];
            max_out_id[cur] = max_id;
        }
        for (i = 0; i < g->node[cur].in_edge_n; ++i) {
            int in_id = g->node[cur].in_id[i];
            if (--deg[in_id] == 0) q[sp_tail++] = in_id;
        }
    }
    /* walk src -> sink through the chosen edges (abpoa_set_hb_cons, :376-392) */
    int j = 0;
    cur = max_out_id[src_id];
    while (cur != sink_id) {
        abc->cons_node_ids[0][j] = cur;
        abc->cons_base[0][j] = g->node[cur].base;
        abc->cons_cov[0][j] = g->node[cur].n_read;
        abc->cons_phred_score[0][j] = cons_phred_score(abc->cons_cov[0][j], abc->clu_n_seq[0]);
        ++j;
        cur = max_out_id[cur];
    }
    abc->cons_len[0] = j;
    abc->n_cons = 1;
    free(deg); free(score); free(max_out_id); free(q);
}

/* node msa rank = max over its aligned group (abpoa_output.c:136-144) */
static int group_msa_rank(abpoa_graph_t *g, int id) {
    int k, rank = g->node_id_to_msa_rank[id];
    for (k = 0; k < g->node[id].aligned_node_n; ++k) {
        int a = g->node[id].aligned_node_id[k];
        rank = AB_MAX2(rank, g->node_id_to_msa_rank[a]);
    }
    return rank;
}

/* majority-vote consensus over MSA columns (abpoa_output.c:394-452, 550-587) */
static void most_frequent(abpoa_graph_t *g, abpoa_para_t *abpt, int src_id, int sink_id, abpoa_cons_t *abc) {
    int use_span = abpt->sub_aln;
    abamd_set_msa_rank(g, src_id, sink_id);
    int m = abpt->m, i, j;
    int msa_l = g->node_id_to_msa_rank[sink_id] - 1;
    int *rc_weight = (int*)abamd_calloc((size_t)msa_l * m, sizeof(int));
    int *msa_node_id = (int*)abamd_calloc((size_t)msa_l * m, sizeof(int));
    for (i = 0; i < msa_l; ++i) rc_weight[i * m + m - 1] = abc->n_seq;
    abc->n_cons = 1;
    abc->clu_n_seq[0] = abc->n_seq;
    for (i = 0; i < abc->n_seq; ++i) abc->clu_read_ids[0][i] = i;

    /* per-column per-base weights; the gap count starts at n_seq and is
     * decremented by every base weight placed in the column (:427-452) */
    for (i = 2; i < g->node_n; ++i) {
        int rank = group_msa_rank(g, i);
        int node_w = g->node[i].n_read; /* n_clu==1: out coverage = n_read */
        msa_node_id[(rank - 1) * m + g->node[i].base] = i;
        rc_weight[(rank - 1) * m + g->node[i].base] = node_w;
        rc_weight[(rank - 1) * m + m - 1] -= node_w;
    }
    int cons_l = 0;
    for (i = 0; i < msa_l; ++i) {
        int max_c = 0, total_c = 0, max_base = m, gap_c, c;
        for (j = 0; j < m - 1; ++j) {
            c = rc_weight[i * m + j];
            if (c > max_c) { max_c = c; max_base = j; }
            total_c += c;
        }
        if (use_span) gap_c = g->node[msa_node_id[i * m + max_base]].n_span_read - total_c;
        else gap_c = abc->clu_n_seq[0] - total_c;
        if (max_c >= gap_c) {
            int cur_id = msa_node_id[i * m + max_base];
            abc->cons_node_ids[0][cons_l] = cur_id;
            abc->cons_base[0][cons_l] = (uint8_t)max_base;
            abc->cons_cov[0][cons_l] = max_c;
            abc->cons_phred_score[0][cons_l] = cons_phred_score(max_c, abc->clu_n_seq[0]);
            cons_l++;
        }
    }
    abc->cons_len[0] = cons_l;
    free(rc_weight); free(msa_node_id);
}

void abpoa_generate_consensus(abpoa_t *ab, abpoa_para_t *abpt) {
    if (ab->abg->is_called_cons == 1) return;
    abpoa_graph_t *g = ab->abg;
    if (g->node_n <= 2) return;
    if (abpt->max_n_cons > 1)
        abamd_fatal("abpoa_generate_consensus", "multi-consensus clustering (-d >= 2) is not implemented in abpoa_amd yet");
    int i, *out_degree = (int*)abamd_malloc((size_t)g->node_n * sizeof(int));
    for (i = 0; i < g->node_n; ++i) out_degree[i] = g->node[i].out_edge_n;
    abpoa_cons_t *abc = ab->abc;
    allocate_cons(abc, g->node_n, ab->abs->n_seq, 1);
    if (abpt->cons_algrm == ABPOA_HB)
        heaviest_bundling(g, abpt, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID, out_degree, abc);
    else
        most_frequent(g, abpt, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID, abc);
    g->is_called_cons = 1;
    free(out_degree);
}

void abpoa_output_fx_consensus(abpoa_t *ab, abpoa_para_t *abpt, FILE *fp) {
    if (!fp) return;
    int ci, j;
    abpoa_cons_t *abc = ab->abc;
    for (ci = 0; ci < abc->n_cons; ++ci) {
        fprintf(fp, "%cConsensus_sequence", abpt->out_fq ? '@' : '>');
        if (abpt->batch_index > 0) fprintf(fp, "_%d", abpt->batch_index);
        if (abc->n_cons > 1) {
            fprintf(fp, "_%d ", ci + 1);
            for (j = 0; j < abc->clu_n_seq[ci]; ++j)
                fprintf(fp, j ? ",%d" : "%d", abc->clu_read_ids[ci][j]);
        }
        fputc('\n', fp);
        for (j = 0; j < abc->cons_len[ci]; ++j) fputc(ab_amd_char256_table[abc->cons_base[ci][j]], fp);
        fputc('\n', fp);
        if (abpt->out_fq) {
            fprintf(fp, "+Consensus_sequence");
            if (abpt->batch_index > 0) fprintf(fp, "_%d", abpt->batch_index);
            if (abc->n_cons > 1) {
                fprintf(fp, "_%d ", ci + 1);
                for (j = 0; j < abc->clu_n_seq[ci]; ++j)
                    fprintf(fp, j ? ",%d" : "%d", abc->clu_read_ids[ci][j]);
            }
            fputc('\n', fp);
            for (j = 0; j < abc->cons_len[ci]; ++j) fputc((char)abc->cons_phred_score[ci][j], fp);
            fputc('\n', fp);
        }
    }
}

/* place every node's base into its read rows at the node's msa column
 * (abpoa_set_msa_seq, abpoa_output.c:106-123) */
static void msa_place_node_full(const abpoa_node_t *v, int rank, uint8_t **msa_base) {
    int i, j, b = 0;
    for (i = 0; i < v->read_ids_n; ++i) {
        for (j = 0; j < v->out_edge_n; ++j) {
            uint64_t num = v->read_ids[j][i];
            while (num) {
                uint64_t low = num & (~num + 1);
                int read_id = ab_amd_ilog2_64(low);
                msa_base[b + read_id][rank - 1] = v->base;
                num ^= low;
            }
        }
        b += 64;
    }
}

void abpoa_generate_rc_msa(abpoa_t *ab, abpoa_para_t *abpt) {
    abpoa_graph_t *g = ab->abg;
    if (g->node_n <= 2) return;
    abamd_set_msa_rank(g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID);
    if (abpt->out_cons) abpoa_generate_consensus(ab, abpt);
    abpoa_seq_t *abs = ab->abs; abpoa_cons_t *abc = ab->abc;
    int i, j, n_seq = abs->n_seq;
    int msa_len = g->node_id_to_msa_rank[ABPOA_SINK_NODE_ID] - 1;
    abc->n_seq = n_seq; abc->msa_len = msa_len;
    abc->msa_base = (uint8_t**)abamd_malloc((size_t)(n_seq + abc->n_cons) * sizeof(uint8_t*));
    for (i = 0; i < n_seq + abc->n_cons; ++i)
        abc->msa_base[i] = (uint8_t*)abamd_malloc((size_t)msa_len);
    for (i = 0; i < n_seq; ++i)
        for (j = 0; j < msa_len; ++j) abc->msa_base[i][j] = (uint8_t)abpt->m;
    for (i = 2; i < g->node_n; ++i)
        msa_place_node_full(&g->node[i], group_msa_rank(g, i), abc->msa_base);
    if (abpt->out_cons) {
        int ci;
        for (ci = 0; ci < abc->n_cons; ++ci) {
            for (i = 0; i < msa_len; ++i) abc->msa_base[n_seq + ci][i] = (uint8_t)abpt->m;
            for (i = 0; i < abc->cons_len[ci]; ++i) {
                int cur_id = abc->cons_node_ids[ci][i];
                abc->msa_base[n_seq + ci][group_msa_rank(g, cur_id) - 1] = abc->cons_base[ci][i];
            }
        }
    }
}

void abpoa_output_rc_msa(abpoa_t *ab, abpoa_para_t *abpt, FILE *fp) {
    if (!fp) return;
    int i, j;
    abpoa_seq_t *abs = ab->abs; abpoa_cons_t *abc = ab->abc;
    if (abc->msa_len <= 0) return;
    for (i = 0; i < abs->n_seq; ++i) {
        if (abs->name[i].l > 0)
            fprintf(fp, abs->is_rc[i] ? ">%s_reverse_complement\n" : ">%s\n", abs->name[i].s);
        else fprintf(fp, ">Seq_%d\n", i + 1);
        for (j = 0; j < abc->msa_len; ++j) fputc(ab_amd_char256_table[abc->msa_base[i][j]], fp);
        fputc('\n', fp);
    }
    if (abpt->out_cons) {
        int ci;
        for (ci = 0; ci < abc->n_cons; ++ci) {
            fprintf(fp, ">Consensus_sequence");
            if (abc->n_cons > 1) {
                fprintf(fp, "_%d ", ci + 1);
                for (j = 0; j < abc->clu_n_seq[ci]; ++j)
                    fprintf(fp, j ? ",%d" : "%d", abc->clu_read_ids[ci][j]);
            }
            fputc('\n', fp);
            for (i = 0; i < abc->msa_len; ++i) fputc(ab_amd_char256_table[abc->msa_base[abc->n_seq + ci][i]], fp);
            fputc('\n', fp);
        }
    }
}

void abpoa_generate_gfa(abpoa_t *ab, abpoa_para_t *abpt, FILE *fp) {
    if (!fp) return;
    abpoa_seq_t *abs = ab->abs; abpoa_graph_t *g = ab->abg;
    if (g->node_n <= 2) return;
    int n_seq = abs->n_seq;
    int *in_deg = (int*)abamd_malloc((size_t)g->node_n * sizeof(int));
    int **read_paths = (int**)abamd_malloc((size_t)n_seq * sizeof(int*));
    int *read_path_i = (int*)abamd_calloc(n_seq, sizeof(int));
    int i, j;
    for (i = 0; i < g->node_n; ++i) in_deg[i] = g->node[i].in_edge_n;
    for (i = 0; i < n_seq; ++i) read_paths[i] = (int*)abamd_malloc((size_t)g->node_n * sizeof(int));

    int nl = 0;
    for (i = 2; i < g->node_n; ++i) nl += g->node[i].in_edge_n;
    fprintf(fp, "H\tVN:Z:1.0\tNS:i:%d\tNL:i:%d\tNP:i:%d\n", g->node_n - 2,
            nl - g->node[ABPOA_SRC_NODE_ID].out_edge_n, n_seq + abpt->out_cons);

    int cap = g->node_n, head = 0, tail = 0;
    int *q = (int*)abamd_malloc((size_t)cap * sizeof(int));
    q[tail++] = ABPOA_SRC_NODE_ID;
    while (head < tail) {
        int cur = q[head++];
        if (cur == ABPOA_SINK_NODE_ID) break;
        if (cur != ABPOA_SRC_NODE_ID) {
            fprintf(fp, "S\t%d\t%c\n", cur - 1, ab_amd_char256_table[g->node[cur].base]);
            for (i = 0; i < g->node[cur].in_edge_n; ++i) {
                int pre = g->node[cur].in_id[i];
                if (pre != ABPOA_SRC_NODE_ID)
                    fprintf(fp, "L\t%d\t+\t%d\t+\t0M\n", pre - 1, cur - 1);
            }
            int b = 0;
            for (i = 0; i < g->node[cur].read_ids_n; ++i) {
                for (j = 0; j < g->node[cur].out_edge_n; ++j) {
                    uint64_t num = g->node[cur].read_ids[j][i];
                    while (num) {
                        uint64_t low = num & (~num + 1);
                        int rid = ab_amd_ilog2_64(low);
                        read_paths[b + rid][read_path_i[b + rid]++] = cur - 1;
                        num ^= low;
                    }
                }
                b += 64;
            }
        }
        for (i = 0; i < g->node[cur].out_edge_n; ++i) {
            int out = g->node[cur].out_id[i];
            if (--in_deg[out] == 0) q[tail++] = out;
        }
    }
    for (i = 0; i < n_seq; ++i) {
        if (abs->name[i].l > 0) fprintf(fp, "P\t%s\t", abs->name[i].s);
        else fprintf(fp, "P\t%d\t", i + 1);
        if (abs->is_rc[i]) {
            for (j = read_path_i[i] - 1; j >= 0; --j)
                fprintf(fp, j != 0 ? "%d-," : "%d-\t*\n", read_paths[i][j]);
        } else {
            for (j = 0; j < read_path_i[i]; ++j)
                fprintf(fp, j != read_path_i[i] - 1 ? "%d+," : "%d+\t*\n", read_paths[i][j]);
        }
    }
    if (abpt->out_cons) {
        abpoa_generate_consensus(ab, abpt);
        abpoa_cons_t *abc = ab->abc;
        int ci;
        for (ci = 0; ci < abc->n_cons; ++ci) {
            fprintf(fp, "P\tConsensus_sequence");
            if (abc->n_cons > 1) fprintf(fp, "_%d", ci + 1);
            fputc('\t', fp);
            for (i = 0; i < abc->cons_len[ci]; ++i)
                fprintf(fp, i != abc->cons_len[ci] - 1 ? "%d+," : "%d+\t*\n", abc->cons_node_ids[ci][i] - 1);
        }
    }
    free(in_deg); free(q);
    for (i = 0; i < n_seq; ++i) free(read_paths[i]);
    free(read_paths); free(read_path_i);
}

void abpoa_output(abpoa_t *ab, abpoa_para_t *abpt, FILE *fp) {
    if (abpt->out_gfa) abpoa_generate_gfa(ab, abpt, fp);
    else {
        if (abpt->out_msa) abpoa_generate_rc_msa(ab, abpt);
        if (abpt->out_cons) {
            abpoa_generate_consensus(ab, abpt);
            if (ab->abg->is_called_cons == 0)
                fprintf(stderr, "Warning: no consensus sequence generated.\n");
        }
        if (abpt->out_msa) abpoa_output_rc_msa(ab, abpt, fp);
        else if (abpt->out_cons) abpoa_output_fx_consensus(ab, abpt, fp);
    }
    if (abpt->out_pog)
        fprintf(stderr, "[abpoa_amd] graph plotting (--out-pog) is not implemented in this build; skipping.\n");
}

void abpoa_dump_pog(abpoa_t *ab, abpoa_para_t *abpt) {
    (void)ab; (void)abpt;
    fprintf(stderr, "[abpoa_amd] abpoa_dump_pog is not implemented in this build.\n");
}
