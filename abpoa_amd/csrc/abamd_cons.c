/* Consensus extraction and result emission.
 *   heaviest-bundle consensus:  abpoa_output.c:476-548 (tie-breaks preserved)
 *   majority-vote consensus:    abpoa_output.c:394-452, 550-587
 *   multi-consensus clustering: abpoa_output.c:650-1182 (candidate het
 *     positions from the MSA matrix -> per-pair distance matrix ->
 *     k-medoids -> per-cluster read-id bitsets -> cluster-filtered
 *     consensus walks; exact tie-breaks and iteration orders preserved)
 *   RC-MSA:                     abpoa_output.c:106-193
 *   GFA:                        abpoa_output.c:196-295
 *   FASTA/FASTQ emission:       abpoa_output.c:297-303, 589-628
 */
#include <limits.h>
#include <math.h>
#include <string.h>
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

/* ---------------- multi-consensus read clustering (-d >= 2) -------------
 * Restates abpoa_output.c:650-1182. The distance space is the candidate
 * heterozygous MSA columns; reads are clustered around k medoids and each
 * cluster gets its own consensus walk with read-filtered edge weights. */

static int group_msa_rank(abpoa_graph_t *g, int id);
static void msa_place_node_full(const abpoa_node_t *v, int rank, uint8_t **msa_base);

typedef struct {
    int pos, depth, var_type, count, n_uniq_alles;
    int *n_clu_reads;          /* [m+1] reads carrying each allele */
    int **clu_read_ids;        /* [m+1][...] read ids per allele */
    int *read_id_to_allele_idx;/* [n_seq] -> allele index or -1 */
    uint8_t *alle_bases;       /* [n_uniq_alles] allele codes, first-seen order */
} het_pos_t;

/* full per-read MSA matrix, gap code = m (abpoa_collect_msa, :125-148) */
static int collect_msa_matrix(abpoa_graph_t *g, abpoa_para_t *abpt, uint8_t **msa, int n_seq) {
    if (g->node_n <= 2) return 0;
    abamd_set_msa_rank(g, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID);
    int msa_l = g->node_id_to_msa_rank[ABPOA_SINK_NODE_ID] - 1;
    int i, j;
    for (i = 0; i < n_seq; ++i) {
        msa[i] = (uint8_t*)abamd_malloc((size_t)msa_l);
        for (j = 0; j < msa_l; ++j) msa[i][j] = (uint8_t)abpt->m;
    }
    for (i = 2; i < g->node_n; ++i)
        msa_place_node_full(&g->node[i], group_msa_rank(g, i), msa);
    return msa_l;
}

/* a candidate position whose allele->read partition already appeared is
 * only counted, scanning newest-first (allele_clu_exist, :650-674) */
static int het_clu_exists(het_pos_t *hp, int n_het, int n_uniq, const int *alleles,
                          const int *n_clu_reads, int *const *clu_read_ids) {
    int i, j, k;
    for (i = n_het - 1; i >= 0; --i) {
        if (hp[i].n_uniq_alles != n_uniq) continue;
        int exist = 1;
        for (j = 0; j < n_uniq; ++j) {
            int ax = hp[i].alle_bases[j], ay = alleles[j];
            if (hp[i].n_clu_reads[ax] != n_clu_reads[ay]) { exist = 0; break; }
            for (k = 0; k < n_clu_reads[ay]; ++k)
                if (hp[i].clu_read_ids[ax][k] != clu_read_ids[ay][k]) { exist = 0; break; }
            if (!exist) break;
        }
        if (exist) return i;
    }
    return -1;
}

/* candidate het columns: >= 2 alleles each covering [min_het, n_seq-min_het]
 * reads; alleles ordered by first appearance; priorities bubble-sorted by
 * (count desc, depth desc, SNP-before-indel) (abpoa_collect_cand_het_pos,
 * :676-829) */
static int collect_cand_het_pos(uint8_t **msa, int msa_l, int n_seq, int m, int min_het,
                                het_pos_t *hp, int *prio) {
    int n_het = 0, i, j, k;
    min_het = AB_MAX2(2, min_het / 2);
    int min_hom = n_seq - min_het;
    int *depth = (int*)abamd_malloc((size_t)(m + 1) * sizeof(int));
    int *alleles = (int*)abamd_malloc((size_t)(m + 1) * sizeof(int));
    int *allele_to_idx = (int*)abamd_malloc((size_t)(m + 1) * sizeof(int));
    int *first_seen = (int*)abamd_malloc((size_t)(m + 1) * sizeof(int));
    int *n_clu_reads = (int*)abamd_malloc((size_t)(m + 1) * sizeof(int));
    int **clu_read_ids = (int**)abamd_malloc((size_t)(m + 1) * sizeof(int*));
    for (j = 0; j < m + 1; ++j) clu_read_ids[j] = (int*)abamd_malloc((size_t)n_seq * sizeof(int));
    for (i = 0; i < msa_l; ++i) {
        int n_uniq = 0, var_type = 0, total_depth = 0;
        memset(depth, 0, (size_t)(m + 1) * sizeof(int));
        memset(n_clu_reads, 0, (size_t)(m + 1) * sizeof(int));
        for (j = 0; j < n_seq; ++j) {
            if (++depth[msa[j][i]] == 1) first_seen[msa[j][i]] = j;
        }
        for (j = 0; j < m + 1; ++j) {
            if (depth[j] >= min_het && depth[j] <= min_hom) {
                alleles[n_uniq++] = j;
                total_depth += depth[j];
                if (j == m) var_type = 1; /* gap allele => indel column */
            }
        }
        if (n_uniq < 2) continue;
        for (j = 0; j < n_uniq - 1; ++j)
            for (k = j + 1; k < n_uniq; ++k)
                if (first_seen[alleles[j]] > first_seen[alleles[k]]) {
                    int t = alleles[j]; alleles[j] = alleles[k]; alleles[k] = t;
                }
        for (j = 0; j < n_uniq; ++j) allele_to_idx[alleles[j]] = j;
        for (j = 0; j < n_seq; ++j)
            for (k = 0; k < n_uniq; ++k)
                if (msa[j][i] == alleles[k]) {
                    clu_read_ids[alleles[k]][n_clu_reads[alleles[k]]++] = j;
                    break;
                }
        int het_i = het_clu_exists(hp, n_het, n_uniq, alleles, n_clu_reads, clu_read_ids);
        if (het_i >= 0) {
            hp[het_i].count++;
            if (var_type == 0) hp[het_i].var_type = 0; /* SNP wins if any column is a SNP */
            continue;
        }
        hp[n_het].pos = i;
        hp[n_het].depth = total_depth;
        hp[n_het].var_type = var_type;
        hp[n_het].count = 1;
        hp[n_het].n_uniq_alles = n_uniq;
        hp[n_het].n_clu_reads = (int*)abamd_malloc((size_t)(m + 1) * sizeof(int));
        hp[n_het].clu_read_ids = (int**)abamd_malloc((size_t)(m + 1) * sizeof(int*));
        hp[n_het].read_id_to_allele_idx = (int*)abamd_malloc((size_t)n_seq * sizeof(int));
        for (j = 0; j < n_seq; ++j) hp[n_het].read_id_to_allele_idx[j] = -1;
        for (j = 0; j < m + 1; ++j) {
            hp[n_het].n_clu_reads[j] = n_clu_reads[j];
            hp[n_het].clu_read_ids[j] = (int*)abamd_malloc((size_t)n_seq * sizeof(int));
            memcpy(hp[n_het].clu_read_ids[j], clu_read_ids[j], (size_t)n_clu_reads[j] * sizeof(int));
            for (k = 0; k < n_clu_reads[j]; ++k)
                hp[n_het].read_id_to_allele_idx[clu_read_ids[j][k]] = allele_to_idx[j];
        }
        hp[n_het].alle_bases = (uint8_t*)abamd_malloc((size_t)n_uniq);
        for (j = 0; j < n_uniq; ++j) hp[n_het].alle_bases[j] = (uint8_t)alleles[j];
        prio[n_het] = n_het;
        n_het++;
    }
    for (i = 0; i < n_het; ++i) prio[i] = i;
    int swapped;
    do {
        swapped = 0;
        for (j = 0; j < n_het - 1; ++j) {
            int a = prio[j], b = prio[j + 1];
            if (hp[a].count < hp[b].count ||
                (hp[a].count == hp[b].count && hp[a].depth < hp[b].depth) ||
                (hp[a].count == hp[b].count && hp[a].depth == hp[b].depth && hp[a].var_type > hp[b].var_type)) {
                prio[j] = b; prio[j + 1] = a; swapped = 1;
            }
        }
    } while (swapped);
    for (j = 0; j < m + 1; ++j) free(clu_read_ids[j]);
    free(clu_read_ids); free(n_clu_reads); free(alleles); free(allele_to_idx);
    free(first_seen); free(depth);
    return n_het;
}

/* pairwise read distance over het columns only: SNP columns weigh 2x, gap
 * columns 1x, scaled by the column's collapse count; positions where either
 * read carries a non-candidate allele are skipped (:824-864) */
static int **collect_dis_matrix(uint8_t **msa, int n_seq, const het_pos_t *hp, int n_het) {
    int i, j, k, a;
    int **dm = (int**)abamd_malloc((size_t)n_seq * sizeof(int*));
    for (i = 0; i < n_seq; ++i) dm[i] = (int*)abamd_calloc(n_seq, sizeof(int));
    for (i = 0; i < n_seq; ++i)
        for (j = i + 1; j < n_seq; ++j) {
            int dis = 0;
            for (k = 0; k < n_het; ++k) {
                int pos = hp[k].pos;
                int w = hp[k].var_type == 0 ? 2 : 1;
                uint8_t b1 = msa[i][pos], b2 = msa[j][pos];
                int v1 = 0, v2 = 0;
                for (a = 0; a < hp[k].n_uniq_alles; ++a) {
                    if (b1 == hp[k].alle_bases[a]) v1 = 1;
                    if (b2 == hp[k].alle_bases[a]) v2 = 1;
                }
                if (!v1 || !v2) continue;
                if (b1 != b2) dis += w * hp[k].count;
            }
            dm[i][j] = dm[j][i] = dis;
        }
    return dm;
}

/* first two medoids: the max-distance read pair across the top het column's
 * allele clusters (:866-891) */
static int collect_2medoids(const het_pos_t *hp, int het_i, int **dm, int *med) {
    int max_dis = 0, max_i = -1, max_j = -1, i, j, r1, r2;
    for (i = 0; i < hp[het_i].n_uniq_alles - 1; ++i) {
        int ai = hp[het_i].alle_bases[i];
        for (j = i + 1; j < hp[het_i].n_uniq_alles; ++j) {
            int aj = hp[het_i].alle_bases[j];
            for (r1 = 0; r1 < hp[het_i].n_clu_reads[ai]; ++r1) {
                int ri = hp[het_i].clu_read_ids[ai][r1];
                for (r2 = 0; r2 < hp[het_i].n_clu_reads[aj]; ++r2) {
                    int rj = hp[het_i].clu_read_ids[aj][r2];
                    if (dm[ri][rj] > max_dis) { max_dis = dm[ri][rj]; max_i = ri; max_j = rj; }
                }
            }
        }
    }
    if (max_dis > 0) { med[0] = max_i; med[1] = max_j; return 2; }
    return 0;
}

/* partition index of a read over het columns 0..het_i (:893-902) */
static int partition_index(const het_pos_t *hp, int het_i, int read_i) {
    int k, idx = 0;
    for (k = 0; k <= het_i; ++k) {
        idx *= hp[k].n_uniq_alles + 1;
        idx += hp[k].read_id_to_allele_idx[read_i] + 1;
    }
    return idx;
}

/* add one medoid: prefer a read from a partition no medoid occupies (most
 * populated partition, then max of min-distance-to-medoids); else the
 * max-min-distance read among the top column's allele clusters (:904-971) */
static int collect_1medoid(const het_pos_t *hp, int het_i, int **dm, int n_seq,
                           int *med, int n_medoids) {
    int n_partitions = 1, k, i, j;
    for (k = 0; k <= het_i; ++k) n_partitions *= hp[k].n_uniq_alles + 1;
    int *pcount = (int*)abamd_calloc(n_partitions, sizeof(int));
    for (i = 0; i < n_seq; ++i) pcount[partition_index(hp, het_i, i)] += 1;
    int max_dis = 0, max_read_i = -1, max_ptotal = -1;
    for (i = 0; i < n_seq; ++i) {
        int found = 0;
        int pi = partition_index(hp, het_i, i);
        int ptotal = pcount[pi];
        for (j = 0; j < n_medoids; ++j)
            if (pi == partition_index(hp, het_i, med[j])) { found = 1; break; }
        if (!found) {
            int min_dis = INT_MAX;
            for (j = 0; j < n_medoids; ++j)
                if (dm[i][med[j]] < min_dis) min_dis = dm[i][med[j]];
            if (ptotal > max_ptotal || (ptotal == max_ptotal && min_dis > max_dis)) {
                max_dis = min_dis; max_read_i = i; max_ptotal = ptotal;
            }
        }
    }
    if (max_read_i == -1) {
        for (i = 0; i < hp[het_i].n_uniq_alles; ++i) {
            int allele = hp[het_i].alle_bases[i];
            int r;
            for (r = 0; r < hp[het_i].n_clu_reads[allele]; ++r) {
                int ri = hp[het_i].clu_read_ids[allele][r];
                int min_dis = INT_MAX, skip = 0;
                for (j = 0; j < n_medoids; ++j) {
                    if (med[j] == ri) { skip = 1; continue; }
                    if (dm[ri][med[j]] < min_dis) min_dis = dm[ri][med[j]];
                }
                if (min_dis > max_dis && !skip) { max_dis = min_dis; max_read_i = ri; }
            }
        }
    }
    free(pcount);
    if (max_read_i != -1) { med[n_medoids] = max_read_i; return 1; }
    return 0;
}

static int collect_multi_medoids(const het_pos_t *hp, int het_i, int **dm, int n_seq,
                                 int max_n_cons, int *med, int n_medoids) {
    int n_to_collect = AB_MIN2(hp[het_i].n_uniq_alles, max_n_cons);
    while (n_medoids < n_to_collect) {
        int got = n_medoids == 0 ? collect_2medoids(hp, het_i, dm, med)
                                 : collect_1medoid(hp, het_i, dm, n_seq, med, n_medoids);
        if (got == 0) break;
        n_medoids += got;
    }
    return n_medoids;
}

static int init_kmedoids(const het_pos_t *hp, const int *prio, int n_het, int **dm,
                         int n_seq, int max_n_cons, int *med) {
    int n_medoids = 0, het_i = 0;
    while (n_medoids < max_n_cons) {
        if (n_medoids == 0)
            n_medoids += collect_multi_medoids(hp, prio[het_i], dm, n_seq, max_n_cons, med, n_medoids);
        else
            n_medoids += collect_1medoid(hp, prio[het_i], dm, n_seq, med, n_medoids);
        het_i++;
        if (het_i >= n_het) break;
    }
    return n_medoids;
}

/* re-pick each cluster's medoid = member with min summed distance, then sort
 * medoids ascending (:1003-1029) */
static void kmedoids_repick(int **dm, int max_n_cons, int **clu_reads,
                            const int *n_clu_seqs, int *med) {
    int i, j, k;
    for (i = 0; i < max_n_cons; ++i) {
        int best = INT_MAX, best_read = -1;
        for (j = 0; j < n_clu_seqs[i]; ++j) {
            int sum = 0, ri = clu_reads[i][j];
            for (k = 0; k < n_clu_seqs[i]; ++k) {
                if (j == k) continue;
                sum += dm[ri][clu_reads[i][k]];
            }
            if (sum < best) { best = sum; best_read = ri; }
        }
        if (best_read != -1) med[i] = best_read;
    }
    for (i = 0; i < max_n_cons - 1; ++i)
        for (j = i + 1; j < max_n_cons; ++j)
            if (med[i] > med[j]) { int t = med[i]; med[i] = med[j]; med[j] = t; }
}

/* one assignment + medoid-update round; ties between medoids send the read
 * to the currently smaller of clusters 0/1, exactly as the reference does
 * even for k > 2 (:1031-1087) */
static int kmedoids_update(int **dm, int n_seq, int max_n_cons, int **med,
                           int **clu_reads, int *n_clu_seqs) {
    int i, j;
    int *new_med = (int*)abamd_malloc((size_t)max_n_cons * sizeof(int));
    for (i = 0; i < max_n_cons; ++i) new_med[i] = -1;
    memset(n_clu_seqs, 0, (size_t)max_n_cons * sizeof(int));
    for (i = 0; i < n_seq; ++i) {
        int min_dis = INT_MAX, min_clu = -1, tied = 0;
        for (j = 0; j < max_n_cons; ++j) {
            if (dm[i][(*med)[j]] < min_dis) { min_dis = dm[i][(*med)[j]]; min_clu = j; tied = 0; }
            else if (dm[i][(*med)[j]] == min_dis) tied = 1;
        }
        if (min_clu == -1) continue;
        if (tied == 1) min_clu = n_clu_seqs[0] < n_clu_seqs[1] ? 0 : 1;
        clu_reads[min_clu][n_clu_seqs[min_clu]++] = i;
    }
    kmedoids_repick(dm, max_n_cons, clu_reads, n_clu_seqs, new_med);
    int changed = 0;
    for (i = 0; i < max_n_cons; ++i) {
        if (new_med[i] == -1) { changed = 0; break; } /* empty cluster: stop */
        if (new_med[i] != (*med)[i]) changed = 1;
    }
    free(*med);
    *med = new_med;
    return changed;
}

/* iterate k-medoids, shrinking k while a cluster stays below min_het reads
 * or <80% of reads are clustered (:1089-1134) */
static int clu_reads_kmedoids(const het_pos_t *hp, const int *prio, int n_het, int **dm,
                              int n_seq, int min_het, int max_n_cons, uint64_t ***clu_read_ids) {
    int i, j;
    int *med = (int*)abamd_malloc((size_t)max_n_cons * sizeof(int));
    int **clu_reads = (int**)abamd_malloc((size_t)max_n_cons * sizeof(int*));
    int *n_clu_seqs = (int*)abamd_malloc((size_t)max_n_cons * sizeof(int));
    for (i = 0; i < max_n_cons; ++i) clu_reads[i] = (int*)abamd_malloc((size_t)n_seq * sizeof(int));
    int to_collect = max_n_cons, n_clusters = 1;
    while (1) {
        if (init_kmedoids(hp, prio, n_het, dm, n_seq, to_collect, med) <= 0) break;
        int iter = 0;
        while (1) {
            int changed = kmedoids_update(dm, n_seq, to_collect, &med, clu_reads, n_clu_seqs);
            if (changed == 0 || ++iter >= 10) break;
        }
        int n_clu = 0, n_clustered = 0;
        for (i = 0; i < to_collect; ++i) {
            if (n_clu_seqs[i] >= min_het) n_clu++;
            n_clustered += n_clu_seqs[i];
        }
        if (n_clu != to_collect || n_clustered < (int)ceil(n_seq * 0.8)) {
            if (--to_collect < 2) break;
        } else { n_clusters = n_clu; break; }
    }
    if (n_clusters != 1) {
        int rid_n = (n_seq - 1) / 64 + 1;
        *clu_read_ids = (uint64_t**)abamd_malloc((size_t)n_clusters * sizeof(uint64_t*));
        for (i = 0; i < n_clusters; ++i) {
            (*clu_read_ids)[i] = (uint64_t*)abamd_calloc(rid_n, sizeof(uint64_t));
            for (j = 0; j < n_clu_seqs[i]; ++j) {
                int ri = clu_reads[i][j];
                (*clu_read_ids)[i][ri / 64] |= 1ull << (ri & 0x3f);
            }
        }
    }
    for (i = 0; i < max_n_cons; ++i) free(clu_reads[i]);
    free(clu_reads); free(n_clu_seqs); free(med);
    return n_clusters;
}

/* MSA -> het positions -> distances -> k-medoids clusters
 * (abpoa_multip_read_clu_kmedoids, :1136-1182) */
static int multip_read_clu(abpoa_graph_t *g, abpoa_para_t *abpt, int n_seq,
                           uint64_t ***clu_read_ids) {
    int i, j, n_clu;
    uint8_t **msa = (uint8_t**)abamd_malloc((size_t)n_seq * sizeof(uint8_t*));
    int msa_l = collect_msa_matrix(g, abpt, msa, n_seq);
    int min_w = AB_MAX2(2, (int)ceil(n_seq * abpt->min_freq));
    het_pos_t *hp = (het_pos_t*)abamd_malloc((size_t)msa_l * sizeof(het_pos_t));
    int *prio = (int*)abamd_malloc((size_t)msa_l * sizeof(int));
    int n_het = collect_cand_het_pos(msa, msa_l, n_seq, abpt->m, min_w, hp, prio);
    if (n_het < 1) n_clu = 1;
    else {
        int **dm = collect_dis_matrix(msa, n_seq, hp, n_het);
        n_clu = clu_reads_kmedoids(hp, prio, n_het, dm, n_seq, min_w, abpt->max_n_cons, clu_read_ids);
        for (i = 0; i < n_seq; ++i) free(dm[i]);
        free(dm);
    }
    for (i = 0; i < n_het; ++i) {
        free(hp[i].alle_bases);
        for (j = 0; j < abpt->m + 1; ++j) free(hp[i].clu_read_ids[j]);
        free(hp[i].clu_read_ids); free(hp[i].n_clu_reads); free(hp[i].read_id_to_allele_idx);
    }
    free(hp); free(prio);
    for (i = 0; i < n_seq; ++i) free(msa[i]);
    free(msa);
    return n_clu;
}

/* ---------------- cluster-filtered edge weights (:305-374) ---------------- */

static int edge_inclu_read_count(const abpoa_node_t *v, int edge_i, int cons_i,
                                 uint64_t *const *clu_read_ids) {
    int n = 0, i;
    for (i = 0; i < v->read_ids_n; ++i)
        n += ab_amd_popcnt64(v->read_ids[edge_i][i] & clu_read_ids[cons_i][i]);
    return n;
}

static int inclu_edge_weight(const abpoa_node_t *v, int edge_i, int cons_i,
                             uint64_t *const *clu_read_ids, int use_qv) {
    if (use_qv == 0) return edge_inclu_read_count(v, edge_i, cons_i, clu_read_ids);
    int w = 0, i;
    for (i = 0; i < v->m_read; ++i) {
        if (v->read_weight[i] > 0) {
            uint64_t c = v->read_ids[edge_i][i / 64] & clu_read_ids[cons_i][i / 64];
            if (c & (1ull << (i & 0x3f))) w += v->read_weight[i];
        }
    }
    return w;
}

/* single-cluster edge weight is just the stored weight (abpoa_output.c:330-336) */
static int edge_weight_clu(const abpoa_node_t *v, int edge_i, int cons_i,
                           uint64_t *const *clu_read_ids, int use_qv, int n_clu) {
    if (n_clu == 1) return v->out_edge_weight[edge_i];
    return inclu_edge_weight(v, edge_i, cons_i, clu_read_ids, use_qv);
}

static int node_out_cov(const abpoa_node_t *nodes, int id, uint64_t *const *clu_read_ids,
                        int cons_i, int n_cons) {
    if (n_cons == 1) return nodes[id].n_read;
    int i, cov = 0;
    for (i = 0; i < nodes[id].out_edge_n; ++i)
        cov += edge_inclu_read_count(&nodes[id], i, cons_i, clu_read_ids);
    return cov;
}

/* NOTE: replicates the reference exactly, including its in-cov loop bound of
 * nodes[0].in_edge_n (the SRC node, in-degree 0), so the in-side term is
 * always 0 and multi-cluster coverage is effectively the out-side coverage
 * (abpoa_node_in_cov/abpoa_node_cov, :356-374). */
static int node_cov_clu(const abpoa_node_t *nodes, int id, uint64_t *const *clu_read_ids,
                        int cons_i, int n_cons) {
    if (n_cons == 1) return nodes[id].n_read;
    int i, j, in_cov = 0;
    for (i = 0; i < nodes[0].in_edge_n; ++i) {
        int in_id = nodes[id].in_id[i];
        for (j = 0; j < nodes[in_id].out_edge_n; ++j)
            if (nodes[in_id].out_id[j] == id) {
                in_cov += edge_inclu_read_count(&nodes[in_id], j, cons_i, clu_read_ids);
                break;
            }
    }
    int out_cov = node_out_cov(nodes, id, clu_read_ids, cons_i, n_cons);
    return AB_MAX2(in_cov, out_cov);
}

/* Reverse-BFS heaviest-bundle walk, one pass per cluster. Per node choose
 * the out-edge with max (cluster-filtered) weight; ties prefer the
 * later-scanned edge when its downstream score is >= (abpoa_output.c:521-535);
 * the src node instead prefers strictly higher weight with score as secondary
 * (:507-517). */
static void heaviest_bundling(abpoa_graph_t *g, abpoa_para_t *abpt, int src_id, int sink_id,
                              const int *out_degree, int n_clu, int read_ids_n,
                              uint64_t **clu_read_ids, abpoa_cons_t *abc) {
    int i, cur, cons_i;
    int n = g->node_n;
    int *deg = (int*)abamd_malloc((size_t)n * sizeof(int));
    int *score = (int*)abamd_malloc((size_t)n * sizeof(int));
    int **max_out_id = (int**)abamd_malloc((size_t)n_clu * sizeof(int*));
    for (i = 0; i < n_clu; ++i) max_out_id[i] = (int*)abamd_malloc((size_t)n * sizeof(int));
    if (n_clu == 1) {
        abc->clu_n_seq[0] = abc->n_seq;
        for (i = 0; i < abc->n_seq; ++i) abc->clu_read_ids[0][i] = i;
    } else {
        for (cons_i = 0; cons_i < n_clu; ++cons_i) {
            int cnt = 0, r;
            for (i = 0; i < read_ids_n; ++i) cnt += ab_amd_popcnt64(clu_read_ids[cons_i][i]);
            abc->clu_n_seq[cons_i] = cnt;
            for (r = 0, cnt = 0; r < abc->n_seq; ++r)
                if (clu_read_ids[cons_i][r / 64] & (1ull << (r & 0x3f)))
                    abc->clu_read_ids[cons_i][cnt++] = r;
        }
    }

    int cap = n;
    int *q = (int*)abamd_malloc((size_t)cap * sizeof(int));
    for (cons_i = 0; cons_i < n_clu; ++cons_i) {
        int sp_head = 0, sp_tail = 0;
        for (i = 0; i < n; ++i) deg[i] = out_degree[i];
        q[sp_tail++] = sink_id;
        while (sp_head < sp_tail) {
            cur = q[sp_head++];
            if (cur == sink_id) {
                max_out_id[cons_i][cur] = -1;
                score[cur] = 0;
            } else if (cur == src_id) {
                int max_id = -1, path_score = -1, path_max_w = -1;
                for (i = 0; i < g->node[cur].out_edge_n; ++i) {
                    int out_id = g->node[cur].out_id[i];
                    int w = edge_weight_clu(&g->node[cur], i, cons_i, clu_read_ids, abpt->use_qv, n_clu);
                    if (w > path_max_w || (w == path_max_w && score[out_id] > path_score)) {
                        max_id = out_id; path_score = score[out_id]; path_max_w = w;
                    }
                }
                max_out_id[cons_i][cur] = max_id;
                break;
            } else {
                int max_id = -1, max_w = INT32_MIN;
                for (i = 0; i < g->node[cur].out_edge_n; ++i) {
                    int out_id = g->node[cur].out_id[i];
                    int w = edge_weight_clu(&g->node[cur], i, cons_i, clu_read_ids, abpt->use_qv, n_clu);
                    if (max_w < w) { max_w = w; max_id = out_id; }
                    else if (max_w == w && score[max_id] <= score[out_id]) max_id = out_id;
                }
                score[cur] = max_w + score[max_id];
                max_out_id[cons_i][cur] = max_id;
            }
            for (i = 0; i < g->node[cur].in_edge_n; ++i) {
                int in_id = g->node[cur].in_id[i];
                if (--deg[in_id] == 0) q[sp_tail++] = in_id;
            }
        }
    }
    /* walk src -> sink through the chosen edges (abpoa_set_hb_cons, :376-392) */
    abc->n_cons = n_clu;
    for (cons_i = 0; cons_i < n_clu; ++cons_i) {
        int j = 0;
        cur = max_out_id[cons_i][src_id];
        while (cur != sink_id) {
            abc->cons_node_ids[cons_i][j] = cur;
            abc->cons_base[cons_i][j] = g->node[cur].base;
            abc->cons_cov[cons_i][j] = node_cov_clu(g->node, cur, clu_read_ids, cons_i, n_clu);
            abc->cons_phred_score[cons_i][j] = cons_phred_score(abc->cons_cov[cons_i][j], abc->clu_n_seq[cons_i]);
            ++j;
            cur = max_out_id[cons_i][cur];
        }
        abc->cons_len[cons_i] = j;
    }
    free(deg); free(score); free(q);
    for (i = 0; i < n_clu; ++i) free(max_out_id[i]);
    free(max_out_id);
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

/* majority-vote consensus over MSA columns, one pass per cluster
 * (abpoa_output.c:394-452, 550-587) */
static void most_frequent(abpoa_graph_t *g, abpoa_para_t *abpt, int src_id, int sink_id,
                          int n_clu, int read_ids_n, uint64_t **clu_read_ids, abpoa_cons_t *abc) {
    int use_span = abpt->sub_aln;
    abamd_set_msa_rank(g, src_id, sink_id);
    int m = abpt->m, i, j, cons_i;
    int msa_l = g->node_id_to_msa_rank[sink_id] - 1;
    int *rc_weight = (int*)abamd_calloc((size_t)n_clu * msa_l * m, sizeof(int));
    int *msa_node_id = (int*)abamd_calloc((size_t)msa_l * m, sizeof(int));
    abc->n_cons = n_clu;
    if (n_clu == 1) {
        abc->clu_n_seq[0] = abc->n_seq;
        for (i = 0; i < abc->n_seq; ++i) abc->clu_read_ids[0][i] = i;
    } else {
        for (cons_i = 0; cons_i < n_clu; ++cons_i) {
            int cnt = 0, r;
            for (i = 0; i < read_ids_n; ++i) cnt += ab_amd_popcnt64(clu_read_ids[cons_i][i]);
            abc->clu_n_seq[cons_i] = cnt;
            for (r = 0, cnt = 0; r < abc->n_seq; ++r)
                if (clu_read_ids[cons_i][r / 64] & (1ull << (r & 0x3f)))
                    abc->clu_read_ids[cons_i][cnt++] = r;
        }
    }
    for (cons_i = 0; cons_i < n_clu; ++cons_i)
        for (i = 0; i < msa_l; ++i)
            rc_weight[(cons_i * (size_t)msa_l + i) * m + m - 1] = abc->clu_n_seq[cons_i];

    /* per-column per-base weights; the gap count starts at clu_n_seq and is
     * decremented by every base weight placed in the column
     * (abpoa_set_row_column_weight, :427-452) */
    for (i = 2; i < g->node_n; ++i) {
        int rank = group_msa_rank(g, i);
        msa_node_id[(rank - 1) * m + g->node[i].base] = i;
        for (cons_i = 0; cons_i < n_clu; ++cons_i) {
            int node_w = node_out_cov(g->node, i, clu_read_ids, cons_i, n_clu);
            rc_weight[(cons_i * (size_t)msa_l + rank - 1) * m + g->node[i].base] = node_w;
            rc_weight[(cons_i * (size_t)msa_l + rank - 1) * m + m - 1] -= node_w;
        }
    }
    for (cons_i = 0; cons_i < n_clu; ++cons_i) {
        int cons_l = 0;
        for (i = 0; i < msa_l; ++i) {
            int max_c = 0, total_c = 0, max_base = m, gap_c, c;
            for (j = 0; j < m - 1; ++j) {
                c = rc_weight[(cons_i * (size_t)msa_l + i) * m + j];
                if (c > max_c) { max_c = c; max_base = j; }
                total_c += c;
            }
            if (use_span) gap_c = g->node[msa_node_id[i * m + max_base]].n_span_read - total_c;
            else gap_c = abc->clu_n_seq[cons_i] - total_c;
            if (max_c >= gap_c) {
                int cur_id = msa_node_id[i * m + max_base];
                abc->cons_node_ids[cons_i][cons_l] = cur_id;
                abc->cons_base[cons_i][cons_l] = (uint8_t)max_base;
                abc->cons_cov[cons_i][cons_l] = max_c;
                abc->cons_phred_score[cons_i][cons_l] = cons_phred_score(max_c, abc->clu_n_seq[cons_i]);
                cons_l++;
            }
        }
        abc->cons_len[cons_i] = cons_l;
    }
    free(rc_weight); free(msa_node_id);
}

void abpoa_generate_consensus(abpoa_t *ab, abpoa_para_t *abpt) {
    if (ab->abg->is_called_cons == 1) return;
    abpoa_graph_t *g = ab->abg;
    if (g->node_n <= 2) return;
    int i, *out_degree = (int*)abamd_malloc((size_t)g->node_n * sizeof(int));
    for (i = 0; i < g->node_n; ++i) out_degree[i] = g->node[i].out_edge_n;
    int n_seq = ab->abs->n_seq;
    int read_ids_n = (n_seq - 1) / 64 + 1;
    int n_clu = 1;
    uint64_t **clu_read_ids = NULL;
    if (abpt->max_n_cons > 1)
        n_clu = multip_read_clu(g, abpt, n_seq, &clu_read_ids);
    abpoa_cons_t *abc = ab->abc;
    allocate_cons(abc, g->node_n, n_seq, n_clu);
    if (abpt->cons_algrm == ABPOA_HB)
        heaviest_bundling(g, abpt, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID, out_degree,
                          n_clu, read_ids_n, clu_read_ids, abc);
    else
        most_frequent(g, abpt, ABPOA_SRC_NODE_ID, ABPOA_SINK_NODE_ID,
                      n_clu, read_ids_n, clu_read_ids, abc);
    if (n_clu > 1) {
        for (i = 0; i < n_clu; ++i) free(clu_read_ids[i]);
        free(clu_read_ids);
    }
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
    if (abpt->out_pog) abpoa_dump_pog(ab, abpt);
}

/* graphviz DOT dump + `dot` render (abpoa_plot.c:33-123): the .dot file is
 * byte-identical to the reference's; rendering shells out to graphviz and
 * fails with the same message when it is unavailable */
void abpoa_dump_pog(abpoa_t *ab, abpoa_para_t *abpt) {
    const char *PROG = "abpoa"; int font_size = 24;
    abpoa_graph_t *g = ab->abg;
    if (g->is_topological_sorted == 0) abpoa_topological_sort(g, abpt);
    static const char node_color[5][10] = {"pink1", "red1", "gold2", "seagreen4", "gray"};
    float node_width = 1;
    const char *rankdir = "LR", *node_style = "filled", *node_fixedsize = "true", *node_shape = "circle";
    int show_aligned_mismatch = 1;

    int i, j, id, index, out_id; char base;
    char **node_label = (char**)abamd_malloc((size_t)g->node_n * sizeof(char*));
    for (i = 0; i < g->node_n; ++i) node_label[i] = (char*)abamd_malloc(128);

    char *dot_fn = (char*)abamd_malloc(strlen(abpt->out_pog) + 10);
    strcpy(dot_fn, abpt->out_pog);
    strcat(dot_fn, ".dot");
    FILE *fp = fopen(dot_fn, "w");
    if (!fp) abamd_fatal("abpoa_dump_pog", "cannot open %s", dot_fn);
    fprintf(fp, "// %s graph dot file.\n// %d nodes.\n", PROG, g->node_n);
    fprintf(fp, "digraph ABPOA_graph {\n\tgraph [rankdir=\"%s\"];\n\tnode [width=%f, style=%s, fixedsize=%s, shape=%s];\n",
            rankdir, node_width, node_style, node_fixedsize, node_shape);
    for (i = 0; i < g->node_n; ++i) {
        id = g->index_to_node_id[i];
        index = i;
        if (id == ABPOA_SRC_NODE_ID) {
            base = 'S';
            sprintf(node_label[id], "\"%c\n%d\"", base, index);
            fprintf(fp, "%s [color=%s, fontsize=%d]\n", node_label[id], node_color[4], font_size);
        } else if (id == ABPOA_SINK_NODE_ID) {
            base = 'E';
            sprintf(node_label[id], "\"%c\n%d\"", base, index);
            fprintf(fp, "%s [color=%s, fontsize=%d]\n", node_label[id], node_color[4], font_size);
        } else {
            base = ab_amd_nt256_table[g->node[id].base];
            sprintf(node_label[id], "\"%c\n%d\"", base, index);
            fprintf(fp, "%s [color=%s, fontsize=%d]\n", node_label[id], node_color[g->node[id].base], font_size);
        }
    }
    int x_index = -1;
    for (i = 0; i < g->node_n; ++i) {
        id = g->index_to_node_id[i];
        for (j = 0; j < g->node[id].out_edge_n; ++j) {
            out_id = g->node[id].out_id[j];
            fprintf(fp, "\t%s -> %s [label=\"%d\", fontsize=20, fontcolor=red, penwidth=%d]\n",
                    node_label[id], node_label[out_id], g->node[id].out_edge_weight[j], g->node[id].out_edge_weight[j] + 1);
        }
        if (g->node[id].aligned_node_n > 0) {
            fprintf(fp, "\t{rank=same; %s ", node_label[id]);
            for (j = 0; j < g->node[id].aligned_node_n; ++j)
                fprintf(fp, "%s ", node_label[g->node[id].aligned_node_id[j]]);
            fprintf(fp, "};\n");
            if (show_aligned_mismatch) {
                if (i > x_index) {
                    x_index = i;
                    fprintf(fp, "\t{ edge [style=dashed, arrowhead=none]; %s ", node_label[id]);
                    for (j = 0; j < g->node[id].aligned_node_n; ++j) {
                        fprintf(fp, "-> %s ", node_label[g->node[id].aligned_node_id[j]]);
                        index = g->node_id_to_index[g->node[id].aligned_node_id[j]];
                        x_index = index > x_index ? index : x_index;
                    }
                    fprintf(fp, "}\n");
                }
            }
        }
    }
    fprintf(fp, "}\n");
    for (i = 0; i < g->node_n; ++i) free(node_label[i]);
    free(node_label);
    fclose(fp);

    char cmd[1024];
    char *type = strrchr(abpt->out_pog, '.');
    if (type == NULL || (strcmp(type + 1, "pdf") != 0 && strcmp(type + 1, "png") != 0))
        abamd_fatal("abpoa_dump_pog", "POG can only be dump to .pdf/.png file");
    snprintf(cmd, sizeof(cmd), "dot %s -T%s > %s", dot_fn, type + 1, abpt->out_pog);
    free(dot_fn);
    if (system(cmd) != 0) abamd_fatal("abpoa_dump_pog", "Fail to plot %s DAG.", PROG);
}
