/* Incremental-MSA graph restore (-i): rebuild a POA graph from a previous
 * run's output before aligning new reads into it.
 *
 * Restates abpoa_seq.c:340-673. Two input formats, auto-detected line by
 * line exactly as the reference does:
 *   - MSA FASTA ('>' lines): each row threads through a column->node map;
 *     '-' columns are skipped, mismatching bases join/extend the column's
 *     aligned-node group (abpoa_fa_parse_seq, :573-608);
 *   - GFA: 'S' lines register segments by name, 'P' lines walk +/- oriented
 *     segment chains into nodes and edges ('-' paths are threaded
 *     sink-to-source and mark the read as reverse-complement)
 *     (abpoa_gfa_parse_S/P, :381-571).
 * Edge bookkeeping uses per-read growing read-id bitsets: read i is added
 * with read_ids_n = 1+((i+1-1)>>6), matching the reference's incremental
 * p_n. The name->index maps are plain open-addressing string hashes; only
 * lookups depend on them, never iteration order, so graph construction
 * order is identical to the reference's.
 */
#include <ctype.h>
#include <string.h>
#include <zlib.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

abpoa_seq_t *abamd_realloc_seq(abpoa_seq_t *abs);
void abamd_cpy_str(abpoa_str_t *dst, const char *s, int l);
int abamd_get_aligned_id(abpoa_graph_t *g, int node_id, uint8_t base);
void abamd_add_aligned_pair(abpoa_graph_t *g, int node_id, int new_id);

/* ---- tiny growable string ---- */
typedef struct { char *s; int l, m; } rstr_t;
static void rstr_putsn(rstr_t *d, const char *s, int l) {
    if (d->l + l + 1 > d->m) {
        d->m = d->l + l + 1;
        int p = 16; while (p < d->m) p <<= 1; d->m = p;
        d->s = (char*)abamd_realloc(d->s, (size_t)d->m);
    }
    memcpy(d->s + d->l, s, (size_t)l);
    d->l += l;
    d->s[d->l] = 0;
}

/* ---- string -> int open-addressing hash (lookup only; order-free) ---- */
typedef struct { char **keys; int *vals; int cap, n; } strmap_t;
static unsigned long sm_hash(const char *s) {
    unsigned long h = 1469598103934665603ull;
    while (*s) { h ^= (unsigned char)*s++; h *= 1099511628211ull; }
    return h;
}
static void sm_init(strmap_t *m) { m->cap = 64; m->n = 0;
    m->keys = (char**)abamd_calloc(m->cap, sizeof(char*));
    m->vals = (int*)abamd_calloc(m->cap, sizeof(int)); }
static void sm_grow(strmap_t *m) {
    int old_cap = m->cap; char **ok = m->keys; int *ov = m->vals;
    m->cap <<= 1;
    m->keys = (char**)abamd_calloc(m->cap, sizeof(char*));
    m->vals = (int*)abamd_calloc(m->cap, sizeof(int));
    for (int i = 0; i < old_cap; ++i) {
        if (!ok[i]) continue;
        unsigned long h = sm_hash(ok[i]) & (m->cap - 1);
        while (m->keys[h]) h = (h + 1) & (m->cap - 1);
        m->keys[h] = ok[i]; m->vals[h] = ov[i];
    }
    free(ok); free(ov);
}
/* returns 1 if inserted, 0 if key already present (*val = existing) */
static int sm_put(strmap_t *m, const char *key, int val, int *existing) {
    if (m->n * 2 >= m->cap) sm_grow(m);
    unsigned long h = sm_hash(key) & (m->cap - 1);
    while (m->keys[h]) {
        if (strcmp(m->keys[h], key) == 0) { if (existing) *existing = m->vals[h]; return 0; }
        h = (h + 1) & (m->cap - 1);
    }
    { size_t kl = strlen(key) + 1; char *cp = (char*)abamd_malloc(kl); memcpy(cp, key, kl); m->keys[h] = cp; }
    m->vals[h] = val; m->n++;
    return 1;
}
static int sm_get(strmap_t *m, const char *key, int *val) {
    unsigned long h = sm_hash(key) & (m->cap - 1);
    while (m->keys[h]) {
        if (strcmp(m->keys[h], key) == 0) { *val = m->vals[h]; return 1; }
        h = (h + 1) & (m->cap - 1);
    }
    return 0;
}
static void sm_free(strmap_t *m) {
    for (int i = 0; i < m->cap; ++i) free(m->keys[i]);
    free(m->keys); free(m->vals);
}

/* ---- segment store (seg_seq_t, abpoa_seq.c:340-379) ---- */
typedef struct { rstr_t *name, *seq; int n, m; strmap_t h; } segs_t;
static void segs_grow(segs_t *g) {
    if (g->n + 1 > g->m) {
        int old = g->m;
        g->m = g->m ? g->m << 1 : 16;
        g->name = (rstr_t*)abamd_realloc(g->name, (size_t)g->m * sizeof(rstr_t));
        g->seq = (rstr_t*)abamd_realloc(g->seq, (size_t)g->m * sizeof(rstr_t));
        memset(g->name + old, 0, (size_t)(g->m - old) * sizeof(rstr_t));
        memset(g->seq + old, 0, (size_t)(g->m - old) * sizeof(rstr_t));
    }
}

/* one MSA-FASTA row into the graph (abpoa_fa_parse_seq, :573-608) */
static int fa_parse_seq(abpoa_graph_t *g, abpoa_seq_t *abs, rstr_t *seq, rstr_t *name,
                        int add_read_id, int p_i, int p_n, int **rank2node_id,
                        int *rank2cap) {
    if (*rank2node_id == 0) {
        *rank2node_id = (int*)abamd_calloc((size_t)seq->l, sizeof(int));
        *rank2cap = seq->l;
    }
    /* a valid MSA has equal-width rows; a longer row would walk off the
     * column map (the reference reads out of bounds here) */
    if (seq->l > *rank2cap)
        abamd_fatal("abpoa_restore_graph",
                    "MSA rows of unequal length (%d > %d) in restore file", seq->l, *rank2cap);
    char *s = seq->s;
    int read_ids_n = 1 + ((p_n - 1) >> 6);
    int i, cur_id, aln_id, last_id = ABPOA_SRC_NODE_ID;
    uint8_t base;
    for (i = 0; s[i]; ++i) {
        if (s[i] == '-') continue;
        base = (uint8_t)ab_amd_char26_table[(int)s[i]];
        cur_id = (*rank2node_id)[i];
        if (cur_id == 0) {
            cur_id = abpoa_add_graph_node(g, base);
            (*rank2node_id)[i] = cur_id;
        } else if (g->node[cur_id].base != base) {
            aln_id = abamd_get_aligned_id(g, cur_id, base);
            if (aln_id == -1) {
                aln_id = abpoa_add_graph_node(g, base);
                abamd_add_aligned_pair(g, cur_id, aln_id);
            }
            cur_id = aln_id;
        }
        abpoa_add_graph_edge(g, last_id, cur_id, 1, 1, (uint8_t)add_read_id, 0, p_i, read_ids_n, p_n);
        last_id = cur_id;
    }
    abpoa_add_graph_edge(g, last_id, ABPOA_SINK_NODE_ID, 1, 1, (uint8_t)add_read_id, 0, p_i, read_ids_n, p_n);
    abamd_realloc_seq(abs);
    abamd_cpy_str(&abs->name[abs->n_seq], name->s, name->l);
    abs->n_seq++;
    return 0;
}

/* S line: register segment name + seq (abpoa_gfa_parse_S, :381-417) */
static int gfa_parse_S(segs_t *segs, char *s) {
    if (s[1] != '\t' || s[2] == '\0') return -1;
    char *deli, *info, *seg_name = 0, *seq = 0;
    int i, name_len = 0, seq_len = 0, ok = 0;
    for (i = 0, deli = info = s + 2;; ++deli) {
        if (*deli == 0 || *deli == '\t') {
            int c = *deli;
            *deli = 0;
            if (i == 0) { seg_name = info; name_len = (int)(deli - info); }
            else if (i == 1) { seq = info; seq_len = (int)(deli - info); ok = 1; break; }
            if (c == 0) break;
            ++i; info = deli + 1;
        }
    }
    if (!ok) abamd_fatal("abpoa_restore_graph", "no seq in GFA segment line (%s)", seg_name ? seg_name : "?");
    segs_grow(segs);
    rstr_putsn(&segs->name[segs->n], seg_name, name_len);
    rstr_putsn(&segs->seq[segs->n], seq, seq_len);
    if (!sm_put(&segs->h, segs->name[segs->n].s, segs->n, NULL))
        abamd_fatal("abpoa_restore_graph", "duplicated segment: \"%s\"", seg_name);
    ++segs->n;
    return 0;
}

/* P line: walk the oriented segment chain (abpoa_gfa_parse_P, :467-571) */
static int gfa_parse_P(abpoa_graph_t *g, abpoa_seq_t *abs, segs_t *segs, int add_read_id,
                       int p_i, int p_n, strmap_t *in_map, strmap_t *out_map, char *s) {
    if (s[1] != '\t' || s[2] == '\0') return -1;
    char *deli, *info, *path = 0, *path_name = 0;
    int i, ok = 0, is_rc = -1, path_name_len = 0;
    int read_ids_n = 1 + ((p_n - 1) >> 6);
    for (i = 0, deli = info = s + 2;; ++deli) {
        if (*deli == 0 || *deli == '\t') {
            int c = *deli;
            *deli = 0;
            if (i == 0) { path_name = info; path_name_len = (int)(deli - info); }
            else if (i == 1) { path = info; ok = 1; break; }
            if (c == 0) break;
            ++i; info = deli + 1;
        }
    }
    if (!ok) abamd_fatal("abpoa_restore_graph", "no path in GFA path line (%s)", path_name ? path_name : "?");
    {
        char *seg_name;
        int id, seg_idx, in_id = -1, out_id = -1;
        int last_id = ABPOA_SRC_NODE_ID, next_id = ABPOA_SINK_NODE_ID;
        for (deli = info = path;; ++deli) {
            if (*deli == '+') {
                if (is_rc == 1) abamd_fatal("abpoa_restore_graph", "path has both + and - segs (%s)", path_name);
                is_rc = 0; *deli = 0; seg_name = info;
                if (!sm_get(&segs->h, seg_name, &seg_idx))
                    abamd_fatal("abpoa_restore_graph", "seg (%s) does not exist", seg_name);
                rstr_t *sseq = &segs->seq[seg_idx];
                char *canon = segs->name[seg_idx].s;
                if (!sm_get(in_map, canon, &in_id)) { /* first time: add nodes */
                    for (i = 0; i < sseq->l; ++i) {
                        id = abpoa_add_graph_node(g, (uint8_t)ab_amd_char26_table[(int)sseq->s[i]]);
                        if (i == 0) in_id = id;
                        if (i == sseq->l - 1) out_id = id;
                    }
                    sm_put(in_map, canon, in_id, NULL);
                    sm_put(out_map, canon, out_id, NULL);
                } else {
                    sm_get(out_map, canon, &out_id);
                }
                abpoa_add_graph_edge(g, last_id, in_id, 1, 1, (uint8_t)add_read_id, 0, p_i, read_ids_n, p_n);
                if (in_id < out_id) {
                    for (i = 0; i < out_id - in_id; ++i)
                        abpoa_add_graph_edge(g, in_id + i, in_id + i + 1, 1, 1, (uint8_t)add_read_id, 0, p_i, read_ids_n, p_n);
                } else if (in_id > out_id)
                    abamd_fatal("abpoa_restore_graph", "in_id (%d) > out_id (%d)", in_id, out_id);
                last_id = out_id;
                info = deli + 2;
            } else if (*deli == '-') {
                if (is_rc == 0) abamd_fatal("abpoa_restore_graph", "path has both + and - segs (%s)", path_name);
                is_rc = 1; *deli = 0; seg_name = info;
                if (!sm_get(&segs->h, seg_name, &seg_idx))
                    abamd_fatal("abpoa_restore_graph", "seg (%s) does not exist", seg_name);
                rstr_t *sseq = &segs->seq[seg_idx];
                char *canon = segs->name[seg_idx].s;
                if (!sm_get(in_map, canon, &in_id)) {
                    for (i = 0; i < sseq->l; ++i) {
                        id = abpoa_add_graph_node(g, (uint8_t)ab_amd_char26_table[(int)sseq->s[i]]);
                        if (i == 0) in_id = id;
                        if (i == sseq->l - 1) out_id = id;
                    }
                    sm_put(in_map, canon, in_id, NULL);
                    sm_put(out_map, canon, out_id, NULL);
                } else {
                    sm_get(out_map, canon, &out_id);
                }
                abpoa_add_graph_edge(g, out_id, next_id, 1, 1, (uint8_t)add_read_id, 0, p_i, read_ids_n, p_n);
                if (in_id < out_id) {
                    for (i = 0; i < out_id - in_id; ++i)
                        abpoa_add_graph_edge(g, in_id + i, in_id + i + 1, 1, 1, (uint8_t)add_read_id, 0, p_i, read_ids_n, p_n);
                } else if (in_id > out_id)
                    abamd_fatal("abpoa_restore_graph", "in_id (%d) > out_id (%d)", in_id, out_id);
                next_id = in_id;
                info = deli + 2;
            } else if (*deli == 0 || *deli == '\t') break;
        }
        if (is_rc) abpoa_add_graph_edge(g, ABPOA_SRC_NODE_ID, next_id, 1, 1, (uint8_t)add_read_id, 0, p_i, read_ids_n, p_n);
        else abpoa_add_graph_edge(g, last_id, ABPOA_SINK_NODE_ID, 1, 1, (uint8_t)add_read_id, 0, p_i, read_ids_n, p_n);
        abamd_realloc_seq(abs);
        abamd_cpy_str(&abs->name[abs->n_seq], path_name, path_name_len);
        abs->is_rc[abs->n_seq] = (uint8_t)is_rc;
        abs->n_seq++;
    }
    return 0;
}

abpoa_t *abpoa_restore_graph(abpoa_t *ab, abpoa_para_t *abpt) {
    char *fn = abpt->incr_fn;
    if (fn == NULL) return ab;
    gzFile fp = strcmp(fn, "-") ? gzopen(fn, "r") : gzdopen(0, "r");
    if (fp == 0) return NULL;
    segs_t segs; memset(&segs, 0, sizeof(segs)); sm_init(&segs.h);
    strmap_t in_map, out_map; sm_init(&in_map); sm_init(&out_map);
    int add_read_id = abpt->use_read_ids;
    int p_i = -1, is_fa = 0, *rank2node_id = 0, rank2cap = 0;
    long line_n = 0;
    abpoa_graph_t *g = ab->abg;
    abpoa_seq_t *abs = ab->abs;

    char *buf = (char*)abamd_malloc(1 << 16);
    rstr_t line; memset(&line, 0, sizeof(line));
    for (;;) {
        line.l = 0; if (line.s) line.s[0] = 0;
        int got = 0;
        while (gzgets(fp, buf, 1 << 16)) {
            got = 1;
            int bl = (int)strlen(buf);
            if (bl > 0 && buf[bl - 1] == '\n') {
                if (bl > 1 && buf[bl - 2] == '\r') bl -= 1;
                rstr_putsn(&line, buf, bl - 1);
                break;
            }
            rstr_putsn(&line, buf, bl);
        }
        if (!got) break;
        line_n++;
        char *s = line.s ? line.s : (char*)"";
        int sl = line.l;
        if (is_fa) {
            if (sl > 0 && s[0] == '>') {
                if (segs.seq[segs.n].l > 0) {
                    fa_parse_seq(g, abs, &segs.seq[segs.n], &segs.name[segs.n],
                                 add_read_id, p_i, p_i + 1, &rank2node_id, &rank2cap);
                    segs.n++;
                }
                int e = 1;
                while (e < sl && !isspace((unsigned char)s[e])) e++;
                segs_grow(&segs);
                rstr_putsn(&segs.name[segs.n], s + 1, e - 1);
                p_i++;
            } else {
                rstr_putsn(&segs.seq[segs.n], s, sl);
            }
        } else {
            if (sl > 0 && s[0] == '>') {
                int e = 1;
                while (e < sl && !isspace((unsigned char)s[e])) e++;
                segs_grow(&segs);
                rstr_putsn(&segs.name[segs.n], s + 1, e - 1);
                is_fa = 1; p_i++;
            } else if (sl > 0 && s[0] == 'S') {
                if (gfa_parse_S(&segs, s) < 0)
                    abamd_fatal("abpoa_restore_graph", "error in S-line at line %ld", line_n);
            } else if (sl > 0 && s[0] == 'P') {
                p_i++;
                if (gfa_parse_P(g, abs, &segs, add_read_id, p_i, p_i + 1, &in_map, &out_map, s) < 0)
                    abamd_fatal("abpoa_restore_graph", "error in P-line at line %ld", line_n);
            }
        }
    }
    if (is_fa && segs.seq[segs.n].l > 0) {
        fa_parse_seq(g, abs, &segs.seq[segs.n], &segs.name[segs.n],
                     add_read_id, p_i, p_i + 1, &rank2node_id, &rank2cap);
        segs.n++;
    }
    free(buf);
    free(line.s);
    gzclose(fp);
    {
        int i;
        for (i = 0; i < segs.m; ++i) { free(segs.name[i].s); free(segs.seq[i].s); }
        free(segs.name); free(segs.seq); sm_free(&segs.h);
    }
    sm_free(&in_map); sm_free(&out_map);
    if (rank2node_id) free(rank2node_id);
    if (abs->n_seq == 0) {
        fprintf(stderr, "[abpoa_restore_graph] Warning: no graph/sequence restored from file '%s'.\n", fn);
        g->node_n = 2;
    }
    g->is_called_cons = g->is_set_msa_rank = g->is_topological_sorted = 0;
    return ab;
}
