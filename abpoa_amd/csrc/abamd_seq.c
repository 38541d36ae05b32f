/* Sequence store + FASTA/FASTQ reading (gz-capable).
 * Own reader implementation with kseq-equivalent field semantics
 * (name = first whitespace-delimited token, comment = rest of header line,
 * multi-line sequences joined); store mirrors abpoa_seq.c:100-193. */
#include <zlib.h>
#include <ctype.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

#define ABAMD_CHUNK_READ_N 1024

abpoa_seq_t *abamd_seq_new(void) {
    abpoa_seq_t *abs = (abpoa_seq_t*)abamd_malloc(sizeof(abpoa_seq_t));
    abs->n_seq = 0; abs->m_seq = ABAMD_CHUNK_READ_N;
    abs->seq = (abpoa_str_t*)abamd_calloc(abs->m_seq, sizeof(abpoa_str_t));
    abs->name = (abpoa_str_t*)abamd_calloc(abs->m_seq, sizeof(abpoa_str_t));
    abs->comment = (abpoa_str_t*)abamd_calloc(abs->m_seq, sizeof(abpoa_str_t));
    abs->qual = (abpoa_str_t*)abamd_calloc(abs->m_seq, sizeof(abpoa_str_t));
    abs->is_rc = (uint8_t*)abamd_calloc(abs->m_seq, sizeof(uint8_t));
    return abs;
}

void abamd_seq_destroy(abpoa_seq_t *abs) {
    int i;
    for (i = 0; i < abs->m_seq; ++i) {
        if (abs->seq[i].m > 0) free(abs->seq[i].s);
        if (abs->name[i].m > 0) free(abs->name[i].s);
        if (abs->comment[i].m > 0) free(abs->comment[i].s);
        if (abs->qual[i].m > 0) free(abs->qual[i].s);
    }
    free(abs->seq); free(abs->name); free(abs->comment); free(abs->qual);
    free(abs->is_rc); free(abs);
}

void abamd_cpy_str(abpoa_str_t *dst, const char *s, int l) {
    if (l > 0) {
        if (dst->m != 0) dst->s = (char*)abamd_realloc(dst->s, (size_t)(l + 1));
        else dst->s = (char*)abamd_malloc((size_t)(l + 1));
        dst->l = l; dst->m = l + 1;
        memcpy(dst->s, s, (size_t)l);
        dst->s[l] = 0;
    }
}

abpoa_seq_t *abamd_realloc_seq(abpoa_seq_t *abs) {
    if (abs->n_seq >= abs->m_seq) {
        int i, m = AB_MAX2(abs->n_seq, abs->m_seq << 1);
        abs->seq = (abpoa_str_t*)abamd_realloc(abs->seq, (size_t)m * sizeof(abpoa_str_t));
        abs->name = (abpoa_str_t*)abamd_realloc(abs->name, (size_t)m * sizeof(abpoa_str_t));
        abs->comment = (abpoa_str_t*)abamd_realloc(abs->comment, (size_t)m * sizeof(abpoa_str_t));
        abs->qual = (abpoa_str_t*)abamd_realloc(abs->qual, (size_t)m * sizeof(abpoa_str_t));
        abs->is_rc = (uint8_t*)abamd_realloc(abs->is_rc, (size_t)m * sizeof(uint8_t));
        for (i = abs->m_seq; i < m; ++i) {
            abs->seq[i].l = abs->seq[i].m = 0;
            abs->name[i].l = abs->name[i].m = 0;
            abs->comment[i].l = abs->comment[i].m = 0;
            abs->qual[i].l = abs->qual[i].m = 0;
            abs->is_rc[i] = 0;
        }
        abs->m_seq = m;
    }
    return abs;
}

/* ---------------- buffered gz FASTA/FASTQ parser ---------------- */

typedef struct {
    gzFile fp;
    unsigned char buf[65536];
    int len, pos;
    int peeked; /* -2 = none */
} ab_gzreader_t;

static int gr_getc(ab_gzreader_t *r) {
    if (r->peeked != -2) { int c = r->peeked; r->peeked = -2; return c; }
    if (r->pos >= r->len) {
        r->len = gzread(r->fp, r->buf, sizeof(r->buf));
        r->pos = 0;
        if (r->len <= 0) return -1;
    }
    return r->buf[r->pos++];
}
static void gr_ungetc(ab_gzreader_t *r, int c) { r->peeked = c; }

typedef struct { char *s; int l, m; } ab_dynstr_t;
static void ds_push(ab_dynstr_t *d, int c) {
    if (d->l + 1 >= d->m) {
        d->m = d->m ? d->m << 1 : 64;
        d->s = (char*)abamd_realloc(d->s, (size_t)d->m);
    }
    d->s[d->l++] = (char)c;
    d->s[d->l] = 0;
}
static void ds_clear(ab_dynstr_t *d) { d->l = 0; if (d->s) d->s[0] = 0; }

typedef struct {
    ab_gzreader_t rd;
    ab_dynstr_t name, comment, seq, qual;
    int last_hdr; /* '>' or '@' already consumed, or 0 */
} abamd_fx_t;

abamd_fx_t *abamd_fx_open(const char *fn) {
    gzFile fp = (fn && strcmp(fn, "-") != 0) ? gzopen(fn, "r") : gzdopen(0, "r");
    if (!fp) abamd_fatal("abamd_fx_open", "cannot open '%s'", fn ? fn : "-");
    abamd_fx_t *x = (abamd_fx_t*)abamd_calloc(1, sizeof(abamd_fx_t));
    x->rd.fp = fp; x->rd.peeked = -2;
    return x;
}
void abamd_fx_close(abamd_fx_t *x) {
    gzclose(x->rd.fp);
    free(x->name.s); free(x->comment.s); free(x->seq.s); free(x->qual.s);
    free(x);
}

/* returns sequence length on success, -1 on EOF */
int abamd_fx_read(abamd_fx_t *x) {
    int c;
    ds_clear(&x->name); ds_clear(&x->comment); ds_clear(&x->seq); ds_clear(&x->qual);
    if (!x->last_hdr) {
        while ((c = gr_getc(&x->rd)) != -1 && c != '>' && c != '@') ;
        if (c == -1) return -1;
        x->last_hdr = c;
    }
    int hdr = x->last_hdr; x->last_hdr = 0;
    /* header: name up to first space/tab, comment = remainder of the line */
    while ((c = gr_getc(&x->rd)) != -1 && c != '\n' && c != ' ' && c != '\t' && c != '\r')
        ds_push(&x->name, c);
    if (c == ' ' || c == '\t') {
        while ((c = gr_getc(&x->rd)) != -1 && c != '\n') {
            if (c == '\r') continue;
            if (x->comment.l == 0 && (c == ' ' || c == '\t')) continue;
            ds_push(&x->comment, c);
        }
    } else if (c == '\r') {
        while ((c = gr_getc(&x->rd)) != -1 && c != '\n') ;
    }
    /* sequence lines until next header or '+' */
    while ((c = gr_getc(&x->rd)) != -1) {
        if (c == '>' || c == '@') { x->last_hdr = c; break; }
        if (c == '+') break;
        if (c == '\n' || c == '\r') continue;
        ds_push(&x->seq, c);
    }
    if (hdr == '@' && c == '+') {
        while ((c = gr_getc(&x->rd)) != -1 && c != '\n') ; /* skip '+' line */
        while (x->qual.l < x->seq.l && (c = gr_getc(&x->rd)) != -1) {
            if (c == '\n' || c == '\r') continue;
            ds_push(&x->qual, c);
        }
        /* position at next record */
        while ((c = gr_getc(&x->rd)) != -1) {
            if (c == '>' || c == '@') { x->last_hdr = c; break; }
            if (c != '\n' && c != '\r') { gr_ungetc(&x->rd, c); break; }
        }
        /* kseq semantics: truncated quality is a malformed record (-2);
         * the record is dropped and reading stops */
        if (x->qual.l < x->seq.l) return -2;
    }
    return x->seq.l;
}

/* read every record from the stream into abs; returns number read
 * (abpoa_seq.c:184-193) */
int abamd_read_seq(abpoa_seq_t *abs, abamd_fx_t *x) {
    int n = 0;
    while (abamd_fx_read(x) >= 0) {
        abamd_realloc_seq(abs);
        int i = abs->n_seq;
        abamd_cpy_str(&abs->seq[i], x->seq.s, x->seq.l);
        abamd_cpy_str(&abs->name[i], x->name.s, x->name.l);
        abamd_cpy_str(&abs->comment[i], x->comment.s, x->comment.l);
        abamd_cpy_str(&abs->qual[i], x->qual.s, x->qual.l);
        /* zero-length record fields keep l = 0 */
        if (x->seq.l == 0) abs->seq[i].l = 0;
        if (x->name.l == 0) abs->name[i].l = 0;
        if (x->comment.l == 0) abs->comment[i].l = 0;
        if (x->qual.l == 0) abs->qual[i].l = 0;
        abs->is_rc[i] = 0;
        abs->n_seq++; n++;
    }
    return n;
}
