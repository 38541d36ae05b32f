#include "abamd_util.h"
#include <stdarg.h>
#include <sys/time.h>
#include <sys/resource.h>

void abamd_fatal(const char *where, const char *fmt, ...) {
    va_list ap;
    va_start(ap, fmt);
    fprintf(stderr, "[%s] ", where);
    vfprintf(stderr, fmt, ap);
    fprintf(stderr, "\n");
    va_end(ap);
    exit(EXIT_FAILURE);
}

void *abamd_malloc(size_t n) {
    void *p = malloc(n ? n : 1);
    if (!p) abamd_fatal("abamd_malloc", "out of memory (%zu bytes)", n);
    return p;
}
void *abamd_calloc(size_t n, size_t sz) {
    void *p = calloc(n ? n : 1, sz ? sz : 1);
    if (!p) abamd_fatal("abamd_calloc", "out of memory (%zu x %zu bytes)", n, sz);
    return p;
}
void *abamd_realloc(void *q, size_t n) {
    void *p = realloc(q, n ? n : 1);
    if (!p) abamd_fatal("abamd_realloc", "out of memory (%zu bytes)", n);
    return p;
}

/* Residue encodings. Same value maps as the reference (abpoa_seq.c:15-98):
 * nucleotides: A/a->0 C/c->1 G/g->2 T/t/U/u->3, everything else->4 ('N'),
 * and the 0..5 -> character map "ACGTN-" (plus byte 27 -> '-').
 * Built programmatically rather than as literal tables. */
unsigned char ab_amd_nt4_table[256];
char ab_amd_nt256_char_storage[256];
const char *ab_amd_nt256_ptr; /* unused; kept simple below */
char ab_amd_nt256_mut[256];
unsigned char ab_amd_aa26_table[256];
char ab_amd_aa256_mut[256];
char ab_amd_char26_table[256];
char ab_amd_char256_table[256];

static int tables_ready = 0;

static void build_nt_tables(void) {
    int i;
    for (i = 0; i < 256; ++i) ab_amd_nt4_table[i] = 4;
    /* low bytes 0..3 map to themselves so already-encoded input passes through */
    ab_amd_nt4_table[0]=0; ab_amd_nt4_table[1]=1; ab_amd_nt4_table[2]=2; ab_amd_nt4_table[3]=3;
    ab_amd_nt4_table['A']=0; ab_amd_nt4_table['a']=0;
    ab_amd_nt4_table['C']=1; ab_amd_nt4_table['c']=1;
    ab_amd_nt4_table['G']=2; ab_amd_nt4_table['g']=2;
    ab_amd_nt4_table['T']=3; ab_amd_nt4_table['t']=3;
    ab_amd_nt4_table['U']=3; ab_amd_nt4_table['u']=3;

    for (i = 0; i < 256; ++i) ab_amd_nt256_mut[i] = 'N';
    ab_amd_nt256_mut[0]='A'; ab_amd_nt256_mut[1]='C'; ab_amd_nt256_mut[2]='G'; ab_amd_nt256_mut[3]='T';
    ab_amd_nt256_mut[4]='N'; ab_amd_nt256_mut[5]='-'; ab_amd_nt256_mut[27]='-';
    for (i = 'A'; i <= 'Z'; ++i) ab_amd_nt256_mut[i] = ab_amd_nt256_mut[i+32] = 'N';
    ab_amd_nt256_mut['A']='A'; ab_amd_nt256_mut['C']='C'; ab_amd_nt256_mut['G']='G';
    ab_amd_nt256_mut['T']='T'; ab_amd_nt256_mut['U']='T';
    ab_amd_nt256_mut['a']='A'; ab_amd_nt256_mut['c']='C'; ab_amd_nt256_mut['g']='G';
    ab_amd_nt256_mut['t']='T'; ab_amd_nt256_mut['u']='T';
}

/* amino-acid codes: the reference assigns ranks over the alphabet with N at 4
 * and T at 3 (i.e. the nt codes embed into the aa codes); order of the rest
 * follows ABCDEFGHIJKLMNOPQRSTUVWXYZ skipping the five nt letters, starting
 * at 5 (abpoa_seq.c:57-95). */
static void build_aa_tables(void) {
    int i;
    static const char order[] = "ACGTNBDEFHIJKLMOPQRSUVWXYZ"; /* code 0..25 */
    for (i = 0; i < 256; ++i) ab_amd_aa26_table[i] = 26;
    for (i = 0; i < 32; ++i) ab_amd_aa26_table[i] = (unsigned char)i; /* pass-through codes */
    ab_amd_aa26_table[27] = 26; ab_amd_aa26_table[28] = 26; ab_amd_aa26_table[29] = 26;
    ab_amd_aa26_table[30] = 26; ab_amd_aa26_table[31] = 26;
    for (i = 26; i < 32; ++i) ab_amd_aa26_table[i] = 26;
    for (i = 0; i < 26; ++i) {
        ab_amd_aa26_table[(int)order[i]] = (unsigned char)i;
        ab_amd_aa26_table[(int)order[i] + 32] = (unsigned char)i; /* lowercase */
    }
    for (i = 0; i < 256; ++i) ab_amd_aa256_mut[i] = '*';
    for (i = 0; i < 26; ++i) ab_amd_aa256_mut[i] = order[i];
    ab_amd_aa256_mut[26] = '*'; ab_amd_aa256_mut[27] = '-';
    for (i = 0; i < 26; ++i) { ab_amd_aa256_mut['A'+i] = (char)('A'+i); ab_amd_aa256_mut['a'+i] = (char)('A'+i); }
}

void ab_amd_init_tables(void) {
    if (tables_ready) return;
    build_nt_tables();
    build_aa_tables();
    tables_ready = 1;
}

/* expose const aliases declared in the header */
extern const char ab_amd_nt256_table[256] __attribute__((alias("ab_amd_nt256_mut")));
extern const char ab_amd_aa256_table[256] __attribute__((alias("ab_amd_aa256_mut")));

char ab_amd_bit_table16[65536];
char ab_amd_log_table65536[65536];

static int ilog2_32(uint32_t v) {
    int r = -1;
    while (v) { v >>= 1; ++r; }
    return r;
}
void ab_amd_set_bit_table16(void) {
    int i; ab_amd_bit_table16[0] = 0;
    for (i = 1; i < 65536; ++i) ab_amd_bit_table16[i] = (char)((i&1) + ab_amd_bit_table16[i>>1]);
}
void ab_amd_set_65536_table(void) {
    int i;
    for (i = 0; i < 65536; ++i) ab_amd_log_table65536[i] = (char)ilog2_32((uint32_t)i);
}
int ab_amd_ilog2_64(uint64_t v) {
    uint64_t t, tt;
    if ((tt = v >> 32)) return (t = tt >> 16) ? 48 + ab_amd_log_table65536[t] : 32 + ab_amd_log_table65536[tt];
    return (t = v >> 16) ? 16 + ab_amd_log_table65536[t] : ab_amd_log_table65536[v];
}

double abamd_realtime(void) {
    struct timeval tp;
    gettimeofday(&tp, NULL);
    return tp.tv_sec + tp.tv_usec * 1e-6;
}
double abamd_cputime(void) {
    struct rusage r;
    getrusage(RUSAGE_SELF, &r);
    return r.ru_utime.tv_sec + r.ru_stime.tv_sec + 1e-6 * (r.ru_utime.tv_usec + r.ru_stime.tv_usec);
}
double abamd_peakrss(void) {
    struct rusage r;
    getrusage(RUSAGE_SELF, &r);
    return r.ru_maxrss; /* KB on Linux */
}
