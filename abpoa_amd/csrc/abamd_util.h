/* Internal helpers for the abpoa_amd host library.
 * Namespaced abamd_/ab_amd_ so the aligner shim can be linked next to the
 * reference's own host objects (hybrid parity binary) without collisions. */
#ifndef ABAMD_UTIL_H
#define ABAMD_UTIL_H

#include <stdio.h>
#include <stdlib.h>
#include <stdint.h>
#include <string.h>

#ifdef __cplusplus
extern "C" {
#endif

#define AB_MIN2(a, b) ((a) < (b) ? (a) : (b))
#define AB_MAX2(a, b) ((a) > (b) ? (a) : (b))
#define AB_MIN3(a, b, c) AB_MIN2(AB_MIN2(a, b), (c))
#define AB_MAX3(a, b, c) AB_MAX2(AB_MAX2(a, b), (c))

/* round v up to the next power of two (matches klib kroundup semantics) */
static inline int ab_round_up_pow2_32(int v) {
    uint32_t x = (uint32_t)v;
    if (x == 0) return 0;
    x--; x |= x>>1; x |= x>>2; x |= x>>4; x |= x>>8; x |= x>>16;
    return (int)(x + 1);
}
static inline uint64_t ab_round_up_pow2_64(uint64_t x) {
    if (x == 0) return 0;
    x--; x |= x>>1; x |= x>>2; x |= x>>4; x |= x>>8; x |= x>>16; x |= x>>32;
    return x + 1;
}

void abamd_fatal(const char *where, const char *fmt, ...) __attribute__((noreturn, format(printf, 2, 3)));
void *abamd_malloc(size_t n);
void *abamd_calloc(size_t n, size_t sz);
void *abamd_realloc(void *p, size_t n);

/* grow array `arr` of element type T to hold at least n+1 entries; cap is the
 * current capacity variable (updated). */
#define AB_GROW(T, arr, n, cap) do { \
    if ((n) >= (cap)) { \
        int _newc = (cap) ? (cap) : 4; \
        while (_newc <= (n)) _newc <<= 1; \
        (arr) = (T*)abamd_realloc((arr), (size_t)_newc * sizeof(T)); \
        (cap) = _newc; \
    } } while (0)

/* residue encoding tables (values mirror the reference tables,
 * abpoa_seq.c:15-98: nt 0..4, aa 0..26, '-' handling included) */
extern unsigned char ab_amd_nt4_table[256];
extern const char ab_amd_nt256_table[256];
extern unsigned char ab_amd_aa26_table[256];
extern const char ab_amd_aa256_table[256];
extern char ab_amd_char26_table[256];
extern char ab_amd_char256_table[256];
void ab_amd_init_tables(void);

/* popcount-by-16-bit table + helpers used by the MSA/read-id machinery
 * (abpoa_output.c:14-61) */
extern char ab_amd_bit_table16[65536];
extern char ab_amd_log_table65536[65536];
void ab_amd_set_bit_table16(void);
void ab_amd_set_65536_table(void);
int ab_amd_ilog2_64(uint64_t v);
static inline int ab_amd_popcnt64(uint64_t b) {
    return ab_amd_bit_table16[b & 0xffff] + ab_amd_bit_table16[(b>>16) & 0xffff]
         + ab_amd_bit_table16[(b>>32) & 0xffff] + ab_amd_bit_table16[(b>>48) & 0xffff];
}

double abamd_realtime(void);
double abamd_cputime(void);
double abamd_peakrss(void);

/* minimizer seeding + anchor-windowed POA (abamd_seed.c); the functions are
 * declared only when abpoa_amd.h's types are visible */
typedef struct { size_t n, m; uint64_t *a; } abamd_u64v_t;
#ifdef ABPOA_AMD_H
int abamd_build_guide_tree_partition(uint8_t **seqs, int *seq_lens, int n_seq,
                                     abpoa_para_t *abpt, int *read_id_map,
                                     abamd_u64v_t *par_anchors, int *par_c);
int abamd_anchor_poa(abpoa_t *ab, abpoa_para_t *abpt, uint8_t **seqs, int **weights,
                     int *seq_lens, abamd_u64v_t par_anchors, int *par_c,
                     int *tpos_to_node_id, int *qpos_to_node_id, int *read_id_map,
                     int exist_n_seq, int n_seq);
#endif

/* persistent host thread pool (abamd_pool.c): fn(arg, tid, nthr) runs on
 * workers tid = 0..nthr-1; returns when all are done. Calls must not nest. */
typedef void (*abamd_pool_fn)(void *arg, int tid, int nthr);
void abamd_pool_run(abamd_pool_fn fn, void *arg, int nthr);
int abamd_pool_size(void);

#ifdef __cplusplus
}
#endif

#endif
