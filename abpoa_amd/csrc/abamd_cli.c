/* abpoa_amd command-line interface.
 * Option surface mirrors the reference CLI (abpoa.c:170-250) so parity tests
 * can drive both binaries with identical argv; stdout framing is identical,
 * diagnostics go to stderr. */
#include <getopt.h>
#include <string.h>
#include "abpoa_amd.h"
#include "abamd_util.h"

#define ABPOA_AMD_VERSION "0.1.0-r1"

static int usage(void) {
    fprintf(stderr, "Usage: abpoa_amd [options] <in.fa/fq> > cons.fa\n");
    fprintf(stderr, "  MI355X-native abPOA-compatible POA. Options match the reference abpoa CLI;\n");
    fprintf(stderr, "  see the reference usage for details. Unsupported in this build: -S -p -i -d>=2 -s -g.\n");
    return 1;
}

static int abpoa_amd_main(char *file_fn, int is_list, abpoa_para_t *abpt) {
    double t0 = abamd_realtime();
    abpoa_t *ab = abpoa_init();
    if (is_list) {
        FILE *list_fp = fopen(file_fn, "r"); char read_fn[1024];
        if (!list_fp) abamd_fatal("abpoa_amd", "cannot open list file '%s'", file_fn);
        int batch_index = 1;
        while (fgets(read_fn, sizeof(read_fn), list_fp)) {
            size_t l = strlen(read_fn);
            if (l && read_fn[l-1] == '\n') read_fn[l-1] = 0;
            abpt->batch_index = batch_index;
            abpoa_msa1(ab, abpt, read_fn, stdout);
            batch_index++;
        }
        fclose(list_fp);
    } else abpoa_msa1(ab, abpt, file_fn, stdout);
    abpoa_free(ab);
    fprintf(stderr, "[abpoa_amd_main] Real time: %.3f sec; CPU: %.3f sec; Peak RSS: %.3f GB.\n",
            abamd_realtime() - t0, abamd_cputime(), abamd_peakrss() / 1024.0 / 1024.0);
    return 0;
}

int main(int argc, char **argv) {
    int c, m, in_list = 0; char *s;
    abpoa_para_t *abpt = abpoa_init_para();
    while ((c = getopt(argc, argv, "m:M:X:t:O:E:b:f:z:e:GLRJQSk:w:n:i:clpso:r:g:a:d:q:hvV:")) >= 0) {
        switch (c) {
            case 'm': m = atoi(optarg);
                      if (m != ABPOA_GLOBAL_MODE && m != ABPOA_EXTEND_MODE && m != ABPOA_LOCAL_MODE) {
                          fprintf(stderr, "Unknown alignment mode: %d.\n", m); return 1;
                      } abpt->align_mode = m; break;
            case 'M': abpt->match = atoi(optarg); break;
            case 'X': abpt->mismatch = atoi(optarg); break;
            case 't': abpt->use_score_matrix = 1; abpt->mat_fn = strdup(optarg); break;
            case 'O': abpt->gap_open1 = (int)strtol(optarg, &s, 10);
                      abpt->gap_open2 = (*s == ',') ? (int)strtol(s + 1, &s, 10) : 0; break;
            case 'E': abpt->gap_ext1 = (int)strtol(optarg, &s, 10);
                      abpt->gap_ext2 = (*s == ',') ? (int)strtol(s + 1, &s, 10) : 0; break;
            case 'G': abpt->inc_path_score = 1; break;
            case 'L': abpt->sort_input_seq = 1; break;
            case 'R': abpt->put_gap_on_right = 1; break;
            case 'J': abpt->put_gap_at_end = 1; break;
            case 'b': abpt->wb = atoi(optarg); break;
            case 'f': abpt->wf = (float)atof(optarg); break;
            case 'z': abpt->zdrop = atoi(optarg); break;
            case 'e': abpt->end_bonus = atoi(optarg); break;
            case 'Q': abpt->use_qv = 1; break;
            case 'S': abpt->disable_seeding = 0; break;
            case 'k': abpt->k = atoi(optarg); break;
            case 'w': abpt->w = atoi(optarg); break;
            case 'n': abpt->min_w = atoi(optarg); break;
            case 'c': abpt->m = 27;
                      abpt->mat = (int*)abamd_realloc(abpt->mat, (size_t)abpt->m * abpt->m * sizeof(int)); break;
            case 'i': abpt->incr_fn = strdup(optarg); break;
            case 'l': in_list = 1; break;
            case 'p': abpt->progressive_poa = 1; break;
            case 's': abpt->amb_strand = 1; break;
            case 'o': if (strcmp(optarg, "-") != 0) {
                          if (freopen(optarg, "wb", stdout) == NULL)
                              abamd_fatal("abpoa_amd", "failed to open output file %s", optarg);
                      } break;
            case 'r': if (atoi(optarg) == ABPOA_OUT_CONS) abpt->out_cons = 1, abpt->out_msa = 0;
                      else if (atoi(optarg) == ABPOA_OUT_MSA) abpt->out_cons = 0, abpt->out_msa = 1;
                      else if (atoi(optarg) == ABPOA_OUT_CONS_MSA) abpt->out_cons = abpt->out_msa = 1;
                      else if (atoi(optarg) == ABPOA_OUT_GFA) abpt->out_cons = 0, abpt->out_gfa = 1;
                      else if (atoi(optarg) == ABPOA_OUT_CONS_GFA) abpt->out_cons = 1, abpt->out_gfa = 1;
                      else if (atoi(optarg) == ABPOA_OUT_CONS_FQ) abpt->out_cons = 1, abpt->out_fq = 1;
                      else fprintf(stderr, "Error: unknown output result mode: %s.\n", optarg);
                      break;
            case 'g': abpt->out_pog = strdup(optarg); break;
            case 'a': abpt->cons_algrm = atoi(optarg); break;
            case 'd': abpt->max_n_cons = atoi(optarg);
                      if (abpt->max_n_cons < 1 || abpt->max_n_cons > 10) {
                          fprintf(stderr, "Error: max number of consensus sequences should be 1~10.\n");
                          return 1;
                      } break;
            case 'q': abpt->min_freq = atof(optarg); break;
            case 'h': return usage();
            case 'V': abpt->verbose = atoi(optarg); break;
            case 'v': printf("%s\n", ABPOA_AMD_VERSION); abpoa_free_para(abpt); return 0;
            default: return usage();
        }
    }
    if (argc - optind != 1) return usage();
    abpoa_post_set_para(abpt);
    fprintf(stderr, "[abpoa_amd] CMD:");
    for (c = 0; c < argc; ++c) fprintf(stderr, " %s", argv[c]);
    fprintf(stderr, "\n");
    abpoa_amd_main(argv[optind], in_list, abpt);
    abpoa_free_para(abpt);
    return 0;
}
