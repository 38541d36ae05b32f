"""pyabpoa-compatible Python API over the MI355X-native core.

Drop-in for the reference's Python package (python/pyabpoa.pyx): the same
``msa_aligner`` / ``msa_result`` classes with the same constructor options,
``msa`` / ``msa_align`` / ``msa_add`` / ``msa_output`` methods and result
attributes. Use ``import abpoa_amd.pyabpoa as pa`` where the reference uses
``import pyabpoa as pa``.

Alignment math runs in libabpoa_amd.so's HIP/CDNA4 core (GPU-only; the
library aborts loudly without one). This module only marshals sequences and
results through ctypes — mirroring pyabpoa.pyx's per-sequence
``abpoa_align_sequence_to_graph`` + ``abpoa_add_graph_alignment`` calls, so
results match the reference binding exactly.
"""
import ctypes
import ctypes.util
from collections import defaultdict as dd

_libc = ctypes.CDLL(None)
_libc.free.argtypes = [ctypes.c_void_p]
_libc.free.restype = None
_libc.strdup.argtypes = [ctypes.c_char_p]
_libc.strdup.restype = ctypes.c_void_p


def _set_cstr(para, attr, value):
    """Assign a C-owned string to an abpoa_para_t filename field.

    abpoa_free_para() (abamd_para.c:131-136, matching the reference) calls
    free() on mat_fn/out_pog/incr_fn, so these fields must never point into
    Python-owned ctypes buffers — they are C-malloc'd copies (libc strdup),
    exactly like the reference pyx's malloc+strcpy. Frees any previous value.
    """
    _libc.free(getattr(para, attr))
    setattr(para, attr, _libc.strdup(value) if value is not None else None)

from . import lib as _load_lib
from . import ConsT

ABPOA_GLOBAL_MODE, ABPOA_LOCAL_MODE, ABPOA_EXTEND_MODE = 0, 1, 2
ABPOA_HB, ABPOA_MF = 0, 1

_cigar_t = ctypes.c_uint64


class ParaT(ctypes.Structure):
    """Mirror of abpoa_para_t (include/abpoa_amd.h); layout verified at
    import against abpoa_init_para()'s known defaults."""
    _fields_ = [
        ("m", ctypes.c_int), ("mat", ctypes.POINTER(ctypes.c_int)), ("mat_fn", ctypes.c_void_p),
        ("use_score_matrix", ctypes.c_int),
        ("match", ctypes.c_int), ("max_mat", ctypes.c_int), ("mismatch", ctypes.c_int),
        ("min_mis", ctypes.c_int), ("gap_open1", ctypes.c_int), ("gap_open2", ctypes.c_int),
        ("gap_ext1", ctypes.c_int), ("gap_ext2", ctypes.c_int), ("inf_min", ctypes.c_int),
        ("sort_input_seq", ctypes.c_int),
        ("inc_path_score", ctypes.c_int),
        ("k", ctypes.c_int), ("w", ctypes.c_int), ("min_w", ctypes.c_int),
        ("wb", ctypes.c_int), ("wf", ctypes.c_float),
        ("zdrop", ctypes.c_int), ("end_bonus", ctypes.c_int),
        ("ret_cigar", ctypes.c_uint8, 1), ("rev_cigar", ctypes.c_uint8, 1),
        ("out_msa", ctypes.c_uint8, 1), ("out_cons", ctypes.c_uint8, 1),
        ("out_gfa", ctypes.c_uint8, 1), ("out_fq", ctypes.c_uint8, 1),
        ("use_read_ids", ctypes.c_uint8, 1), ("amb_strand", ctypes.c_uint8, 1),
        ("sub_aln", ctypes.c_uint8, 1), ("use_qv", ctypes.c_uint8, 1),
        ("disable_seeding", ctypes.c_uint8, 1), ("progressive_poa", ctypes.c_uint8, 1),
        ("put_gap_on_right", ctypes.c_uint8, 1), ("put_gap_at_end", ctypes.c_uint8, 1),
        ("incr_fn", ctypes.c_void_p), ("out_pog", ctypes.c_void_p),
        ("align_mode", ctypes.c_int), ("gap_mode", ctypes.c_int),
        ("max_n_cons", ctypes.c_int), ("cons_algrm", ctypes.c_int),
        ("min_freq", ctypes.c_double),
        ("verbose", ctypes.c_int),
        ("batch_index", ctypes.c_int),
    ]


class StrT(ctypes.Structure):
    _fields_ = [("l", ctypes.c_int), ("m", ctypes.c_int), ("s", ctypes.c_char_p)]


class SeqT(ctypes.Structure):
    _fields_ = [
        ("n_seq", ctypes.c_int), ("m_seq", ctypes.c_int),
        ("seq", ctypes.POINTER(StrT)), ("name", ctypes.POINTER(StrT)),
        ("comment", ctypes.POINTER(StrT)), ("qual", ctypes.POINTER(StrT)),
        ("is_rc", ctypes.POINTER(ctypes.c_uint8)),
    ]


class AbpoaT(ctypes.Structure):
    _fields_ = [
        ("abg", ctypes.c_void_p), ("abs", ctypes.POINTER(SeqT)),
        ("abm", ctypes.c_void_p), ("abc", ctypes.POINTER(ConsT)),
    ]


class ResT(ctypes.Structure):
    _fields_ = [
        ("n_cigar", ctypes.c_int), ("m_cigar", ctypes.c_int),
        ("graph_cigar", ctypes.POINTER(_cigar_t)),
        ("node_s", ctypes.c_int), ("node_e", ctypes.c_int),
        ("query_s", ctypes.c_int), ("query_e", ctypes.c_int),
        ("n_aln_bases", ctypes.c_int), ("n_matched_bases", ctypes.c_int),
        ("best_score", ctypes.c_int32),
    ]


_checked = False


def _lib():
    global _checked
    L = _load_lib()
    if not _checked:
        L.abpoa_init_para.restype = ctypes.POINTER(ParaT)
        L.abpoa_free_para.argtypes = [ctypes.POINTER(ParaT)]
        L.abpoa_post_set_para.argtypes = [ctypes.POINTER(ParaT)]
        L.abpoa_init.restype = ctypes.POINTER(AbpoaT)
        L.abpoa_free.argtypes = [ctypes.POINTER(AbpoaT)]
        L.abpoa_reset.argtypes = [ctypes.POINTER(AbpoaT), ctypes.POINTER(ParaT), ctypes.c_int]
        L.abpoa_restore_graph.argtypes = [ctypes.POINTER(AbpoaT), ctypes.POINTER(ParaT)]
        L.abpoa_align_sequence_to_graph.argtypes = [
            ctypes.POINTER(AbpoaT), ctypes.POINTER(ParaT),
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_int, ctypes.POINTER(ResT)]
        L.abpoa_add_graph_alignment.argtypes = [
            ctypes.POINTER(AbpoaT), ctypes.POINTER(ParaT),
            ctypes.POINTER(ctypes.c_uint8), ctypes.POINTER(ctypes.c_int), ctypes.c_int,
            ctypes.POINTER(ctypes.c_int), ResT, ctypes.c_int, ctypes.c_int, ctypes.c_int]
        L.abpoa_generate_rc_msa.argtypes = [ctypes.POINTER(AbpoaT), ctypes.POINTER(ParaT)]
        L.abpoa_generate_consensus.argtypes = [ctypes.POINTER(AbpoaT), ctypes.POINTER(ParaT)]
        L.abpoa_dump_pog.argtypes = [ctypes.POINTER(AbpoaT), ctypes.POINTER(ParaT)]
        # verify the ctypes mirror against abpoa_init_para's known defaults
        p = L.abpoa_init_para()
        d = p.contents
        ok = (d.m == 5 and d.match == 2 and d.mismatch == 4 and d.gap_open1 == 4
              and d.gap_open2 == 24 and d.gap_ext1 == 2 and d.gap_ext2 == 1
              and d.k == 19 and d.w == 10 and d.min_w == 500 and d.wb == 10
              and abs(d.wf - 0.01) < 1e-6 and d.zdrop == -1 and d.end_bonus == -1
              and d.align_mode == ABPOA_GLOBAL_MODE and d.max_n_cons == 1
              and abs(d.min_freq - 0.25) < 1e-9)
        L.abpoa_free_para(p)
        if not ok:
            raise RuntimeError("abpoa_para_t ctypes mirror does not match the native layout")
        _checked = True
    return L


class msa_result:
    def __init__(self, n_seq, n_cons, clu_n_seq, clu_read_ids, cons_len, cons_seq,
                 cons_cov, cons_qv, msa_len, msa_seq):
        self._n_seq, self._n_cons = n_seq, n_cons
        self._clu_n_seq, self._clu_read_ids = clu_n_seq, clu_read_ids
        self._cons_len, self._cons_seq = cons_len, cons_seq
        self._cons_cov, self._cons_qv = cons_cov, cons_qv
        self._msa_len, self._msa_seq = msa_len, msa_seq

    @property
    def n_seq(self): return self._n_seq
    @property
    def n_cons(self): return self._n_cons
    @property
    def clu_n_seq(self): return self._clu_n_seq
    @property
    def clu_read_ids(self): return self._clu_read_ids
    @property
    def cons_len(self): return self._cons_len
    @property
    def cons_seq(self): return self._cons_seq
    @property
    def cons_cov(self): return self._cons_cov
    @property
    def cons_qv(self): return self._cons_qv
    @property
    def msa_len(self): return self._msa_len
    @property
    def msa_seq(self): return self._msa_seq

    def print_msa(self):
        if not self._msa_seq:
            return
        for i, s in enumerate(self._msa_seq):
            if i < self._n_seq:
                print(">Seq_%d" % (i + 1))
            else:
                if self._n_cons > 1:
                    cons_id = "_%d %s" % (i - self._n_seq + 1,
                                          ",".join(map(str, self._clu_read_ids[i - self._n_seq])))
                else:
                    cons_id = ""
                print(">Consensus_sequence%s" % cons_id)
            print(s)


def _set_seq_int_dict(m):
    if m == 5:
        seqs, ints = "ACGUTN", [0, 1, 2, 3, 3, 4]
    elif m == 27:
        seqs = "ACGTNBDEFHIJKLMOPQRSUVWXYZ*"
        ints = list(range(27))
    else:
        raise Exception("Unexpected m: %d" % m)
    seq2int = dd(lambda: m - 1)
    int2seq = dd(lambda: "-")
    for s, i in zip(seqs, ints):
        seq2int[s] = i
        seq2int[s.lower()] = i
        int2seq[i] = s
    return seq2int, int2seq


class msa_aligner:
    def __init__(self, aln_mode='g', is_aa=False,
                 match=2, mismatch=4, score_matrix=b'', gap_open1=4, gap_open2=24,
                 gap_ext1=2, gap_ext2=1, extra_b=10, extra_f=0.01, cons_algrm='HB'):
        L = _lib()
        self._L = L
        self.ab = L.abpoa_init()
        # start from library defaults, then apply exactly the fields
        # pyabpoa.pyx sets (python/pyabpoa.pyx:93-150)
        self._para = L.abpoa_init_para()
        p = self._para.contents
        if aln_mode == 'g':
            p.align_mode = ABPOA_GLOBAL_MODE
        elif aln_mode == 'l':
            p.align_mode = ABPOA_LOCAL_MODE
        elif aln_mode == 'e':
            p.align_mode = ABPOA_EXTEND_MODE
        else:
            raise Exception("Unknown align mode: %s" % aln_mode)
        p.m = 27 if is_aa else 5
        p.match = match
        p.mismatch = mismatch
        if score_matrix:
            if isinstance(score_matrix, str):
                score_matrix = score_matrix.encode()
            p.use_score_matrix = 1
            _set_cstr(p, "mat_fn", score_matrix)
        p.gap_open1, p.gap_open2 = gap_open1, gap_open2
        p.gap_ext1, p.gap_ext2 = gap_ext1, gap_ext2
        p.ret_cigar = 1
        p.wb = extra_b
        p.wf = extra_f
        p.use_qv = 0
        p.end_bonus = -1
        p.zdrop = -1
        p.disable_seeding = 1
        p.progressive_poa = 0
        if cons_algrm.upper() == 'MF':
            p.cons_algrm = ABPOA_MF
        elif cons_algrm.upper() == 'HB':
            p.cons_algrm = ABPOA_HB
        else:
            raise Exception("Unknown conseneus calling mode: %s" % cons_algrm)
        self.seq2int_dict, self.int2seq_dict = _set_seq_int_dict(p.m)
        self._keepalive = []

    def __del__(self):
        try:
            if getattr(self, "ab", None):
                self._L.abpoa_free(self.ab)
            if getattr(self, "_para", None):
                self._L.abpoa_free_para(self._para)
        except Exception:
            pass

    def __bool__(self):
        return bool(self.ab)

    def _add_sequences(self, seqs, qscores, exist_n, tot_n):
        L = self._L
        if qscores is not None and len(qscores) != len(seqs):
            raise ValueError("qscores must contain one entry per input sequence.")
        for read_i, seq in enumerate(seqs):
            seq_l = len(seq)
            codes = bytes(self.seq2int_dict[c] for c in seq)
            bseq = (ctypes.c_uint8 * seq_l).from_buffer_copy(codes) if seq_l else (ctypes.c_uint8 * 1)()
            weights = None
            if qscores is not None:
                q = qscores[read_i]
                if len(q) != seq_l:
                    raise ValueError("Each qscore array must have the same length as its sequence.")
                vals = [int(x) for x in q]
                if any(v < 0 for v in vals):
                    raise ValueError("Qscores must be non-negative integers.")
                weights = (ctypes.c_int * seq_l)(*vals)
            res = ResT()
            res.n_cigar = 0
            L.abpoa_align_sequence_to_graph(self.ab, self._para, bseq, seq_l, ctypes.byref(res))
            L.abpoa_add_graph_alignment(self.ab, self._para, bseq, weights, seq_l,
                                        None, res, exist_n + read_i, tot_n, 1)
            if res.n_cigar:
                _libc.free(ctypes.cast(res.graph_cigar, ctypes.c_void_p))

    def _collect(self, tot_n):
        abc = self.ab.contents.abc.contents
        n_cons = abc.n_cons
        clu_n_seq, clu_read_ids, cons_len = [], [], []
        cons_seq, cons_cov, cons_qv, msa_seq = [], [], [], []
        for i in range(n_cons):
            clu_n_seq.append(abc.clu_n_seq[i])
            cons_len.append(abc.cons_len[i])
            ids1, seq1, cov1, qv1 = [], "", [], ""
            for j in range(abc.clu_n_seq[i]):
                ids1.append(abc.clu_read_ids[i][j])
            clu_read_ids.append(ids1)
            for j in range(abc.cons_len[i]):
                seq1 += self.int2seq_dict[abc.cons_base[i][j]]
                cov1.append(abc.cons_cov[i][j])
                if abc.cons_phred_score:
                    qv1 += chr(abc.cons_phred_score[i][j])
            cons_seq.append(seq1)
            cons_cov.append(cov1)
            cons_qv.append(qv1)
        msa_len = abc.msa_len
        if msa_len > 0:
            for i in range(abc.n_seq + n_cons):
                msa_seq.append("".join(self.int2seq_dict[abc.msa_base[i][j]] for j in range(msa_len)))
        return msa_result(tot_n, n_cons, clu_n_seq, clu_read_ids, cons_len, cons_seq,
                          cons_cov, cons_qv, msa_len, msa_seq)

    def msa(self, seqs, out_cons, out_msa, max_n_cons=1, min_freq=0.25,
            out_pog=b'', incr_fn=b'', qscores=None):
        L = self._L
        p = self._para.contents
        seq_n = len(seqs)
        exist_n, tot_n = 0, seq_n
        p.out_cons = 1 if out_cons else 0
        p.out_msa = 1 if out_msa else 0
        if max_n_cons < 1 or max_n_cons > 2:
            raise Exception("Error: max number of consensus sequences should be 1 or 2.")
        p.max_n_cons = max_n_cons
        p.min_freq = min_freq
        p.use_qv = 1 if qscores is not None else 0
        if out_pog:
            if isinstance(out_pog, str):
                out_pog = out_pog.encode()
            _set_cstr(p, "out_pog", out_pog)
        else:
            _set_cstr(p, "out_pog", None)
        L.abpoa_post_set_para(self._para)
        L.abpoa_reset(self.ab, self._para, len(seqs[0]))
        if incr_fn:
            if isinstance(incr_fn, str):
                incr_fn = incr_fn.encode()
            _set_cstr(p, "incr_fn", incr_fn)
            L.abpoa_restore_graph(self.ab, self._para)
            exist_n = self.ab.contents.abs.contents.n_seq
            tot_n += exist_n
        else:
            _set_cstr(p, "incr_fn", None)
        self.ab.contents.abs.contents.n_seq += seq_n
        self._add_sequences(seqs, qscores, exist_n, tot_n)
        if p.out_msa:
            L.abpoa_generate_rc_msa(self.ab, self._para)
        elif p.out_cons:
            L.abpoa_generate_consensus(self.ab, self._para)
        result = self._collect(tot_n)
        if p.out_pog:
            L.abpoa_dump_pog(self.ab, self._para)
        return result

    def msa_align(self, seqs, out_cons, out_msa, max_n_cons=1, min_freq=0.25,
                  incr_fn=b'', qscores=None):
        L = self._L
        p = self._para.contents
        seq_n = len(seqs)
        exist_n, tot_n = 0, seq_n
        p.out_cons = 1 if out_cons else 0
        p.out_msa = 1 if out_msa else 0
        if max_n_cons < 1 or max_n_cons > 2:
            raise Exception("Error: max number of consensus sequences should be 1 or 2.")
        p.max_n_cons = max_n_cons
        p.min_freq = min_freq
        p.use_qv = 1 if qscores is not None else 0
        L.abpoa_post_set_para(self._para)
        L.abpoa_reset(self.ab, self._para, len(seqs[0]))
        if incr_fn:
            if isinstance(incr_fn, str):
                incr_fn = incr_fn.encode()
            _set_cstr(p, "incr_fn", incr_fn)
            L.abpoa_restore_graph(self.ab, self._para)
            exist_n = self.ab.contents.abs.contents.n_seq
            tot_n += exist_n
        else:
            _set_cstr(p, "incr_fn", None)
        self.ab.contents.abs.contents.n_seq += seq_n
        self._add_sequences(seqs, qscores, exist_n, tot_n)
        return self

    def msa_add(self, new_seqs, qscores=None):
        if isinstance(new_seqs, str):
            raise TypeError('Expected a list of strings. If you want to add a single '
                            'sequence, pass it as a list: ["ACGT..."]')
        exist_n = self.ab.contents.abs.contents.n_seq
        if exist_n == 0:
            raise Exception("Error: no existing sequences in the graph. "
                            "Please run msa() or msa_align() first.")
        seq_n = len(new_seqs)
        tot_n = seq_n + exist_n
        if qscores is not None:
            self._para.contents.use_qv = 1
        self.ab.contents.abs.contents.n_seq += seq_n
        self._add_sequences(new_seqs, qscores, exist_n, tot_n)
        return self

    def msa_output(self):
        L = self._L
        p = self._para.contents
        if p.out_msa:
            L.abpoa_generate_rc_msa(self.ab, self._para)
        elif p.out_cons:
            L.abpoa_generate_consensus(self.ab, self._para)
        return self._collect(self.ab.contents.abs.contents.n_seq)
