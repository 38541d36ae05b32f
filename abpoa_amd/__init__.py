"""abpoa_amd — MI355X-native partial order alignment (abPOA-compatible).

Thin ctypes access to the native library for tests and bench. The compute
path is the HIP/CDNA4 core inside libabpoa_amd.so; this package performs no
alignment math in Python.
"""
import ctypes
import os

_CSRC = os.path.join(os.path.dirname(os.path.abspath(__file__)), "csrc")
LIB_PATH = os.path.join(_CSRC, "libabpoa_amd.so")
CLI_PATH = os.path.join(_CSRC, "abpoa_amd")
CLI_CPUTEST_PATH = os.path.join(_CSRC, "abpoa_amd_cputest")
ORACLE_SO = os.path.join(os.path.dirname(_CSRC), os.pardir, "oracle", "liboracle.so")

_lib = None


def lib():
    """Load (once) and return the native library handle."""
    global _lib
    if _lib is None:
        path = os.environ.get("ABPOA_AMD_LIB", LIB_PATH)
        if not os.path.exists(path):
            raise RuntimeError("libabpoa_amd.so not built; run __graft_entry__.build()")
        _lib = ctypes.CDLL(path)
        _lib.abpoa_amd_get_stats.argtypes = [ctypes.POINTER(ctypes.c_uint64)] * 3
        _lib.abpoa_amd_get_stats.restype = None
        _lib.abpoa_amd_reset_stats.restype = None
    return _lib


def get_stats():
    """(dp_cells, kernel_ns, n_launches) accumulated by the native core."""
    a, b, c = ctypes.c_uint64(), ctypes.c_uint64(), ctypes.c_uint64()
    lib().abpoa_amd_get_stats(ctypes.byref(a), ctypes.byref(b), ctypes.byref(c))
    return a.value, b.value, c.value


def reset_stats():
    lib().abpoa_amd_reset_stats()


class ConsT(ctypes.Structure):
    """Mirror of abpoa_cons_t (include/abpoa_amd.h)."""
    _fields_ = [
        ("n_cons", ctypes.c_int), ("n_seq", ctypes.c_int), ("msa_len", ctypes.c_int),
        ("clu_n_seq", ctypes.POINTER(ctypes.c_int)),
        ("clu_read_ids", ctypes.POINTER(ctypes.POINTER(ctypes.c_int))),
        ("cons_len", ctypes.POINTER(ctypes.c_int)),
        ("cons_node_ids", ctypes.POINTER(ctypes.POINTER(ctypes.c_int))),
        ("cons_base", ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        ("msa_base", ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        ("cons_cov", ctypes.POINTER(ctypes.POINTER(ctypes.c_int))),
        ("cons_phred_score", ctypes.POINTER(ctypes.POINTER(ctypes.c_int))),
    ]


CONS_CB = ctypes.CFUNCTYPE(None, ctypes.c_int, ctypes.POINTER(ConsT), ctypes.c_void_p)


def msa_batch_consensus(sets, n_threads=4, cons_algrm=None):
    """Run the batched GPU driver over `sets` (list of list of bytes, codes
    0..3) and return each set's consensus as an ACGT string.
    cons_algrm: None = library default (HB), "MF" = most-frequent (exercises
    the per-edge read-id bitsets on the device-resident path)."""
    L = lib()
    L.abpoa_init_para.restype = ctypes.c_void_p
    L.abpoa_post_set_para.argtypes = [ctypes.c_void_p]
    L.abpoa_free_para.argtypes = [ctypes.c_void_p]
    L.abpoa_amd_msa_batch.argtypes = [
        ctypes.c_void_p, ctypes.c_int,
        ctypes.POINTER(ctypes.c_int),
        ctypes.POINTER(ctypes.POINTER(ctypes.c_int)),
        ctypes.POINTER(ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        CONS_CB, ctypes.c_void_p, ctypes.c_int]
    para = L.abpoa_init_para()
    if cons_algrm is not None:
        from .pyabpoa import ParaT, ABPOA_HB, ABPOA_MF
        p = ctypes.cast(para, ctypes.POINTER(ParaT)).contents
        p.cons_algrm = {"HB": ABPOA_HB, "MF": ABPOA_MF}[cons_algrm.upper()]
    L.abpoa_post_set_para(para)

    n_sets = len(sets)
    NSeqs = (ctypes.c_int * n_sets)(*[len(s) for s in sets])
    lens_keep, ptrs_keep, bufs_keep = [], [], []
    for s in sets:
        lens = (ctypes.c_int * len(s))(*[len(r) for r in s])
        lens_keep.append(lens)
        bufs = [ctypes.create_string_buffer(r, len(r)) for r in s]
        bufs_keep.append(bufs)
        ptrs = (ctypes.POINTER(ctypes.c_uint8) * len(s))(
            *[ctypes.cast(b, ctypes.POINTER(ctypes.c_uint8)) for b in bufs])
        ptrs_keep.append(ptrs)
    LensTop = (ctypes.POINTER(ctypes.c_int) * n_sets)(*lens_keep)
    SeqsTop = (ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8)) * n_sets)(*ptrs_keep)

    out = [None] * n_sets
    ACGT = "ACGTN-"

    @CONS_CB
    def cb(idx, cons_p, _user):
        c = cons_p.contents
        seqs = []
        for ci in range(c.n_cons):
            ln = c.cons_len[ci]
            seqs.append("".join(ACGT[c.cons_base[ci][j]] for j in range(ln)))
        out[idx] = seqs[0] if len(seqs) == 1 else seqs

    rc = L.abpoa_amd_msa_batch(para, n_sets, NSeqs, LensTop, SeqsTop, cb, None, n_threads)
    L.abpoa_free_para(para)
    assert rc == 0
    return out
