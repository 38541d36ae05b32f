#!/usr/bin/env python3
"""Benchmark: read-sets/sec (and Gcells/sec) for 50x10 kbp global convex-gap
POA on MI355X — the reference's headline workload (BASELINE.json).

One "step" = one batched pass of the hot path over --sets-per-step synthetic
read sets (each 50 reads x 10 kbp at 10% ONT-style error, generated on the
fly, resident in HBM when the timed kernels run). Whole-job value is the
aggregate over all ranks (one process per GPU; sets shard across ranks with
no data-path collective: weak scaling).

Emits ONE JSON line from rank 0, including:
  roofline:     dominant kernel (cg_global_mw_kernel) algorithmic HBM bytes/s
                from HIP events on the library streams (6 B per DP cell at
                int16: 3 planes H/E1/E2 written once; F planes recomputed at
                backtrack), plus counter-measured traffic from two separate
                rocprofv3 --pmc probe passes (gfx950 FETCH correction)
  cpu_baseline: the unmodified reference binary (oracle/_ref/abpoa) on this
                box's host cores: 1-core measured, one-socket linear
                extrapolation, and (ABPOA_BENCH_CPU_SOCKET) a measured
                concurrent batch run.
"""
import argparse
import ctypes
import json
import os
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

WORKLOAD = "1000 synthetic sets x 50 reads x 10 kbp, 10% ONT-style error, global convex gap, 1 MI355X (BASELINE.json configs[1])"
# BASELINE.json configs[2]/[3] variants (non-default --workload values emit
# their own JSON line; the driver's default run is always configs[1])
WORKLOADS = {
    "cg":     ("global convex gap (BASELINE configs[1])", 0, None),
    "affine": ("global affine gap -O4 -E2 (BASELINE configs[2])", 0, "affine"),
    "linear": ("global linear gap -O0 (BASELINE configs[2])", 0, "linear"),
    "local":  ("local convex gap -m1 (BASELINE configs[2])", 1, None),
    "extend": ("extension convex gap -m2 (BASELINE configs[2])", 2, None),
    "aa":     ("amino-acid 30x2 kaa BLOSUM62 global affine (BASELINE configs[3])", 0, "aa"),
}
HBM_PEAK_GBPS = 8000.0  # gfx950 spec peak (MI355X_MICROARCH.md)
ALG_BYTES_PER_CELL = 6.0  # 3 planes (H,E1,E2) x int16, written once; F planes recomputed at backtrack


def gen_sets(rng, n_sets, depth, qlen, p_sub=0.045, p_del=0.03, p_ins=0.025, alphabet=4):
    """Generate encoded (0..3) read sets with numpy; returns list of list of bytes.
    Error model matches tests/make_synth.py: per base, substitute with p_sub,
    delete with p_del, keep-and-insert-one with p_ins, else keep."""
    import numpy as np
    sets = []
    for _ in range(n_sets):
        ref = rng.integers(0, alphabet, size=qlen, dtype=np.uint8)
        reads = []
        for _ in range(depth):
            r = rng.random(qlen)
            sub = r < p_sub
            mask_del = (r >= p_sub) & (r < p_sub + p_del)
            ins = (r >= p_sub + p_del) & (r < p_sub + p_del + p_ins)
            out = ref.copy()
            nsub = int(sub.sum())
            if nsub:
                out[sub] = (ref[sub] + rng.integers(1, alphabet, size=nsub, dtype=np.uint8)) % alphabet
            kept = out[~mask_del]
            ins_pos = np.nonzero(ins[~mask_del])[0]
            if len(ins_pos):
                ins_bases = rng.integers(0, alphabet, size=len(ins_pos), dtype=np.uint8)
                kept = np.insert(kept, ins_pos + 1, ins_bases)
            reads.append(np.ascontiguousarray(kept).tobytes())
        sets.append(reads)
    return sets


def build_batch_args(lib, sets):
    n_sets = len(sets)
    NSeqs = (ctypes.c_int * n_sets)(*[len(s) for s in sets])
    seq_len_arrays = []
    seq_ptr_arrays = []
    keepalive = []
    for s in sets:
        lens = (ctypes.c_int * len(s))(*[len(r) for r in s])
        seq_len_arrays.append(lens)
        bufs = [ctypes.create_string_buffer(r, len(r)) for r in s]
        keepalive.append(bufs)
        ptrs = (ctypes.POINTER(ctypes.c_uint8) * len(s))(
            *[ctypes.cast(b, ctypes.POINTER(ctypes.c_uint8)) for b in bufs])
        seq_ptr_arrays.append(ptrs)
    LensTop = (ctypes.POINTER(ctypes.c_int) * n_sets)(*seq_len_arrays)
    SeqsTop = (ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8)) * n_sets)(*seq_ptr_arrays)
    return NSeqs, LensTop, SeqsTop, (seq_len_arrays, seq_ptr_arrays, keepalive)


class Para(ctypes.Structure):
    pass  # opaque; created/freed by the library


def make_para(lib, workload):
    """abpoa_para_t for the requested workload (field pokes mirror the
    reference CLI flags; the default path touches nothing)."""
    para = lib.abpoa_init_para()
    if workload != "cg":
        from abpoa_amd.pyabpoa import ParaT
        p = ctypes.cast(para, ctypes.POINTER(ParaT)).contents
        _, align_mode, gapk = WORKLOADS[workload]
        p.align_mode = align_mode
        if gapk == "affine":
            p.gap_open1, p.gap_ext1, p.gap_open2, p.gap_ext2 = 4, 2, 0, 0
        elif gapk == "linear":
            p.gap_open1, p.gap_ext1, p.gap_open2, p.gap_ext2 = 0, 2, 0, 0
        elif gapk == "aa":
            p.m = 27
            libc0 = ctypes.CDLL(None)
            libc0.realloc.restype = ctypes.c_void_p
            libc0.realloc.argtypes = [ctypes.c_void_p, ctypes.c_size_t]
            newmat = libc0.realloc(ctypes.cast(p.mat, ctypes.c_void_p), 27 * 27 * 4)
            p.mat = ctypes.cast(newmat, ctypes.POINTER(ctypes.c_int))
            p.use_score_matrix = 1
            p.gap_open1, p.gap_ext1, p.gap_open2, p.gap_ext2 = 4, 2, 0, 0
            mtx = os.path.join(ROOT, "tests", "golden", "BLOSUM62.mtx")
            libc = ctypes.CDLL(None)
            libc.strdup.restype = ctypes.c_void_p
            libc.strdup.argtypes = [ctypes.c_char_p]
            p.mat_fn = libc.strdup(mtx.encode())
        if workload == "extend":
            p.zdrop, p.end_bonus = 100, -1
    lib.abpoa_post_set_para(para)
    return para


def native_setup():
    import abpoa_amd
    lib = abpoa_amd.lib()
    lib.abpoa_init_para.restype = ctypes.c_void_p
    lib.abpoa_post_set_para.argtypes = [ctypes.c_void_p]
    lib.abpoa_free_para.argtypes = [ctypes.c_void_p]
    lib.abpoa_amd_msa_batch.argtypes = [
        ctypes.c_void_p, ctypes.c_int,
        ctypes.POINTER(ctypes.c_int),
        ctypes.POINTER(ctypes.POINTER(ctypes.c_int)),
        ctypes.POINTER(ctypes.POINTER(ctypes.POINTER(ctypes.c_uint8))),
        ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int]
    lib.abpoa_amd_msa_batch.restype = ctypes.c_int
    return lib


def run_step(lib, para, sets, n_threads):
    NSeqs, LensTop, SeqsTop, keep = build_batch_args(lib, sets)
    rc = lib.abpoa_amd_msa_batch(para, len(sets), NSeqs, LensTop, SeqsTop,
                                 None, None, n_threads)
    assert rc == 0
    del keep


def _cpu_model():
    try:
        with open("/proc/cpuinfo") as f:
            for line in f:
                if line.startswith("model name"):
                    return line.split(":", 1)[1].strip()
    except OSError:
        pass
    return "unknown"


def cpu_baseline_leg(depth, qlen):
    """Time the unmodified reference binary (bounded sample).

    Reports the measured single-core figure AND the full-socket framing the
    north star prices against: a linear extrapolation to every core of this
    host (the reference binary is single-threaded), plus, when
    ABPOA_BENCH_CPU_SOCKET is set, an ACTUAL concurrent run of one process
    per set across the socket (bounded to ~30 s)."""
    ref_bin = os.path.join(ROOT, "oracle", "_ref", "abpoa")
    if not os.path.exists(ref_bin) or os.environ.get("ABPOA_BENCH_SKIP_CPU"):
        return None
    import tempfile
    nthreads = os.cpu_count() or 1
    # one physical socket (the north star's framing): cores/socket from cpuinfo
    socket_cores = None
    try:
        with open("/proc/cpuinfo") as f:
            for ln in f:
                if ln.startswith("cpu cores"):
                    socket_cores = int(ln.split(":")[1])
                    break
    except (OSError, ValueError):
        pass
    ncores = socket_cores or nthreads
    out = None
    with tempfile.TemporaryDirectory() as td:
        fa = os.path.join(td, "cpu.fa")
        subprocess.run([sys.executable, os.path.join(ROOT, "tests", "make_synth.py"), fa,
                        "--seed", "99", "--len", str(qlen), "--depth", str(depth)],
                       check=True, stderr=subprocess.DEVNULL)
        t0 = time.monotonic()
        subprocess.run([ref_bin, fa], check=True, stdout=subprocess.DEVNULL,
                       stderr=subprocess.DEVNULL)
        dt = time.monotonic() - t0
        out = {"value": 1.0 / dt, "unit": "sets/s", "cores": 1, "kind": "reference",
               "sample": "1 set (%d reads x %d bp), reference binary, cold arena, host: %s"
                         % (depth, qlen, _cpu_model()),
               "socket_cores": ncores,
               "socket_value_linear_extrapolation": round(ncores / dt, 4),
               "machine_threads": nthreads,
               "socket_note": "reference is single-threaded; socket figure is value x %d "
                              "physical cores of one socket (machine has %d logical CPUs)"
                              % (ncores, nthreads)}
        if os.environ.get("ABPOA_BENCH_CPU_SOCKET"):
            # actual concurrent socket sample: one reference process per
            # core, each given a 3-set list (-l) so its quadratic arena
            # amortizes past the first set (batch mode, abpoa.c:152-161)
            nproc = min(ncores, 32)
            per = 3
            lists = []
            for i in range(nproc):
                fs = []
                for j in range(per):
                    f2 = os.path.join(td, "cpu%d_%d.fa" % (i, j))
                    subprocess.run([sys.executable,
                                    os.path.join(ROOT, "tests", "make_synth.py"),
                                    f2, "--seed", str(100 + i * per + j), "--len", str(qlen),
                                    "--depth", str(depth)], check=True,
                                   stderr=subprocess.DEVNULL)
                    fs.append(f2)
                lf = os.path.join(td, "list%d.txt" % i)
                with open(lf, "w") as f:
                    f.write("\n".join(fs) + "\n")
                lists.append(lf)
            t0 = time.monotonic()
            procs = [subprocess.Popen([ref_bin, "-l", lf], stdout=subprocess.DEVNULL,
                                      stderr=subprocess.DEVNULL) for lf in lists]
            for p in procs:
                assert p.wait() == 0
            dt2 = time.monotonic() - t0
            out["socket_value_measured"] = round(nproc * per / dt2, 4)
            out["socket_measured_procs"] = nproc
            out["socket_measured_sets_per_proc"] = per
    return out


def traffic_probe(args):
    """Counter-measured HBM bytes per DP-kernel launch (roofline.traffic).

    Two separate rocprofv3 --pmc passes over a small probe run (FETCH_SIZE
    and WRITE_SIZE cannot share a TCC pass), collected per the MI355X guide:
    FETCH_SIZE on gfx950 reports half the bytes of wide coalesced reads, so
    it is doubled; WRITE_SIZE is used as reported. Returns bytes per
    DP-kernel launch (all gap-mode kernels counted), or None."""
    import csv as _csv
    import glob as _glob
    import shutil
    import tempfile
    if os.environ.get("ABPOA_BENCH_NO_TRAFFIC") or not shutil.which("rocprofv3"):
        return None
    n_probe_sets = 16
    res = {}
    with tempfile.TemporaryDirectory(dir="/tmp") as td:
        for pmc in ("FETCH_SIZE", "WRITE_SIZE"):
            env = dict(os.environ)
            env["TMPDIR"] = "/tmp"
            env["ABPOA_BENCH_NO_TRAFFIC"] = "1"
            env["ABPOA_BENCH_SKIP_CPU"] = "1"
            odir = os.path.join(td, pmc)
            try:
                r = subprocess.run(
                    ["rocprofv3", "--pmc", pmc, "--output-format", "csv",
                     "-d", odir, "-o", "probe", "--",
                     sys.executable, os.path.abspath(__file__),
                     "--sets-per-step", str(n_probe_sets), "--steps", "1", "--warmup", "0",
                     "--depth", str(args.depth), "--qlen", str(args.qlen),
                     "--workload", args.workload],
                    cwd="/tmp", env=env, stdout=subprocess.PIPE,
                    stderr=subprocess.DEVNULL, timeout=420)
            except (OSError, subprocess.TimeoutExpired):
                return None
            if r.returncode != 0:
                return None
            probe_cells = 0
            for ln in (r.stdout or b"").decode(errors="replace").splitlines():
                if ln.startswith("{"):
                    try:
                        probe_cells = json.loads(ln)["config"]["total_cells"]
                    except (ValueError, KeyError):
                        pass
            total = 0.0
            n_disp = 0
            for fn in _glob.glob(os.path.join(odir, "**", "*counter_collection.csv"),
                                 recursive=True):
                with open(fn) as f:
                    for row in _csv.DictReader(f):
                        kname = row.get("Kernel_Name", "")
                        if "global_mw_kernel" in kname or "ag_global_kernel" in kname                                 or "lg_global_kernel" in kname:
                            total += float(row.get("Counter_Value", 0) or 0)
                            n_disp += 1
            if n_disp == 0:
                return None
            res[pmc] = (total, n_disp, probe_cells)
    fetch, nf, pcells = res["FETCH_SIZE"]
    write, nw, _ = res["WRITE_SIZE"]
    if pcells <= 0:
        return None
    # *_SIZE counters report kilobytes; FETCH_SIZE doubled per the gfx950
    # calibration (MI355X_MICROARCH.md: wide coalesced reads tally at half)
    bytes_per_cell = (2.0 * fetch + write) * 1024.0 / pcells
    return {"bytes_per_cell": bytes_per_cell, "probe_sets": n_probe_sets,
            "fetch_corrected_x2": True, "dispatches": max(nf, nw)}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=1)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--sets-per-step", type=int, default=1000)
    ap.add_argument("--depth", type=int, default=50)
    ap.add_argument("--qlen", type=int, default=10000)
    ap.add_argument("--threads", type=int, default=max(2, (os.cpu_count() or 8) - 2))
    ap.add_argument("--dry-run", action="store_true",
                    help="exercise the distributed harness without a GPU (CPU tests)")
    ap.add_argument("--workload", choices=sorted(WORKLOADS), default="cg",
                    help="BASELINE config variant; default = configs[1] (the driver's contract)")
    args = ap.parse_args()
    if args.workload == "aa":
        # configs[3]: 30 reads x 2 kaa unless explicitly overridden
        if args.depth == 50: args.depth = 30
        if args.qlen == 10000: args.qlen = 2000

    import numpy as np
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    # One process per GPU: pin the device BEFORE any HIP init. torch's bundled
    # ROCm runtime must never initialize in this process (it conflicts with
    # the system ROCm the native library links), so rank coordination uses
    # gloo and device work is synchronized inside the library
    # (hipStreamSynchronize at every batch end).
    if "HIP_VISIBLE_DEVICES" not in os.environ and "LOCAL_RANK" in os.environ:
        os.environ["HIP_VISIBLE_DEVICES"] = os.environ["LOCAL_RANK"]
    dist = None
    if world > 1 or os.environ.get("MASTER_ADDR"):
        import torch.distributed as tdist
        tdist.init_process_group(backend="gloo")
        dist = tdist
        rank = tdist.get_rank()
        world = tdist.get_world_size()

    if args.dry_run:
        lib = para = None
    else:
        lib = native_setup()
        para = make_para(lib, args.workload)
        import abpoa_amd

    rng = np.random.default_rng(4242 + 1000 * rank)
    alphabet = 20 if args.workload == "aa" else 4

    # Pregenerate every step's input OUTSIDE the timed region (the metric is
    # quoted with inputs resident; generation is not part of the hot path) and
    # prebuild the ctypes argument trees so the timed loop only runs the
    # native driver.
    prepared = []
    if not args.dry_run:
        for _ in range(args.warmup + args.steps):
            sets = gen_sets(rng, args.sets_per_step, args.depth, args.qlen, alphabet=alphabet)
            prepared.append(build_batch_args(lib, sets))

    def one_step(step_idx):
        if args.dry_run:
            time.sleep(0.01)
            return
        NSeqs, LensTop, SeqsTop, _keep = prepared[step_idx]
        rc = lib.abpoa_amd_msa_batch(para, args.sets_per_step, NSeqs, LensTop,
                                     SeqsTop, None, None, args.threads)
        assert rc == 0

    def barrier_sync():
        # device work is already drained: the native driver ends every batch
        # with hipStreamSynchronize on its own stream
        if dist is not None:
            dist.barrier()

    for wstep in range(args.warmup):
        one_step(wstep)

    if not args.dry_run:
        import abpoa_amd
        abpoa_amd.reset_stats()
        try:
            abpoa_amd.lib().abpoa_amd_reset_gpu_spans()
        except AttributeError:
            pass
    barrier_sync()
    t0 = time.monotonic()
    for k in range(args.steps):
        one_step(args.warmup + k)
    barrier_sync()
    elapsed = time.monotonic() - t0

    # max over ranks
    if dist is not None:
        import torch
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        total_sets = args.steps * args.sets_per_step * world
        value = total_sets / elapsed
        if args.dry_run:
            line = {"metric": "read-sets/sec, 50x10 kbp global convex-gap POA (DRY RUN)",
                    "value": value, "unit": "sets/s", "n_gpus": world,
                    "steps": args.steps, "warmup": args.warmup,
                    "ms_per_step": elapsed / args.steps * 1e3,
                    "higher_is_better": True, "scaling": "weak", "dry_run": True}
            print(json.dumps(line), flush=True)
            return
        import abpoa_amd
        cells, kns, launches = abpoa_amd.get_stats()
        import ctypes as _ct
        _ab = _ct.c_uint64()
        abpoa_amd.lib().abpoa_amd_get_stats2(_ct.byref(_ab))
        alg_bytes = _ab.value or cells * ALG_BYTES_PER_CELL
        gcells_s = cells / elapsed / 1e9
        # kernel time, split: summed per-launch event time per
        # kernel kind AND the wall-clock union of kernel spans across the
        # overlapped streams (the sums exceed the wall under overlap; the
        # union never does)
        dpms = _ct.c_double(); foldms = _ct.c_double(); busyms = _ct.c_double()
        try:
            abpoa_amd.lib().abpoa_amd_get_gpu_spans(
                _ct.byref(busyms), _ct.byref(dpms), _ct.byref(foldms))
            dp_s, fold_s, busy_s = dpms.value / 1e3, foldms.value / 1e3, busyms.value / 1e3
        except AttributeError:
            dp_s = kns / 1e9; fold_s = 0.0; busy_s = None
        ach = alg_bytes / max(kns, 1)  # bytes/ns == GB/s (per summed DP event time)
        try:
            traffic = traffic_probe(args) if world == 1 else None
        except Exception:
            traffic = None  # traffic is evidence, never a reason to fail the run
        tval = None
        if traffic:
            # counter-measured bytes/cell (probe) x this run's cells/launch
            tval = round(traffic["bytes_per_cell"] * cells / max(launches, 1))
        roofline = {"bound": "hbm", "achieved": round(ach, 1), "peak": HBM_PEAK_GBPS,
                    "unit": "GB/s", "frac": round(ach / HBM_PEAK_GBPS, 4),
                    "traffic": tval,
                    "traffic_note": ("PMC FETCH_SIZE(x2 gfx950 correction)+WRITE_SIZE per DP-kernel launch, "
                                     "separate --pmc passes on a %d-set probe, scaled by cells"
                                     % traffic["probe_sets"]) if traffic else None}
        cpu = cpu_baseline_leg(args.depth, args.qlen) if world == 1 else None
        wl_desc = WORKLOAD if args.workload == "cg" else (
            "%d synthetic sets x %d reads x %d bp, %s, 1 MI355X"
            % (args.sets_per_step, args.depth, args.qlen, WORKLOADS[args.workload][0]))
        metric = ("read-sets/sec, 50x10 kbp global convex-gap POA" if args.workload == "cg"
                  else "read-sets/sec, %s" % WORKLOADS[args.workload][0])
        line = {
            "metric": metric,
            "value": round(value, 4), "unit": "sets/s",
            "n_gpus": world, "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 1),
            "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
            "dtype": "int16+int32 rescore", "data": "synthetic",
            "config": {"workload": wl_desc, "sets_per_step": args.sets_per_step,
                       "depth": args.depth, "qlen": args.qlen,
                       "gcells_per_s": round(gcells_s, 3),
                       "total_cells": cells,
                       "kernel_s": round(kns / 1e9, 3),
                       "dp_kernel_s": round(dp_s, 3),
                       "fold_kernel_s": round(fold_s, 3),
                       "gpu_busy_s": round(busy_s, 3) if busy_s is not None else None,
                       "launches": launches,
                       "parallelism": "dp%d independent read-set shards (no data-path collective)" % world},
            "roofline": roofline,
            "cpu_baseline": cpu,
        }
        print(json.dumps(line), flush=True)


if __name__ == "__main__":
    main()
