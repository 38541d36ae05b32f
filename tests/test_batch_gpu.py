"""Batched driver parity (-m gpu): the lockstep multi-set batch path must
produce exactly the consensus the sequential CLI path produces."""
import os
import subprocess
import sys
import pytest

from conftest import ROOT, GPU_BIN, run_stdout

pytestmark = pytest.mark.gpu


def test_batch_matches_cli(tmp_path):
    sys.path.insert(0, ROOT)
    import abpoa_amd
    import numpy as np
    rng = np.random.default_rng(123)
    sys.path.insert(0, ROOT)
    import importlib
    bench = importlib.import_module("bench")
    sets = bench.gen_sets(rng, 4, depth=20, qlen=800)
    # batched GPU consensus (device-resident driver: graphs fold on the GPU)
    cons = abpoa_amd.msa_batch_consensus(sets, n_threads=2)
    # sequential CLI on the same sets
    code2ch = "ACGT"
    for i, s in enumerate(sets):
        fa = tmp_path / ("s%d.fa" % i)
        with open(fa, "w") as f:
            for r_i, r in enumerate(s):
                f.write(">r%d\n%s\n" % (r_i, "".join(code2ch[b] for b in r)))
        out = run_stdout([GPU_BIN, str(fa)]).decode()
        seq = "".join(out.splitlines()[1:])
        assert cons[i] == seq, "batch/CLI consensus mismatch on set %d" % i


def test_resident_matches_hostfold():
    """The device-resident batch driver (GPU fold + device-built CSR) must be
    byte-identical to the round-1 host-fold driver on the same inputs —
    including the MF-consensus config, which exercises the per-edge read-id
    bitsets maintained on device."""
    sys.path.insert(0, ROOT)
    import abpoa_amd
    import importlib
    import numpy as np
    bench = importlib.import_module("bench")
    rng = np.random.default_rng(321)
    sets = bench.gen_sets(rng, 6, depth=15, qlen=600)
    for algrm in (None, "MF"):
        os.environ.pop("ABPOA_AMD_HOST_FOLD", None)
        resident = abpoa_amd.msa_batch_consensus(sets, n_threads=2, cons_algrm=algrm)
        os.environ["ABPOA_AMD_HOST_FOLD"] = "1"
        try:
            hostfold = abpoa_amd.msa_batch_consensus(sets, n_threads=2, cons_algrm=algrm)
        finally:
            os.environ.pop("ABPOA_AMD_HOST_FOLD", None)
        assert resident == hostfold, "resident/host-fold mismatch (cons_algrm=%s)" % algrm


def test_resident_multiword_readids():
    """Depth-80 MF-consensus sets: per-edge read-id bitsets span TWO 64-bit
    words (rid_n = 2) in the device-resident pools — byte-compare against
    the host-fold driver."""
    sys.path.insert(0, ROOT)
    import abpoa_amd
    import importlib
    import numpy as np
    bench = importlib.import_module("bench")
    rng = np.random.default_rng(88)
    sets = bench.gen_sets(rng, 3, depth=80, qlen=400)
    resident = abpoa_amd.msa_batch_consensus(sets, n_threads=2, cons_algrm="MF")
    os.environ["ABPOA_AMD_HOST_FOLD"] = "1"
    try:
        hostfold = abpoa_amd.msa_batch_consensus(sets, n_threads=2, cons_algrm="MF")
    finally:
        os.environ.pop("ABPOA_AMD_HOST_FOLD", None)
    assert resident == hostfold, "multi-word read-id bitset divergence"


def test_resident_pool_expansion():
    """Tiny ABPOA_AMD_NODE_ALPHA forces graph-pool overflows mid-run: the
    expand + device-to-device move + refold path must still be byte-exact."""
    sys.path.insert(0, ROOT)
    import abpoa_amd
    import importlib
    import numpy as np
    bench = importlib.import_module("bench")
    rng = np.random.default_rng(77)
    sets = bench.gen_sets(rng, 3, depth=20, qlen=500)
    os.environ["ABPOA_AMD_NODE_ALPHA"] = "0.02"
    try:
        tight = abpoa_amd.msa_batch_consensus(sets, n_threads=2)
    finally:
        os.environ.pop("ABPOA_AMD_NODE_ALPHA", None)
    roomy = abpoa_amd.msa_batch_consensus(sets, n_threads=2)
    assert tight == roomy, "pool-expansion path changed the consensus"


def test_two_rank_one_gpu_smoke(tmp_path):
    """Sharding-path hardware smoke: two ranks
    time-slice ONE GPU (gloo rendezvous, HIP_VISIBLE_DEVICES pinned to 0 for
    both), each processing its shard through the resident driver — the exact
    code path the driver's 8-GPU scaling run takes, minus the extra devices."""
    import json
    import subprocess
    env = dict(os.environ)
    env["HIP_VISIBLE_DEVICES"] = "0"
    env["ABPOA_BENCH_SKIP_CPU"] = "1"
    env["ABPOA_BENCH_NO_TRAFFIC"] = "1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29417", os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--steps", "1", "--warmup", "0",
         "--sets-per-step", "6", "--qlen", "600", "--depth", "10"],
        env=env, cwd=ROOT, stdout=subprocess.PIPE, stderr=subprocess.PIPE, timeout=420)
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    line = [l for l in r.stdout.decode().splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2 and out["value"] > 0
