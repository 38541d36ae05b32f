"""CPU parity for the batched pipeline driver (no GPU needed).

The batchtest tool runs abpoa_amd_msa_batch with the oracle injected through
the dispatch seam; its output must be byte-identical across every pipeline
shape (ABPOA_AMD_GROUPS=1/2/3 changes group count and launch lookahead) and
must equal the sequential CLI consensus on each set."""
import os
import subprocess

from conftest import ROOT, CPUTEST_BIN as CPU_BIN, ORACLE_SO, run_stdout

BATCH_BIN = os.path.join(ROOT, "abpoa_amd", "csrc", "abpoa_amd_batchtest")


def _gen_sets(tmp_path, n_sets=5, depth=8, length=300, seed=77):
    import random
    paths = []
    for s in range(n_sets):
        rng = random.Random(seed + s)
        ref = "".join(rng.choice("ACGT") for _ in range(length))
        p = tmp_path / ("set%d.fa" % s)
        with open(p, "w") as f:
            for i in range(depth):
                out = []
                for ch in ref:
                    r = rng.random()
                    if r < 0.05:
                        out.append(rng.choice("ACGT"))
                    elif r < 0.08:
                        pass
                    elif r < 0.11:
                        out.extend((ch, rng.choice("ACGT")))
                    else:
                        out.append(ch)
                f.write(">r%d\n%s\n" % (i, "".join(out)))
        paths.append(str(p))
    return paths


def test_batch_pipeline_orders_and_cli_parity(tmp_path):
    paths = _gen_sets(tmp_path)
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    outs = {}
    for g in ("1", "2", "3", "4", "5", "6"):
        e = dict(env)
        e["ABPOA_AMD_GROUPS"] = g
        outs[g] = subprocess.run([BATCH_BIN] + paths, env=e, check=True,
                                 stdout=subprocess.PIPE,
                                 stderr=subprocess.DEVNULL).stdout
    assert all(outs[g] == outs["1"] for g in outs), \
        "batch consensus differs across pipeline group counts"

    # per-set parity with the sequential CLI path
    recs = outs["1"].decode().split(">")[1:]
    batch_cons = ["".join(r.splitlines()[1:]) for r in recs]
    for i, p in enumerate(paths):
        cli = run_stdout([CPU_BIN, p], env=env).decode()
        seq = "".join(cli.splitlines()[1:])
        assert batch_cons[i] == seq, "batch/CLI mismatch on set %d" % i


def test_batch_single_set(tmp_path):
    """n_sets=1 exercises the degenerate LA=0 pipeline (strict
    build->launch->finish->fold per round)."""
    paths = _gen_sets(tmp_path, n_sets=1, depth=12, length=400, seed=9)
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    out = subprocess.run([BATCH_BIN] + paths, env=env, check=True,
                         stdout=subprocess.PIPE,
                         stderr=subprocess.DEVNULL).stdout
    batch_cons = "".join(out.decode().split(">")[1].splitlines()[1:])
    cli = run_stdout([CPU_BIN, paths[0]], env=env).decode()
    assert batch_cons == "".join(cli.splitlines()[1:])
