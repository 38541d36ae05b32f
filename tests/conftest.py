import os
import subprocess
import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CSRC = os.path.join(ROOT, "abpoa_amd", "csrc")
GOLDEN = os.path.join(ROOT, "tests", "golden")
CPUTEST_BIN = os.path.join(CSRC, "abpoa_amd_cputest")
GPU_BIN = os.path.join(CSRC, "abpoa_amd")
ORACLE_SO = os.path.join(ROOT, "oracle", "liboracle.so")
REF_BIN = os.path.join(ROOT, "oracle", "_ref", "abpoa")


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an AMD GPU (run via gpurun / on an MI355X box)")


def _build_cpu_artifacts():
    if not (os.path.exists(CPUTEST_BIN) and os.path.exists(ORACLE_SO)):
        subprocess.run(["make", "-j4", "cputest"], cwd=CSRC, check=True,
                       stdout=subprocess.DEVNULL)
        subprocess.run(["make", "liboracle.so"], cwd=os.path.join(ROOT, "oracle"),
                       check=True, stdout=subprocess.DEVNULL)


@pytest.fixture(scope="session")
def cputest_bin():
    _build_cpu_artifacts()
    return CPUTEST_BIN


@pytest.fixture(scope="session")
def oracle_env(cputest_bin):
    env = dict(os.environ)
    env["ABPOA_AMD_TEST_ALIGNER_SO"] = ORACLE_SO
    return env


@pytest.fixture(scope="session")
def ref_bin():
    """The reference binary compiled from /root/reference (dev container only)."""
    if not os.path.exists(REF_BIN):
        if os.path.isdir("/root/reference"):
            subprocess.run(["make", "-j4", "all"], cwd=os.path.join(ROOT, "oracle"),
                           check=True, stdout=subprocess.DEVNULL)
    if not os.path.exists(REF_BIN):
        pytest.skip("reference binary unavailable (no /root/reference here)")
    return REF_BIN


def run_stdout(cmd, env=None):
    return subprocess.run(cmd, env=env, check=True, stdout=subprocess.PIPE,
                          stderr=subprocess.DEVNULL).stdout
