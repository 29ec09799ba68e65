"""Tool smoke coverage: ucc_info introspection and ucc_perftest modes
(reference tools/info + tools/perf roles). These run the built
binaries on the host path; heavier sweeps live in `make asan`."""

import os
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
INFO = os.path.join(REPO, "build", "ucc_info")
PERF = os.path.join(REPO, "build", "ucc_perftest")

pytestmark = pytest.mark.skipif(
    not (os.path.exists(INFO) and os.path.exists(PERF)),
    reason="tools not built")


def _run(cmd, timeout=300, env_extra=None):
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    if env_extra:
        env.update(env_extra)
    p = subprocess.run(cmd, capture_output=True, text=True,
                       timeout=timeout, env=env, cwd=REPO)
    return p


def test_info_version_configs_scoremap():
    p = _run([INFO, "-v", "-c", "-s"])
    assert p.returncode == 0, p.stderr[-500:]
    out = p.stdout
    assert "UCC-AMD" in out and "gfx950" in out
    # config dump lists TL knobs with docs
    assert "UCC_TL_TCP_KN_RADIX=" in out
    assert "UCC_TL_CDNA4_CHUNK_SIZE=" in out
    assert "UCC_TL_SHM_SOCKET_STAGING=" in out
    # simulated 2-rank team score map
    assert "allreduce:host" in out


@pytest.mark.parametrize("coll", ["allreduce", "allgather", "bcast",
                                  "reduce_scatter", "reduce_scatterv",
                                  "gatherv", "scatterv",
                                  "allgatherv"])
def test_perftest_validated(coll):
    """-C runs a golden-pattern validation iteration per size."""
    p = _run([PERF, "-c", coll, "-j", "3", "-b", "64", "-e", "16384",
              "-n", "2", "-w", "1", "-C"])
    assert p.returncode == 0, (p.stdout[-300:], p.stderr[-500:])
    assert "bytes" in p.stdout


def test_perftest_root_shift_and_inplace():
    p = _run([PERF, "-c", "bcast", "-j", "4", "-b", "256", "-e",
              "4096", "-n", "4", "-w", "1", "-R"])
    assert p.returncode == 0, p.stderr[-500:]
    p = _run([PERF, "-c", "allreduce", "-j", "3", "-b", "256", "-e",
              "4096", "-n", "2", "-w", "1", "-i", "-C"])
    assert p.returncode == 0, (p.stdout[-300:], p.stderr[-500:])


def test_perftest_persistent_and_skewed():
    p = _run([PERF, "-c", "allreduce", "-j", "3", "-b", "256", "-e",
              "8192", "-n", "2", "-w", "1", "-F"])
    assert p.returncode == 0, p.stderr[-500:]
    p = _run([PERF, "-c", "alltoallv", "-j", "4", "-b", "1024", "-e",
              "16384", "-n", "2", "-w", "1", "-M"])
    assert p.returncode == 0, p.stderr[-500:]
