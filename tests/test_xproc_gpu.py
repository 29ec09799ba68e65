"""Cross-process device tests: 2 OS processes (torchrun) sharing the GPUs
of the box (both on cuda:0 on a 1-GPU box). Validates hipIpc peer
mappings, the cross-process fused kernel, triggered post, and hipGraph
capture/replay — the production topology bench.py uses at N>1."""

import os
import subprocess
import sys

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_xproc_device_colls():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         "--master-port=29517", os.path.join(REPO, "tests",
                                             "xproc_worker.py")],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600)
    sys.stdout.write(proc.stdout[-3000:])
    sys.stderr.write(proc.stderr[-3000:])
    assert proc.returncode == 0
    assert proc.stdout.count("XPROC_OK") == 2


def test_xproc_stress():
    """Mixed-collective cross-process stress (slot rotation, parity and
    counter continuity across fused/gated/staged paths)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         "--master-port=29527", os.path.join(REPO, "tests",
                                             "stress_worker.py"), "60"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600)
    sys.stdout.write(proc.stdout[-2000:])
    sys.stderr.write(proc.stderr[-2000:])
    assert proc.returncode == 0
    assert proc.stdout.count("STRESS_OK") == 2


def test_xproc_spin_timeout_recovery():
    """A dead peer degrades to UCC_ERR_TIMED_OUT via the bounded kernel
    spin — the GPU stays usable (docs/GATED_PIPELINE.md safety)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["UCC_TL_CDNA4_SPIN_LIMIT"] = "3000000"  # ~ms-scale bound
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         "--master-port=29533", os.path.join(REPO, "tests",
                                             "timeout_worker.py")],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300)
    sys.stdout.write(proc.stdout[-2000:])
    sys.stderr.write(proc.stderr[-2000:])
    assert proc.returncode == 0
    assert proc.stdout.count("TIMEOUT_OK") == 2
