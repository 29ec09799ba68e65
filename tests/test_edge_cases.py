"""Edge cases: zero-count contributions in v-collectives, 1-element
messages, non-power-of-two teams — the shapes that break staging
protocols (round-count agreement, empty-fragment publication)."""

import numpy as np
import pytest

from ucc_amd import dtypes
from ucc_amd.testing import LocalJob


@pytest.fixture(scope="module", params=[3, 5])
def job(request):
    return LocalJob(request.param)


def test_allgatherv_zero_count(job):
    n = job.n
    rng = np.random.default_rng(31)
    cnts = [0 if r == 1 else (r + 1) * 50 for r in range(n)]
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).tolist()
    total = sum(cnts)
    srcs = [rng.random(max(c, 1)).astype(np.float32) for c in cnts]
    dsts = [np.zeros(max(total, 1), np.float32) for _ in range(n)]
    reqs = job.coll("allgatherv", [
        dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
             count=cnts[r], dt=dtypes.FLOAT32,
             dst_counts=cnts, dst_displs=dsps) for r in range(n)])
    job.run(reqs)
    exp = np.concatenate([srcs[r][:cnts[r]] for r in range(n)])
    for d in dsts:
        np.testing.assert_array_equal(d[:total], exp)


def test_alltoallv_zero_row(job):
    n = job.n
    rng = np.random.default_rng(32)
    # rank 0 sends nothing at all; others send (r)*(d+1)*20
    scnt = [[0] * n if r == 0 else [(r) * (d + 1) * 20 for d in range(n)]
            for r in range(n)]
    rcnt = [[scnt[s][r] for s in range(n)] for r in range(n)]
    sdsp = [np.concatenate([[0], np.cumsum(c)[:-1]]).tolist()
            for c in scnt]
    rdsp = [np.concatenate([[0], np.cumsum(c)[:-1]]).tolist()
            for c in rcnt]
    srcs = [rng.random(max(sum(scnt[r]), 1)).astype(np.float32)
            for r in range(n)]
    dsts = [np.zeros(max(sum(rcnt[r]), 1), np.float32) for r in range(n)]
    reqs = job.coll("alltoallv", [
        dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data, count=0,
             dt=dtypes.FLOAT32,
             src_counts=scnt[r], src_displs=sdsp[r],
             dst_counts=rcnt[r], dst_displs=rdsp[r])
        for r in range(n)])
    job.run(reqs)
    for r in range(n):
        for s in range(n):
            got = dsts[r][rdsp[r][s]:rdsp[r][s] + rcnt[r][s]]
            exp = srcs[s][sdsp[s][r]:sdsp[s][r] + scnt[s][r]]
            np.testing.assert_array_equal(got, exp)


def test_reduce_scatterv_zero_count(job):
    n = job.n
    rng = np.random.default_rng(33)
    cnts = [0 if r == n - 1 else 40 * (r + 1) for r in range(n)]
    total = sum(cnts)
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).tolist()
    srcs = [rng.random(total).astype(np.float32) for _ in range(n)]
    dsts = [np.zeros(max(cnts[r], 1), np.float32) for r in range(n)]
    reqs = job.coll("reduce_scatterv", [
        dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data, count=0,
             dt=dtypes.FLOAT32, dst_counts=cnts) for r in range(n)])
    job.run(reqs)
    exp = np.sum(srcs, axis=0)
    for r in range(n):
        if cnts[r]:
            np.testing.assert_allclose(
                dsts[r][:cnts[r]], exp[dsps[r]:dsps[r] + cnts[r]],
                rtol=1e-5, atol=1e-5)


def test_one_element_colls(job):
    n = job.n
    arrs = [np.array([float(r + 1)], np.float32) for r in range(n)]
    outs = job.allreduce_np(arrs)
    for o in outs:
        np.testing.assert_allclose(o, [sum(range(1, n + 1))])
    # 1-element bcast
    bufs = [np.array([-1.0], np.float64) for _ in range(n)]
    bufs[0][0] = 42.0
    reqs = job.coll("bcast", [
        dict(src=b.ctypes.data, dst=0, count=1, dt=dtypes.FLOAT64, root=0)
        for b in bufs])
    job.run(reqs)
    for b in bufs:
        assert b[0] == 42.0


def test_zero_count_allreduce(job):
    """count=0 fixed-count colls complete via the stub fast path."""
    n = job.n
    reqs = job.coll("allreduce", [
        dict(src=0, dst=0, count=0, dt=dtypes.FLOAT32)
        for _ in range(n)])
    job.run(reqs)
