"""Host-path mixed-collective stress: random coll/size/dtype sequence
through the in-process jig (shm TL; tcp variant via env in
test_multiproc). Complements the per-coll tests with slot-rotation and
protocol-interleaving coverage."""

import random

import numpy as np
import pytest

from ucc_amd import dtypes
from ucc_amd.testing import LocalJob


@pytest.mark.parametrize("n", [2, 5])
def test_host_mixed_stress(n):
    job = LocalJob(n)
    c = job.c
    rng = random.Random(7)
    for it in range(40):
        coll = rng.choice(["allreduce", "reduce_scatter", "allgather",
                           "alltoall", "bcast", "barrier"])
        per = rng.choice([16, 1000, 50_000])
        g = np.random.default_rng(50_000 + it)
        if coll == "allreduce":
            arrs = [(g.random(per) - 0.5).astype(np.float32)
                    for _ in range(n)]
            outs = job.allreduce_np(arrs)
            exp = np.sum(arrs, axis=0)
            for o in outs:
                np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-5)
        elif coll == "reduce_scatter":
            srcs = [(g.random(per * n) - 0.5).astype(np.float32)
                    for _ in range(n)]
            dsts = [np.zeros(per, np.float32) for _ in range(n)]
            reqs = job.coll(coll, [
                dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                     count=per, dt=dtypes.FLOAT32) for r in range(n)])
            job.run(reqs)
            exp = np.sum(srcs, axis=0)
            for r in range(n):
                np.testing.assert_allclose(
                    dsts[r], exp[r * per:(r + 1) * per], rtol=1e-5,
                    atol=1e-5)
        elif coll == "allgather":
            srcs = [(g.random(per)).astype(np.float64) for _ in range(n)]
            dsts = [np.zeros(per * n, np.float64) for _ in range(n)]
            reqs = job.coll(coll, [
                dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                     count=per * n, dt=dtypes.FLOAT64) for r in range(n)])
            job.run(reqs)
            exp = np.concatenate(srcs)
            for d in dsts:
                np.testing.assert_array_equal(d, exp)
        elif coll == "alltoall":
            srcs = [(g.random(per * n)).astype(np.float32)
                    for _ in range(n)]
            dsts = [np.zeros(per * n, np.float32) for _ in range(n)]
            reqs = job.coll(coll, [
                dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                     count=per * n, dt=dtypes.FLOAT32) for r in range(n)])
            job.run(reqs)
            for d in range(n):
                for s in range(n):
                    np.testing.assert_array_equal(
                        dsts[d][s * per:(s + 1) * per],
                        srcs[s][d * per:(d + 1) * per])
        elif coll == "bcast":
            root = it % n
            bufs = [np.zeros(per, np.float32) for _ in range(n)]
            bufs[root][:] = g.random(per).astype(np.float32)
            exp = bufs[root].copy()
            reqs = job.coll(coll, [
                dict(src=b.ctypes.data, dst=0, count=per,
                     dt=dtypes.FLOAT32, root=root) for b in bufs])
            job.run(reqs)
            for b in bufs:
                np.testing.assert_array_equal(b, exp)
        else:
            reqs = job.coll("barrier", [
                dict(src=0, dst=0, count=0, dt=dtypes.FLOAT32)
                for _ in range(n)])
            job.run(reqs)
