"""Seeded randomized collective-sequence stress over the tcp mesh:
random collective types, sizes spanning every score band (so bruck /
sparbit / knomial / linear / ring / SRA / DBT / halving all interleave
on the same per-team tag space), random roots — each op validated
against numpy. Catches cross-algorithm tag/sequence interactions that
single-collective tests cannot."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

FUZZ = r"""
import sys
import numpy as np
sys.path.insert(0, %r)
from ucc_amd import core, dtypes
from ucc_amd.testing import LocalJob

seed = int(sys.argv[1])
n = [2, 3, 4, 6, 7, 8][seed %% 6]  # cover folds/even-n/tree shapes
job = LocalJob(n)
rng = np.random.default_rng(seed)
c = core()

# a persistent allreduce re-posted throughout the sequence (slot and
# tag-space reuse interleaved with every other algorithm)
P = 900
psrc = [np.zeros(P, np.float32) for _ in range(n)]
pdst = [np.zeros(P, np.float32) for _ in range(n)]
preq = job.coll("allreduce", [
    dict(src=psrc[r].ctypes.data, dst=pdst[r].ctypes.data, count=P,
         dt=dtypes.FLOAT32, flags=c.FLAG_PERSISTENT)
    for r in range(n)])

for opno in range(60):
    coll = rng.choice(["allreduce", "allgather", "bcast", "reduce",
                       "reduce_scatter", "alltoall", "barrier",
                       "gather", "scatter", "alltoallv",
                       "reduce_scatterv"])
    # sizes span the latency/mid/bandwidth bands (elements)
    per = int(rng.choice([1, 7, 333, 4096, 20000, 70000]))
    root = int(rng.integers(0, n))
    if coll == "barrier":
        reqs = job.coll("barrier", [
            dict(src=0, dst=0, count=0, dt=dtypes.INT8)
            for _ in range(n)])
        job.run(reqs)
        continue
    if opno %% 9 == 4:  # interleave a persistent re-post
        for r in range(n):
            psrc[r][:] = rng.random(P).astype(np.float32) + opno
        pexp = np.sum(psrc, axis=0)
        job.run(preq)
        for o in pdst:
            np.testing.assert_allclose(o, pexp, rtol=1e-5, atol=1e-4,
                                       err_msg=f"persistent@op{opno}")
    if coll == "allreduce":
        inplace = bool(rng.integers(0, 2))
        arrs = [(rng.random(per) - 0.5).astype(np.float32)
                for _ in range(n)]
        exp = np.sum(arrs, axis=0)
        if inplace:
            reqs = job.coll("allreduce", [
                dict(src=0, dst=arrs[r].ctypes.data, count=per,
                     dt=dtypes.FLOAT32, flags=c.FLAG_IN_PLACE)
                for r in range(n)])
            job.run(reqs)
            outs = arrs
        else:
            outs = [np.zeros(per, np.float32) for _ in range(n)]
            reqs = job.coll("allreduce", [
                dict(src=arrs[r].ctypes.data, dst=outs[r].ctypes.data,
                     count=per, dt=dtypes.FLOAT32)
                for r in range(n)])
            job.run(reqs)
        for o in outs:
            np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-5,
                                       err_msg=f"op{opno}")
    elif coll == "allgather":
        inplace = bool(rng.integers(0, 2))
        blks = [rng.standard_normal(per).astype(np.float32)
                for _ in range(n)]
        dsts = [np.zeros(per * n, np.float32) for _ in range(n)]
        if inplace:
            for r in range(n):
                dsts[r][r * per:(r + 1) * per] = blks[r]
            reqs = job.coll("allgather", [
                dict(src=0, dst=dsts[r].ctypes.data, count=per * n,
                     dt=dtypes.FLOAT32, flags=c.FLAG_IN_PLACE)
                for r in range(n)])
        else:
            reqs = job.coll("allgather", [
                dict(src=blks[r].ctypes.data, dst=dsts[r].ctypes.data,
                     count=per * n, dt=dtypes.FLOAT32)
                for r in range(n)])
        job.run(reqs)
        exp = np.concatenate(blks)
        for d in dsts:
            np.testing.assert_array_equal(d, exp, err_msg=f"op{opno}")
    elif coll == "bcast":
        bufs = [np.zeros(per, np.float64) for _ in range(n)]
        bufs[root][:] = rng.random(per)
        exp = bufs[root].copy()
        reqs = job.coll("bcast", [
            dict(src=b.ctypes.data, dst=0, count=per,
                 dt=dtypes.FLOAT64, root=root) for b in bufs])
        job.run(reqs)
        for b in bufs:
            np.testing.assert_array_equal(b, exp, err_msg=f"op{opno}")
    elif coll == "reduce":
        srcs = [(rng.random(per) - 0.5).astype(np.float32)
                for _ in range(n)]
        dst = np.zeros(per, np.float32)
        reqs = job.coll("reduce", [
            dict(src=srcs[r].ctypes.data,
                 dst=dst.ctypes.data if r == root else 0, count=per,
                 dt=dtypes.FLOAT32, root=root) for r in range(n)])
        job.run(reqs)
        np.testing.assert_allclose(dst, np.sum(srcs, axis=0),
                                   rtol=1e-5, atol=1e-5,
                                   err_msg=f"op{opno}")
    elif coll == "reduce_scatter":
        srcs = [(rng.random(per * n) - 0.5).astype(np.float32)
                for _ in range(n)]
        dsts = [np.zeros(per, np.float32) for _ in range(n)]
        reqs = job.coll("reduce_scatter", [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=per, dt=dtypes.FLOAT32) for r in range(n)])
        job.run(reqs)
        exp = np.sum(srcs, axis=0)
        for r in range(n):
            np.testing.assert_allclose(
                dsts[r], exp[r * per:(r + 1) * per], rtol=1e-5,
                atol=1e-5, err_msg=f"op{opno}")
    elif coll == "alltoall":
        srcs = [rng.standard_normal(per * n).astype(np.float32)
                for _ in range(n)]
        dsts = [np.zeros(per * n, np.float32) for _ in range(n)]
        reqs = job.coll("alltoall", [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=per * n, dt=dtypes.FLOAT32) for r in range(n)])
        job.run(reqs)
        for d in range(n):
            for s in range(n):
                np.testing.assert_array_equal(
                    dsts[d][s * per:(s + 1) * per],
                    srcs[s][d * per:(d + 1) * per],
                    err_msg=f"op{opno}")
    elif coll == "gather":
        srcs = [rng.standard_normal(per).astype(np.float32)
                for _ in range(n)]
        gdst = np.zeros(per * n, np.float32)
        reqs = job.coll("gather", [
            dict(src=srcs[r].ctypes.data,
                 dst=gdst.ctypes.data if r == root else 0,
                 count=per * n if r == root else per,
                 dt=dtypes.FLOAT32, root=root) for r in range(n)])
        job.run(reqs)
        np.testing.assert_array_equal(gdst, np.concatenate(srcs),
                                      err_msg=f"op{opno}")
    elif coll == "scatter":
        big = rng.standard_normal(per * n).astype(np.float32)
        sdst = [np.zeros(per, np.float32) for _ in range(n)]
        reqs = job.coll("scatter", [
            dict(src=big.ctypes.data if r == root else 0,
                 dst=sdst[r].ctypes.data,
                 count=per * n if r == root else per,
                 dt=dtypes.FLOAT32, root=root) for r in range(n)])
        job.run(reqs)
        for r in range(n):
            np.testing.assert_array_equal(
                sdst[r], big[r * per:(r + 1) * per],
                err_msg=f"op{opno}")
    elif coll == "alltoallv":
        scnt = [[int(rng.integers(0, max(per // 2, 2)))
                 for _ in range(n)] for _ in range(n)]
        rcnt = [[scnt[src][dd] for src in range(n)]
                for dd in range(n)]
        sdsp = [np.cumsum([0] + row[:-1]).tolist() for row in scnt]
        rdsp = [np.cumsum([0] + row[:-1]).tolist() for row in rcnt]
        stot = [max(sum(row), 1) for row in scnt]
        rtot = [max(sum(row), 1) for row in rcnt]
        srcs = [rng.standard_normal(stot[r]).astype(np.float32)
                for r in range(n)]
        dsts = [np.zeros(rtot[r], np.float32) for r in range(n)]
        reqs = job.coll("alltoallv", [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=0, dt=dtypes.FLOAT32,
                 src_counts=scnt[r], src_displs=sdsp[r],
                 dst_counts=rcnt[r], dst_displs=rdsp[r])
            for r in range(n)])
        job.run(reqs)
        for dd in range(n):
            for src in range(n):
                cq = rcnt[dd][src]
                np.testing.assert_array_equal(
                    dsts[dd][rdsp[dd][src]:rdsp[dd][src] + cq],
                    srcs[src][sdsp[src][dd]:sdsp[src][dd] + cq],
                    err_msg=f"op{opno}")
    elif coll == "reduce_scatterv":
        cnts = [int(rng.integers(0, per + 1)) for _ in range(n)]
        tot = max(sum(cnts), 1)
        dsp = np.cumsum([0] + cnts[:-1]).tolist()
        srcs = [(rng.random(tot) - 0.5).astype(np.float32)
                for _ in range(n)]
        dsts = [np.zeros(max(cnts[r], 1), np.float32)
                for r in range(n)]
        reqs = job.coll("reduce_scatterv", [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=cnts[r], dt=dtypes.FLOAT32,
                 dst_counts=cnts, dst_displs=dsp)
            for r in range(n)])
        job.run(reqs)
        exp = np.sum(srcs, axis=0)
        for r in range(n):
            if cnts[r]:
                np.testing.assert_allclose(
                    dsts[r][:cnts[r]],
                    exp[dsp[r]:dsp[r] + cnts[r]], rtol=1e-5,
                    atol=1e-5, err_msg=f"op{opno}")
print("FUZZ_OK")
""" % (REPO,)


@pytest.mark.parametrize("seed", [11, 42])
def test_tcp_random_sequence(seed):
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    p = subprocess.run([sys.executable, "-c", FUZZ, str(seed)],
                       env=env, capture_output=True, text=True,
                       timeout=600)
    sys.stdout.write(p.stdout[-1500:])
    sys.stderr.write(p.stderr[-2500:])
    assert p.returncode == 0 and "FUZZ_OK" in p.stdout


def test_shm_socket_random_sequence():
    """Same randomized sequence with the shm TL serving and the
    socket-relay staging active (pseudo-sockets) — slot/parity/relay
    interleavings across mixed collectives."""
    env = dict(os.environ)
    env["UCC_FAKE_SOCKET_SPLIT"] = "3"
    env["UCC_TL_SHM_CHUNK_SIZE"] = "65536"
    p = subprocess.run([sys.executable, "-c", FUZZ, "7"],
                       env=env, capture_output=True, text=True,
                       timeout=600)
    sys.stdout.write(p.stdout[-1500:])
    sys.stderr.write(p.stderr[-2500:])
    assert p.returncode == 0 and "FUZZ_OK" in p.stdout


def test_hier_random_sequence():
    """Randomized sequence over the hier composition (fake 2-node
    split): RAB allreduce, 2step bcast/reduce and the hierarchical
    barrier interleave with flat tcp colls on shared sub-team
    sequence spaces."""
    env = dict(os.environ)
    env["UCC_FAKE_NODE_SPLIT"] = "2"
    env["UCC_TL_SHM_ENABLE"] = "0"
    p = subprocess.run([sys.executable, "-c", FUZZ, "23"],
                       env=env, capture_output=True, text=True,
                       timeout=600)
    sys.stdout.write(p.stdout[-1500:])
    sys.stderr.write(p.stderr[-2500:])
    assert p.returncode == 0 and "FUZZ_OK" in p.stdout
