"""Host-path collective correctness through the full public API
(in-process multi-rank jig, golden numpy references).

Mirrors the reference test strategy: per-coll
parameterized over team sizes x dtypes x ops, validated against host
golden buffers.
"""

import os
import sys

import numpy as np
import pytest

from ucc_amd import dtypes
from ucc_amd.testing import LocalJob

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


SIZES = [1, 2, 3, 8]  # 1 = self TL (zero-transport rank)
COUNTS = [1, 7, 1024, 70000]


@pytest.fixture(scope="module", params=SIZES)
def job(request):
    return LocalJob(request.param)


def _rand(rng, count, np_dt):
    if np.issubdtype(np.dtype(np_dt), np.integer):
        return rng.integers(-50, 50, size=count).astype(np_dt)
    return (rng.random(count) - 0.5).astype(np_dt)


@pytest.mark.parametrize("count", COUNTS)
@pytest.mark.parametrize("np_dt", [np.float32, np.float64, np.int32, np.int64])
def test_allreduce_sum(job, count, np_dt):
    rng = np.random.default_rng(42)
    arrays = [_rand(rng, count, np_dt) for _ in range(job.n)]
    expected = sum(a.astype(np.float64) for a in arrays)
    outs = job.allreduce_np(arrays)
    for o in outs:
        np.testing.assert_allclose(
            o.astype(np.float64), expected, rtol=1e-5, atol=1e-5
        )


@pytest.mark.parametrize("op,npop", [
    (dtypes.OP_MAX, np.maximum),
    (dtypes.OP_MIN, np.minimum),
    (dtypes.OP_PROD, np.multiply),
])
def test_allreduce_ops(job, op, npop):
    rng = np.random.default_rng(1)
    arrays = [(rng.random(513) + 0.5).astype(np.float32) for _ in range(job.n)]
    expected = arrays[0].copy()
    for a in arrays[1:]:
        expected = npop(expected, a)
    outs = job.allreduce_np(arrays, op=op)
    for o in outs:
        np.testing.assert_allclose(o, expected, rtol=1e-5)


def test_allreduce_avg(job):
    rng = np.random.default_rng(2)
    arrays = [(rng.random(4096) - 0.5).astype(np.float32) for _ in range(job.n)]
    expected = sum(a.astype(np.float64) for a in arrays) / job.n
    outs = job.allreduce_np(arrays, op=dtypes.OP_AVG)
    for o in outs:
        np.testing.assert_allclose(o.astype(np.float64), expected, rtol=1e-5,
                                   atol=1e-6)


def test_allreduce_inplace(job):
    rng = np.random.default_rng(3)
    bufs = [(rng.random(2048) - 0.5).astype(np.float32) for _ in range(job.n)]
    expected = sum(b.astype(np.float64) for b in bufs)
    c = job.c
    reqs = job.coll(
        "allreduce",
        [
            dict(src=0, dst=bufs[r].ctypes.data, count=bufs[r].size,
                 dt=dtypes.FLOAT32, flags=c.FLAG_IN_PLACE)
            for r in range(job.n)
        ],
    )
    job.run(reqs)
    for b in bufs:
        np.testing.assert_allclose(b.astype(np.float64), expected, rtol=1e-5,
                                   atol=1e-5)


@pytest.mark.parametrize("count", [4, 4096])
def test_bcast(job, count):
    rng = np.random.default_rng(4)
    root = job.n - 1
    bufs = [np.zeros(count, np.float32) for _ in range(job.n)]
    bufs[root] = (rng.random(count) - 0.5).astype(np.float32)
    expected = bufs[root].copy()
    reqs = job.coll(
        "bcast",
        [
            dict(src=bufs[r].ctypes.data, dst=0, count=count,
                 dt=dtypes.FLOAT32, root=root)
            for r in range(job.n)
        ],
    )
    job.run(reqs)
    for b in bufs:
        np.testing.assert_array_equal(b, expected)


def test_barrier(job):
    reqs = job.coll("barrier", [dict(src=0, dst=0, count=0, dt=dtypes.INT8)
                                for _ in range(job.n)])
    job.run(reqs)


@pytest.mark.parametrize("count", [8, 100000])
def test_allgather(job, count):
    rng = np.random.default_rng(5)
    srcs = [(rng.random(count) - 0.5).astype(np.float32)
            for _ in range(job.n)]
    dsts = [np.zeros(count * job.n, np.float32) for _ in range(job.n)]
    expected = np.concatenate(srcs)
    reqs = job.coll(
        "allgather",
        [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=count * job.n, dt=dtypes.FLOAT32)
            for r in range(job.n)
        ],
    )
    job.run(reqs)
    for d in dsts:
        np.testing.assert_array_equal(d, expected)


def test_reduce(job):
    rng = np.random.default_rng(6)
    root = 0
    srcs = [(rng.random(3000) - 0.5).astype(np.float32)
            for _ in range(job.n)]
    dsts = [np.zeros(3000, np.float32) for _ in range(job.n)]
    expected = sum(s.astype(np.float64) for s in srcs)
    reqs = job.coll(
        "reduce",
        [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=3000, dt=dtypes.FLOAT32, root=root)
            for r in range(job.n)
        ],
    )
    job.run(reqs)
    np.testing.assert_allclose(dsts[root].astype(np.float64), expected,
                               rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("per", [16, 30000])
def test_reduce_scatter(job, per):
    rng = np.random.default_rng(7)
    total = per * job.n
    srcs = [(rng.random(total) - 0.5).astype(np.float32)
            for _ in range(job.n)]
    dsts = [np.zeros(per, np.float32) for _ in range(job.n)]
    expected = sum(s.astype(np.float64) for s in srcs)
    reqs = job.coll(
        "reduce_scatter",
        [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=per, dt=dtypes.FLOAT32)
            for r in range(job.n)
        ],
    )
    job.run(reqs)
    for r in range(job.n):
        np.testing.assert_allclose(
            dsts[r].astype(np.float64), expected[r * per:(r + 1) * per],
            rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("per", [4, 5000])
def test_alltoall(job, per):
    rng = np.random.default_rng(8)
    n = job.n
    srcs = [(rng.random(per * n) - 0.5).astype(np.float32) for _ in range(n)]
    dsts = [np.zeros(per * n, np.float32) for _ in range(n)]
    reqs = job.coll(
        "alltoall",
        [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=per * n, dt=dtypes.FLOAT32)
            for r in range(n)
        ],
    )
    job.run(reqs)
    for d in range(n):
        for s in range(n):
            np.testing.assert_array_equal(
                dsts[d][s * per:(s + 1) * per],
                srcs[s][d * per:(d + 1) * per])


def test_gather_scatter(job):
    rng = np.random.default_rng(9)
    n, per, root = job.n, 1000, min(1, job.n - 1)
    srcs = [(rng.random(per) - 0.5).astype(np.float32) for _ in range(n)]
    gdst = np.zeros(per * n, np.float32)
    reqs = job.coll(
        "gather",
        [
            dict(src=srcs[r].ctypes.data,
                 dst=gdst.ctypes.data if r == root else 0,
                 count=per if r != root else per * n,
                 dt=dtypes.FLOAT32, root=root)
            for r in range(n)
        ],
    )
    job.run(reqs)
    np.testing.assert_array_equal(gdst, np.concatenate(srcs))

    sdsts = [np.zeros(per, np.float32) for _ in range(n)]
    reqs = job.coll(
        "scatter",
        [
            dict(src=gdst.ctypes.data if r == root else 0,
                 dst=sdsts[r].ctypes.data,
                 count=per * n if r == root else per,
                 dt=dtypes.FLOAT32, root=root)
            for r in range(n)
        ],
    )
    job.run(reqs)
    for r in range(n):
        np.testing.assert_array_equal(sdsts[r], srcs[r])


def test_alltoallv(job):
    rng = np.random.default_rng(10)
    n = job.n
    # skewed counts: rank r sends (r+1)*(d+2) elements to rank d
    scnt = [[(r + 1) * (d + 2) for d in range(n)] for r in range(n)]
    rcnt = [[scnt[s][r] for s in range(n)] for r in range(n)]
    sdsp = [np.concatenate([[0], np.cumsum(scnt[r])[:-1]]).astype(np.uint64)
            for r in range(n)]
    rdsp = [np.concatenate([[0], np.cumsum(rcnt[r])[:-1]]).astype(np.uint64)
            for r in range(n)]
    srcs = [(rng.random(int(sum(scnt[r]))) - 0.5).astype(np.float32)
            for r in range(n)]
    dsts = [np.zeros(int(sum(rcnt[r])), np.float32) for r in range(n)]
    reqs = job.coll(
        "alltoallv",
        [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,
                 count=0, dt=dtypes.FLOAT32,
                 src_counts=[int(x) for x in scnt[r]],
                 src_displs=[int(x) for x in sdsp[r]],
                 dst_counts=[int(x) for x in rcnt[r]],
                 dst_displs=[int(x) for x in rdsp[r]])
            for r in range(n)
        ],
    )
    job.run(reqs)
    for d in range(n):
        for s in range(n):
            got = dsts[d][int(rdsp[d][s]):int(rdsp[d][s]) + rcnt[d][s]]
            exp = srcs[s][int(sdsp[s][d]):int(sdsp[s][d]) + scnt[s][d]]
            np.testing.assert_array_equal(got, exp)


def test_persistent_allreduce(job):
    rng = np.random.default_rng(11)
    c = job.c
    bufs = [(rng.random(512) - 0.5).astype(np.float32) for _ in range(job.n)]
    outs = [np.zeros(512, np.float32) for _ in range(job.n)]
    reqs = job.coll(
        "allreduce",
        [
            dict(src=bufs[r].ctypes.data, dst=outs[r].ctypes.data,
                 count=512, dt=dtypes.FLOAT32, flags=c.FLAG_PERSISTENT)
            for r in range(job.n)
        ],
    )
    for it in range(3):
        for b in bufs:
            b += 1.0
        expected = sum(b.astype(np.float64) for b in bufs)
        job.run(reqs)
        for o in outs:
            np.testing.assert_allclose(o.astype(np.float64), expected,
                                       rtol=1e-5, atol=1e-5)


def test_concurrent_colls(job):
    """several collectives in flight on one team (slot pipelining)"""
    rng = np.random.default_rng(12)
    n = job.n
    n_colls = 6
    all_reqs, all_outs, all_exp = [], [], []
    for k in range(n_colls):
        arrays = [(rng.random(257) - 0.5).astype(np.float32)
                  for _ in range(n)]
        outs = [np.zeros(257, np.float32) for _ in range(n)]
        all_exp.append(sum(a.astype(np.float64) for a in arrays))
        all_outs.append(outs)
        reqs = job.coll(
            "allreduce",
            [
                dict(src=arrays[r].ctypes.data, dst=outs[r].ctypes.data,
                     count=257, dt=dtypes.FLOAT32)
                for r in range(n)
            ],
        )
        all_reqs.append((reqs, arrays))
    flat = [r for reqs, _ in all_reqs for r in reqs]
    job.run(flat)
    for k in range(n_colls):
        for o in all_outs[k]:
            np.testing.assert_allclose(o.astype(np.float64), all_exp[k],
                                       rtol=1e-5, atol=1e-5)


def test_topo_sbgps(job):
    """Single-node placement: NODE == FULL, one leader (SURVEY topo/sbgp
    parity, reference ucc_sbgp.c subgroup discovery)."""
    from ucc_amd import core
    c = core()
    info = c.topo_sbgps(job.teams[0])
    assert info["node_size"] == job.n
    assert info["node_idx"] == 0
    assert info["leaders_size"] == 1
    assert info["leaders_idx"] == 0
    # SOCKET/NUMA sbgps (reference ucc_sbgp.h:10-41 kinds): the whole
    # in-process jig runs on one machine, so the subgroup sizes sum to
    # the node and my_idx is a valid membership; unknown ids (-1) must
    # degrade to a full-node group, never an empty one.
    assert 1 <= info["socket_size"] <= job.n
    assert 0 <= info["socket_idx"] < info["socket_size"]
    assert 1 <= info["numa_size"] <= job.n
    assert 0 <= info["numa_idx"] < info["numa_size"]
    assert info["socket_leaders_size"] >= 1
    assert info["numa_leaders_size"] >= 1
    # every rank of the jig is the same process => same CPU model
    assert info["same_cpu"] is True


@pytest.mark.parametrize("npdt,dt", [
    (np.int32, dtypes.INT32), (np.int64, dtypes.INT64),
    (np.uint8, dtypes.UINT8), (np.float64, dtypes.FLOAT64),
])
def test_allreduce_int_float_dtypes(job, npdt, dt):
    rng = np.random.default_rng(21)
    n = job.n
    if np.issubdtype(npdt, np.integer):
        arrs = [rng.integers(0, 50, 777).astype(npdt) for _ in range(n)]
    else:
        arrs = [(rng.random(777) - 0.5).astype(npdt) for _ in range(n)]
    outs = [np.zeros(777, npdt) for _ in range(n)]
    reqs = job.coll("allreduce", [
        dict(src=arrs[r].ctypes.data, dst=outs[r].ctypes.data, count=777,
             dt=dt) for r in range(n)])
    job.run(reqs)
    exp = np.sum(arrs, axis=0, dtype=npdt)
    for o in outs:
        if np.issubdtype(npdt, np.integer):
            np.testing.assert_array_equal(o, exp)
        else:
            np.testing.assert_allclose(o, exp, rtol=1e-12, atol=1e-12)


def test_allreduce_bf16_host(job):
    """bf16 software reduction on the host path (reference ec/cpu bf16)."""
    from ucc_amd import core  # noqa: F401
    n = job.n
    rng = np.random.default_rng(22)
    f32 = [(rng.random(1024) - 0.5).astype(np.float32) for _ in range(n)]
    # encode to bf16 (truncate-to-nearest-even like the library)
    def to_bf16(a):
        u = a.view(np.uint32)
        r = 0x7FFF + ((u >> 16) & 1)
        return ((u + r) >> 16).astype(np.uint16)
    def from_bf16(b):
        return (b.astype(np.uint32) << 16).view(np.float32)
    srcs = [to_bf16(a) for a in f32]
    outs = [np.zeros(1024, np.uint16) for _ in range(n)]
    reqs = job.coll("allreduce", [
        dict(src=srcs[r].ctypes.data, dst=outs[r].ctypes.data, count=1024,
             dt=dtypes.BFLOAT16) for r in range(n)])
    job.run(reqs)
    exp = np.sum([from_bf16(s) for s in srcs], axis=0)
    for o in outs:
        np.testing.assert_allclose(from_bf16(o), exp, rtol=3e-2,
                                   atol=1e-1)


def test_allreduce_logical_bitwise(job):
    n = job.n
    rng = np.random.default_rng(23)
    arrs = [rng.integers(0, 2 ** 16, 512).astype(np.uint32)
            for _ in range(n)]
    for op, fn in ((dtypes.OP_BAND, np.bitwise_and),
                   (dtypes.OP_BOR, np.bitwise_or),
                   (dtypes.OP_BXOR, np.bitwise_xor)):
        outs = [np.zeros(512, np.uint32) for _ in range(n)]
        reqs = job.coll("allreduce", [
            dict(src=arrs[r].ctypes.data, dst=outs[r].ctypes.data,
                 count=512, dt=dtypes.UINT32, op=op) for r in range(n)])
        job.run(reqs)
        exp = arrs[0]
        for a in arrs[1:]:
            exp = fn(exp, a)
        for o in outs:
            np.testing.assert_array_equal(o, exp)
    # logical ops (land/lor/lxor): C truth semantics on integers
    arrs = [rng.integers(0, 2, 512).astype(np.int32) for _ in range(n)]
    for op, fn in ((dtypes.OP_LAND, lambda a, b:
                    np.logical_and(a, b).astype(np.int32)),
                   (dtypes.OP_LOR, lambda a, b:
                    np.logical_or(a, b).astype(np.int32)),
                   (dtypes.OP_LXOR, lambda a, b:
                    np.logical_xor(a, b).astype(np.int32))):
        outs = [np.zeros(512, np.int32) for _ in range(n)]
        reqs = job.coll("allreduce", [
            dict(src=arrs[r].ctypes.data, dst=outs[r].ctypes.data,
                 count=512, dt=dtypes.INT32, op=op) for r in range(n)])
        job.run(reqs)
        exp = arrs[0].copy()
        for a in arrs[1:]:
            exp = fn(exp, a)
        for o in outs:
            np.testing.assert_array_equal(o, exp)


def test_gatherv_scatterv(job):
    """v-counts rooted colls (reference gatherv/scatterv coverage)."""
    rng = np.random.default_rng(41)
    n, root = job.n, 0
    cnts = [(r + 2) * 37 for r in range(n)]
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).astype(np.uint64)
    total = int(sum(cnts))
    srcs = [(rng.random(cnts[r])).astype(np.float32) for r in range(n)]
    gdst = np.zeros(total, np.float32)
    reqs = job.coll("gatherv", [
        dict(src=srcs[r].ctypes.data,
             dst=gdst.ctypes.data if r == root else 0,
             count=cnts[r], dt=dtypes.FLOAT32, root=root,
             dst_counts=cnts, dst_displs=dsps.tolist())
        for r in range(n)])
    job.run(reqs)
    np.testing.assert_array_equal(gdst, np.concatenate(srcs))

    sdsts = [np.zeros(cnts[r], np.float32) for r in range(n)]
    reqs = job.coll("scatterv", [
        dict(src=gdst.ctypes.data if r == root else 0,
             dst=sdsts[r].ctypes.data, count=cnts[r], dt=dtypes.FLOAT32,
             root=root, src_counts=cnts, src_displs=dsps.tolist())
        for r in range(n)])
    job.run(reqs)
    for r in range(n):
        np.testing.assert_array_equal(sdsts[r], srcs[r])


def test_gatherv_rootonly_vargs(job):
    """UCC semantics: v-args are significant at the ROOT only — non-root
    ranks pass no counts (regression: union misread crashed/flaked)."""
    rng = np.random.default_rng(42)
    n, root = job.n, 0
    cnts = [(r + 1) * 21 for r in range(n)]
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).tolist()
    total = int(sum(cnts))
    srcs = [(rng.random(cnts[r])).astype(np.float32) for r in range(n)]
    gdst = np.zeros(total, np.float32)
    reqs = job.coll("gatherv", [
        dict(src=srcs[r].ctypes.data,
             dst=gdst.ctypes.data if r == root else 0,
             count=cnts[r], dt=dtypes.FLOAT32, root=root,
             **({"dst_counts": cnts, "dst_displs": dsps}
                if r == root else {}))
        for r in range(n)])
    job.run(reqs)
    np.testing.assert_array_equal(gdst, np.concatenate(srcs))

    sdsts = [np.zeros(cnts[r], np.float32) for r in range(n)]
    reqs = job.coll("scatterv", [
        dict(src=gdst.ctypes.data if r == root else 0,
             dst=sdsts[r].ctypes.data, count=cnts[r], dt=dtypes.FLOAT32,
             root=root,
             **({"src_counts": cnts, "src_displs": dsps}
                if r == root else {}))
        for r in range(n)])
    job.run(reqs)
    for r in range(n):
        np.testing.assert_array_equal(sdsts[r], srcs[r])


def test_shm_socket_staged_bcast():
    """Socket-aware shm bcast (reference ucc_sbgp SOCKET consumption):
    with UCC_FAKE_SOCKET_SPLIT the node team spans pseudo-sockets, so
    chunks relay through one leader per non-root socket. Correctness
    across roots, sizes spanning multiple chunks, and a mid-team
    leader; plus a run with staging disabled for identical results."""
    import subprocess

    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "n = 6\n"
        "job = LocalJob(n)\n"
        "info = core().topo_sbgps(job.teams[0])\n"
        "assert info['socket_leaders_size'] == 3, info\n"
        "rng = np.random.default_rng(5)\n"
        "for root in (0, 1, 4):\n"
        "    for count in (63, 5000, 300_000):\n"
        "        bufs = [np.zeros(count, np.float64) for _ in range(n)]\n"
        "        bufs[root][:] = rng.random(count)\n"
        "        exp = bufs[root].copy()\n"
        "        reqs = job.coll('bcast', [\n"
        "            dict(src=b.ctypes.data, dst=0, count=count,\n"
        "                 dt=dtypes.FLOAT64, root=root) for b in bufs])\n"
        "        job.run(reqs)\n"
        "        for b in bufs:\n"
        "            np.testing.assert_array_equal(b, exp)\n"
        "# allreduce: phase-C result reads relay through socket leaders\n"
        "for count in (63, 5000, 300_000):\n"
        "    arrs = [(rng.random(count) - 0.5).astype(np.float32)\n"
        "            for _ in range(n)]\n"
        "    outs = [np.zeros(count, np.float32) for _ in range(n)]\n"
        "    reqs = job.coll('allreduce', [\n"
        "        dict(src=arrs[r].ctypes.data, dst=outs[r].ctypes.data,\n"
        "             count=count, dt=dtypes.FLOAT32)\n"
        "        for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    exp = np.sum(arrs, axis=0)\n"
        "    for o in outs:\n"
        "        np.testing.assert_allclose(o, exp, rtol=1e-5,\n"
        "                                   atol=1e-5)\n"
        "print('SCK_BCAST_OK')\n"
    ) % (REPO,)
    for staging in ("1", "0"):
        env = dict(os.environ)
        env["UCC_FAKE_SOCKET_SPLIT"] = "3"
        env["UCC_TL_SHM_SOCKET_STAGING"] = staging
        env["UCC_TL_SHM_CHUNK_SIZE"] = "65536"
        p = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, text=True, timeout=300)
        sys.stdout.write(p.stdout[-300:])
        sys.stderr.write(p.stderr[-2000:])
        assert p.returncode == 0 and "SCK_BCAST_OK" in p.stdout, staging
