"""tl/tcp correctness: run the in-process jig with the shm TL disabled so
every host collective goes over the TCP mesh (127.0.0.1 sockets) — the
inter-node transport path (reference tl/ucp role), validated on CPU."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import sys
import numpy as np
sys.path.insert(0, %r)
from ucc_amd import core, dtypes
from ucc_amd.testing import LocalJob

n = 5  # non-power-of-2: exercises the fold path of recursive doubling
job = LocalJob(n)
rng = np.random.default_rng(3)

# allreduce fp32 + bf16-ish dtype via int path
for count in (7, 1024, 100_003):
    arrs = [(rng.random(count) - 0.5).astype(np.float32) for _ in range(n)]
    outs = job.allreduce_np(arrs)
    exp = np.sum(arrs, axis=0)
    for o in outs:
        np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-5)

# SRA path (>=64 KiB) with AVG and MAX
arrs = [(rng.random(120_000) - 0.5).astype(np.float32) for _ in range(n)]
outs = job.allreduce_np(arrs, op=dtypes.OP_AVG)
exp = np.mean(arrs, axis=0)
for o in outs:
    np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-6)
outs = job.allreduce_np(arrs, op=dtypes.OP_MAX)
exp = np.max(arrs, axis=0)
for o in outs:
    np.testing.assert_array_equal(o, exp)

# bcast large (SAG ring path) from a non-zero root
big = [np.zeros(60_000, np.float64) for _ in range(n)]
big[3][:] = rng.random(60_000)
exp = big[3].copy()
reqs = job.coll("bcast", [
    dict(src=b.ctypes.data, dst=0, count=60_000, dt=dtypes.FLOAT64,
         root=3) for b in big])
job.run(reqs)
for b in big:
    np.testing.assert_array_equal(b, exp)

# bcast
bufs = [np.zeros(5000, np.float64) for _ in range(n)]
bufs[2][:] = rng.random(5000)
exp = bufs[2].copy()
reqs = job.coll("bcast", [
    dict(src=b.ctypes.data, dst=0, count=5000, dt=dtypes.FLOAT64, root=2)
    for b in bufs])
job.run(reqs)
for b in bufs:
    np.testing.assert_array_equal(b, exp)

# alltoall
per = 321
srcs = [(rng.random(per * n)).astype(np.float32) for _ in range(n)]
dsts = [np.zeros(per * n, np.float32) for _ in range(n)]
reqs = job.coll("alltoall", [
    dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data, count=per * n,
         dt=dtypes.FLOAT32) for r in range(n)])
job.run(reqs)
for d in range(n):
    for s in range(n):
        np.testing.assert_array_equal(
            dsts[d][s * per:(s + 1) * per], srcs[s][d * per:(d + 1) * per])

# allgatherv (uneven)
cnts = [(r + 1) * 77 for r in range(n)]
dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).astype(np.uint64)
total = int(sum(cnts))
srcs = [(rng.random(cnts[r])).astype(np.float32) for r in range(n)]
dsts = [np.zeros(total, np.float32) for _ in range(n)]
reqs = job.coll("allgatherv", [
    dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data, count=cnts[r],
         dt=dtypes.FLOAT32, dst_counts=cnts, dst_displs=dsps.tolist())
    for r in range(n)])
job.run(reqs)
exp = np.concatenate(srcs)
for d in dsts:
    np.testing.assert_array_equal(d, exp)

# reduce_scatter
per = 500
srcs = [(rng.random(per * n)).astype(np.float32) for _ in range(n)]
dsts = [np.zeros(per, np.float32) for _ in range(n)]
reqs = job.coll("reduce_scatter", [
    dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data, count=per,
         dt=dtypes.FLOAT32) for r in range(n)])
job.run(reqs)
exp = np.sum(srcs, axis=0)
for r in range(n):
    np.testing.assert_allclose(dsts[r], exp[r * per:(r + 1) * per],
                               rtol=1e-5, atol=1e-5)

# reduce large (SRG ring path) to a non-zero root
rsrcs = [(rng.random(90_000) - 0.5).astype(np.float32) for _ in range(n)]
rdst = np.zeros(90_000, np.float32)
reqs = job.coll("reduce", [
    dict(src=rsrcs[r].ctypes.data,
         dst=rdst.ctypes.data if r == 2 else 0, count=90_000,
         dt=dtypes.FLOAT32, root=2) for r in range(n)])
job.run(reqs)
np.testing.assert_allclose(rdst, np.sum(rsrcs, axis=0), rtol=1e-5,
                           atol=1e-5)

# barrier
reqs = job.coll("barrier", [dict(src=0, dst=0, count=0, dt=dtypes.FLOAT32)
                            for _ in range(n)])
job.run(reqs)

print("TCP_TL_OK")
""" % (REPO,)


def test_tcp_tl_colls():
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    p = subprocess.run([sys.executable, "-c", WORKER], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0
    assert "TCP_TL_OK" in p.stdout


def test_alltoall_bruck():
    """Bruck alltoall (log2 n aggregated rounds) picked for small
    messages over tcp; correctness incl. non-power-of-two ranks."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (3, 5, 8):\n"
        "    job = LocalJob(n)\n"
        "    c = core()\n"
        "    assert '@tcp/bruck' in c.score_map_str(job.teams[0])\n"
        "    rng = np.random.default_rng(n)\n"
        "    for per in (1, 33, 500):\n"
        "        srcs = [rng.standard_normal(per * n).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        dsts = [np.zeros(per * n, np.float32) for _ in range(n)]\n"
        "        reqs = job.coll('alltoall', [\n"
        "            dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,\n"
        "                 count=per * n, dt=dtypes.FLOAT32)\n"
        "            for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        for r in range(n):\n"
        "            for s in range(n):\n"
        "                np.testing.assert_array_equal(\n"
        "                    dsts[r][s * per:(s + 1) * per],\n"
        "                    srcs[s][r * per:(r + 1) * per])\n"
        "print('BRUCK_OK')\n" % (REPO,))
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-1000:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "BRUCK_OK" in p.stdout


def test_bcast_knomial_and_dbt():
    """k-nomial radix-k bcast (reference recursive_knomial.h role) and
    double-binary-tree bcast/reduce (double_binary_tree.h role): all
    roots x odd/even team sizes x sizes straddling the score bands,
    validated against numpy references over the tcp mesh."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (2, 3, 6, 8):\n"
        "    job = LocalJob(n)\n"
        "    c = core()\n"
        "    smap = c.score_map_str(job.teams[0])\n"
        "    assert '@tcp/knomial' in smap, smap\n"
        "    assert '@tcp/dbt' in smap, smap\n"
        "    rng = np.random.default_rng(n)\n"
        "    # knomial band (<8KB) and dbt band (8KB..4MB)\n"
        "    for count in (17, 1000, 5000, 60_000):\n"
        "        for root in range(n):\n"
        "            bufs = [np.zeros(count, np.float32)\n"
        "                    for _ in range(n)]\n"
        "            bufs[root][:] = rng.standard_normal(count)\n"
        "            exp = bufs[root].copy()\n"
        "            reqs = job.coll('bcast', [\n"
        "                dict(src=b.ctypes.data, dst=0, count=count,\n"
        "                     dt=dtypes.FLOAT32, root=root)\n"
        "                for b in bufs])\n"
        "            job.run(reqs)\n"
        "            for b in bufs:\n"
        "                np.testing.assert_array_equal(b, exp)\n"
        "    # dbt reduce: every root, odd count (uneven halves)\n"
        "    for count in (4097, 30_001):\n"
        "        for root in range(n):\n"
        "            srcs = [rng.standard_normal(count)\n"
        "                    .astype(np.float32) for _ in range(n)]\n"
        "            dst = np.zeros(count, np.float32)\n"
        "            reqs = job.coll('reduce', [\n"
        "                dict(src=srcs[r].ctypes.data,\n"
        "                     dst=dst.ctypes.data if r == root else 0,\n"
        "                     count=count, dt=dtypes.FLOAT32,\n"
        "                     root=root) for r in range(n)])\n"
        "            job.run(reqs)\n"
        "            np.testing.assert_allclose(\n"
        "                dst, np.sum(srcs, axis=0), rtol=1e-5,\n"
        "                atol=1e-4)\n"
        "print('KNOM_DBT_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=600)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "KNOM_DBT_OK" in p.stdout


def test_bcast_knomial_radix_sweep():
    """The radix knob changes the tree shape, not the result."""
    for radix in (2, 3, 8):
        code = (
            "import sys; sys.path.insert(0, %r)\n"
            "import numpy as np\n"
            "from ucc_amd import core, dtypes\n"
            "from ucc_amd.testing import LocalJob\n"
            "n = 7\n"
            "job = LocalJob(n)\n"
            "rng = np.random.default_rng(5)\n"
            "for root in (0, 3, 6):\n"
            "    bufs = [np.zeros(999, np.float32) for _ in range(n)]\n"
            "    bufs[root][:] = rng.standard_normal(999)\n"
            "    exp = bufs[root].copy()\n"
            "    reqs = job.coll('bcast', [\n"
            "        dict(src=b.ctypes.data, dst=0, count=999,\n"
            "             dt=dtypes.FLOAT32, root=root) for b in bufs])\n"
            "    job.run(reqs)\n"
            "    for b in bufs:\n"
            "        np.testing.assert_array_equal(b, exp)\n"
            "print('RADIX_OK')\n"
        ) % (REPO,)
        env = dict(os.environ)
        env["UCC_TL_SHM_ENABLE"] = "0"
        env["UCC_TL_TCP_KN_RADIX"] = str(radix)
        env["UCC_TUNE"] = "bcast:@knomial:99"
        p = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, text=True, timeout=300)
        sys.stdout.write(p.stdout[-500:])
        sys.stderr.write(p.stderr[-2000:])
        assert p.returncode == 0 and "RADIX_OK" in p.stdout, \
            f"radix {radix} failed"


def test_allreduce_sliding_window():
    """Sliding-window allreduce (reference allreduce_sliding_window.c
    role): windows of the message run SRA rings concurrently with a
    bounded depth. Threshold lowered so the CPU test exercises many
    windows; AVG covers the per-window final-round scale."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (2, 5):\n"
        "    job = LocalJob(n)\n"
        "    c = core()\n"
        "    assert '@tcp/sliding_window' in "
        "c.score_map_str(job.teams[0])\n"
        "    rng = np.random.default_rng(n)\n"
        "    count = 700_001  # ~2.8MB fp32 -> ~11 windows of 256KB\n"
        "    arrs = [(rng.random(count) - 0.5).astype(np.float32)\n"
        "            for _ in range(n)]\n"
        "    outs = job.allreduce_np(arrs)\n"
        "    exp = np.sum(arrs, axis=0)\n"
        "    for o in outs:\n"
        "        np.testing.assert_allclose(o, exp, rtol=1e-5,\n"
        "                                   atol=1e-4)\n"
        "    outs = job.allreduce_np(arrs, op=dtypes.OP_AVG)\n"
        "    exp = np.mean(arrs, axis=0)\n"
        "    for o in outs:\n"
        "        np.testing.assert_allclose(o, exp, rtol=1e-5,\n"
        "                                   atol=1e-5)\n"
        "print('SLIDING_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TL_TCP_SLIDING_MIN"] = "1048576"
    env["UCC_TL_TCP_SLIDING_WINDOW"] = "262144"
    env["UCC_TL_TCP_SLIDING_DEPTH"] = "3"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=600)
    sys.stdout.write(p.stdout[-1000:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "SLIDING_OK" in p.stdout


def test_allgather_bruck():
    """Bruck allgather: ceil(log2 n) rounds + final rotation, any n
    (reference tl/ucp allgather bruck role)."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (2, 3, 5, 8):\n"
        "    job = LocalJob(n)\n"
        "    c = core()\n"
        "    smap = c.score_map_str(job.teams[0])\n"
        "    assert 'allgather:host:0-65536:@tcp/bruck' in smap, smap\n"
        "    rng = np.random.default_rng(n)\n"
        "    for per in (1, 77, 4000):\n"
        "        blks = [rng.standard_normal(per).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        dsts = [np.zeros(per * n, np.float32)\n"
        "                for _ in range(n)]\n"
        "        reqs = job.coll('allgather', [\n"
        "            dict(src=blks[r].ctypes.data,\n"
        "                 dst=dsts[r].ctypes.data, count=per * n,\n"
        "                 dt=dtypes.FLOAT32) for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        exp = np.concatenate(blks)\n"
        "        for d in dsts:\n"
        "            np.testing.assert_array_equal(d, exp)\n"
        "print('AG_BRUCK_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TUNE"] = "allgather:@bruck:99"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "AG_BRUCK_OK" in p.stdout


def test_allgather_knomial_radix():
    """Radix-k knomial allgather (reference tl/ucp allgather knomial
    role): radix-k dissemination at absolute offsets, radices 2-5 over
    odd/even/prime team sizes."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (3, 8, 9, 13):\n"
        "    job = LocalJob(n)\n"
        "    assert '@tcp/knomial' in core().score_map_str(\n"
        "        job.teams[0])\n"
        "    rng = np.random.default_rng(n)\n"
        "    for per in (1, 77, 900):\n"
        "        blks = [rng.standard_normal(per).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        dsts = [np.zeros(per * n, np.float32)\n"
        "                for _ in range(n)]\n"
        "        reqs = job.coll('allgather', [\n"
        "            dict(src=blks[r].ctypes.data,\n"
        "                 dst=dsts[r].ctypes.data, count=per * n,\n"
        "                 dt=dtypes.FLOAT32) for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        exp = np.concatenate(blks)\n"
        "        for d in dsts:\n"
        "            np.testing.assert_array_equal(d, exp)\n"
        "print('AG_KN_OK')\n"
    ) % (REPO,)
    for radix in ("2", "3", "4"):
        env = dict(os.environ)
        env["UCC_TL_SHM_ENABLE"] = "0"
        env["UCC_TUNE"] = "allgather:@knomial:99"
        env["UCC_TL_TCP_KN_RADIX"] = radix
        p = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, text=True, timeout=300)
        sys.stdout.write(p.stdout[-300:])
        sys.stderr.write(p.stderr[-2000:])
        assert p.returncode == 0 and "AG_KN_OK" in p.stdout, radix


def test_allgather_sparbit():
    """Sparbit-role allgather (reference tl/ucp allgather sparbit):
    ceil(log2 n) rounds, data-ordered (blocks land at absolute dst
    positions, no work buffer or rotation), any n including the wrap
    split of circular runs."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (2, 3, 5, 6, 8, 11):\n"
        "    job = LocalJob(n)\n"
        "    smap = core().score_map_str(job.teams[0])\n"
        "    assert 'allgather:host:0-262144:@tcp/sparbit' in smap, smap\n"
        "    rng = np.random.default_rng(n)\n"
        "    for per in (1, 77, 4000):\n"
        "        blks = [rng.standard_normal(per).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        dsts = [np.zeros(per * n, np.float32)\n"
        "                for _ in range(n)]\n"
        "        reqs = job.coll('allgather', [\n"
        "            dict(src=blks[r].ctypes.data,\n"
        "                 dst=dsts[r].ctypes.data, count=per * n,\n"
        "                 dt=dtypes.FLOAT32) for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        exp = np.concatenate(blks)\n"
        "        for d in dsts:\n"
        "            np.testing.assert_array_equal(d, exp)\n"
        "print('AG_SPARBIT_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "AG_SPARBIT_OK" in p.stdout


def test_allgather_neighbor():
    """Neighbor-exchange allgather (reference tl/ucp allgather neighbor
    role): n/2 rounds of 2-block swaps with alternating direction, even
    n only; odd n must fall through to ring/bruck."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (4, 6, 8, 3, 5):\n"  # odd n: selection falls back
        "    job = LocalJob(n)\n"
        "    if n %% 2 == 0:\n"
        "        smap = core().score_map_str(job.teams[0])\n"
        "        assert '@tcp/neighbor' in smap, smap\n"
        "    rng = np.random.default_rng(n)\n"
        "    for per in (1, 77, 5000):\n"
        "        blks = [rng.standard_normal(per).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        dsts = [np.zeros(per * n, np.float32)\n"
        "                for _ in range(n)]\n"
        "        reqs = job.coll('allgather', [\n"
        "            dict(src=blks[r].ctypes.data,\n"
        "                 dst=dsts[r].ctypes.data, count=per * n,\n"
        "                 dt=dtypes.FLOAT32) for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        exp = np.concatenate(blks)\n"
        "        for d in dsts:\n"
        "            np.testing.assert_array_equal(d, exp)\n"
        "print('AG_NBR_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TUNE"] = "allgather:@neighbor:99"
    env["UCC_TL_TCP_AG_NEIGHBOR_MIN"] = "0"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "AG_NBR_OK" in p.stdout


def test_allgather_linear_batched():
    """Linear batched allgather(v) (reference tl/ucp allgather linear /
    batched_num_posts role): direct one-hop sends to all peers with a
    posting-window throttle; also exercises the A2A_NUM_POSTS window
    on the pairwise alltoall."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (3, 5, 8):\n"
        "    job = LocalJob(n)\n"
        "    smap = core().score_map_str(job.teams[0])\n"
        "    assert '@tcp/linear' in smap, smap\n"
        "    rng = np.random.default_rng(n)\n"
        "    for per in (77, 4000):\n"
        "        blks = [rng.standard_normal(per).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        dsts = [np.zeros(per * n, np.float32)\n"
        "                for _ in range(n)]\n"
        "        reqs = job.coll('allgather', [\n"
        "            dict(src=blks[r].ctypes.data,\n"
        "                 dst=dsts[r].ctypes.data, count=per * n,\n"
        "                 dt=dtypes.FLOAT32) for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        exp = np.concatenate(blks)\n"
        "        for d in dsts:\n"
        "            np.testing.assert_array_equal(d, exp)\n"
        "    # ragged allgatherv through the same linear task\n"
        "    cnts = [13 * (r + 1) for r in range(n)]\n"
        "    dspl = np.cumsum([0] + cnts[:-1]).tolist()\n"
        "    tot = sum(cnts)\n"
        "    blks = [rng.standard_normal(cnts[r]).astype(np.float32)\n"
        "            for r in range(n)]\n"
        "    dsts = [np.zeros(tot, np.float32) for _ in range(n)]\n"
        "    reqs = job.coll('allgatherv', [\n"
        "        dict(src=blks[r].ctypes.data, dst=dsts[r].ctypes.data,\n"
        "             count=cnts[r], dt=dtypes.FLOAT32,\n"
        "             dst_counts=cnts, dst_displs=dspl)\n"
        "        for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    exp = np.concatenate(blks)\n"
        "    for d in dsts:\n"
        "        np.testing.assert_array_equal(d, exp)\n"
        "    # pairwise alltoall with a narrow posting window\n"
        "    per = 501\n"
        "    srcs = [rng.standard_normal(per * n).astype(np.float32)\n"
        "            for _ in range(n)]\n"
        "    dsts = [np.zeros(per * n, np.float32) for _ in range(n)]\n"
        "    reqs = job.coll('alltoall', [\n"
        "        dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,\n"
        "             count=per * n, dt=dtypes.FLOAT32)\n"
        "        for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    for d in range(n):\n"
        "        exp = np.concatenate([\n"
        "            srcs[s][d * per:(d + 1) * per] for s in range(n)])\n"
        "        np.testing.assert_array_equal(dsts[d], exp)\n"
        "print('AG_LINEAR_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TUNE"] = "allgather:@linear:99,allgatherv:@linear:99"
    env["UCC_TL_TCP_AG_LINEAR_NUM_POSTS"] = "2"
    env["UCC_TL_TCP_A2A_NUM_POSTS"] = "2"
    env["UCC_TL_TCP_BRUCK_MAX"] = "0"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "AG_LINEAR_OK" in p.stdout


def test_alltoallv_hybrid():
    """Hybrid a2av (reference alltoallv_hybrid.c role): small pairs
    aggregate through the Bruck digit exchange, large pairs go direct.
    Skewed counts straddling the threshold, zero pairs, odd/even n."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (2, 3, 6, 8):\n"
        "    job = LocalJob(n)\n"
        "    c = core()\n"
        "    assert '@tcp/hybrid' in c.score_map_str(job.teams[0])\n"
        "    rng = np.random.default_rng(n)\n"
        "    # counts: mix of 0, tiny (<thr=1024B), and big (>thr)\n"
        "    scnt = [[0 if (r + d) %% 5 == 0 else\n"
        "             (7 + r + d if (r * d) %% 3 else 900 + 100 * d)\n"
        "             for d in range(n)] for r in range(n)]\n"
        "    rcnt = [[scnt[s][r] for s in range(n)]\n"
        "            for r in range(n)]\n"
        "    def dsp(cs):\n"
        "        out, off = [], 3\n"
        "        for cq in cs:\n"
        "            out.append(off)\n"
        "            off += cq + 2\n"
        "        return out, off\n"
        "    sdsp = [dsp(cq)[0] for cq in scnt]\n"
        "    rdsp = [dsp(cq)[0] for cq in rcnt]\n"
        "    stot = [dsp(cq)[1] for cq in scnt]\n"
        "    rtot = [dsp(cq)[1] for cq in rcnt]\n"
        "    srcs = [rng.standard_normal(stot[r]).astype(np.float32)\n"
        "            for r in range(n)]\n"
        "    dsts = [np.zeros(rtot[r], np.float32) for r in range(n)]\n"
        "    reqs = job.coll('alltoallv', [\n"
        "        dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,\n"
        "             count=0, dt=dtypes.FLOAT32, src_counts=scnt[r],\n"
        "             src_displs=sdsp[r], dst_counts=rcnt[r],\n"
        "             dst_displs=rdsp[r]) for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    for r in range(n):\n"
        "        for s in range(n):\n"
        "            np.testing.assert_array_equal(\n"
        "                dsts[r][rdsp[r][s]:rdsp[r][s] + rcnt[r][s]],\n"
        "                srcs[s][sdsp[s][r]:sdsp[s][r] + scnt[s][r]])\n"
        "print('HYBRID_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TL_TCP_A2AV_HYBRID_THRESH"] = "1024"
    env["UCC_TUNE"] = "alltoallv:@hybrid:99"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "HYBRID_OK" in p.stdout


def test_gather_scatter_knomial():
    """k-nomial gather/scatter (reference tl/ucp knomial role):
    virtual-rank contiguous subtree ranges, log_k latency. All roots x
    odd/even n x radix 2/4, root in-place variants."""
    for radix in (2, 4):
        code = (
            "import sys; sys.path.insert(0, %r)\n"
            "import numpy as np\n"
            "from ucc_amd import core, dtypes\n"
            "from ucc_amd.testing import LocalJob\n"
            "c = core()\n"
            "for n in (2, 5, 8):\n"
            "    job = LocalJob(n)\n"
            "    smap = c.score_map_str(job.teams[0])\n"
            "    assert 'gather:host:0-inf:@tcp/knomial' in smap, smap\n"
            "    assert 'scatter:host:0-inf:@tcp/knomial' in smap, smap\n"
            "    rng = np.random.default_rng(n)\n"
            "    per = 257\n"
            "    for root in range(n):\n"
            "        # gather\n"
            "        srcs = [rng.standard_normal(per)\n"
            "                .astype(np.float32) for _ in range(n)]\n"
            "        dst = np.zeros(per * n, np.float32)\n"
            "        reqs = job.coll('gather', [\n"
            "            dict(src=srcs[r].ctypes.data,\n"
            "                 dst=(dst.ctypes.data if r == root else 0),\n"
            "                 count=(per * n if r == root else per),\n"
            "                 dt=dtypes.FLOAT32, root=root)\n"
            "            for r in range(n)])\n"
            "        job.run(reqs)\n"
            "        np.testing.assert_array_equal(\n"
            "            dst, np.concatenate(srcs))\n"
            "        # scatter\n"
            "        big = rng.standard_normal(per * n)\n"
            "        big = big.astype(np.float32)\n"
            "        outs = [np.zeros(per, np.float32)\n"
            "                for _ in range(n)]\n"
            "        reqs = job.coll('scatter', [\n"
            "            dict(src=(big.ctypes.data if r == root else 0),\n"
            "                 dst=outs[r].ctypes.data,\n"
            "                 count=(per * n if r == root else per),\n"
            "                 dt=dtypes.FLOAT32, root=root)\n"
            "            for r in range(n)])\n"
            "        job.run(reqs)\n"
            "        for r in range(n):\n"
            "            np.testing.assert_array_equal(\n"
            "                outs[r], big[r * per:(r + 1) * per])\n"
            "print('KN_GS_OK')\n"
        ) % (REPO,)
        env = dict(os.environ)
        env["UCC_TL_SHM_ENABLE"] = "0"
        env["UCC_TL_TCP_KN_RADIX"] = str(radix)
        env["UCC_TUNE"] = "gather:@knomial:99,scatter:@knomial:99"
        p = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, text=True, timeout=600)
        sys.stdout.write(p.stdout[-500:])
        sys.stderr.write(p.stderr[-3000:])
        assert p.returncode == 0 and "KN_GS_OK" in p.stdout, \
            f"radix {radix}"


def test_allreduce_dbt():
    """DBT allreduce (reference allreduce dbt role): DBT reduce to 0
    composed with DBT bcast; odd/even n, AVG, odd counts (uneven
    message halves)."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (2, 3, 6, 8):\n"
        "    job = LocalJob(n)\n"
        "    smap = core().score_map_str(job.teams[0])\n"
        "    assert 'allreduce' in smap and '@tcp/dbt' in smap, smap\n"
        "    rng = np.random.default_rng(n)\n"
        "    for count in (3001, 40_001):\n"
        "        arrs = [(rng.random(count) - 0.5).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        outs = [np.zeros(count, np.float32) for _ in range(n)]\n"
        "        reqs = job.coll('allreduce', [\n"
        "            dict(src=arrs[r].ctypes.data,\n"
        "                 dst=outs[r].ctypes.data, count=count,\n"
        "                 dt=dtypes.FLOAT32) for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        exp = np.sum(arrs, axis=0)\n"
        "        for o in outs:\n"
        "            np.testing.assert_allclose(o, exp, rtol=1e-5,\n"
        "                                       atol=1e-5)\n"
        "    arrs = [(rng.random(7001) - 0.5).astype(np.float32)\n"
        "            for _ in range(n)]\n"
        "    outs = [np.zeros(7001, np.float32) for _ in range(n)]\n"
        "    reqs = job.coll('allreduce', [\n"
        "        dict(src=arrs[r].ctypes.data, dst=outs[r].ctypes.data,\n"
        "             count=7001, dt=dtypes.FLOAT32, op=dtypes.OP_AVG)\n"
        "        for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    exp = np.mean(arrs, axis=0)\n"
        "    for o in outs:\n"
        "        np.testing.assert_allclose(o, exp, rtol=1e-5,\n"
        "                                   atol=1e-6)\n"
        "print('AR_DBT_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TUNE"] = "allreduce:@dbt:99"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-400:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "AR_DBT_OK" in p.stdout


def test_reduce_knomial_radix():
    """Radix-k knomial reduce (reference tl/ucp reduce knomial role):
    children's subtree sums climb the reversed bcast tree; every root,
    radices 2/3/4, odd/even n, AVG and in-place root."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (2, 5, 7, 8):\n"
        "    job = LocalJob(n)\n"
        "    rng = np.random.default_rng(n)\n"
        "    count = 1003\n"
        "    for root in range(n):\n"
        "        srcs = [(rng.random(count) - 0.5).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        dst = np.zeros(count, np.float32)\n"
        "        reqs = job.coll('reduce', [\n"
        "            dict(src=srcs[r].ctypes.data,\n"
        "                 dst=dst.ctypes.data if r == root else 0,\n"
        "                 count=count, dt=dtypes.FLOAT32, root=root)\n"
        "            for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        np.testing.assert_allclose(dst, np.sum(srcs, axis=0),\n"
        "                                   rtol=1e-5, atol=1e-5)\n"
        "    # AVG at root 1\n"
        "    srcs = [(rng.random(count) - 0.5).astype(np.float32)\n"
        "            for _ in range(n)]\n"
        "    dst = np.zeros(count, np.float32)\n"
        "    reqs = job.coll('reduce', [\n"
        "        dict(src=srcs[r].ctypes.data,\n"
        "             dst=dst.ctypes.data if r == 1 %% n else 0,\n"
        "             count=count, dt=dtypes.FLOAT32, root=1 %% n,\n"
        "             op=dtypes.OP_AVG) for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    np.testing.assert_allclose(dst, np.mean(srcs, axis=0),\n"
        "                               rtol=1e-5, atol=1e-6)\n"
        "print('REDUCE_KN_OK')\n"
    ) % (REPO,)
    for radix in ("2", "3", "4"):
        env = dict(os.environ)
        env["UCC_TL_SHM_ENABLE"] = "0"
        env["UCC_TUNE"] = "reduce:@knomial:99"
        env["UCC_TL_TCP_KN_RADIX"] = radix
        p = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, text=True, timeout=300)
        sys.stdout.write(p.stdout[-300:])
        sys.stderr.write(p.stderr[-2000:])
        assert p.returncode == 0 and "REDUCE_KN_OK" in p.stdout, radix


def test_allreduce_knomial_radix():
    """Radix-k knomial allreduce (reference allreduce_knomial +
    recursive_knomial.h PROXY/EXTRA role): non-power-of-k team sizes
    fold extras into proxies; result identical to the reference sum."""
    for radix in (3, 4):
        code = (
            "import sys; sys.path.insert(0, %r)\n"
            "import numpy as np\n"
            "from ucc_amd import core, dtypes\n"
            "from ucc_amd.testing import LocalJob\n"
            "c = core()\n"
            "for n in (2, 3, 5, 7, 9):\n"
            "    job = LocalJob(n)\n"
            "    assert '@tcp/knomial' in c.score_map_str(\n"
            "        job.teams[0]), n\n"
            "    rng = np.random.default_rng(n)\n"
            "    for cnt in (1, 501, 9_001):\n"
            "        arrs = [(rng.random(cnt) - 0.5)\n"
            "                .astype(np.float32) for _ in range(n)]\n"
            "        outs = job.allreduce_np(arrs)\n"
            "        exp = np.sum(arrs, axis=0)\n"
            "        for o in outs:\n"
            "            np.testing.assert_allclose(\n"
            "                o, exp, rtol=1e-5, atol=1e-5)\n"
            "    # AVG through the knomial path\n"
            "    arrs = [np.full(777, float(r + 1), np.float32)\n"
            "            for r in range(n)]\n"
            "    outs = job.allreduce_np(arrs, op=dtypes.OP_AVG)\n"
            "    exp = np.full(777, sum(range(1, n + 1)) / n,\n"
            "                  np.float32)\n"
            "    for o in outs:\n"
            "        np.testing.assert_allclose(o, exp, rtol=1e-5,\n"
            "                                   atol=1e-5)\n"
            "print('KN_AR_OK')\n"
        ) % (REPO,)
        env = dict(os.environ)
        env["UCC_TL_SHM_ENABLE"] = "0"
        env["UCC_TL_TCP_KN_RADIX"] = str(radix)
        env["UCC_TUNE"] = "allreduce:@knomial:99"
        p = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, text=True, timeout=600)
        sys.stdout.write(p.stdout[-500:])
        sys.stderr.write(p.stderr[-3000:])
        assert p.returncode == 0 and "KN_AR_OK" in p.stdout, \
            f"radix {radix}"


def test_reduce_scatterv_ring():
    """Ring reduce_scatterv (reference reduce_scatterv ring role): the
    ring engine with per-block v sizes — ragged counts incl. zero
    blocks, AVG, odd/even n."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "for n in (2, 3, 5, 8):\n"
        "    job = LocalJob(n)\n"
        "    rng = np.random.default_rng(n)\n"
        "    cnts = [137 * (r + 1) for r in range(n)]\n"
        "    cnts[n // 2] = 0\n"
        "    tot = sum(cnts)\n"
        "    dsp = np.cumsum([0] + cnts[:-1]).tolist()\n"
        "    srcs = [(rng.random(tot) - 0.5).astype(np.float32)\n"
        "            for _ in range(n)]\n"
        "    dsts = [np.zeros(max(cnts[r], 1), np.float32)\n"
        "            for r in range(n)]\n"
        "    for op, scale in ((dtypes.OP_SUM, 1.0),\n"
        "                      (dtypes.OP_AVG, 1.0 / n)):\n"
        "        reqs = job.coll('reduce_scatterv', [\n"
        "            dict(src=srcs[r].ctypes.data,\n"
        "                 dst=dsts[r].ctypes.data, count=cnts[r],\n"
        "                 dt=dtypes.FLOAT32, op=op,\n"
        "                 dst_counts=cnts, dst_displs=dsp)\n"
        "            for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        exp = np.sum(srcs, axis=0) * scale\n"
        "        for r in range(n):\n"
        "            if cnts[r]:\n"
        "                np.testing.assert_allclose(\n"
        "                    dsts[r][:cnts[r]],\n"
        "                    exp[dsp[r]:dsp[r] + cnts[r]],\n"
        "                    rtol=1e-5, atol=1e-5)\n"
        "print('RSV_RING_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TUNE"] = "reduce_scatterv:@ring:99"
    env["UCC_TL_TCP_RS_RING_MIN"] = "0"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-400:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "RSV_RING_OK" in p.stdout


def test_reduce_scatter_halving():
    """Recursive-halving reduce_scatter (reference tl/ucp rs knomial /
    Rabenseifner role): log2(m) halving rounds with a fold for
    non-power-of-2 n and a final block redistribution; sum + AVG."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "c = core()\n"
        "for n in (2, 3, 5, 6, 8, 11):\n"
        "    job = LocalJob(n)\n"
        "    smap = c.score_map_str(job.teams[0])\n"
        "    assert '@tcp/knomial' in smap, smap\n"
        "    rng = np.random.default_rng(n)\n"
        "    for per in (1, 501, 4001):\n"
        "        srcs = [(rng.random(per * n) - 0.5).astype(np.float32)\n"
        "                for _ in range(n)]\n"
        "        dsts = [np.zeros(per, np.float32) for _ in range(n)]\n"
        "        reqs = job.coll('reduce_scatter', [\n"
        "            dict(src=srcs[r].ctypes.data,\n"
        "                 dst=dsts[r].ctypes.data,\n"
        "                 count=per, dt=dtypes.FLOAT32)\n"
        "            for r in range(n)])\n"
        "        job.run(reqs)\n"
        "        exp = np.sum(srcs, axis=0)\n"
        "        for r in range(n):\n"
        "            np.testing.assert_allclose(\n"
        "                dsts[r], exp[r * per:(r + 1) * per],\n"
        "                rtol=1e-5, atol=1e-5)\n"
        "    # AVG (receiver-side 1/n scaling)\n"
        "    per = 777\n"
        "    srcs = [(rng.random(per * n) - 0.5).astype(np.float32)\n"
        "            for _ in range(n)]\n"
        "    dsts = [np.zeros(per, np.float32) for _ in range(n)]\n"
        "    reqs = job.coll('reduce_scatter', [\n"
        "        dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,\n"
        "             count=per, dt=dtypes.FLOAT32, op=dtypes.OP_AVG)\n"
        "        for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    exp = np.sum(srcs, axis=0)\n"
        "    for r in range(n):\n"
        "        np.testing.assert_allclose(\n"
        "            dsts[r], exp[r * per:(r + 1) * per] / n,\n"
        "            rtol=1e-5, atol=1e-5)\n"
        "print('RS_HALVING_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TUNE"] = "reduce_scatter:@knomial:99"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=600)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-2000:])
    assert p.returncode == 0 and "RS_HALVING_OK" in p.stdout


def test_reduce_scatter_ring():
    """Ring reduce-scatter (reference tl/ucp rs ring role): n-1 rounds,
    each rank ends owning its fully reduced block; in-place + AVG."""
    code = (
        "import sys; sys.path.insert(0, %r)\n"
        "import numpy as np\n"
        "from ucc_amd import core, dtypes\n"
        "from ucc_amd.testing import LocalJob\n"
        "c = core()\n"
        "for n in (2, 3, 5, 8):\n"
        "    job = LocalJob(n)\n"
        "    assert 'reduce_scatter:host' in c.score_map_str(\n"
        "        job.teams[0])\n"
        "    rng = np.random.default_rng(n)\n"
        "    per = 4001\n"
        "    srcs = [(rng.random(per * n) - 0.5).astype(np.float32)\n"
        "            for _ in range(n)]\n"
        "    dsts = [np.zeros(per, np.float32) for _ in range(n)]\n"
        "    reqs = job.coll('reduce_scatter', [\n"
        "        dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,\n"
        "             count=per, dt=dtypes.FLOAT32)\n"
        "        for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    exp = np.sum(srcs, axis=0)\n"
        "    for r in range(n):\n"
        "        np.testing.assert_allclose(\n"
        "            dsts[r], exp[r * per:(r + 1) * per], rtol=1e-5,\n"
        "            atol=1e-5)\n"
        "    # AVG\n"
        "    reqs = job.coll('reduce_scatter', [\n"
        "        dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data,\n"
        "             count=per, dt=dtypes.FLOAT32, op=dtypes.OP_AVG)\n"
        "        for r in range(n)])\n"
        "    job.run(reqs)\n"
        "    for r in range(n):\n"
        "        np.testing.assert_allclose(\n"
        "            dsts[r], exp[r * per:(r + 1) * per] / n,\n"
        "            rtol=1e-5, atol=1e-5)\n"
        "print('RS_RING_OK')\n"
    ) % (REPO,)
    env = dict(os.environ)
    env["UCC_TL_SHM_ENABLE"] = "0"
    env["UCC_TL_TCP_RS_RING_MIN"] = "1024"
    env["UCC_TL_TCP_RS_RING_BIDIR"] = "1"  # both ring directions
    env["UCC_TUNE"] = "reduce_scatter:@ring:99"
    p = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=600)
    sys.stdout.write(p.stdout[-500:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "RS_RING_OK" in p.stdout
