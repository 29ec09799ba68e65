"""Hierarchical composition (cl/hier role): node reduce -> leader
allreduce -> node bcast over internal sub-teams, exercised on one
machine via UCC_FAKE_NODE_SPLIT (contexts spread over pseudo-nodes).
Reference parity: cl/hier allreduce RAB (allreduce_rab.c)."""

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

n = 6  # 2 pseudo-nodes x 3 ranks
job = LocalJob(n)
c = core()
smap = c.score_map_str(job.teams[0])
assert "@hier/rab" in smap, "hier not installed:\n" + smap
assert "@hier/split_rail" in smap, "split_rail not installed:\n" + smap

rng = np.random.default_rng(11)
for count in (64, 5000, 100_001):
    arrs = [(rng.random(count) - 0.5).astype(np.float32) for _ in range(n)]
    outs = job.allreduce_np(arrs)
    exp = np.sum(arrs, axis=0)
    for o in outs:
        np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-5)

# AVG (post-scaled)
arrs = [np.full(1000, float(r + 1), np.float32) for r in range(n)]
outs = job.allreduce_np(arrs, op=dtypes.OP_AVG)
exp = np.full(1000, sum(range(1, n + 1)) / n, np.float32)
for o in outs:
    np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-5)

# hier 2step bcast from a non-leader root
root = 4  # rank 4: non-leader in both 2- and 3-way splits (n=6)
bufs = [np.zeros(30_000, np.float32) for _ in range(n)]
bufs[root][:] = rng.random(30_000).astype(np.float32)
exp2 = bufs[root].copy()
reqs = job.coll("bcast", [
    dict(src=b.ctypes.data, dst=0, count=30_000, dt=dtypes.FLOAT32,
         root=root) for b in bufs])
job.run(reqs)
for b in bufs:
    np.testing.assert_array_equal(b, exp2)

# split_rail explicitly (>=64KB auto-picks it; also force via tuning
# check on a large in-place vector)
arrs = [(rng.random(300_000) - 0.5).astype(np.float32) for _ in range(n)]
exp3 = np.sum(arrs, axis=0)
reqs = job.coll("allreduce", [
    dict(src=0, dst=a.ctypes.data, count=300_000, dt=dtypes.FLOAT32,
         flags=c.FLAG_IN_PLACE) for a in arrs])
job.run(reqs)
for a in arrs:
    np.testing.assert_allclose(a, exp3, rtol=1e-5, atol=1e-4)

# hier 2step reduce: leader root, non-leader root, AVG, in-place
for root, op in ((0, dtypes.OP_SUM), (4, dtypes.OP_SUM),
                 (4, dtypes.OP_AVG)):
    srcs = [rng.random(12_345).astype(np.float32) for _ in range(n)]
    dst = np.zeros(12_345, np.float32)
    exp4 = np.sum(srcs, axis=0)
    if op == dtypes.OP_AVG:
        exp4 = exp4 / n
    reqs = job.coll("reduce", [
        dict(src=srcs[r].ctypes.data, dst=(dst.ctypes.data if r == root
                                           else 0),
             count=12_345, dt=dtypes.FLOAT32, root=root, op=op)
        for r in range(n)])
    job.run(reqs)
    np.testing.assert_allclose(dst, exp4, rtol=1e-5, atol=1e-5)

# in-place reduce at root (root's contribution read from dst)
srcs = [rng.random(999).astype(np.float32) for _ in range(n)]
root = 3
dst = srcs[root].copy()
exp5 = np.sum(srcs, axis=0)
reqs = job.coll("reduce", [
    dict(src=(0 if r == root else srcs[r].ctypes.data),
         dst=(dst.ctypes.data if r == root else 0), count=999,
         dt=dtypes.FLOAT32, root=root,
         flags=(c.FLAG_IN_PLACE if r == root else 0))
    for r in range(n)])
job.run(reqs)
np.testing.assert_allclose(dst, exp5, rtol=1e-5, atol=1e-5)

# hier allgatherv (node-packed relay + unpack), ragged counts incl. 0
assert "@hier/node_packed" in smap, smap
cnts = [137, 0, 5001, 64, 2999, 1][:n]
while len(cnts) < n:
    cnts.append(71 * len(cnts))
dsps = []
off = 17  # leading gap: displacements need not be packed
for cq in cnts:
    dsps.append(off)
    off += cq + 3  # gaps between blocks
tot = off + 5
srcs6 = [rng.random(max(cnts[r], 1)).astype(np.float64)[:cnts[r]]
         for r in range(n)]
dsts6 = [np.zeros(tot, np.float64) for _ in range(n)]
reqs = job.coll("allgatherv", [
    dict(src=(srcs6[r].ctypes.data if cnts[r] else 0),
         dst=dsts6[r].ctypes.data, count=cnts[r], dt=dtypes.FLOAT64,
         dst_counts=cnts, dst_displs=dsps) for r in range(n)])
job.run(reqs)
for d in dsts6:
    for r in range(n):
        np.testing.assert_array_equal(
            d[dsps[r]:dsps[r] + cnts[r]], srcs6[r])

# hier node-aggregated alltoallv: skewed counts incl. zero pairs
assert "@hier/node_aggregated" in smap, smap
scnt = [[((r * 3 + d * 5) %% 7) * 23 + (0 if (r + d) %% 4 == 0 else 11)
         for d in range(n)] for r in range(n)]
rcnt = [[scnt[s][r] for s in range(n)] for r in range(n)]
def _dsp(cs):
    out, off = [], 2  # leading gap
    for cq in cs:
        out.append(off)
        off += cq + 1  # inter-block gaps
    return out, off
sdsp = [_dsp(c)[0] for c in scnt]
rdsp = [_dsp(c)[0] for c in rcnt]
stot = [_dsp(c)[1] for c in scnt]
rtot = [_dsp(c)[1] for c in rcnt]
srcs7 = [rng.random(stot[r]).astype(np.float32) for r in range(n)]
dsts7 = [np.zeros(rtot[r], np.float32) for r in range(n)]
reqs = job.coll("alltoallv", [
    dict(src=srcs7[r].ctypes.data, dst=dsts7[r].ctypes.data, count=0,
         dt=dtypes.FLOAT32, src_counts=scnt[r], src_displs=sdsp[r],
         dst_counts=rcnt[r], dst_displs=rdsp[r]) for r in range(n)])
job.run(reqs)
for r in range(n):
    for s in range(n):
        np.testing.assert_array_equal(
            dsts7[r][rdsp[r][s]:rdsp[r][s] + rcnt[r][s]],
            srcs7[s][sdsp[s][r]:sdsp[s][r] + scnt[s][r]])

# hier barrier (fanin -> leaders barrier -> fanout), repeated
for _ in range(3):
    reqs = job.coll("barrier", [dict(src=0, dst=0, count=0, dt=dtypes.INT8)
                              for _ in range(n)])
    job.run(reqs)

# persistent allreduce over hier (re-post re-runs the composition)
pa = [np.zeros(4000, np.float32) for _ in range(n)]
pout = [np.zeros(4000, np.float32) for _ in range(n)]
preq = job.coll("allreduce", [
    dict(src=pa[r].ctypes.data, dst=pout[r].ctypes.data, count=4000,
         dt=dtypes.FLOAT32, flags=c.FLAG_PERSISTENT) for r in range(n)])
for it in range(3):
    for r in range(n):
        pa[r][:] = rng.random(4000).astype(np.float32) + it
    expp = np.sum(pa, axis=0)
    job.run(preq)
    for o in pout:
        np.testing.assert_allclose(o, expp, rtol=1e-5, atol=1e-4)

# repeated (sub-team/slot reuse)
for it in range(10):
    arrs = [np.full(257, float(r * it + 1), np.float64) for r in range(n)]
    outs = job.allreduce_np(arrs)
    exp = np.sum(arrs, axis=0)
    for o in outs:
        np.testing.assert_allclose(o, exp)
print("HIER_OK")
""" % (REPO,)


def _run(env_extra):
    env = dict(os.environ)
    env.update(env_extra)
    p = subprocess.run([sys.executable, "-c", WORKER], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "HIER_OK" in p.stdout


def test_hier_allreduce_fake_nodes():
    _run({"UCC_FAKE_NODE_SPLIT": "2"})


def test_hier_allreduce_three_nodes():
    _run({"UCC_FAKE_NODE_SPLIT": "3"})


NONUNIFORM_WORKER = r"""
import sys
import numpy as np
sys.path.insert(0, %r)
from ucc_amd import core, dtypes
from ucc_amd.testing import LocalJob

n = 5  # split 2 -> pseudo-nodes of 3 and 2 (NON-uniform)
job = LocalJob(n)
c = core()
smap = c.score_map_str(job.teams[0])
assert "@hier/rab" in smap, smap
assert "@hier/split_rail" not in smap, smap  # needs uniform nodes
assert "@hier/node_aggregated" in smap, smap

rng = np.random.default_rng(21)
# RAB allreduce
arrs = [(rng.random(40_000) - 0.5).astype(np.float32) for _ in range(n)]
outs = job.allreduce_np(arrs)
exp = np.sum(arrs, axis=0)
for o in outs:
    np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-5)

# node-aggregated a2av across non-uniform nodes
scnt = [[(r + 2 * d + 1) %% 5 * 13 for d in range(n)] for r in range(n)]
rcnt = [[scnt[s][r] for s in range(n)] for r in range(n)]
def _d(cs):
    out, off = [], 0
    for cq in cs:
        out.append(off)
        off += cq
    return out, off
sdsp = [_d(cq)[0] for cq in scnt]
rdsp = [_d(cq)[0] for cq in rcnt]
srcs = [rng.random(max(_d(scnt[r])[1], 1)).astype(np.float32)
        for r in range(n)]
dsts = [np.zeros(max(_d(rcnt[r])[1], 1), np.float32) for r in range(n)]
reqs = job.coll("alltoallv", [
    dict(src=srcs[r].ctypes.data, dst=dsts[r].ctypes.data, count=0,
         dt=dtypes.FLOAT32, src_counts=scnt[r], src_displs=sdsp[r],
         dst_counts=rcnt[r], dst_displs=rdsp[r]) for r in range(n)])
job.run(reqs)
for r in range(n):
    for s in range(n):
        np.testing.assert_array_equal(
            dsts[r][rdsp[r][s]:rdsp[r][s] + rcnt[r][s]],
            srcs[s][sdsp[s][r]:sdsp[s][r] + scnt[s][r]])

# 2step reduce with a non-leader root on the smaller node
srcs = [rng.random(7_777).astype(np.float32) for _ in range(n)]
dst = np.zeros(7_777, np.float32)
root = n - 1
reqs = job.coll("reduce", [
    dict(src=srcs[r].ctypes.data,
         dst=(dst.ctypes.data if r == root else 0), count=7_777,
         dt=dtypes.FLOAT32, root=root) for r in range(n)])
job.run(reqs)
np.testing.assert_allclose(dst, np.sum(srcs, axis=0), rtol=1e-5,
                           atol=1e-5)
print("HIER_NU_OK")
""" % (REPO,)


def test_hier_nonuniform_nodes():
    env = dict(os.environ)
    env["UCC_FAKE_NODE_SPLIT"] = "2"
    p = subprocess.run([sys.executable, "-c", NONUNIFORM_WORKER], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "HIER_NU_OK" in p.stdout


PIPELINE_WORKER = r"""
import os
import sys
import numpy as np
sys.path.insert(0, %r)
from ucc_amd import core, dtypes
from ucc_amd.testing import LocalJob

n = 6
job = LocalJob(n)
c = core()

rng = np.random.default_rng(31)
count = 200_000  # 800 KB fp32, FRAG_SIZE=64k -> 13 fragments
arrs = [(rng.random(count) - 0.5).astype(np.float32) for _ in range(n)]
outs = job.allreduce_np(arrs)
exp = np.sum(arrs, axis=0)
for o in outs:
    np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-5)

# AVG through the pipelined path (per-fragment post-scale)
arrs = [np.full(count, float(r + 1), np.float32) for r in range(n)]
outs = job.allreduce_np(arrs, op=dtypes.OP_AVG)
expa = np.full(count, sum(range(1, n + 1)) / n, np.float32)
for o in outs:
    np.testing.assert_allclose(o, expa, rtol=1e-5, atol=1e-5)
print("HIER_PIPE_OK")
"""  % (REPO,)


import pytest


@pytest.mark.parametrize("alg", ["rab", "split_rail"])
def test_hier_pipelined_allreduce_interleaves(tmp_path, alg):
    """fragments of a composed hier collective must
    OVERLAP stages — stage-2 (node bcast / node allgatherv) of fragment
    0 must run before stage-1 (inter-node phase) of the last fragment
    completes, per pipeline task (reference
    ucc_schedule_pipelined.h:36-78 role)."""
    trace = tmp_path / "pipe_trace.txt"
    env = dict(os.environ)
    env.update({
        "UCC_FAKE_NODE_SPLIT": "2",
        "UCC_CL_HIER_FRAG_SIZE": "65536",
        "UCC_CL_HIER_PIPELINE_TRACE": str(trace),
        "UCC_TUNE": f"allreduce:@{alg}:99",
    })
    p = subprocess.run([sys.executable, "-c", PIPELINE_WORKER], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "HIER_PIPE_OK" in p.stdout
    assert trace.exists(), "pipeline trace file missing: pipelined path" \
                           " was not taken"
    # group events by task instance; each line: "P|C <taskid> frag stage"
    per_task = {}
    gen = {}
    for ln in trace.read_text().splitlines():
        ev, tid, f, s = ln.split()
        f, s = int(f), int(s)
        # task addresses are reused across collectives: a P(0,0) event
        # opens a fresh stream for that address
        if ev == "P" and f == 0 and s == 0:
            gen[tid] = gen.get(tid, 0) + 1
        key = (tid, gen.get(tid, 0))
        per_task.setdefault(key, []).append((ev, f, s))
    checked = 0
    for tid, evs in per_task.items():
        frags = {f for _, f, _ in evs}
        if len(frags) < 3:
            continue
        last = max(frags)
        # index of stage-2 post of frag 0 and stage-1 completion of the
        # last frag within THIS task's event order
        i_p02 = next(i for i, (ev, f, s) in enumerate(evs)
                     if ev == "P" and f == 0 and s == 2)
        i_cl1 = next(i for i, (ev, f, s) in enumerate(evs)
                     if ev == "C" and f == last and s == 1)
        assert i_p02 < i_cl1, (
            f"no interleave in task {tid}: P(0,2)@{i_p02} "
            f">= C({last},1)@{i_cl1}")
        # shared-team post order must be CANONICAL (rank-deterministic):
        # stages 0 and 2 both post on the node team; required sequence
        # is E(0), E(1), L(0), E(2), L(1), ..., L(last)
        node_posts = [(f, st) for ev, f, st in evs
                      if ev == "P" and st in (0, 2)]
        nf = last + 1
        pos = {}
        for i2, fs in enumerate(node_posts):
            pos.setdefault(fs, i2)
        for f in range(nf):
            need = (min(f + 1, nf - 1), 0)
            assert pos[(f, 2)] > pos[need], (
                f"canonical interleave violated: L({f}) before "
                f"E({need[0]}) in {node_posts}")
        # pipeline depth respected: a frag is "open" from its first post
        # until its own final-stage completion; never more than pdepth=2
        # open at once
        final_stage = {}
        for ev, f, s in evs:
            final_stage[f] = max(final_stage.get(f, 0), s)
        open_frags = set()
        max_open = 0
        for ev, f, s in evs:
            if ev == "P":
                open_frags.add(f)
            if ev == "C" and s == final_stage[f]:
                open_frags.discard(f)
            max_open = max(max_open, len(open_frags))
        assert max_open <= 2, f"pdepth exceeded: {max_open}"
        checked += 1
    assert checked >= 1, "no multi-fragment pipeline task traced"


DEVICE_WORKER = r"""
import sys
import numpy as np
sys.path.insert(0, %r)
import torch
from ucc_amd import core, dtypes
from ucc_amd.testing import LocalJob

torch.cuda.set_device(0)
n = 4  # 2 pseudo-nodes x 2 ranks
job = LocalJob(n)
c = core()
smap = c.score_map_str(job.teams[0])
assert "@hier/rab_dev" in smap, smap

torch.manual_seed(77)
# multi-fragment device RAB: node phases on cdna4, leader phase staged
# D2H over the host transports
for count in (5000, 3_000_000):
    srcs = [torch.randn(count, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(count, device="cuda") for _ in range(n)]
    exp = sum(s.cpu() for s in srcs)
    reqs = job.coll("allreduce", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(),
             count=count, dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        for r in range(n)])
    job.run(reqs)
    torch.cuda.synchronize()
    for d in dsts:
        torch.testing.assert_close(d.cpu(), exp, rtol=1e-5, atol=1e-4)

# AVG on device (stage-3 device scale)
srcs = [torch.full((40_000,), float(r + 1), device="cuda")
        for r in range(n)]
dsts = [torch.zeros(40_000, device="cuda") for _ in range(n)]
exp = torch.full((40_000,), sum(range(1, n + 1)) / n)
reqs = job.coll("allreduce", [
    dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=40_000,
         dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA, op=dtypes.OP_AVG)
    for r in range(n)])
job.run(reqs)
torch.cuda.synchronize()
for d in dsts:
    torch.testing.assert_close(d.cpu(), exp, rtol=1e-5, atol=1e-5)
# 2step bcast on device (root-node dev bcast -> staged leaders hop ->
# node dev bcast); non-leader root
root = 3
bufs = [torch.zeros(123_456, device="cuda") for _ in range(n)]
bufs[root].normal_()
exp_b = bufs[root].cpu().clone()
reqs = job.coll("bcast", [
    dict(src=b.data_ptr(), dst=0, count=123_456, dt=dtypes.FLOAT32,
         mem_type=dtypes.MEM_CUDA, root=root) for b in bufs])
job.run(reqs)
torch.cuda.synchronize()
for b in bufs:
    torch.testing.assert_close(b.cpu(), exp_b)

# 2step reduce on device: leader and non-leader roots, AVG
for root, op in ((0, dtypes.OP_SUM), (3, dtypes.OP_SUM),
                 (3, dtypes.OP_AVG)):
    srcs = [torch.randn(77_777, device="cuda") for _ in range(n)]
    dst = torch.zeros(77_777, device="cuda")
    exp_r = sum(s2.cpu() for s2 in srcs)
    if op == dtypes.OP_AVG:
        exp_r = exp_r / n
    reqs = job.coll("reduce", [
        dict(src=srcs[r].data_ptr(),
             dst=(dst.data_ptr() if r == root else 0), count=77_777,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA, root=root,
             op=op) for r in range(n)])
    job.run(reqs)
    torch.cuda.synchronize()
    torch.testing.assert_close(dst.cpu(), exp_r, rtol=1e-5, atol=1e-4)

# device allgatherv (node-packed): ragged counts incl. a zero block;
# node gatherv + node bcast run on the device TL, the leaders hop is
# staged D2H/H2D
cnts = [1000 * (r + 1) for r in range(n)]
cnts[1] = 0
dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).astype(np.uint64)
total = int(sum(cnts))
g_srcs = [torch.randn(max(cnts[r], 1), device="cuda")
          for r in range(n)]
g_dsts = [torch.zeros(total, device="cuda") for _ in range(n)]
reqs = job.coll("allgatherv", [
    dict(src=g_srcs[r].data_ptr(), dst=g_dsts[r].data_ptr(),
         count=cnts[r], dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
         dst_counts=cnts, dst_displs=dsps.tolist())
    for r in range(n)])
job.run(reqs)
torch.cuda.synchronize()
exp_g = torch.cat([g_srcs[r][:cnts[r]].cpu() for r in range(n)])
for d in g_dsts:
    torch.testing.assert_close(d.cpu(), exp_g)

print("HIER_DEV_OK")
""" % (REPO,)


@pytest.mark.gpu
def test_hier_device_allreduce_fake_nodes():
    """Device-memory hier RAB: node phases over the cdna4 TL, leader
    phase staged D2H over host transports (multi-node GPU composition,
    testable on one box via UCC_FAKE_NODE_SPLIT)."""
    env = dict(os.environ)
    env.update({"UCC_FAKE_NODE_SPLIT": "2",
                "UCC_CL_HIER_FRAG_SIZE": "1048576"})
    p = subprocess.run([sys.executable, "-c", DEVICE_WORKER], env=env,
                       capture_output=True, text=True, timeout=300)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "HIER_DEV_OK" in p.stdout


SCALE_WORKER = r"""
import sys
import numpy as np
sys.path.insert(0, %r)
from ucc_amd import core, dtypes
from ucc_amd.testing import LocalJob

n = 16  # 4 pseudo-nodes x 4 ranks: 5 sub-team bootstraps per rank
job = LocalJob(n)
c = core()
smap = c.score_map_str(job.teams[0])
assert "@hier/rab" in smap and "@hier/split_rail" in smap, smap
rng = np.random.default_rng(161)

# allreduce through RAB (small) and split_rail (>=64k) bands
for count in (5000, 40_000):
    arrs = [(rng.random(count) - 0.5).astype(np.float32)
            for _ in range(n)]
    outs = [np.zeros(count, np.float32) for _ in range(n)]
    reqs = job.coll("allreduce", [
        dict(src=arrs[r].ctypes.data, dst=outs[r].ctypes.data,
             count=count, dt=dtypes.FLOAT32) for r in range(n)])
    job.run(reqs)
    exp = np.sum(arrs, axis=0)
    for o in outs:
        np.testing.assert_allclose(o, exp, rtol=1e-5, atol=1e-4)

# 2step bcast from a non-leader root on the last node
root = n - 1
bufs = [np.zeros(9001, np.float64) for _ in range(n)]
bufs[root][:] = rng.random(9001)
exp_b = bufs[root].copy()
reqs = job.coll("bcast", [
    dict(src=b.ctypes.data, dst=0, count=9001, dt=dtypes.FLOAT64,
         root=root) for b in bufs])
job.run(reqs)
for b in bufs:
    np.testing.assert_array_equal(b, exp_b)

# 2step reduce to a mid-team non-leader root
root = 6
rsrcs = [(rng.random(7777) - 0.5).astype(np.float32) for _ in range(n)]
rdst = np.zeros(7777, np.float32)
reqs = job.coll("reduce", [
    dict(src=rsrcs[r].ctypes.data,
         dst=rdst.ctypes.data if r == root else 0, count=7777,
         dt=dtypes.FLOAT32, root=root) for r in range(n)])
job.run(reqs)
np.testing.assert_allclose(rdst, np.sum(rsrcs, axis=0), rtol=1e-5,
                           atol=1e-4)

# node-packed allgatherv with ragged counts
cnts = [97 * ((r %% 5) + 1) for r in range(n)]
dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).astype(np.uint64)
total = int(sum(cnts))
gs = [(rng.random(cnts[r])).astype(np.float32) for r in range(n)]
gd = [np.zeros(total, np.float32) for _ in range(n)]
reqs = job.coll("allgatherv", [
    dict(src=gs[r].ctypes.data, dst=gd[r].ctypes.data, count=cnts[r],
         dt=dtypes.FLOAT32, dst_counts=cnts,
         dst_displs=dsps.tolist()) for r in range(n)])
job.run(reqs)
exp_g = np.concatenate(gs)
for d in gd:
    np.testing.assert_array_equal(d, exp_g)

# barrier
reqs = job.coll("barrier", [dict(src=0, dst=0, count=0,
                                 dt=dtypes.FLOAT32) for _ in range(n)])
job.run(reqs)
print("HIER_SCALE_OK")
""" % (REPO,)


def test_hier_scale_16_ranks_4_nodes():
    """Hier bootstrap + collectives at 16 ranks over 4 pseudo-nodes:
    every rank drives node/leaders/rail sub-team creation through the
    padded parent-OOB rounds (the largest in-process hier scale; the
    bootstrap alignment bugs are O(ranks x rounds))."""
    env = dict(os.environ)
    env["UCC_FAKE_NODE_SPLIT"] = "4"
    p = subprocess.run([sys.executable, "-c", SCALE_WORKER], env=env,
                       capture_output=True, text=True, timeout=600)
    sys.stdout.write(p.stdout[-2000:])
    sys.stderr.write(p.stderr[-3000:])
    assert p.returncode == 0 and "HIER_SCALE_OK" in p.stdout
