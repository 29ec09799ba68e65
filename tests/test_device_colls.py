"""Device-path collective correctness on one MI355X: the in-process
multi-rank jig with all ranks on cuda:0 exercises the cdna4 TL end to end
(staged-linear kernels + fused single-kernel allreduce + HIP peer reads),
validated against fp32 torch references computed on host.
"""

import numpy as np
import pytest

torch = pytest.importorskip("torch")

from ucc_amd import core, dtypes
from ucc_amd.testing import LocalJob

pytestmark = pytest.mark.gpu

SIZES = [2, 4, 8]


@pytest.fixture(scope="module", params=SIZES)
def job(request):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.cuda.set_device(0)
    return LocalJob(request.param)


def _run_device(job, coll, per_rank):
    reqs = job.coll(coll, per_rank)
    job.run(reqs)
    torch.cuda.synchronize()


TOL = {
    torch.float32: (1e-5, 1e-5),
    torch.bfloat16: (2e-2, 2e-2),
    torch.float16: (1e-2, 1e-2),
    torch.float64: (1e-12, 1e-12),
}


@pytest.mark.parametrize("count", [8, 1024, 200_000, 3_000_000])
@pytest.mark.parametrize("tdt", [torch.float32, torch.bfloat16])
def test_allreduce_device(job, count, tdt):
    torch.manual_seed(7)
    n = job.n
    srcs = [torch.randn(count, dtype=torch.float32).to(tdt).cuda()
            for _ in range(n)]
    dsts = [torch.zeros(count, dtype=tdt, device="cuda") for _ in range(n)]
    expected = sum(s.cpu().float() for s in srcs)
    _run_device(job, "allreduce", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=count,
             dt=dtypes.from_torch(tdt), mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    rtol, atol = TOL[tdt]
    for d in dsts:
        torch.testing.assert_close(d.cpu().float(), expected,
                                   rtol=rtol, atol=atol * job.n * 3)


@pytest.mark.parametrize("count", [1024, 1_000_000])
def test_allreduce_device_avg(job, count):
    torch.manual_seed(8)
    n = job.n
    srcs = [torch.randn(count, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(count, device="cuda") for _ in range(n)]
    expected = sum(s.cpu() for s in srcs) / n
    _run_device(job, "allreduce", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=count,
             dt=dtypes.FLOAT32, op=dtypes.OP_AVG, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    for d in dsts:
        torch.testing.assert_close(d.cpu(), expected, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("per", [1000, 500_000])
def test_reduce_scatter_device(job, per):
    torch.manual_seed(9)
    n = job.n
    srcs = [torch.randn(per * n, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(per, device="cuda") for _ in range(n)]
    expected = sum(s.cpu() for s in srcs)
    _run_device(job, "reduce_scatter", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=per,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    for r in range(n):
        torch.testing.assert_close(dsts[r].cpu(),
                                   expected[r * per:(r + 1) * per],
                                   rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("per", [999, 400_000])
def test_allgather_device(job, per):
    torch.manual_seed(10)
    n = job.n
    srcs = [torch.randn(per, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(per * n, device="cuda") for _ in range(n)]
    expected = torch.cat([s.cpu() for s in srcs])
    _run_device(job, "allgather", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=per * n,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    for d in dsts:
        torch.testing.assert_close(d.cpu(), expected)


def test_bcast_device(job):
    torch.manual_seed(11)
    n = job.n
    root = n - 1
    bufs = [torch.zeros(123_457, device="cuda") for _ in range(n)]
    bufs[root] = torch.randn(123_457, device="cuda")
    expected = bufs[root].cpu().clone()
    _run_device(job, "bcast", [
        dict(src=bufs[r].data_ptr(), dst=0, count=123_457,
             dt=dtypes.FLOAT32, root=root, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    for b in bufs:
        torch.testing.assert_close(b.cpu(), expected)


def test_reduce_device(job):
    torch.manual_seed(12)
    n = job.n
    srcs = [torch.randn(77_777, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(77_777, device="cuda") for _ in range(n)]
    expected = sum(s.cpu() for s in srcs)
    _run_device(job, "reduce", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=77_777,
             dt=dtypes.FLOAT32, root=0, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    torch.testing.assert_close(dsts[0].cpu(), expected, rtol=1e-5, atol=1e-4)


def test_allreduce_device_fp8(job):
    """fp8 e4m3 sum: upconvert-accumulate-downconvert in f32 (SURVEY
    fp8 semantics note); reference computed the same way on host."""
    if not hasattr(torch, "float8_e4m3fn"):
        pytest.skip("torch without fp8")
    torch.manual_seed(13)
    n = job.n
    count = 4096
    srcs_f = [torch.randn(count) * 0.25 for _ in range(n)]
    srcs = [s.to(torch.float8_e4m3fn).cuda() for s in srcs_f]
    dsts = [torch.zeros(count, dtype=torch.float8_e4m3fn, device="cuda")
            for _ in range(n)]
    expected = sum(s.cpu().float() for s in srcs)
    _run_device(job, "allreduce", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=count,
             dt=dtypes.FLOAT8_E4M3, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    for d in dsts:
        torch.testing.assert_close(d.cpu().float(), expected,
                                   rtol=0.15, atol=0.5)


def test_persistent_allreduce_device(job):
    torch.manual_seed(14)
    n = job.n
    c = job.c
    count = 100_000
    srcs = [torch.randn(count, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(count, device="cuda") for _ in range(n)]
    reqs = job.coll("allreduce", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=count,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
             flags=c.FLAG_PERSISTENT)
        for r in range(n)
    ])
    for it in range(4):
        for s in srcs:
            s.add_(1.0)
        expected = sum(s.cpu() for s in srcs)
        job.run(reqs)
        torch.cuda.synchronize()
        for d in dsts:
            torch.testing.assert_close(d.cpu(), expected, rtol=1e-5,
                                       atol=1e-4)


@pytest.mark.parametrize("per", [512, 300_000])
def test_alltoall_device(job, per):
    torch.manual_seed(15)
    n = job.n
    srcs = [torch.randn(per * n, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(per * n, device="cuda") for _ in range(n)]
    _run_device(job, "alltoall", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=per * n,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    for r in range(n):
        expected = torch.cat(
            [srcs[s].cpu()[r * per:(r + 1) * per] for s in range(n)])
        torch.testing.assert_close(dsts[r].cpu(), expected)


def test_alltoallv_device_skewed(job):
    """MoE expert-parallel pattern: skewed per-peer splits (config #4)."""
    torch.manual_seed(16)
    n = job.n
    # rank r sends (r+1)*(d+1)*37 elements to rank d
    scnt = [[(r + 1) * (d + 1) * 37 for d in range(n)] for r in range(n)]
    rcnt = [[scnt[s][r] for s in range(n)] for r in range(n)]
    sdsp = [np.concatenate([[0], np.cumsum(c)[:-1]]).tolist() for c in scnt]
    rdsp = [np.concatenate([[0], np.cumsum(c)[:-1]]).tolist() for c in rcnt]
    srcs = [torch.randn(sum(scnt[r]), device="cuda") for r in range(n)]
    dsts = [torch.zeros(sum(rcnt[r]), device="cuda") for r in range(n)]
    _run_device(job, "alltoallv", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=0,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
             src_counts=scnt[r], src_displs=sdsp[r],
             dst_counts=rcnt[r], dst_displs=rdsp[r])
        for r in range(n)
    ])
    for r in range(n):
        for s in range(n):
            got = dsts[r].cpu()[rdsp[r][s]:rdsp[r][s] + rcnt[r][s]]
            exp = srcs[s].cpu()[sdsp[s][r]:sdsp[s][r] + scnt[s][r]]
            torch.testing.assert_close(got, exp)


def test_oneshot_zerocopy_allreduce(job):
    """Non-persistent allreduce >= ZCOPY_ONESHOT_MIN (16 MiB): the gated
    pipeline runs its zero-copy exchange per post (imports served by the
    team IPC cache) and reduces straight from user buffers."""
    torch.manual_seed(33)
    n = job.n
    count = 6 * 1024 * 1024  # 24 MiB fp32 > one-shot threshold
    srcs = [torch.randn(count, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(count, device="cuda") for _ in range(n)]
    expected = sum(s.cpu() for s in srcs)
    for _ in range(2):  # second pass hits the IPC-import cache
        _run_device(job, "allreduce", [
            dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(),
                 count=count, dt=dtypes.FLOAT32,
                 mem_type=dtypes.MEM_CUDA)
            for r in range(n)
        ])
    for d in dsts:
        torch.testing.assert_close(d.cpu(), expected, rtol=1e-5,
                                   atol=1e-3 * n)
    del srcs, dsts
    torch.cuda.empty_cache()


def test_asymm_memtype_allreduce(job):
    """Host src + device dst: core stages src into a device scratch at
    post (reference ucc_coll.c:236-246 role) and the device TL runs."""
    torch.manual_seed(31)
    n = job.n
    count = 40_000
    srcs = [np.random.default_rng(r).standard_normal(count)
            .astype(np.float32) for r in range(n)]
    dsts = [torch.zeros(count, device="cuda") for _ in range(n)]
    expected = torch.from_numpy(np.sum(srcs, axis=0))
    _run_device(job, "allreduce", [
        dict(src=srcs[r].ctypes.data, dst=dsts[r].data_ptr(), count=count,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
             src_mem_type=dtypes.MEM_HOST)
        for r in range(n)
    ])
    for d in dsts:
        torch.testing.assert_close(d.cpu(), expected, rtol=1e-5,
                                   atol=1e-4)


def test_asymm_scratch_mpool_reuse(job):
    """Repeated asymmetric-staging colls must serve their device scratch
    from the mc mpool (reference mc_rocm.c:97-108), not a fresh
    hipMalloc per collective: after a warm first pass, raw allocation
    count stays flat across further passes."""
    n = job.n
    count = 30_000  # 120 KB < the 1 MiB pool elem
    srcs = [np.random.default_rng(100 + r).standard_normal(count)
            .astype(np.float32) for r in range(n)]
    dsts = [torch.zeros(count, device="cuda") for _ in range(n)]
    expected = torch.from_numpy(np.sum(srcs, axis=0))

    def one_pass():
        _run_device(job, "allreduce", [
            dict(src=srcs[r].ctypes.data, dst=dsts[r].data_ptr(),
                 count=count, dt=dtypes.FLOAT32,
                 mem_type=dtypes.MEM_CUDA,
                 src_mem_type=dtypes.MEM_HOST)
            for r in range(n)
        ])

    one_pass()  # warm the pool
    before = core().scratch_raw_allocs()
    for _ in range(4):
        one_pass()
    after = core().scratch_raw_allocs()
    assert after == before, (
        f"scratch mpool miss: {after - before} raw allocs in 4 warm "
        "passes")
    for d in dsts:
        torch.testing.assert_close(d.cpu(), expected, rtol=1e-5,
                                   atol=1e-4)


def test_asymm_memtype_gather_root(job):
    """Rooted asymm: the root's src is host while everyone's data side is
    device — only the root stages, keeping TL choice consistent."""
    torch.manual_seed(32)
    n, per, root = job.n, 5000, 1
    srcs_dev = [torch.randn(per, device="cuda") for _ in range(n)]
    src_host = srcs_dev[root].cpu().numpy()
    dst = torch.zeros(per * n, device="cuda")
    args = []
    for r in range(n):
        if r == root:
            args.append(dict(src=src_host.ctypes.data,
                             dst=dst.data_ptr(), count=per * n,
                             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
                             src_mem_type=dtypes.MEM_HOST, root=root))
        else:
            args.append(dict(src=srcs_dev[r].data_ptr(), dst=0,
                             count=per, dt=dtypes.FLOAT32,
                             mem_type=dtypes.MEM_CUDA, root=root))
    _run_device(job, "gather", args)
    expected = torch.cat([s.cpu() for s in srcs_dev])
    torch.testing.assert_close(dst.cpu(), expected)


def test_alltoallv_device_multifrag(job):
    """Gated a2av crossing the staging-cell boundary: every pair moves
    more than one cell (chunk/n bytes), so the global-max exchange and
    the multi-fragment pipeline both run (>1 fragment per pair)."""
    torch.manual_seed(29)
    n = job.n
    cell_elems = ((32 * 1024 * 1024 // n) & ~255) // 4
    scnt = [[cell_elems + (d * 7919 + r * 101) % 50_000
             for d in range(n)] for r in range(n)]
    rcnt = [[scnt[s][r] for s in range(n)] for r in range(n)]
    sdsp = [np.concatenate([[0], np.cumsum(c)[:-1]]).tolist() for c in scnt]
    rdsp = [np.concatenate([[0], np.cumsum(c)[:-1]]).tolist() for c in rcnt]
    srcs = [torch.randn(sum(scnt[r]), device="cuda") for r in range(n)]
    dsts = [torch.zeros(sum(rcnt[r]), device="cuda") for r in range(n)]
    _run_device(job, "alltoallv", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=0,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
             src_counts=scnt[r], src_displs=sdsp[r],
             dst_counts=rcnt[r], dst_displs=rdsp[r])
        for r in range(n)
    ])
    for r in range(n):
        for s in range(n):
            got = dsts[r].cpu()[rdsp[r][s]:rdsp[r][s] + rcnt[r][s]]
            exp = srcs[s].cpu()[sdsp[s][r]:sdsp[s][r] + scnt[s][r]]
            torch.testing.assert_close(got, exp)
    del srcs, dsts
    torch.cuda.empty_cache()


def test_allgatherv_device(job):
    torch.manual_seed(17)
    n = job.n
    cnts = [(r + 1) * 1001 for r in range(n)]
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).tolist()
    total = sum(cnts)
    srcs = [torch.randn(cnts[r], device="cuda") for r in range(n)]
    dsts = [torch.zeros(total, device="cuda") for _ in range(n)]
    expected = torch.cat([s.cpu() for s in srcs])
    _run_device(job, "allgatherv", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=cnts[r],
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
             dst_counts=cnts, dst_displs=dsps)
        for r in range(n)
    ])
    for d in dsts:
        torch.testing.assert_close(d.cpu(), expected)


def test_reduce_scatterv_device(job):
    torch.manual_seed(18)
    n = job.n
    cnts = [(r + 1) * 777 for r in range(n)]
    total = sum(cnts)
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).tolist()
    srcs = [torch.randn(total, device="cuda") for _ in range(n)]
    dsts = [torch.zeros(cnts[r], device="cuda") for r in range(n)]
    expected = sum(s.cpu() for s in srcs)
    _run_device(job, "reduce_scatterv", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=0,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
             dst_counts=cnts)
        for r in range(n)
    ])
    for r in range(n):
        torch.testing.assert_close(
            dsts[r].cpu(), expected[dsps[r]:dsps[r] + cnts[r]],
            rtol=1e-5, atol=1e-5)


def test_gather_scatter_device(job):
    torch.manual_seed(19)
    n = job.n
    per = 123_000
    root = 0
    srcs = [torch.randn(per, device="cuda") for _ in range(n)]
    gdst = [torch.zeros(per * n, device="cuda") if r == root
            else torch.zeros(1, device="cuda") for r in range(n)]
    _run_device(job, "gather", [
        dict(src=srcs[r].data_ptr(), dst=gdst[r].data_ptr(),
             count=per * n if r == root else per,
             dt=dtypes.FLOAT32, root=root, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    expected = torch.cat([s.cpu() for s in srcs])
    torch.testing.assert_close(gdst[root].cpu(), expected)

    sdst = [torch.zeros(per, device="cuda") for _ in range(n)]
    big = torch.randn(per * n, device="cuda")
    _run_device(job, "scatter", [
        dict(src=big.data_ptr() if r == root else 0,
             dst=sdst[r].data_ptr(),
             count=per * n if r == root else per,
             dt=dtypes.FLOAT32, root=root, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    for r in range(n):
        torch.testing.assert_close(sdst[r].cpu(),
                                   big.cpu()[r * per:(r + 1) * per])


def test_allreduce_device_complex(job):
    """complex64 SUM: elementwise float pairs (reference ec complex
    reduce role, SUM/AVG subset on device)."""
    torch.manual_seed(20)
    n = job.n
    count = 4096
    srcs = [torch.randn(count, dtype=torch.complex64, device="cuda")
            for _ in range(n)]
    dsts = [torch.zeros(count, dtype=torch.complex64, device="cuda")
            for _ in range(n)]
    expected = sum(s.cpu() for s in srcs)
    _run_device(job, "allreduce", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=count,
             dt=dtypes.FLOAT32_COMPLEX, mem_type=dtypes.MEM_CUDA)
        for r in range(n)
    ])
    for d in dsts:
        torch.testing.assert_close(d.cpu(), expected, rtol=1e-5, atol=1e-4)


def test_allgatherv_device_zero_count(job):
    torch.manual_seed(21)
    n = job.n
    cnts = [0 if r == 1 % n else (r + 1) * 500 for r in range(n)]
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).tolist()
    total = int(sum(cnts))
    srcs = [torch.randn(max(cnts[r], 1), device="cuda") for r in range(n)]
    dsts = [torch.zeros(max(total, 1), device="cuda") for _ in range(n)]
    _run_device(job, "allgatherv", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(),
             count=cnts[r], dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
             dst_counts=cnts, dst_displs=dsps)
        for r in range(n)
    ])
    expected = torch.cat([srcs[r].cpu()[:cnts[r]] for r in range(n)])
    for d in dsts:
        torch.testing.assert_close(d.cpu()[:total], expected)


def test_alltoallv_device_zero_row(job):
    torch.manual_seed(22)
    n = job.n
    scnt = [[0] * n if r == 0 else [(r) * (d + 1) * 300 for d in range(n)]
            for r in range(n)]
    rcnt = [[scnt[s][r] for s in range(n)] for r in range(n)]
    sdsp = [np.concatenate([[0], np.cumsum(c)[:-1]]).tolist()
            for c in scnt]
    rdsp = [np.concatenate([[0], np.cumsum(c)[:-1]]).tolist()
            for c in rcnt]
    srcs = [torch.randn(max(sum(scnt[r]), 1), device="cuda")
            for r in range(n)]
    dsts = [torch.zeros(max(sum(rcnt[r]), 1), device="cuda")
            for r in range(n)]
    _run_device(job, "alltoallv", [
        dict(src=srcs[r].data_ptr(), dst=dsts[r].data_ptr(), count=0,
             dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
             src_counts=scnt[r], src_displs=sdsp[r],
             dst_counts=rcnt[r], dst_displs=rdsp[r])
        for r in range(n)
    ])
    for r in range(n):
        for s in range(n):
            got = dsts[r].cpu()[rdsp[r][s]:rdsp[r][s] + rcnt[r][s]]
            exp = srcs[s].cpu()[sdsp[s][r]:sdsp[s][r] + scnt[s][r]]
            torch.testing.assert_close(got, exp)


def test_gatherv_scatterv_device(job):
    torch.manual_seed(23)
    n, root = job.n, 0
    cnts = [(r + 2) * 4321 for r in range(n)]
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).tolist()
    total = int(sum(cnts))
    srcs = [torch.randn(cnts[r], device="cuda") for r in range(n)]
    gdst = torch.zeros(total, device="cuda")
    _run_device(job, "gatherv", [
        dict(src=srcs[r].data_ptr(),
             dst=gdst.data_ptr() if r == root else 0,
             count=cnts[r], dt=dtypes.FLOAT32, root=root,
             mem_type=dtypes.MEM_CUDA,
             dst_counts=cnts, dst_displs=dsps)
        for r in range(n)
    ])
    expected = torch.cat([s.cpu() for s in srcs])
    torch.testing.assert_close(gdst.cpu(), expected)

    sdsts = [torch.zeros(cnts[r], device="cuda") for r in range(n)]
    _run_device(job, "scatterv", [
        dict(src=gdst.data_ptr() if r == root else 0,
             dst=sdsts[r].data_ptr(), count=cnts[r], dt=dtypes.FLOAT32,
             root=root, mem_type=dtypes.MEM_CUDA,
             src_counts=cnts, src_displs=dsps)
        for r in range(n)
    ])
    for r in range(n):
        torch.testing.assert_close(sdsts[r].cpu(), srcs[r].cpu())


def test_gatherv_device_rootonly_vargs(job):
    """v-args significant at root only (device staged path)."""
    torch.manual_seed(24)
    n, root = job.n, 0
    cnts = [(r + 1) * 777 for r in range(n)]
    dsps = np.concatenate([[0], np.cumsum(cnts)[:-1]]).tolist()
    total = int(sum(cnts))
    srcs = [torch.randn(cnts[r], device="cuda") for r in range(n)]
    gdst = torch.zeros(total, device="cuda")
    _run_device(job, "gatherv", [
        dict(src=srcs[r].data_ptr(),
             dst=gdst.data_ptr() if r == root else 0,
             count=cnts[r], dt=dtypes.FLOAT32, root=root,
             mem_type=dtypes.MEM_CUDA,
             **({"dst_counts": cnts, "dst_displs": dsps}
                if r == root else {}))
        for r in range(n)
    ])
    torch.testing.assert_close(gdst.cpu(),
                               torch.cat([s.cpu() for s in srcs]))


def test_device_size1_team():
    """Size-1 team on device memory: the self TL serves every coll as a
    device copy/no-op (reference tl/self role for n=1 GPU jobs)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.cuda.set_device(0)
    j1 = LocalJob(1)
    smap = core().score_map_str(j1.teams[0])
    assert "@self" in smap, smap
    src = torch.randn(4096, device="cuda")
    dst = torch.zeros(4096, device="cuda")
    for coll in ("allreduce", "allgather", "alltoall",
                 "reduce_scatter"):
        dst.zero_()
        _run_device(j1, coll, [
            dict(src=src.data_ptr(), dst=dst.data_ptr(), count=4096,
                 dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)])
        torch.testing.assert_close(dst, src)
    # bcast: in-place single rank, must be a no-op that completes
    _run_device(j1, "bcast", [
        dict(src=src.data_ptr(), dst=0, count=4096, dt=dtypes.FLOAT32,
             mem_type=dtypes.MEM_CUDA)])
    # barrier
    _run_device(j1, "barrier", [
        dict(src=0, dst=0, count=0, dt=dtypes.FLOAT32)])
