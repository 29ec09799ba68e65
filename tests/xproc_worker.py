"""Cross-process device-path worker: run under
  python -m torch.distributed.run --nnodes=1 --nproc-per-node=2 \
      --master-addr 127.0.0.1 tests/xproc_worker.py

Both ranks share cuda:0 on a 1-GPU box (or map to their own GPU on an
8-GPU node): this exercises the production path the in-process jig cannot
— hipIpcOpenMemHandle peer mappings across processes, the cross-process
fused single-kernel allreduce, stream-triggered post (ucc_ee_create +
ucc_collective_triggered_post) and hipGraph capture/replay of the fused
kernel (BASELINE config #5).
"""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from ucc_amd import core, dtypes  # noqa: E402


def oob(group, world):
    def allgather(data: bytes):
        n = len(data)
        t = torch.frombuffer(bytearray(data), dtype=torch.uint8).clone()
        outs = [torch.empty(n, dtype=torch.uint8) for _ in range(world)]
        dist.all_gather(outs, t, group=group)
        return [o.numpy().tobytes() for o in outs]

    return allgather


def wait(req, ctx):
    req.post()
    it = 0
    while req.test() == core().INPROGRESS:
        ctx.progress()
        it += 1
        if it > 200_000_000:
            raise TimeoutError("collective stuck")


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    local = int(os.environ["LOCAL_RANK"])
    torch.cuda.set_device(local % torch.cuda.device_count())
    dist.init_process_group("gloo", rank=rank, world_size=world)
    c = core()
    lib = c.Lib()
    ctx = c.Context(lib)
    team = c.team_create_post(ctx, py_allgather=oob(dist.group.WORLD, world),
                              rank=rank, n_ranks=world)
    while True:
        st = c.team_create_test(team)
        if st == c.OK:
            break
        if st < 0:
            raise RuntimeError(f"team create failed: {st}")

    g0 = torch.Generator(device="cpu").manual_seed(1234)
    results = []
    only_basic = os.environ.get("XPROC_BASIC_ONLY", "0") == "1"

    # 1. small fused cross-process allreduce (bf16)
    count = 8192
    full = torch.randn(world, count, generator=g0)
    src = full[rank].to(torch.bfloat16).cuda()
    dst = torch.zeros(count, dtype=torch.bfloat16, device="cuda")
    expected = sum(full[r].to(torch.bfloat16).float() for r in range(world))
    r1 = c.coll_init(team, "allreduce", src=src.data_ptr(),
                     dst=dst.data_ptr(), count=count, dt=dtypes.BFLOAT16,
                     mem_type=dtypes.MEM_CUDA)
    wait(r1, ctx)
    torch.cuda.synchronize()
    torch.testing.assert_close(dst.cpu().float(), expected, rtol=2e-2,
                               atol=2e-1)
    results.append("fused_xproc")

    # 2. large cross-process allreduce (fp32): the device-gated pipeline
    # (zero host round-trips), repeated to exercise parity/counter
    # continuity across collectives; persistent re-post included.
    count = 20_000_000
    full = torch.randn(world, count, generator=g0)
    src = full[rank].cuda()
    dst = torch.zeros(count, device="cuda")
    r2 = c.coll_init(team, "allreduce", src=src.data_ptr(),
                     dst=dst.data_ptr(), count=count, dt=dtypes.FLOAT32,
                     mem_type=dtypes.MEM_CUDA, flags=c.FLAG_PERSISTENT)
    for it in range(3):
        src.copy_(full[rank] + it)
        torch.cuda.synchronize()
        expected = (full + it).sum(0)
        wait(r2, ctx)
        torch.cuda.synchronize()
        torch.testing.assert_close(dst.cpu(), expected, rtol=1e-5,
                                   atol=1e-4)
    results.append("gated_xproc")

    # 2b. one-shot (non-persistent) large allreduce: takes the
    # zero-copy path per post (>= ZCOPY_ONESHOT_MIN); the second call
    # re-imports peers' buffers from the team IPC cache.
    for it in (7, 8):
        src.copy_(full[rank] + it)
        torch.cuda.synchronize()
        expected = (full + it).sum(0)
        r2b = c.coll_init(team, "allreduce", src=src.data_ptr(),
                          dst=dst.data_ptr(), count=count,
                          dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        wait(r2b, ctx)
        torch.cuda.synchronize()
        torch.testing.assert_close(dst.cpu(), expected, rtol=1e-5,
                                   atol=1e-4)
        del r2b
    results.append("oneshot_zc_xproc")

    # 2c. exporter-side free + realloc (reference tl_cuda_cache.c
    # invalidate-on-overlap role): empty_cache() releases the backing
    # allocations, and the next allocations may land on the same VA or
    # recycle the IPC handle identity — the team import cache must
    # re-validate instead of serving a stale mapping (r02 regression:
    # garbage peer blocks in the first zero-copy coll after realloc).
    del r2, src, dst
    torch.cuda.empty_cache()
    count2 = 18_000_000
    full2 = torch.randn(world, count2, generator=g0)
    src = full2[rank].cuda()
    dst = torch.zeros(count2, device="cuda")
    r2c = c.coll_init(team, "allreduce", src=src.data_ptr(),
                      dst=dst.data_ptr(), count=count2,
                      dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
    wait(r2c, ctx)
    torch.cuda.synchronize()
    torch.testing.assert_close(dst.cpu(), full2.sum(0), rtol=1e-5,
                               atol=1e-4)
    del r2c
    results.append("zc_realloc_xproc")

    # 2d. registered-buffer (onesided-role) collective: src/dst memh
    # from ucc_mem_map are honored — the zc exchange consumes the
    # pre-exported handles instead of per-post hipIpcGetMemHandle
    # (reference tl/ucp alltoall_onesided src_memh/dst_memh role).
    count3 = 12_000_000
    full3 = torch.randn(world, count3, generator=g0)
    src3 = full3[rank].cuda()
    dst3 = torch.zeros(count3, device="cuda")
    sh = c.mem_map_export_keep(src3.data_ptr(), count3 * 4)
    dh = c.mem_map_export_keep(dst3.data_ptr(), count3 * 4)
    uses0 = c.cdna4_memh_uses()
    r2d = c.coll_init(team, "allreduce", src=src3.data_ptr(),
                      dst=dst3.data_ptr(), count=count3,
                      dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
                      flags=c.FLAG_PERSISTENT | c.FLAG_MEM_MAPPED,
                      src_memh=sh, dst_memh=dh)
    for it in range(2):
        src3.copy_(full3[rank] * (it + 1))
        torch.cuda.synchronize()
        wait(r2d, ctx)
        torch.cuda.synchronize()
        torch.testing.assert_close(dst3.cpu(), full3.sum(0) * (it + 1),
                                   rtol=1e-5, atol=1e-4)
    assert c.cdna4_memh_uses() > uses0, \
        "registered handles were not consumed by the zc exchange"
    del r2d
    c.mem_unmap_handle(sh)
    c.mem_unmap_handle(dh)
    results.append("memh_onesided_xproc")

    # 3. cross-process alltoallv (skewed, fp16)
    scnt = [[(r + 1) * (d + 1) * 1024 for d in range(world)]
            for r in range(world)]
    rcnt = [[scnt[s][r] for s in range(world)] for r in range(world)]
    def cumsum0(v):
        out, t = [], 0
        for x in v:
            out.append(t)
            t += x
        return out
    sdsp = [cumsum0(x) for x in scnt]
    rdsp = [cumsum0(x) for x in rcnt]
    sfull = [torch.randn(sum(scnt[r]), generator=g0).to(torch.float16)
             for r in range(world)]
    src = sfull[rank].cuda()
    dst = torch.zeros(sum(rcnt[rank]), dtype=torch.float16, device="cuda")
    r3 = c.coll_init(team, "alltoallv", src=src.data_ptr(),
                     dst=dst.data_ptr(), count=0, dt=dtypes.FLOAT16,
                     mem_type=dtypes.MEM_CUDA,
                     src_counts=scnt[rank], src_displs=sdsp[rank],
                     dst_counts=rcnt[rank], dst_displs=rdsp[rank])
    wait(r3, ctx)
    torch.cuda.synchronize()
    for s in range(world):
        got = dst.cpu()[rdsp[rank][s]:rdsp[rank][s] + rcnt[rank][s]]
        exp = sfull[s][sdsp[s][rank]:sdsp[s][rank] + scnt[s][rank]]
        torch.testing.assert_close(got, exp)
    results.append("alltoallv_xproc")

    # 3c. multi-fragment gated alltoallv: per-pair length crosses the
    # staging cell (chunk/world), so the device-gated a2av runs its
    # host global-max exchange AND >1 pipelined fragment per pair.
    cell_elems = ((32 * 1024 * 1024 // world) & ~255) // 4
    mf_scnt = [[cell_elems + 30_000 + 1000 * (r + d)
                for d in range(world)] for r in range(world)]
    mf_rcnt = [mf_scnt[s][rank] for s in range(world)]
    mf_sd, off = [], 0
    for cq in mf_scnt[rank]:
        mf_sd.append(off)
        off += cq
    mf_rd, roff = [], 0
    for cq in mf_rcnt:
        mf_rd.append(roff)
        roff += cq
    g3 = torch.Generator().manual_seed(777)
    mf_full = [torch.randn(sum(mf_scnt[r]), generator=g3)
               for r in range(world)]
    mf_src = mf_full[rank].cuda()
    mf_dst = torch.zeros(roff, device="cuda")
    r3c = c.coll_init(team, "alltoallv", src=mf_src.data_ptr(),
                      dst=mf_dst.data_ptr(), count=0, dt=dtypes.FLOAT32,
                      mem_type=dtypes.MEM_CUDA,
                      src_counts=mf_scnt[rank], src_displs=mf_sd,
                      dst_counts=mf_rcnt, dst_displs=mf_rd)
    wait(r3c, ctx)
    torch.cuda.synchronize()
    for s in range(world):
        got = mf_dst.cpu()[mf_rd[s]:mf_rd[s] + mf_rcnt[s]]
        sod = sum(mf_scnt[s][:rank])
        exp = mf_full[s][sod:sod + mf_rcnt[s]]
        torch.testing.assert_close(got, exp)
    del mf_src, mf_dst, mf_full
    torch.cuda.empty_cache()
    results.append("a2av_multifrag_gated")

    # 3a2. gated reduce_scatter / allgather / alltoall (device-gated
    # pipeline paths, cross-process only), repeated for counter
    # continuity across mixed coll types on shared slots.
    per = 5_000_000
    for it in range(2):
        full = torch.randn(world, per * world, generator=g0)
        src = full[rank].cuda()
        dst = torch.zeros(per, device="cuda")
        rs = c.coll_init(team, "reduce_scatter", src=src.data_ptr(),
                         dst=dst.data_ptr(), count=per, dt=dtypes.FLOAT32,
                         mem_type=dtypes.MEM_CUDA)
        wait(rs, ctx)
        torch.cuda.synchronize()
        exp = full.sum(0)
        torch.testing.assert_close(
            dst.cpu(), exp[rank * per:(rank + 1) * per], rtol=1e-5,
            atol=1e-4)

        blk = torch.randn(per, generator=g0)
        src = blk.cuda()
        agd = torch.zeros(per * world, device="cuda")
        ag = c.coll_init(team, "allgather", src=src.data_ptr(),
                         dst=agd.data_ptr(), count=per * world,
                         dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        # every rank must contribute its own block: use rank-dependent data
        src.copy_(blk + rank)
        torch.cuda.synchronize()
        wait(ag, ctx)
        torch.cuda.synchronize()
        exp = torch.cat([blk + r for r in range(world)])
        torch.testing.assert_close(agd.cpu(), exp)

        a2s = torch.randn(world, per * world, generator=g0)
        src = a2s[rank].cuda()
        a2d = torch.zeros(per * world, device="cuda")
        a2 = c.coll_init(team, "alltoall", src=src.data_ptr(),
                         dst=a2d.data_ptr(), count=per * world,
                         dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        wait(a2, ctx)
        torch.cuda.synchronize()
        exp = torch.cat(
            [a2s[s][rank * per:(rank + 1) * per] for s in range(world)])
        torch.testing.assert_close(a2d.cpu(), exp)
        # v-variants through the gated pipeline (globally-known counts)
        vcnts = [(r + 1) * 3_000_000 for r in range(world)]
        vd = []
        t0 = 0
        for x in vcnts:
            vd.append(t0)
            t0 += x
        vtotal = sum(vcnts)
        # per-rank blocks from a local generator (must NOT consume g0:
        # its stream is shared across ranks and vcnts differ per rank)
        gsrcs = []
        gg = torch.Generator().manual_seed(4242 + it)
        for r in range(world):
            gsrcs.append(torch.randn(vcnts[r], generator=gg))
        src = gsrcs[rank].cuda()
        agd = torch.zeros(vtotal, device="cuda")
        agv = c.coll_init(team, "allgatherv", src=src.data_ptr(),
                          dst=agd.data_ptr(), count=vcnts[rank],
                          dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
                          dst_counts=vcnts, dst_displs=vd)
        wait(agv, ctx)
        torch.cuda.synchronize()
        torch.testing.assert_close(agd.cpu(), torch.cat(gsrcs))

        rsrcs = []
        gg = torch.Generator().manual_seed(5252 + it)
        for r in range(world):
            rsrcs.append(torch.randn(vtotal, generator=gg))
        rssrc = rsrcs[rank].cuda()
        rsd = torch.zeros(vcnts[rank], device="cuda")
        rsv = c.coll_init(team, "reduce_scatterv", src=rssrc.data_ptr(),
                          dst=rsd.data_ptr(), count=0, dt=dtypes.FLOAT32,
                          mem_type=dtypes.MEM_CUDA, dst_counts=vcnts)
        wait(rsv, ctx)
        torch.cuda.synchronize()
        exp = sum(rsrcs)
        torch.testing.assert_close(
            rsd.cpu(), exp[vd[rank]:vd[rank] + vcnts[rank]], rtol=1e-5,
            atol=1e-4)
    # 3a3. zero-copy persistent reduce_scatter / allgather (2 posts:
    # exchange on the first, direct peer reads after)
    per = 4_000_000
    zfull = torch.randn(world, per * world, generator=g0)
    zsrc = zfull[rank].cuda()
    zdst = torch.zeros(per, device="cuda")
    zrs = c.coll_init(team, "reduce_scatter", src=zsrc.data_ptr(),
                      dst=zdst.data_ptr(), count=per, dt=dtypes.FLOAT32,
                      mem_type=dtypes.MEM_CUDA, flags=c.FLAG_PERSISTENT)
    for it in range(2):
        zsrc.copy_(zfull[rank] * (it + 1))
        torch.cuda.synchronize()
        wait(zrs, ctx)
        torch.cuda.synchronize()
        exp = zfull.sum(0) * (it + 1)
        torch.testing.assert_close(
            zdst.cpu(), exp[rank * per:(rank + 1) * per], rtol=1e-5,
            atol=1e-4)

    zgsrc = torch.randn(per, generator=g0)
    asrc = (zgsrc + rank).cuda()
    adst = torch.zeros(per * world, device="cuda")
    zag = c.coll_init(team, "allgather", src=asrc.data_ptr(),
                      dst=adst.data_ptr(), count=per * world,
                      dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
                      flags=c.FLAG_PERSISTENT)
    for it in range(2):
        asrc.copy_(zgsrc + rank + it)
        torch.cuda.synchronize()
        wait(zag, ctx)
        torch.cuda.synchronize()
        exp = torch.cat([zgsrc + r + it for r in range(world)])
        torch.testing.assert_close(adst.cpu(), exp)
    # persistent alltoall through the zero-copy path
    za2s = torch.randn(world, per * world, generator=g0)
    zsrc2 = za2s[rank].cuda()
    zdst2 = torch.zeros(per * world, device="cuda")
    za2 = c.coll_init(team, "alltoall", src=zsrc2.data_ptr(),
                      dst=zdst2.data_ptr(), count=per * world,
                      dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA,
                      flags=c.FLAG_PERSISTENT)
    for it in range(2):
        zsrc2.copy_(za2s[rank] * (it + 1))
        torch.cuda.synchronize()
        wait(za2, ctx)
        torch.cuda.synchronize()
        exp = torch.cat([za2s[s2][rank * per:(rank + 1) * per] * (it + 1)
                         for s2 in range(world)])
        torch.testing.assert_close(zdst2.cpu(), exp)
    results.append("gated_rs_ag_a2a_v_zc")

    # 3b. ucc_mem_map export/import across processes: rank 0 exports a
    # device buffer, rank 1 imports and reads it over IPC.
    if world == 2:
        if rank == 0:
            payload = torch.arange(1024, dtype=torch.float32, device="cuda")
            blob = c.mem_map_export(payload.data_ptr(), payload.numel() * 4)
            dist.broadcast_object_list([bytes(blob)], src=0)
            torch.cuda.synchronize()
            dist.barrier()
        else:
            holder = [None]
            dist.broadcast_object_list(holder, src=0)
            mapped = c.mem_map_import(holder[0])
            got = torch.empty(1024, dtype=torch.float32, device="cuda")
            c.hip_memcpy_d2d(got.data_ptr(), mapped, 1024 * 4)
            torch.cuda.synchronize()
            torch.testing.assert_close(
                got.cpu(), torch.arange(1024, dtype=torch.float32))
            c.mem_map_close(mapped)
            dist.barrier()
        results.append("mem_map")

    if only_basic:
        dist.barrier()
        print(f"XPROC_OK rank={rank} {'+'.join(results)}", flush=True)
        dist.destroy_process_group()
        return

    # 4. triggered post on a user stream (persistent, re-triggered)
    count = 65536
    full = torch.randn(world, count, generator=g0)
    src = full[rank].cuda()
    dst = torch.zeros(count, device="cuda")
    expected = full.sum(0)
    rp = c.coll_init(team, "allreduce", src=src.data_ptr(),
                     dst=dst.data_ptr(), count=count, dt=dtypes.FLOAT32,
                     mem_type=dtypes.MEM_CUDA, flags=c.FLAG_PERSISTENT)
    s = torch.cuda.Stream()
    ee = c.ee_create(team, s.cuda_stream)
    for it in range(3):
        dst.zero_()
        torch.cuda.synchronize()
        dist.barrier()
        c.triggered_post(ee, rp)
        s.synchronize()
        torch.testing.assert_close(dst.cpu(), expected, rtol=1e-5,
                                   atol=1e-4)
        # EE event-queue flow (reference ucc.h:2050-2260): a POST
        # event at launch and a COLLECTIVE_COMPLETE once the stream
        # work finished (the stream is synchronized here, so both
        # must be available).
        evs = []
        while True:
            t = c.ee_pop_event(ee)
            if t < 0:
                break
            evs.append(t)
        assert c.EVENT_COLLECTIVE_POST in evs, evs
        assert c.EVENT_COLLECTIVE_COMPLETE in evs, evs
    results.append("triggered")

    # 5. hipGraph capture + replay of the triggered fused allreduce
    g = torch.cuda.CUDAGraph()
    dist.barrier()
    with torch.cuda.graph(g, stream=s):
        c.triggered_post(ee, rp)
    dist.barrier()
    for it in range(5):
        src.copy_(full[rank] + it)
        dst.zero_()
        torch.cuda.synchronize()
        dist.barrier()
        g.replay()
        torch.cuda.synchronize()
        exp = (full + it).sum(0)
        torch.testing.assert_close(dst.cpu(), exp, rtol=1e-5, atol=1e-4)
    results.append("hipgraph_replay")

    # 5b. hipGraph capture + replay of a LARGE (gated pipeline) persistent
    # allreduce: kernels derive their iteration on device
    count = 12_000_000  # ~48 MB fp32 -> multi-fragment gated path
    gfull = torch.randn(world, count, generator=g0)
    gsrc = gfull[rank].cuda()
    gdst = torch.zeros(count, device="cuda")
    rg = c.coll_init(team, "allreduce", src=gsrc.data_ptr(),
                     dst=gdst.data_ptr(), count=count, dt=dtypes.FLOAT32,
                     mem_type=dtypes.MEM_CUDA, flags=c.FLAG_PERSISTENT)
    # warm (non-captured triggered posts)
    for it in range(2):
        torch.cuda.synchronize()
        dist.barrier()
        c.triggered_post(ee, rg)
        s.synchronize()
        torch.testing.assert_close(gdst.cpu(), gfull.sum(0), rtol=1e-5,
                                   atol=1e-4)
    g2 = torch.cuda.CUDAGraph()
    dist.barrier()
    with torch.cuda.graph(g2, stream=s):
        c.triggered_post(ee, rg)
    dist.barrier()
    for it in range(3):
        gsrc.copy_(gfull[rank] * (it + 1))
        gdst.zero_()
        torch.cuda.synchronize()
        dist.barrier()
        g2.replay()
        torch.cuda.synchronize()
        torch.testing.assert_close(gdst.cpu(), gfull.sum(0) * (it + 1),
                                   rtol=1e-5, atol=1e-3)
    results.append("hipgraph_gated")

    c.ee_destroy(ee)
    dist.barrier()
    print(f"XPROC_OK rank={rank} {'+'.join(results)}", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
