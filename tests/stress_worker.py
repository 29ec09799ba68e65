"""Cross-process mixed-collective stress: random sequence of device
collectives (fused / gated / staged paths, varying sizes and dtypes)
with deterministic per-iteration seeds on every rank — shakes out
slot-rotation, parity, and counter-continuity bugs that single-coll
tests cannot. Run under torchrun like xproc_worker.py:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node=2 \
      --master-addr 127.0.0.1 tests/stress_worker.py [iters]
"""

import os
import random
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
        if it > 500_000_000:
            raise TimeoutError("collective stuck")


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 60
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
    dist.barrier()

    rng = random.Random(42)  # same sequence on every rank
    tdts = [(torch.float32, dtypes.FLOAT32, 1e-5, 1e-4),
            (torch.bfloat16, dtypes.BFLOAT16, 3e-2, 5e-1),
            (torch.float16, dtypes.FLOAT16, 2e-2, 3e-1)]
    colls = ["allreduce", "reduce_scatter", "allgather", "alltoall",
             "alltoallv", "bcast", "reduce"]
    for it in range(iters):
        coll = rng.choice(colls)
        tdt, dt, rtol, atol = rng.choice(tdts)
        per = rng.choice([64, 4096, 100_000, 1_000_000, 9_000_000])
        g = torch.Generator().manual_seed(10_000 + it)
        if coll == "allreduce":
            full = torch.randn(world, per, generator=g)
            src = full[rank].to(tdt).cuda()
            dst = torch.zeros(per, dtype=tdt, device="cuda")
            req = c.coll_init(team, coll, src=src.data_ptr(),
                              dst=dst.data_ptr(), count=per, dt=dt,
                              mem_type=dtypes.MEM_CUDA)
            wait(req, ctx)
            torch.cuda.synchronize()
            exp = sum(full[r].to(tdt).float() for r in range(world))
            torch.testing.assert_close(dst.cpu().float(), exp, rtol=rtol,
                                       atol=atol * world)
        elif coll == "reduce_scatter":
            full = torch.randn(world, per * world, generator=g)
            src = full[rank].to(tdt).cuda()
            dst = torch.zeros(per, dtype=tdt, device="cuda")
            req = c.coll_init(team, coll, src=src.data_ptr(),
                              dst=dst.data_ptr(), count=per, dt=dt,
                              mem_type=dtypes.MEM_CUDA)
            wait(req, ctx)
            torch.cuda.synchronize()
            exp = sum(full[r].to(tdt).float() for r in range(world))
            torch.testing.assert_close(
                dst.cpu().float(), exp[rank * per:(rank + 1) * per],
                rtol=rtol, atol=atol * world)
        elif coll == "allgather":
            full = torch.randn(world, per, generator=g)
            src = full[rank].to(tdt).cuda()
            dst = torch.zeros(per * world, dtype=tdt, device="cuda")
            req = c.coll_init(team, coll, src=src.data_ptr(),
                              dst=dst.data_ptr(), count=per * world,
                              dt=dt, mem_type=dtypes.MEM_CUDA)
            wait(req, ctx)
            torch.cuda.synchronize()
            exp = torch.cat([full[r].to(tdt) for r in range(world)])
            torch.testing.assert_close(dst.cpu(), exp)
        elif coll == "alltoall":
            full = torch.randn(world, per * world, generator=g)
            src = full[rank].to(tdt).cuda()
            dst = torch.zeros(per * world, dtype=tdt, device="cuda")
            req = c.coll_init(team, coll, src=src.data_ptr(),
                              dst=dst.data_ptr(), count=per * world,
                              dt=dt, mem_type=dtypes.MEM_CUDA)
            wait(req, ctx)
            torch.cuda.synchronize()
            exp = torch.cat([full[s].to(tdt)[rank * per:(rank + 1) * per]
                             for s in range(world)])
            torch.testing.assert_close(dst.cpu(), exp)
        elif coll == "alltoallv":
            # skewed pairs (some zero) through the gated a2av path
            base = min(per, 500_000)
            scnt = [[((r * 5 + d * 3 + it) % 4) * (base // 3)
                     for d in range(world)] for r in range(world)]
            rcnt = [scnt[s][rank] for s in range(world)]
            sd, off = [], 0
            for cq in scnt[rank]:
                sd.append(off)
                off += cq
            rd, roff = [], 0
            for cq in rcnt:
                rd.append(roff)
                roff += cq
            full = [torch.randn(sum(scnt[r]), generator=g).to(tdt)
                    for r in range(world)]
            src = full[rank].cuda() if sum(scnt[rank]) else \
                torch.zeros(1, dtype=tdt, device="cuda")
            dst = torch.zeros(max(roff, 1), dtype=tdt, device="cuda")
            req = c.coll_init(team, "alltoallv", src=src.data_ptr(),
                              dst=dst.data_ptr(), count=0, dt=dt,
                              mem_type=dtypes.MEM_CUDA,
                              src_counts=scnt[rank], src_displs=sd,
                              dst_counts=rcnt, dst_displs=rd)
            wait(req, ctx)
            torch.cuda.synchronize()
            for s in range(world):
                got = dst.cpu()[rd[s]:rd[s] + rcnt[s]]
                sod = sum(scnt[s][:rank])
                exp = full[s][sod:sod + rcnt[s]]
                torch.testing.assert_close(got, exp)
        elif coll == "bcast":
            root = it % world
            full = torch.randn(per, generator=g).to(tdt)
            buf = (full.cuda() if rank == root
                   else torch.zeros(per, dtype=tdt, device="cuda"))
            req = c.coll_init(team, coll, src=buf.data_ptr(), dst=0,
                              count=per, dt=dt, root=root,
                              mem_type=dtypes.MEM_CUDA)
            wait(req, ctx)
            torch.cuda.synchronize()
            torch.testing.assert_close(buf.cpu(), full)
        else:  # reduce
            root = it % world
            full = torch.randn(world, per, generator=g)
            src = full[rank].to(tdt).cuda()
            dst = torch.zeros(per, dtype=tdt, device="cuda")
            req = c.coll_init(team, coll, src=src.data_ptr(),
                              dst=dst.data_ptr(), count=per, dt=dt,
                              root=root, mem_type=dtypes.MEM_CUDA)
            wait(req, ctx)
            torch.cuda.synchronize()
            if rank == root:
                exp = sum(full[r].to(tdt).float() for r in range(world))
                torch.testing.assert_close(dst.cpu().float(), exp,
                                           rtol=rtol, atol=atol * world)
        if rank == 0 and (it + 1) % 20 == 0:
            print(f"stress {it + 1}/{iters} ok", flush=True)
    dist.barrier()
    print(f"STRESS_OK rank={rank} iters={iters}", flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
