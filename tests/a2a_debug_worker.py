"""Focused repro for the r02 a2a-zc regression: the 3a2 sequence
(reduce_scatter, allgather, alltoall) x2, cross-process. On mismatch,
re-reads the dst after a delay to distinguish an early read (data fixes
itself once the peer's copy lands) from a stale IPC mapping (stays
garbage). Run under torchrun with 2 ranks.
"""
import os
import sys
import time

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
            raise TimeoutError("stuck")


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(int(os.environ["LOCAL_RANK"])
                          % torch.cuda.device_count())
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
        torch.testing.assert_close(dst.cpu(),
                                   exp[rank * per:(rank + 1) * per],
                                   rtol=1e-5, atol=1e-4)
        print(f"[{rank}] it{it} rs ok", flush=True)

        blk = torch.randn(per, generator=g0)
        src = blk.cuda()
        agd = torch.zeros(per * world, device="cuda")
        ag = c.coll_init(team, "allgather", src=src.data_ptr(),
                         dst=agd.data_ptr(), count=per * world,
                         dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        src.copy_(blk + rank)
        torch.cuda.synchronize()
        wait(ag, ctx)
        torch.cuda.synchronize()
        exp = torch.cat([blk + r for r in range(world)])
        torch.testing.assert_close(agd.cpu(), exp)
        print(f"[{rank}] it{it} ag ok", flush=True)

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
        got = a2d.cpu()
        ok = torch.allclose(got, exp)
        print(f"[{rank}] it{it} a2a ok={ok}", flush=True)
        if not ok:
            bad = (got - exp).abs() > 1e-4
            idx = bad.nonzero()[:4].flatten().tolist()
            print(f"[{rank}] first bad idx {idx}; "
                  f"got {[round(float(got[i]),4) for i in idx]} "
                  f"exp {[round(float(exp[i]),4) for i in idx]}",
                  flush=True)
            nbad = int(bad.sum())
            time.sleep(1.0)
            torch.cuda.synchronize()
            got2 = a2d.cpu()
            nbad2 = int(((got2 - exp).abs() > 1e-4).sum())
            print(f"[{rank}] it{it} bad={nbad} after-resync bad={nbad2} "
                  f"(same->stale mapping or wrong source; fewer->early "
                  "read)", flush=True)
            # is the garbage actually the PREVIOUS tensor at that VA?
            prev = torch.cat(
                [a2s[s][rank * per:(rank + 1) * per]
                 for s in range(world)])
            del prev
            sys.exit(1)
    print(f"A2ADBG_OK {rank}", flush=True)


if __name__ == "__main__":
    main()
