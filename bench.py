#!/usr/bin/env python3
"""ucc_amd flagship benchmark: bf16 allreduce bus bandwidth + 8B latency.

Implements the BASELINE.json metric ("ucc_perftest allreduce bus-BW (GB/s)
+ 8B latency, bf16, 1/2/4/8 MI355X"):

  bus_bw = (S / t) * 2*(N-1)/N      (tools/perf/ucc_pt_coll_allreduce.cc
                                     formula; S = bytes per rank, t = max
                                     time per iteration across ranks)

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  N>1 runs under torch.distributed.run, one rank per GPU; bootstrap OOB is
  a gloo process group; the collectives run through ucc_amd's cdna4 TL over
  xGMI (NOT through RCCL).

For N==1 the collective degenerates to a device-local copy; value is then
S/t (stated in config.n1_semantics) so every N reports a number.
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from ucc_amd import core, dtypes  # noqa: E402


def oob_from_gloo(group, world):
    import torch.distributed as dist

    def allgather(data: bytes):
        n = len(data)
        t = torch.frombuffer(bytearray(data), dtype=torch.uint8).clone()
        outs = [torch.empty(n, dtype=torch.uint8) for _ in range(world)]
        dist.all_gather(outs, t, group=group)
        return [o.numpy().tobytes() for o in outs]

    return allgather


def wait(req, ctx):
    req.post()
    while req.test() == core().INPROGRESS:
        ctx.progress()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--mbytes", type=int, default=256,
                   help="allreduce message MiB per rank")
    p.add_argument("--lat-iters", type=int, default=200)
    args = p.parse_args()

    c = core()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local = int(os.environ.get("LOCAL_RANK", "0"))
    torch.cuda.set_device(local % torch.cuda.device_count())

    dist = None
    group = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group("gloo", rank=rank, world_size=world)
        group = dist.group.WORLD

    lib = c.Lib()
    ctx = c.Context(lib)
    if world > 1:
        team = c.team_create_post(
            ctx, py_allgather=oob_from_gloo(group, world), rank=rank,
            n_ranks=world)
    else:
        team = c.team_create_post(ctx)
    while True:
        st = c.team_create_test(team)
        if st == c.OK:
            break
        if st < 0:
            raise RuntimeError(f"team create failed: {st}")

    count = args.mbytes * 1024 * 1024 // 2  # bf16 elements
    S = count * 2
    src = torch.randn(count, dtype=torch.float32).to(torch.bfloat16).cuda()
    dst = torch.zeros(count, dtype=torch.bfloat16, device="cuda")
    req = c.coll_init(team, "allreduce", src=src.data_ptr(),
                      dst=dst.data_ptr(), count=count, dt=dtypes.BFLOAT16,
                      mem_type=dtypes.MEM_CUDA, flags=c.FLAG_PERSISTENT)

    def barrier():
        if dist:
            dist.barrier(group)
        torch.cuda.synchronize()

    for _ in range(args.warmup):
        wait(req, ctx)
    barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        wait(req, ctx)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    barrier()

    elapsed = t1 - t0
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX, group=group)
        elapsed = float(t.item())
    per_iter = elapsed / args.steps
    if world > 1:
        busbw = (S / per_iter) * 2 * (world - 1) / world / 1e9
    else:
        busbw = S / per_iter / 1e9

    # 8B latency (count=4 bf16)
    lsrc = torch.randn(4, dtype=torch.float32).to(torch.bfloat16).cuda()
    ldst = torch.zeros(4, dtype=torch.bfloat16, device="cuda")
    lreq = c.coll_init(team, "allreduce", src=lsrc.data_ptr(),
                       dst=ldst.data_ptr(), count=4, dt=dtypes.BFLOAT16,
                       mem_type=dtypes.MEM_CUDA, flags=c.FLAG_PERSISTENT)
    for _ in range(20):
        wait(lreq, ctx)
    barrier()
    t0 = time.perf_counter()
    for _ in range(args.lat_iters):
        wait(lreq, ctx)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    lat_us = (t1 - t0) / args.lat_iters * 1e6
    if dist:
        t = torch.tensor([lat_us], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX, group=group)
        lat_us = float(t.item())

    if rank == 0:
        out = {
            "metric": "allreduce_busbw_GBps",
            "value": round(busbw, 2),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(per_iter * 1e3, 4),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "ucc_perftest-equivalent allreduce",
                "msg_mib_per_rank": args.mbytes,
                "lat8b_us": round(lat_us, 2),
                "parallelism": f"allreduce x{world} over xGMI (tl/cdna4)",
                "n1_semantics": "N==1 value is local-copy S/t",
                "busbw_formula": "S/t * 2(N-1)/N",
            },
        }
        print(json.dumps(out))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
