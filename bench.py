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

N==1 (no torchrun): the headline value is a REAL collective — the parent
forks 2 ranks that share the one GPU (gloo rendezvous on 127.0.0.1) and
run the persistent zero-copy cdna4 allreduce cross-process, exactly the
kernels (k_gated_*/k_staged_*) that carry the multi-GPU path; the
device-local copy number (self TL) is reported alongside in config.
"""

import argparse
import json
import os
import random
import subprocess
import sys
import time

import torch

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

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


def make_team(c, ctx, rank, world, group):
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
    return team


def bench_allreduce(c, ctx, team, args, world, dist, group):
    """Timed persistent bf16 allreduce + 8B latency. Returns a dict."""
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
    return {"busbw": busbw, "ms_per_step": per_iter * 1e3,
            "lat8b_us": lat_us}


def run_rank(args):
    """One rank of a multi-rank job (torchrun, or a forked N=1 child)."""
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

    team = None
    lib = c.Lib()
    ctx = c.Context(lib)
    team = make_team(c, ctx, rank, world, group)
    r = bench_allreduce(c, ctx, team, args, world, dist, group)
    if rank == 0:
        if args.child:
            # inner result consumed by the parent process
            print("CHILD_RESULT " + json.dumps(
                {**r, "world": world}), flush=True)
        else:
            out = {
                "metric": "allreduce_busbw_GBps",
                "value": round(r["busbw"], 2),
                "unit": "GB/s",
                "n_gpus": world,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": round(r["ms_per_step"], 4),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": None,
                "dtype": "bf16",
                "data": "synthetic",
                "config": {
                    "model": "ucc_perftest-equivalent allreduce",
                    "msg_mib_per_rank": args.mbytes,
                    "lat8b_us": round(r["lat8b_us"], 2),
                    "parallelism":
                        f"allreduce x{world} over xGMI "
                        "(tl/cdna4 zero-copy gated pipeline)",
                    "busbw_formula": "S/t * 2(N-1)/N",
                },
            }
            print(json.dumps(out))
    if dist:
        dist.destroy_process_group()


def run_local_copy(args):
    """N=1 self-TL measurement (device-local copy S/t + 8B latency)."""
    c = core()
    torch.cuda.set_device(0)
    lib = c.Lib()
    ctx = c.Context(lib)
    team = make_team(c, ctx, 0, 1, None)
    la = argparse.Namespace(**vars(args))
    la.steps = max(5, min(args.steps, 20))
    la.warmup = max(3, min(args.warmup, 5))
    r = bench_allreduce(c, ctx, team, la, 1, None, None)
    return r


def run_parent_n1(args):
    """N=1 driver invocation: fork 2 ranks onto the single GPU so the
    headline number exercises the production cdna4 cross-process path
    (VERDICT r01 item 1), and also record the self-TL local-copy S/t."""
    local_copy = None
    try:
        lc = run_local_copy(args)
        local_copy = {"GBps": round(lc["busbw"], 2),
                      "lat8b_us": round(lc["lat8b_us"], 2)}
    except Exception as e:  # never let the aside break the bench
        local_copy = {"error": str(e)[:200]}

    port = random.randint(29600, 29999)
    env = dict(os.environ)
    env.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
               WORLD_SIZE=str(args.procs))
    child_cmd = [sys.executable, os.path.abspath(__file__), "--child",
                 "--gpus", "1", "--steps", str(args.steps),
                 "--warmup", str(args.warmup),
                 "--mbytes", str(args.mbytes),
                 "--lat-iters", str(args.lat_iters)]
    procs = []
    for r in range(args.procs):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen(
            child_cmd, env=e,
            stdout=subprocess.PIPE if r == 0 else subprocess.DEVNULL,
            stderr=subprocess.STDOUT, text=True))
    out0, _ = procs[0].communicate(timeout=1200)
    for p in procs[1:]:
        p.wait(timeout=120)
    inner = None
    for line in out0.splitlines():
        if line.startswith("CHILD_RESULT "):
            inner = json.loads(line[len("CHILD_RESULT "):])
    if inner is None:
        sys.stderr.write("2-proc child failed; output:\n" + out0[-4000:]
                         + "\n")
        # fall back to the local-copy number so the contract holds
        out = {
            "metric": "allreduce_busbw_GBps",
            "value": local_copy.get("GBps", 0.0) if local_copy else 0.0,
            "unit": "GB/s",
            "n_gpus": 1,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": None,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "ucc_perftest-equivalent allreduce",
                "msg_mib_per_rank": args.mbytes,
                "parallelism": "self-TL local copy (2-proc run FAILED)",
                "n1_semantics": "N==1 value is local-copy S/t",
            },
        }
        print(json.dumps(out))
        return
    out = {
        "metric": "allreduce_busbw_GBps",
        "value": round(inner["busbw"], 2),
        "unit": "GB/s",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(inner["ms_per_step"], 4),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {
            "model": "ucc_perftest-equivalent allreduce",
            "msg_mib_per_rank": args.mbytes,
            "lat8b_us": round(inner["lat8b_us"], 2),
            "parallelism": "2 procs sharing 1 GPU: cross-process "
                           "tl/cdna4 zero-copy gated allreduce "
                           "(k_gated_*/k_staged_* kernels; shared-HBM "
                           "rig, no xGMI at N=1)",
            "busbw_formula": "S/t * 2(N-1)/N with N=2 procs",
            "local_copy_selfTL": local_copy,
            "n1_semantics": "value is the 2-proc cross-process cdna4 "
                            "allreduce busbw on one GPU",
        },
    }
    print(json.dumps(out))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--mbytes", type=int, default=256,
                   help="allreduce message MiB per rank")
    p.add_argument("--lat-iters", type=int, default=200)
    p.add_argument("--procs", type=int, default=2,
                   help="N=1 only: fork this many ranks onto the one GPU")
    p.add_argument("--child", action="store_true",
                   help=argparse.SUPPRESS)
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    if args.child or world > 1:
        run_rank(args)
    elif args.procs > 1:
        run_parent_n1(args)
    else:
        # pure local-copy mode (debug)
        r = run_local_copy(args)
        print(json.dumps({"local_copy_GBps": round(r["busbw"], 2),
                          "lat8b_us": round(r["lat8b_us"], 2)}))


if __name__ == "__main__":
    main()
