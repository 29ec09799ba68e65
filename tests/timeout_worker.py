"""Device spin-timeout recovery: rank 0 posts a fused allreduce that
rank 1 never joins. With UCC_TL_CDNA4_SPIN_LIMIT small, the kernel hits
its bounded spin, sets the error word, ALL block threads exit together,
and the host reports UCC_ERR_TIMED_OUT — the GPU stays healthy and the
processes exit cleanly (the team is consumed, per UCC timeout
semantics). Run under torchrun with 2 procs."""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from ucc_amd import core, dtypes  # noqa: E402


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    torch.cuda.set_device(int(os.environ["LOCAL_RANK"]) %
                          torch.cuda.device_count())
    dist.init_process_group("gloo", rank=rank, world_size=world)
    c = core()
    lib = c.Lib()
    ctx = c.Context(lib)

    def oob(data: bytes):
        t = torch.frombuffer(bytearray(data), dtype=torch.uint8).clone()
        outs = [torch.empty(len(data), dtype=torch.uint8)
                for _ in range(world)]
        dist.all_gather(outs, t)
        return [o.numpy().tobytes() for o in outs]

    team = c.team_create_post(ctx, py_allgather=oob, rank=rank,
                              n_ranks=world)
    while True:
        st = c.team_create_test(team)
        if st == c.OK:
            break
        assert st >= 0
    dist.barrier()

    if rank == 0:
        src = torch.randn(1024, device="cuda")
        dst = torch.zeros(1024, device="cuda")
        req = c.coll_init(team, "allreduce", src=src.data_ptr(),
                          dst=dst.data_ptr(), count=1024,
                          dt=dtypes.FLOAT32, mem_type=dtypes.MEM_CUDA)
        try:
            req.post()
            st = c.INPROGRESS
            import time
            deadline = time.time() + 60
            while st == c.INPROGRESS and time.time() < deadline:
                st = req.test()
                ctx.progress()
            assert st < 0, f"expected a timeout error, got {st}"
            print(f"TIMEOUT_OK rank=0 status={st}", flush=True)
        except RuntimeError as e:
            # post itself may surface the error
            print(f"TIMEOUT_OK rank=0 exc={e}", flush=True)
        torch.cuda.synchronize()  # GPU healthy
    else:
        print("TIMEOUT_OK rank=1 (did not post)", flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
