"""torch.distributed integration: a Communicator that bootstraps a ucc_amd
team over an existing (gloo) process group and runs device collectives on
torch tensors through the cdna4 TL — the ProcessGroup-UCC analog
(reference: pytorch ProcessGroup-UCC sits on the same public API)."""

import torch

from .. import core, dtypes


def _gloo_oob(group, world):
    import torch.distributed as dist

    def allgather(data: bytes):
        n = len(data)
        t = torch.frombuffer(bytearray(data), dtype=torch.uint8).clone()
        outs = [torch.empty(n, dtype=torch.uint8) for _ in range(world)]
        dist.all_gather(outs, t, group=group)
        return [o.numpy().tobytes() for o in outs]

    return allgather


_OPMAP = {
    "sum": dtypes.OP_SUM,
    "prod": dtypes.OP_PROD,
    "max": dtypes.OP_MAX,
    "min": dtypes.OP_MIN,
    "avg": dtypes.OP_AVG,
}


class Communicator:
    """One ucc_amd rank bound to this process (one process per GPU)."""

    def __init__(self, group=None):
        import torch.distributed as dist

        c = core()
        self.c = c
        if group is None:
            group = dist.group.WORLD
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        self.lib = c.Lib()
        self.ctx = c.Context(self.lib)
        self.team = c.team_create_post(
            self.ctx, py_allgather=_gloo_oob(group, self.world),
            rank=self.rank, n_ranks=self.world)
        while True:
            st = c.team_create_test(self.team)
            if st == c.OK:
                break
            if st < 0:
                raise RuntimeError(f"ucc team create failed: {st}")

    # ------------------------------------------------------------- colls
    def _mem(self, t):
        return dtypes.MEM_CUDA if t.is_cuda else dtypes.MEM_HOST

    def _wait(self, req):
        req.post()
        while req.test() == self.c.INPROGRESS:
            self.ctx.progress()

    def coll_init(self, coll, src, dst, count, dt, **kw):
        return self.c.coll_init(self.team, coll, src=src, dst=dst,
                                count=count, dt=dt, **kw)

    def allreduce(self, tensor, op="sum", out=None):
        dt = dtypes.from_torch(tensor.dtype)
        if out is None:
            out = tensor
            flags = self.c.FLAG_IN_PLACE
            src = 0
        else:
            flags = 0
            src = tensor.data_ptr()
        req = self.c.coll_init(self.team, "allreduce", src=src,
                               dst=out.data_ptr(), count=tensor.numel(),
                               dt=dt, op=_OPMAP[op], mem_type=self._mem(out),
                               flags=flags)
        self._wait(req)
        return out

    def reduce_scatter(self, src, dst, op="sum"):
        dt = dtypes.from_torch(src.dtype)
        req = self.c.coll_init(self.team, "reduce_scatter",
                               src=src.data_ptr(), dst=dst.data_ptr(),
                               count=dst.numel(), dt=dt, op=_OPMAP[op],
                               mem_type=self._mem(dst))
        self._wait(req)
        return dst

    def allgather(self, src, dst):
        dt = dtypes.from_torch(src.dtype)
        req = self.c.coll_init(self.team, "allgather", src=src.data_ptr(),
                               dst=dst.data_ptr(), count=dst.numel(), dt=dt,
                               mem_type=self._mem(dst))
        self._wait(req)
        return dst

    def broadcast(self, tensor, root=0):
        dt = dtypes.from_torch(tensor.dtype)
        req = self.c.coll_init(self.team, "bcast", src=tensor.data_ptr(),
                               dst=0, count=tensor.numel(), dt=dt, root=root,
                               mem_type=self._mem(tensor))
        self._wait(req)
        return tensor

    def alltoall(self, src, dst):
        dt = dtypes.from_torch(src.dtype)
        req = self.c.coll_init(self.team, "alltoall",
                               src=src.data_ptr(), dst=dst.data_ptr(),
                               count=dst.numel(), dt=dt,
                               mem_type=self._mem(dst))
        self._wait(req)
        return dst

    def reduce(self, tensor, root=0, op="sum", out=None):
        dt = dtypes.from_torch(tensor.dtype)
        is_root = self.rank == root
        req = self.c.coll_init(
            self.team, "reduce", src=tensor.data_ptr(),
            dst=(out.data_ptr() if (is_root and out is not None)
                 else tensor.data_ptr() if is_root else 0),
            count=tensor.numel(), dt=dt, op=_OPMAP[op], root=root,
            mem_type=self._mem(tensor),
            flags=(self.c.FLAG_IN_PLACE
                   if (is_root and out is None) else 0))
        self._wait(req)
        return out if (is_root and out is not None) else tensor

    def barrier(self):
        req = self.c.coll_init(self.team, "barrier", src=0, dst=0, count=0,
                               dt=dtypes.INT8)
        self._wait(req)
