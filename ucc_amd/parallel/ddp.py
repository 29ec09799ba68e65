"""Data-parallel gradient averaging over ucc_amd collectives.

The SURVEY §2.9 DP workload: bucketed gradient allreduce with persistent
requests (init-once / post-many), overlapping compute where the caller
drives `flush()` after backward. This is the ProcessGroup/DDP-comm-hook
role re-done directly on the library: buckets are flat, persistent
allreduce requests are created once per bucket, and every step re-posts
them (tl/cdna4's gated pipeline at typical bucket sizes).
"""

import torch

from .. import dtypes


class BucketAllreducer:
    """Flat-bucket gradient averaging: register params once, then per
    step call accumulate() after backward and flush() to average."""

    def __init__(self, comm, params, bucket_mb=25):
        self.comm = comm
        self.params = [p for p in params if p.requires_grad]
        self.world = comm.world
        bucket_bytes = bucket_mb << 20
        self.buckets = []  # list of (param_list, flat_tensor, req)
        cur, cur_bytes = [], 0
        for p in self.params:
            nb = p.numel() * p.element_size()
            if cur and cur_bytes + nb > bucket_bytes:
                self._seal(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nb
        if cur:
            self._seal(cur)

    def _seal(self, plist):
        dev = plist[0].device
        dt = plist[0].dtype
        total = sum(p.numel() for p in plist)
        flat = torch.zeros(total, dtype=dt, device=dev)
        c = self.comm.c
        req = c.coll_init(
            self.comm.team, "allreduce", src=0, dst=flat.data_ptr(),
            count=total, dt=dtypes.from_torch(dt),
            op=dtypes.OP_SUM,
            mem_type=(dtypes.MEM_CUDA if flat.is_cuda else dtypes.MEM_HOST),
            flags=c.FLAG_PERSISTENT | c.FLAG_IN_PLACE)
        self.buckets.append((plist, flat, req))

    def step(self):
        """Average all gradients: pack -> persistent allreduce -> unpack
        (divide by world size)."""
        c = self.comm.c
        for plist, flat, req in self.buckets:
            off = 0
            for p in plist:
                n = p.numel()
                if p.grad is not None:
                    flat[off:off + n].copy_(p.grad.detach().reshape(-1))
                else:
                    flat[off:off + n].zero_()
                off += n
        if flat.is_cuda:
            torch.cuda.synchronize()
        # post all buckets, then progress them together (overlap)
        for _, _, req in self.buckets:
            req.post()
        pending = [req for _, _, req in self.buckets]
        while any(r.test() == c.INPROGRESS for r in pending):
            self.comm.ctx.progress()
        if self.buckets and self.buckets[0][1].is_cuda:
            torch.cuda.synchronize()
        inv = 1.0 / self.world
        for plist, flat, _ in self.buckets:
            off = 0
            for p in plist:
                n = p.numel()
                if p.grad is not None:
                    p.grad.detach().reshape(-1).copy_(flat[off:off + n])
                    p.grad.mul_(inv)
                off += n
