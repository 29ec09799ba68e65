"""ZeRO-3 / FSDP-style sharded-parameter communication over ucc_amd.

SURVEY §2.9 FSDP/ZeRO workload: parameters live sharded 1/world per
rank; before use they are allgathered, after backward gradients are
reduce-scattered back to shards. Both phases run as PERSISTENT requests
(init-once / post-many) so tl/cdna4's zero-copy gated pipeline reads
peers' shards straight over xGMI with no staging copies.
"""

import torch

from .. import dtypes


class ShardedParamGroup:
    """One flat sharded buffer for a list of same-dtype parameters.

    Layout: the UNsharded flat vector is padded to world*shard elements;
    rank r owns flat[r*shard:(r+1)*shard]. `gather()` materializes the
    full flat vector on every rank (persistent in-place allgather);
    `reduce_scatter_grads()` averages the flat gradient and leaves this
    rank's shard in `grad_shard`.
    """

    def __init__(self, comm, params, device=None):
        self.comm = comm
        self.params = list(params)
        if not self.params:
            raise ValueError("no parameters")
        dt = self.params[0].dtype
        if any(p.dtype != dt for p in self.params):
            raise ValueError("one ShardedParamGroup per dtype")
        dev = device if device is not None else self.params[0].device
        w = comm.world
        total = sum(p.numel() for p in self.params)
        self.shard_elems = (total + w - 1) // w
        padded = self.shard_elems * w
        self.flat = torch.zeros(padded, dtype=dt, device=dev)
        off = 0
        for p in self.params:
            n = p.numel()
            self.flat[off:off + n].copy_(p.detach().reshape(-1))
            off += n
        self._offsets = self._compute_offsets()
        lo = comm.rank * self.shard_elems
        self.shard = self.flat[lo:lo + self.shard_elems].clone()
        self.grad_flat = torch.zeros_like(self.flat)
        self.grad_shard = torch.zeros(self.shard_elems, dtype=dt,
                                      device=dev)
        c = comm.c
        mem = dtypes.MEM_CUDA if self.flat.is_cuda else dtypes.MEM_HOST
        # persistent in-place allgather: my shard sits at its block in
        # flat; peers' blocks are read zero-copy where eligible
        self._ag = c.coll_init(
            comm.team, "allgather", src=0, dst=self.flat.data_ptr(),
            count=padded, dt=dtypes.from_torch(dt), mem_type=mem,
            flags=c.FLAG_PERSISTENT | c.FLAG_IN_PLACE)
        # persistent reduce_scatter of the flat gradient into my shard
        self._rs = c.coll_init(
            comm.team, "reduce_scatter", src=self.grad_flat.data_ptr(),
            dst=self.grad_shard.data_ptr(), count=self.shard_elems,
            dt=dtypes.from_torch(dt), op=dtypes.OP_AVG, mem_type=mem,
            flags=c.FLAG_PERSISTENT)

    def _compute_offsets(self):
        offs, off = [], 0
        for p in self.params:
            offs.append(off)
            off += p.numel()
        return offs

    # ---------------------------------------------------------- phases
    def gather(self):
        """Materialize full parameters on every rank (forward entry)."""
        lo = self.comm.rank * self.shard_elems
        self.flat[lo:lo + self.shard_elems].copy_(self.shard)
        self.comm._wait(self._ag)
        for p, off in zip(self.params, self._offsets):
            p.detach().reshape(-1).copy_(self.flat[off:off + p.numel()])

    def reduce_scatter_grads(self):
        """Average gradients across ranks; keep only my shard."""
        off = 0
        for p in self.params:
            n = p.numel()
            g = p.grad
            if g is not None:
                self.grad_flat[off:off + n].copy_(g.detach().reshape(-1))
            else:
                self.grad_flat[off:off + n].zero_()
            off += n
        self.grad_flat[off:].zero_()
        self.comm._wait(self._rs)
        return self.grad_shard

    def optimizer_step(self, lr):
        """Toy sharded SGD: update my shard from grad_shard."""
        self.shard.add_(self.grad_shard, alpha=-lr)
