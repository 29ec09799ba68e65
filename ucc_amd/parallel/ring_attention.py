"""Context-parallel / ring-attention KV movement.

SURVEY §2.9 SP/CP/ring-attention workload: each CP rank holds a KV
shard; ring attention rotates the shards around the ranks one hop per
step while queries stay put, and CP prefill variants allgather the
full KV once. The reference maps the rotation onto its p2p ring
pattern (coll_patterns/ring); here the hop IS one sparse alltoallv —
counts are zero except to the destination neighbor — which runs
natively on device teams (cdna4 gated/CE alltoallv over xGMI) and on
host teams (tcp pairwise/hybrid), with no p2p layer needed.
"""

import torch

from .. import dtypes


class RingKV:
    """KV-shard rotation and gathering over a Communicator."""

    def __init__(self, comm):
        self.comm = comm

    def rotate(self, kv, out=None, displacement=1):
        """Send my KV shard to (rank+displacement) mod world; receive
        the shard of (rank-displacement). One ring-attention step is
        displacement=1. Returns the received shard."""
        comm = self.comm
        n = comm.world
        d = displacement % n
        if d == 0:
            return kv if out is None else out.copy_(kv)
        if out is None:
            out = torch.empty_like(kv)
        elems = kv.numel()
        to = (comm.rank + d) % n
        frm = (comm.rank - d) % n
        scnt = [0] * n
        sdsp = [0] * n
        rcnt = [0] * n
        rdsp = [0] * n
        scnt[to] = elems
        rcnt[frm] = elems
        req = comm.c.coll_init(
            comm.team, "alltoallv", src=kv.data_ptr(),
            dst=out.data_ptr(), count=elems,
            dt=dtypes.from_torch(kv.dtype),
            mem_type=dtypes.MEM_CUDA if kv.is_cuda else dtypes.MEM_HOST,
            src_counts=scnt, src_displs=sdsp,
            dst_counts=rcnt, dst_displs=rdsp)
        comm._wait(req)
        return out

    def ring_steps(self, kv, fn):
        """Full ring-attention sweep: call fn(shard, owner_rank) on my
        own shard, then on each received shard as the ring rotates
        world-1 times. Uses double buffering so fn overlaps nothing —
        the communication fast path is the collective itself."""
        comm = self.comm
        n = comm.world
        cur = kv
        nxt = torch.empty_like(kv)
        fn(cur, comm.rank)
        for s in range(1, n):
            self.rotate(cur, out=nxt)
            cur, nxt = nxt, cur
            fn(cur, (comm.rank - s) % n)
        return cur

    def gather(self, kv, out=None):
        """CP prefill: allgather every rank's KV shard (out shape =
        [world * kv.numel()] flat, or pass a preallocated tensor)."""
        comm = self.comm
        if out is None:
            out = torch.empty(comm.world * kv.numel(), dtype=kv.dtype,
                              device=kv.device)
        comm.allgather(kv, out)
        return out
