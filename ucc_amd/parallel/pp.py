"""Pipeline-parallel p2p over active-set subset collectives.

SURVEY §2.9 PP workload: stage-to-stage activation/grad send/recv and
stage-weight bcast. UCC's public API has no raw send/recv; the
reference serves PP through the tl/ucp sendrecv layer and active-set
subset collectives (ucc.h active_set). This helper maps a p2p edge
onto a 2-member active-set bcast — root = sender, set = {sender,
receiver} via {start=sender, stride=receiver-sender, size=2} — with a
per-edge rotating tag so back-to-back transfers on one edge never
share a wire tag.

CUDA tensors are staged through pinned-free host copies here: the p2p
edge is a control-plane-size transfer between pipeline stages (one
activation boundary per microbatch); bulk device traffic belongs on
the collective fast paths (cdna4 alltoall/allgather).
"""

import torch

from .. import dtypes

_TAG_BASE = 0x50000  # keep clear of user tags and MoE/Ulysses traffic


class PipelineEdge:
    """p2p transfers between two ranks of a Communicator's team."""

    def __init__(self, comm):
        self.comm = comm
        self._tags = {}

    def _next_tag(self, a, b):
        key = (a, b)
        t = self._tags.get(key, 0)
        self._tags[key] = (t + 1) % 4096
        return _TAG_BASE + ((a * 131 + b) % 0x1000) * 0x1000 + t

    def _xfer(self, tensor, src_rank, dst_rank):
        if src_rank == dst_rank:
            return tensor
        comm = self.comm
        if not (comm.rank == src_rank or comm.rank == dst_rank):
            return tensor
        host = tensor if not tensor.is_cuda else (
            tensor.cpu() if comm.rank == src_rank
            else torch.empty(tensor.shape, dtype=tensor.dtype))
        tag = self._next_tag(src_rank, dst_rank)
        req = comm.c.coll_init(
            comm.team, "bcast", src=host.data_ptr(), dst=0,
            count=host.numel(), dt=dtypes.from_torch(host.dtype),
            root=src_rank,
            active_set=(src_rank, dst_rank - src_rank, 2), tag=tag)
        self.comm._wait(req)
        if tensor.is_cuda and comm.rank == dst_rank:
            tensor.copy_(host)
        return tensor

    def send(self, tensor, dst_rank):
        """Send tensor to dst_rank (dst_rank posts recv())."""
        return self._xfer(tensor, self.comm.rank, dst_rank)

    def recv(self, tensor, src_rank):
        """Receive into tensor from src_rank (src_rank posts send())."""
        return self._xfer(tensor, src_rank, self.comm.rank)


class PipelineStage:
    """1F1B-style stage helper: forward activations flow rank r ->
    r+1, gradients flow r+1 -> r. Both endpoints call the matching
    method with the same tensor shape."""

    def __init__(self, comm):
        self.comm = comm
        self.edge = PipelineEdge(comm)
        self.is_first = comm.rank == 0
        self.is_last = comm.rank == comm.world - 1

    def send_forward(self, acts):
        if not self.is_last:
            self.edge.send(acts, self.comm.rank + 1)
        return acts

    def recv_forward(self, acts):
        if not self.is_first:
            self.edge.recv(acts, self.comm.rank - 1)
        return acts

    def send_backward(self, grads):
        if not self.is_first:
            self.edge.send(grads, self.comm.rank - 1)
        return grads

    def recv_backward(self, grads):
        if not self.is_last:
            self.edge.recv(grads, self.comm.rank + 1)
        return grads

    def broadcast_stage_weights(self, tensors, root=0):
        """Replicated-stage weight bcast (reference PP mapping: bcast
        of stage weights)."""
        for t in tensors:
            self.comm.broadcast(t, root=root)
        return tensors
