"""Sequence-parallel (Ulysses) head<->sequence exchange over alltoall.

The SURVEY §2.9 SP/Ulysses workload: attention over a sequence sharded
across SP ranks switches between
  sequence-sharded  [seq/P, heads,   dh]   (MLP/layernorm layout) and
  head-sharded      [seq,   heads/P, dh]   (attention layout)
with one alltoall each way (DeepSpeed-Ulysses exchange). This helper is
the thin, correct mapping onto the library's alltoall: contiguous
per-destination blocks in, per-source blocks out, plus the local
permutation that restores the logical layout.
"""

import torch


class HeadSeqExchanger:
    """Ulysses exchange over a ucc_amd Communicator.

    seq_to_heads(x): x is this rank's sequence shard
        [s_local, H, dh]  ->  [s_local * P, H/P, dh]
    (full sequence, this rank's head shard). heads_to_seq inverts it.
    H must be divisible by the SP world size P.
    """

    def __init__(self, comm):
        self.comm = comm

    def _alltoall(self, send, recv):
        from .. import dtypes

        c = self.comm.c
        req = c.coll_init(
            self.comm.team, "alltoall", src=send.data_ptr(),
            dst=recv.data_ptr(), count=recv.numel(),
            dt=dtypes.from_torch(send.dtype),
            mem_type=self.comm._mem(send))
        self.comm._wait(req)

    def seq_to_heads(self, x):
        """[s_local, H, dh] -> [s_local*P, H/P, dh]."""
        P = self.comm.world
        s_local, H, dh = x.shape
        assert H % P == 0, "heads must divide the SP world size"
        hp = H // P
        # destination rank d gets my head-slice d: block-contiguous send
        send = (x.reshape(s_local, P, hp, dh).permute(1, 0, 2, 3)
                .contiguous())
        recv = torch.empty_like(send)
        self._alltoall(send, recv)
        # recv[s] = rank s's sequence shard of MY heads
        return recv.reshape(P * s_local, hp, dh)

    def heads_to_seq(self, x):
        """[s_local*P, H/P, dh] -> [s_local, H, dh] (inverse)."""
        P = self.comm.world
        s_full, hp, dh = x.shape
        assert s_full % P == 0
        s_local = s_full // P
        send = x.reshape(P, s_local, hp, dh).contiguous()
        recv = torch.empty_like(send)
        self._alltoall(send, recv)
        # recv[d] = my sequence rows of head-slice d
        return (recv.reshape(P, s_local, hp, dh).permute(1, 0, 2, 3)
                .reshape(s_local, P * hp, dh).contiguous())
