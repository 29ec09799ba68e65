"""Expert-parallel (MoE) token exchange over ucc_amd alltoallv.

The SURVEY §2.9 EP workload: tokens routed to experts on other ranks
with skewed per-peer counts — a thin, correct dispatch/combine pair on
the library's alltoallv (tl/cdna4 staged per-peer cells on device,
shm/tcp on host)."""

import torch

from .. import dtypes


def _cumsum0(counts):
    out, t = [], 0
    for x in counts:
        out.append(t)
        t += x
    return out


class TokenExchanger:
    """alltoallv dispatch: rank r sends send_counts[d] tokens (rows) to
    each rank d and receives what others routed to it."""

    def __init__(self, comm):
        self.comm = comm

    def exchange_counts(self, send_counts):
        """Symmetric count exchange (alltoall of one int per peer)."""
        import torch

        world = self.comm.world
        src = torch.tensor(send_counts, dtype=torch.int64)
        dst = torch.zeros(world, dtype=torch.int64)
        c = self.comm.c
        req = c.coll_init(self.comm.team, "alltoall",
                          src=src.data_ptr(), dst=dst.data_ptr(),
                          count=world, dt=dtypes.INT64,
                          mem_type=dtypes.MEM_HOST)
        self.comm._wait(req)
        return dst.tolist()  # recv_counts[s] = tokens coming from rank s

    def dispatch(self, tokens, send_counts, recv_counts=None):
        """tokens: [sum(send_counts), hidden] rows grouped by
        destination rank. Returns the received [sum(recv_counts),
        hidden] tensor (and recv_counts)."""
        if recv_counts is None:
            recv_counts = self.exchange_counts(send_counts)
        hidden = tokens.shape[1]
        scnt = [c * hidden for c in send_counts]
        rcnt = [c * hidden for c in recv_counts]
        sdsp = _cumsum0(scnt)
        rdsp = _cumsum0(rcnt)
        out = torch.empty(sum(recv_counts), hidden, dtype=tokens.dtype,
                          device=tokens.device)
        c = self.comm.c
        req = c.coll_init(
            self.comm.team, "alltoallv", src=tokens.data_ptr(),
            dst=out.data_ptr(), count=0,
            dt=dtypes.from_torch(tokens.dtype),
            mem_type=(dtypes.MEM_CUDA if tokens.is_cuda
                      else dtypes.MEM_HOST),
            src_counts=scnt, src_displs=sdsp,
            dst_counts=rcnt, dst_displs=rdsp)
        self.comm._wait(req)
        return out, recv_counts

    def combine(self, expert_out, recv_counts, send_counts):
        """Inverse exchange: return expert outputs to token owners."""
        out, _ = TokenExchanger.dispatch(
            self, expert_out, recv_counts, send_counts)
        return out
