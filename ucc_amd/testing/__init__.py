"""In-process multi-rank test jig.

Mirrors the reference gtest fixture design (test/gtest/common/test_ucc.h:
UccJob = N full lib+context stacks in one OS process, memcpy OOB), so
collective algorithms are debuggable on a single CPU or GPU.
"""

import numpy as np

from .. import core, dtypes


class LocalJob:
    """n simulated ranks (lib+context+team each) in this process."""

    def __init__(self, n):
        c = core()
        self.c = c
        self.n = n
        self.libs = [c.Lib() for _ in range(n)]
        self.ctxs = [c.Context(lib) for lib in self.libs]
        oob = c.LocalOob(n)
        self.teams = [
            c.team_create_post(self.ctxs[r], local_oob=oob, rank=r)
            for r in range(n)
        ]
        for _ in range(100000):
            sts = [c.team_create_test(t) for t in self.teams]
            if all(s == c.OK for s in sts):
                break
            bad = [s for s in sts if s < 0]
            if bad:
                raise RuntimeError(f"team create failed: {bad}")
        else:
            raise RuntimeError("team create did not converge")

    def coll(self, coll, per_rank_kwargs):
        """Init one collective per rank; returns list of requests."""
        c = self.c
        return [
            c.coll_init(self.teams[r], coll, **per_rank_kwargs[r])
            for r in range(self.n)
        ]

    def run(self, reqs, max_iter=2_000_000):
        for r in reqs:
            r.post()
        for _ in range(max_iter):
            if all(r.test() != self.c.INPROGRESS for r in reqs):
                return
            for ctx in self.ctxs:
                ctx.progress()
        raise TimeoutError("collective did not complete")

    # ---------------------------------------------------------- helpers
    def allreduce_np(self, arrays, op=dtypes.OP_SUM):
        """arrays: list of n numpy arrays (same shape/dtype). Returns list
        of result arrays (the library's output per rank)."""
        n = self.n
        outs = [np.zeros_like(a) for a in arrays]
        reqs = self.coll(
            "allreduce",
            [
                dict(
                    src=arrays[r].ctypes.data,
                    dst=outs[r].ctypes.data,
                    count=arrays[r].size,
                    dt=dtypes.from_numpy(arrays[r].dtype),
                    op=op,
                )
                for r in range(n)
            ],
        )
        self.run(reqs)
        return outs
