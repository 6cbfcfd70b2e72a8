"""Bucketed gradient allreduce with compute overlap.

Usage (one process per GPU):

    buck = GradBucketer(dist, buffers)        # list of flat grad tensors
    for b in reversed(range(len(buffers))):   # as backward produces grads
        ...compute...
        buck.start(b)                         # non-blocking allreduce
    buck.wait_all()                           # before optimizer step

Requests are persistent (created once, re-started every iteration) — the
reference's Session::Commit contract (src/mlsl_impl.cpp:567-578).
"""
import mlsl_amd as mx
from mlsl_amd.api import _as_ptr_dtype


class GradBucketer:
    def __init__(self, dist, buffers, op="sum", group="data", dtype=None,
                 outputs=None):
        self.dist = dist
        self.buffers = list(buffers)
        self.outputs = list(outputs) if outputs is not None else self.buffers
        assert len(self.outputs) == len(self.buffers)
        self.op = op
        self.group = group
        self.dtypes = []
        for b in self.buffers:
            _, dt = _as_ptr_dtype(b)
            self.dtypes.append(dtype or dt)
        self._reqs = [None] * len(self.buffers)

    def start(self, idx):
        b = self.buffers[idx]
        n = b.size if hasattr(b, "size") and isinstance(b.size, int) else (
            b.numel() if hasattr(b, "numel") else len(b))
        self._reqs[idx] = self.dist.all_reduce(b, self.outputs[idx], n,
                                               op=self.op,
                                               dtype=self.dtypes[idx],
                                               group=self.group)

    def start_all(self):
        for i in range(len(self.buffers)):
            self.start(i)

    def wait(self, idx):
        if self._reqs[idx] is not None:
            mx.wait(self._reqs[idx])
            self._reqs[idx] = None

    def test(self, idx):
        if self._reqs[idx] is None:
            return True
        done, _ = mx.test(self._reqs[idx])
        if done:
            self._reqs[idx] = None
        return done

    def wait_all(self):
        for i in range(len(self.buffers)):
            self.wait(i)
