"""DistributedData: DDP-style data-parallel wrapper over the mlsl_amd
engine — the "framework integration" role the reference played for
Caffe/Torch (MLSL is the comm library *under* a framework).

Usage (one process per GPU):

    model = Net().cuda()
    dd = DistributedData(model, dist)   # broadcasts initial params
    for batch in loader:
        loss = model(batch).sum()
        loss.backward()                  # hooks start bucketed allreduce
        dd.finish_gradients()            # waits + averages
        opt.step(); opt.zero_grad()

Gradients are bucketed (default 25 MiB) and allreduced non-blocking as
buckets fill during backward — the reference's Backward2/Update overlap
split, driven from torch autograd hooks.
"""
import torch

import mlsl_amd as mx


class DistributedData:
    def __init__(self, module, dist=None, bucket_mb=25, group="data",
                 average=True):
        self.module = module
        self.dist = dist or mx.Distribution(mx.world_size(), 1)
        self.group = group
        self.world = self.dist.process_count(group)
        self.average = average
        self.bucket_bytes = bucket_mb * 1024 * 1024
        self._params = [p for p in module.parameters() if p.requires_grad]
        if torch.cuda.is_available():
            mx.set_compute_stream(torch.cuda.current_stream().cuda_stream)

        # broadcast initial parameters from rank 0 (reference test protocol,
        # mlsl_test.cpp:651-652)
        if self.world > 1:
            for p in self._params:
                flat = p.data.contiguous().view(-1)
                mx.wait(self.dist.bcast(flat, flat.numel(), root=0,
                                        dtype=_dt(p), group=group))
                with torch.no_grad():
                    p.data.copy_(flat.view_as(p.data))

        # bucket assignment in reverse parameter order (grads arrive
        # roughly back-to-front during backward), partitioned by dtype: a
        # bucket must be dtype-uniform because torch.cat would silently
        # promote mixed grads while the allreduce dtype comes from one
        # member param (torch DDP keys its buckets the same way)
        self._buckets = []
        cur, cur_bytes, cur_dtype = [], 0, None
        for p in reversed(self._params):
            if cur and p.dtype != cur_dtype:
                self._buckets.append(cur)
                cur, cur_bytes = [], 0
            cur_dtype = p.dtype
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= self.bucket_bytes:
                self._buckets.append(cur)
                cur, cur_bytes = [], 0
        if cur:
            self._buckets.append(cur)
        self._param_bucket = {}
        for bi, ps in enumerate(self._buckets):
            for p in ps:
                self._param_bucket[id(p)] = bi
        self._pending = [0] * len(self._buckets)
        self._reqs = [None] * len(self._buckets)
        self._flat = [None] * len(self._buckets)

        for p in self._params:
            p.register_post_accumulate_grad_hook(self._hook)
        self._reset_counts()

    def _reset_counts(self):
        for bi, ps in enumerate(self._buckets):
            self._pending[bi] = len(ps)

    def _hook(self, p):
        if self.world <= 1:
            return
        bi = self._param_bucket[id(p)]
        self._pending[bi] -= 1
        if self._pending[bi] == 0:
            ps = self._buckets[bi]
            flat = torch.cat([q.grad.detach().reshape(-1) for q in ps])
            self._flat[bi] = flat
            self._reqs[bi] = self.dist.all_reduce(flat, flat, flat.numel(),
                                                  op="sum", dtype=_dt(p),
                                                  group=self.group)

    def finish_gradients(self):
        """Wait all bucket allreduces and scatter averaged grads back."""
        if self.world <= 1:
            return
        for bi, ps in enumerate(self._buckets):
            if self._reqs[bi] is None:
                continue
            mx.wait(self._reqs[bi])
            flat = self._flat[bi]
            if self.average:
                flat /= self.world
            off = 0
            for q in ps:
                n = q.grad.numel()
                q.grad.copy_(flat[off:off + n].view_as(q.grad))
                off += n
            self._reqs[bi] = None
            self._flat[bi] = None
        self._reset_counts()


def _dt(p):
    return {torch.float32: "f32", torch.float64: "f64",
            torch.bfloat16: "bf16", torch.float16: "f16"}[p.dtype]
