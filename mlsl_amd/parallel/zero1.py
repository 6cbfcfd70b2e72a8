"""ShardedOptimizer: ZeRO-1-style distributed weight update over the
mlsl_amd engine — the torch-level face of the reference's
``distributedUpdate`` ParameterSets (include/mlsl.hpp:306-339,
src/mlsl_impl.cpp:401-435): gradients are reduce-scattered so each data-
parallel rank owns ``1/dp`` of every flat parameter group, the wrapped
optimizer steps only on the owned shard, and the updated shard is
all-gathered back (the reference's StartIncrementComm/WaitIncrementComm
AllGather).

Optimizer state (momentum, Adam moments) therefore lives only for the
owned shard on each rank: state memory scales 1/dp.

Usage (one process per GPU):

    model = Net().cuda()
    dd = DistributedData(model, dist)           # grads stay full-size
    sopt = ShardedOptimizer(model.parameters(), torch.optim.AdamW,
                            dist, lr=1e-3)
    for batch in loader:
        loss = model(batch); loss.backward()
        dd.finish_gradients()                   # averaged full grads
        sopt.step()                             # RS -> shard step -> AG
        sopt.zero_grad()

``ShardedOptimizer`` can also run WITHOUT a DDP wrapper: pass
``reduce="rs"`` and it replaces the gradient allreduce entirely —
reduce_scatter(grad) + sharded step + all_gather(param), one exchange of
each flat buffer per step (the reference's grad RS + increment AG pair).
"""
import torch

import mlsl_amd as mx


def _dt(t):
    return {torch.float32: "f32", torch.float64: "f64",
            torch.bfloat16: "bf16", torch.float16: "f16"}[t.dtype]


class ShardedOptimizer:
    def __init__(self, params, opt_cls, dist=None, group="data",
                 reduce="none", average=True, **opt_kwargs):
        """reduce: "none" (grads already reduced, e.g. by DistributedData)
        or "rs" (this wrapper reduce-scatters raw local grads itself)."""
        self.params = [p for p in params if p.requires_grad]
        assert self.params, "no trainable parameters"
        self.dist = dist or mx.Distribution(mx.world_size(), 1)
        self.group = group
        self.world = self.dist.process_count(group)
        self.rank = self.dist.process_idx(group)
        self.average = average
        assert reduce in ("none", "rs")
        self.reduce = reduce

        if torch.cuda.is_available() and self.params[0].is_cuda:
            mx.set_compute_stream(torch.cuda.current_stream().cuda_stream)
        dev = self.params[0].device
        dt = self.params[0].dtype
        assert all(p.dtype == dt for p in self.params), \
            "one flat group: uniform dtype required"
        total = sum(p.numel() for p in self.params)
        # pad so the shard divides evenly (owned = ceil math in the
        # reference, mlsl_impl.cpp:401-411; padding is the flat analog)
        self.shard = (total + self.world - 1) // self.world
        self.flat = torch.zeros(self.shard * self.world, dtype=dt, device=dev)
        self.flat_grad = torch.zeros_like(self.flat)
        # map parameters onto the flat buffer (they become views)
        off = 0
        with torch.no_grad():
            for p in self.params:
                n = p.numel()
                self.flat[off:off + n].copy_(p.data.reshape(-1))
                p.data = self.flat[off:off + n].view_as(p.data)
                off += n
        lo = self.rank * self.shard
        self.own_param = self.flat[lo:lo + self.shard]
        self.own_grad = self.flat_grad[lo:lo + self.shard]
        # the wrapped optimizer sees ONLY the owned shard (state: 1/dp)
        shard_param = torch.nn.Parameter(self.own_param, requires_grad=False)
        self._shard_holder = shard_param
        self.opt = opt_cls([shard_param], **opt_kwargs)

    def _gather_grads(self):
        off = 0
        for p in self.params:
            n = p.numel()
            g = p.grad
            if g is None:
                self.flat_grad[off:off + n].zero_()
            else:
                self.flat_grad[off:off + n].copy_(g.detach().reshape(-1))
            off += n
        self.flat_grad[off:].zero_()

    def step(self):
        self._gather_grads()
        if self.world > 1:
            if self.reduce == "rs":
                # raw local grads -> reduce_scatter into the owned shard
                mx.wait(self.dist.reduce_scatter(
                    self.flat_grad, self.own_grad, self.shard, op="sum",
                    dtype=_dt(self.flat), group=self.group))
            else:
                # grads already reduced everywhere: owned slice is ready
                self.own_grad.copy_(
                    self.flat_grad[self.rank * self.shard:
                                   (self.rank + 1) * self.shard])
            if self.average and self.reduce == "rs":
                self.own_grad /= self.world
        else:
            self.own_grad.copy_(self.flat_grad[:self.shard])

        self._shard_holder.grad = self.own_grad
        self.opt.step()
        self._shard_holder.grad = None

        if self.world > 1:
            # increment AllGather (reference StartIncrementComm):
            # every rank's updated shard -> full parameter buffer
            mx.wait(self.dist.all_gather(self.own_param, self.shard,
                                         self.flat, dtype=_dt(self.flat),
                                         group=self.group))

    def zero_grad(self, set_to_none=False):
        for p in self.params:
            if p.grad is not None:
                if set_to_none:
                    p.grad = None
                else:
                    p.grad.zero_()

    def state_dict(self):
        return self.opt.state_dict()

    def load_state_dict(self, sd):
        self.opt.load_state_dict(sd)
