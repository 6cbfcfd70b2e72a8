"""Synthetic models for smoke tests and benchmarks.

The library is a communication framework (the reference has no tensors or
autograd — SURVEY.md); "models" here are the communication-pattern workloads
the reference ships as tests/benchmarks:

- SyntheticNet: the 2-layer CC net (tests/examples/mlsl_test shape), driving
  forward/backward activation exchange + gradient allreduce through a
  Session. Compute is a deterministic elementwise transform (torch or numpy)
  so it runs anywhere.
- resnet50_buckets(): the ResNet-50-shaped gradient bucket list used by the
  overlap benchmark (driver config 3, ~25M params).
"""
import numpy as np

import mlsl_amd as mx


def resnet50_buckets(bucket_mb=25, dtype_size=4, total_params=25_557_032):
    """Split ~25.5M parameters into allreduce buckets of ~bucket_mb MB."""
    per = bucket_mb * 1024 * 1024 // dtype_size
    buckets, left = [], total_params
    while left > 0:
        n = min(per, left)
        buckets.append(n)
        left -= n
    return buckets


def transformer_buckets(model="llama8b-ish", dtype_size=2):
    """Gradient bucket lists for transformer-shaped workloads (bf16):
    per-layer fused buckets (attention + MLP weights) — the bucket sizes a
    Ulysses/TP trainer would allreduce per layer."""
    shapes = {
        # (layers, hidden, intermediate)
        "gpt2-l": (36, 1280, 5120),
        "llama8b-ish": (32, 4096, 14336),
    }[model]
    L, H, I = shapes
    per_layer = 4 * H * H + 3 * H * I + 2 * H  # qkv+o proj, gate/up/down, norms
    return [per_layer for _ in range(L)] + [2 * 32000 * H // 2]  # embeddings


class SyntheticNet:
    """Two CC layers on a Distribution grid; one step = forward exchange +
    backward exchange + gradient allreduce + (optional) increment allgather.
    """

    def __init__(self, data_parts, model_parts, global_mb=16, fm=16, fm_size=9,
                 kernel_size=4, dist_update=False, xp=np, device=None):
        self.xp = xp
        self.device = device
        self.sess = mx.Session()
        self.sess.set_global_minibatch_size(global_mb)
        self.dist = mx.Distribution(data_parts, model_parts)
        self.mp = model_parts
        self.du = dist_update

        def op(name, fin, fout):
            info = self.sess.create_op_reg_info("cc")
            info.set_name(name)
            info.add_input(fin, fm_size, "f32")
            info.add_output(fout, fm_size, "f32")
            info.add_parameter_set(fin * fout, kernel_size, "f32",
                                   distributed_update=dist_update)
            info.validate(self.dist)
            return self.sess.operation(self.sess.add_operation(info, self.dist))

        self.op0 = op("fc0", fm * model_parts, 2 * fm * model_parts)
        self.op1 = op("fc1", 2 * fm * model_parts, fm * model_parts)
        self.op0.set_next(self.op1, 0, 0)
        self.sess.commit()

        self.out0 = self.op0.output(0)
        self.in1 = self.op1.input(0)
        self._alloc()

    def _zeros(self, n):
        if self.xp is np:
            return np.zeros(max(n, 1), dtype=np.float32)
        return self.xp.zeros(max(n, 1), dtype=self.xp.float32, device=self.device)

    def _alloc(self):
        self.comm0 = self._zeros(self.out0.comm_buf_size // 4)
        self.comm1 = self._zeros(self.in1.comm_buf_size // 4)
        self.grads = []
        for op in (self.op0, self.op1):
            ps = op.parameter_set(0)
            self.grads.append(self._zeros(ps.local_kernel_count * ps.kernel_size))

    def step(self, it=0):
        """One training iteration's communication schedule."""
        need_fwd = self.mp > 1
        if need_fwd:
            # forward: out0 partial sums -> ReduceScatter -> in1
            self.comm0 += 0  # touch (stands in for pack of computed partials)
            self.out0.start_comm(self.comm0)
            self.in1.wait_comm()
            # backward: in1 grads -> AllGather -> out0
            self.in1.start_comm(self.comm1)
            self.out0.wait_comm()
        # gradients (overlapped Start then Wait in reverse order, the
        # reference's Backward2/Update split)
        for op, g in zip((self.op1, self.op0), reversed(self.grads)):
            op.parameter_set(0).start_gradient_comm(g)
        for op, g in zip((self.op0, self.op1), self.grads):
            ps = op.parameter_set(0)
            ps.wait_gradient_comm()
            if self.du and self.dist.process_count("data") > 1:
                ps.start_increment_comm(g)
                ps.wait_increment_comm()
        return True
