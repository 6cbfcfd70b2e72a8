"""torch.distributed backend "mlsl": run unmodified torch.distributed code
over the mlsl_amd engine.

    import mlsl_amd.torch_backend  # registers the backend
    torch.distributed.init_process_group(backend="mlsl")
    torch.distributed.all_reduce(t)

Collectives complete synchronously (the returned Work is already done) —
the right default for the TCP host transport and for correctness-first
device use; overlap-sensitive code should use the native API
(PersistentRequest / DistributedData) directly.

``destroy_process_group`` does not tear down the mlsl environment (other
process groups may still use it); call ``mlsl_amd.finalize()`` when done,
or let process exit clean it up.

Implementation note: a pure-Python ``dist.ProcessGroup`` subclass (the
pattern torch itself uses for in-process test groups); torch's store/
rendezvous is left untouched — our bootstrap rendezvous uses
MASTER_PORT+1 so both stores coexist.
"""
import torch
import torch.distributed as dist
from torch._C._distributed_c10d import _create_work_from_future
from torch.futures import Future

import mlsl_amd as mx

_DT = {torch.float32: "f32", torch.float64: "f64", torch.bfloat16: "bf16",
       torch.float16: "f16", torch.uint8: "u8", torch.int32: "i32",
       torch.int64: "i64"}


def _op(reduce_op):
    if reduce_op == dist.ReduceOp.SUM:
        return "sum"
    if reduce_op == dist.ReduceOp.MIN:
        return "min"
    if reduce_op == dist.ReduceOp.MAX:
        return "max"
    raise NotImplementedError(f"mlsl backend: ReduceOp {reduce_op}")


def _done(result):
    fut = Future()
    fut.set_result(result)
    return _create_work_from_future(fut)


class MlslProcessGroup(dist.ProcessGroup):
    def __init__(self, rank, world_size):
        super().__init__(rank, world_size)
        if not mx.is_initialized():
            mx.init(rank, world_size)
        self._dist = mx.Distribution(world_size, 1)
        self._rank = rank
        self._world = world_size
        if torch.cuda.is_available():
            mx.set_compute_stream(torch.cuda.current_stream().cuda_stream)

    def getBackendName(self):
        return "mlsl"

    # -- helpers -----------------------------------------------------------
    def _allreduce_one(self, t, op):
        c = t if t.is_contiguous() else t.contiguous()
        mx.wait(self._dist.all_reduce(c, c, c.numel(), op=_op(op),
                                      dtype=_DT[t.dtype], group="data"))
        if c.data_ptr() != t.data_ptr():
            t.copy_(c)

    # -- collectives -------------------------------------------------------
    def allreduce(self, tensor_list, opts=None):
        op = opts.reduceOp if opts is not None else dist.ReduceOp.SUM
        for t in tensor_list:
            self._allreduce_one(t, op)
        return _done(tensor_list)

    def allreduce_coalesced(self, tensor_list, opts=None):
        return self.allreduce(tensor_list, opts)

    def broadcast(self, tensor_list, opts=None):
        root = int(opts.rootRank) if opts is not None else 0
        for t in tensor_list:
            c = t if t.is_contiguous() else t.contiguous()
            mx.wait(self._dist.bcast(c, c.numel(), root=root,
                                     dtype=_DT[t.dtype], group="data"))
            if c.data_ptr() != t.data_ptr():
                t.copy_(c)
        return _done(tensor_list)

    def allgather(self, output_tensors, input_tensor, opts=None):
        outs, inp = output_tensors[0], input_tensor[0]
        c = inp.contiguous()
        flat = torch.empty(self._world * c.numel(), dtype=c.dtype,
                           device=c.device)
        mx.wait(self._dist.all_gather(c, c.numel(), flat,
                                      dtype=_DT[c.dtype], group="data"))
        for r, o in enumerate(outs):
            o.copy_(flat[r * c.numel():(r + 1) * c.numel()].view_as(o))
        return _done([output_tensors])

    def _allgather_base(self, output_tensor, input_tensor, opts=None):
        c = input_tensor.contiguous()
        out = output_tensor if output_tensor.is_contiguous() \
            else output_tensor.contiguous()
        mx.wait(self._dist.all_gather(c, c.numel(), out,
                                      dtype=_DT[c.dtype], group="data"))
        if out.data_ptr() != output_tensor.data_ptr():
            output_tensor.copy_(out)
        return _done(output_tensor)

    def reduce_scatter(self, output_tensors, input_tensor_lists, opts=None):
        op = opts.reduceOp if opts is not None else dist.ReduceOp.SUM
        out, ins = output_tensors[0], input_tensor_lists[0]
        flat = torch.cat([t.reshape(-1) for t in ins])
        res = torch.empty_like(out.reshape(-1))
        mx.wait(self._dist.reduce_scatter(flat, res, out.numel(), op=_op(op),
                                          dtype=_DT[out.dtype], group="data"))
        out.copy_(res.view_as(out))
        return _done([output_tensors])

    def _reduce_scatter_base(self, output_tensor, input_tensor, opts=None):
        op = opts.reduceOp if opts is not None else dist.ReduceOp.SUM
        c = input_tensor.contiguous()
        res = output_tensor if output_tensor.is_contiguous() \
            else output_tensor.contiguous()
        mx.wait(self._dist.reduce_scatter(c, res, output_tensor.numel(),
                                          op=_op(op), dtype=_DT[c.dtype],
                                          group="data"))
        if res.data_ptr() != output_tensor.data_ptr():
            output_tensor.copy_(res)
        return _done(output_tensor)

    def alltoall_base(self, output_buffer, input_buffer, output_split_sizes,
                      input_split_sizes, opts=None):
        if output_split_sizes or input_split_sizes:
            scnt = [int(x) for x in input_split_sizes]
            rcnt = [int(x) for x in output_split_sizes]
            soff = [sum(scnt[:i]) for i in range(self._world)]
            roff = [sum(rcnt[:i]) for i in range(self._world)]
            mx.wait(self._dist.all_to_allv(
                input_buffer.contiguous(), scnt, soff, output_buffer,
                rcnt, roff, dtype=_DT[input_buffer.dtype], group="data"))
        else:
            per = input_buffer.numel() // self._world
            mx.wait(self._dist.all_to_all(
                input_buffer.contiguous(), per, output_buffer,
                dtype=_DT[input_buffer.dtype], group="data"))
        return _done(output_buffer)

    def reduce(self, tensor_list, opts=None):
        op = opts.reduceOp if opts is not None else dist.ReduceOp.SUM
        root = int(opts.rootRank) if opts is not None else 0
        for t in tensor_list:
            c = t if t.is_contiguous() else t.contiguous()
            res = torch.empty_like(c)
            mx.wait(self._dist.reduce(c, res, c.numel(), op=_op(op), root=root,
                                      dtype=_DT[t.dtype], group="data"))
            if self._rank == root:
                t.copy_(res)
        return _done(tensor_list)

    def gather(self, output_tensors, input_tensors, opts=None):
        root = int(opts.rootRank) if opts is not None else 0
        inp = input_tensors[0].contiguous()
        if self._rank == root:
            outs = output_tensors[0]
            flat = torch.empty(self._world * inp.numel(), dtype=inp.dtype,
                               device=inp.device)
        else:
            flat = torch.empty(1, dtype=inp.dtype, device=inp.device)
        mx.wait(self._dist.gather(inp, inp.numel(), flat, root=root,
                                  dtype=_DT[inp.dtype], group="data"))
        if self._rank == root:
            for r, o in enumerate(outs):
                o.copy_(flat[r * inp.numel():(r + 1) * inp.numel()].view_as(o))
        return _done(output_tensors)

    def scatter(self, output_tensors, input_tensors, opts=None):
        root = int(opts.rootRank) if opts is not None else 0
        out = output_tensors[0]
        o = out if out.is_contiguous() else out.contiguous()
        if self._rank == root:
            flat = torch.cat([t.reshape(-1) for t in input_tensors[0]])
        else:
            flat = torch.empty(1, dtype=out.dtype, device=out.device)
        mx.wait(self._dist.scatter(flat, o, o.numel(), root=root,
                                   dtype=_DT[out.dtype], group="data"))
        if o.data_ptr() != out.data_ptr():
            out.copy_(o)
        return _done(output_tensors)

    def send(self, tensor_list, dst_rank, tag=0):
        # matched pairwise with the peer's recv (NCCL-style semantics);
        # implemented over send_recv_list with a send-only pair. Matching is
        # per-directed-edge sequence order only — distinct user tags would
        # silently mismatch payloads, so refuse them loudly (NCCL itself
        # ignores tags; we choose the error over the silent drop).
        if tag != 0:
            raise NotImplementedError("mlsl backend: send/recv tag must be 0 "
                                      "(matching is per-edge sequence order)")
        for t in tensor_list:
            c = t.contiguous()
            mx.wait(self._dist.send_recv_list(
                c, c, [(dst_rank, 0, c.numel(), 0, 0)],
                dtype=_DT[t.dtype], group="data"))
        return _done(tensor_list)

    def recv(self, tensor_list, src_rank, tag=0):
        if tag != 0:
            raise NotImplementedError("mlsl backend: send/recv tag must be 0 "
                                      "(matching is per-edge sequence order)")
        for t in tensor_list:
            c = t if t.is_contiguous() else t.contiguous()
            mx.wait(self._dist.send_recv_list(
                c, c, [(src_rank, 0, 0, 0, c.numel())],
                dtype=_DT[t.dtype], group="data"))
            if c.data_ptr() != t.data_ptr():
                t.copy_(c)
        return _done(tensor_list)

    def barrier(self, opts=None):
        self._dist.barrier("data")
        return _done(True)


def _create(store, rank, world_size, timeout=None, *args, **kwargs):
    return MlslProcessGroup(rank, world_size)


dist.Backend.register_backend("mlsl", _create, devices=["cpu", "cuda"])
