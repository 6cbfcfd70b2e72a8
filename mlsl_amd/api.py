"""Pythonic API over the mlsl_amd C binding.

Accepts numpy arrays, torch tensors (CPU or HIP device), raw integers
(device pointers) or ctypes pointers as buffers. Dtype/reduction/group names
mirror the reference Python binding's enums (include/mlsl/mlsl.py:521-554).
"""
import ctypes

from ._lib import lib, check, MlslError, c_size_t, c_void_p, c_int, c_ull

# enum name maps (values = core/types.hpp)
DTYPE = {"f32": 0, "f64": 1, "u8": 2, "bf16": 3, "f16": 4, "i32": 5, "i64": 6}
DTYPE_SIZE = {"f32": 4, "f64": 8, "u8": 1, "bf16": 2, "f16": 2, "i32": 4, "i64": 8}
REDOP = {"sum": 0, "min": 1, "max": 2}
GROUP = {"data": 0, "model": 1, "global": 2}
OPTYPE = {"cc": 0, "bias": 1, "act": 2, "pool": 3, "split": 4, "concat": 5,
          "bcast": 6, "reduce": 7, "data": 8, "eval": 9}
COMPRESSION = {"none": 0, "int8": 1}

_NUMPY_DTYPE = {"float32": "f32", "float64": "f64", "uint8": "u8",
                "int32": "i32", "int64": "i64", "float16": "f16"}
_TORCH_DTYPE = {"torch.float32": "f32", "torch.float64": "f64", "torch.uint8": "u8",
                "torch.bfloat16": "bf16", "torch.float16": "f16",
                "torch.int32": "i32", "torch.int64": "i64"}


_registered_stream = [None]


def _sync_compute_stream(t):
    """Re-register torch's CURRENT stream as the compute stream whenever a
    CUDA tensor enters a collective. Registering only once (at wrapper
    construction) ordered collectives against a stale stream if the
    application later produced gradients on a different torch stream —
    silent corruption. The ctypes call is skipped when unchanged."""
    import torch
    s = torch.cuda.current_stream(t.device).cuda_stream
    if s != _registered_stream[0]:
        set_compute_stream(s)
        _registered_stream[0] = s


def _as_ptr_dtype(buf, dtype=None):
    """Return (void_ptr, inferred_dtype_name or None)."""
    if buf is None:
        return None, dtype
    if isinstance(buf, int):
        return c_void_p(buf), dtype
    if isinstance(buf, ctypes.c_void_p):
        return buf, dtype
    # numpy
    if hasattr(buf, "__array_interface__"):
        ai = buf.__array_interface__
        return c_void_p(ai["data"][0]), dtype or _NUMPY_DTYPE.get(str(buf.dtype))
    # torch
    if hasattr(buf, "data_ptr"):
        if getattr(buf, "is_cuda", False):
            _sync_compute_stream(buf)
        return c_void_p(buf.data_ptr()), dtype or _TORCH_DTYPE.get(str(buf.dtype))
    raise TypeError(f"unsupported buffer type {type(buf)}")


def _sizes(arr):
    return (c_size_t * len(arr))(*arr)


def init(rank=-1, size=-1):
    """Idempotent: a no-op when the environment is already up (torch-style
    convenience; the C/C++ Init still errors on double-init)."""
    if is_initialized():
        return
    check(lib().mlsl_init(rank, size))


def finalize():
    check(lib().mlsl_finalize())


def configure(config):
    """Tenant re-split: configure("color=N") — ranks sharing N become their
    own world (reference Environment::Configure)."""
    L = lib()
    if not hasattr(L, "_cfg_declared"):
        L.mlsl_configure.argtypes = [ctypes.c_char_p]
        L.mlsl_configure.restype = c_int
        L._cfg_declared = True
    check(L.mlsl_configure(config.encode()))


def is_initialized():
    v = c_int(0)
    check(lib().mlsl_initialized(ctypes.byref(v)))
    return bool(v.value)


def rank():
    v = c_size_t(0)
    check(lib().mlsl_rank(ctypes.byref(v)))
    return v.value


def world_size():
    v = c_size_t(0)
    check(lib().mlsl_world_size(ctypes.byref(v)))
    return v.value


def version():
    v = c_int(0)
    check(lib().mlsl_get_version(ctypes.byref(v)))
    return (v.value >> 16, v.value & 0xFFFF)


def alloc(nbytes, align=64):
    p = c_void_p()
    check(lib().mlsl_alloc(nbytes, align, ctypes.byref(p)))
    return p.value


def free(ptr):
    check(lib().mlsl_dealloc(c_void_p(ptr)))


def memcpy(dst, src, nbytes):
    """Coherent copy to/from Alloc'd (HBM) buffers — the sanctioned way to
    fill or read comm buffers from the host (direct CPU stores over BAR are
    not coherent with the GPU's caches). dst/src: int address, numpy array,
    or torch tensor."""
    dp, _ = _as_ptr_dtype(dst)
    sp, _ = _as_ptr_dtype(src)
    check(lib().mlsl_memcpy(dp, sp, nbytes))


def set_quant_params(block_elems=256):
    check(lib().mlsl_set_quant_params(block_elems))


def set_compute_stream(stream_ptr):
    """Order device collectives after work on this HIP stream (pass
    torch.cuda.current_stream().cuda_stream). Default: the legacy default
    stream, which covers plain torch usage."""
    import ctypes as _c
    L = lib()
    if not hasattr(L, "_scs_declared"):
        L.mlsl_set_compute_stream.argtypes = [_c.c_void_p]
        L.mlsl_set_compute_stream.restype = _c.c_int
        L._scs_declared = True
    check(L.mlsl_set_compute_stream(_c.c_void_p(stream_ptr)))
    _registered_stream[0] = stream_ptr


def wait(req):
    """Complete a generic-collective request; returns the result pointer."""
    out = c_void_p()
    check(lib().mlsl_wait(req, ctypes.byref(out)))
    return out.value


def test(req):
    done = c_int(0)
    out = c_void_p()
    check(lib().mlsl_test(req, ctypes.byref(done), ctypes.byref(out)))
    return bool(done.value), out.value


class Distribution:
    """The data x model process grid (reference Distribution)."""

    def __init__(self, data_parts=None, model_parts=None, *, colors=None, _handle=None):
        self._h = _handle
        if self._h is None:
            h = c_void_p()
            if colors is not None:
                check(lib().mlsl_distribution_create_with_colors(
                    colors[0], colors[1], ctypes.byref(h)))
            else:
                check(lib().mlsl_distribution_create(
                    data_parts, model_parts, ctypes.byref(h)))
            self._h = h

    def close(self):
        if self._h:
            check(lib().mlsl_distribution_free(self._h))
            self._h = None

    def process_idx(self, group="global"):
        v = c_size_t(0)
        check(lib().mlsl_distribution_process_idx(self._h, GROUP[group], ctypes.byref(v)))
        return v.value

    def process_count(self, group="global"):
        v = c_size_t(0)
        check(lib().mlsl_distribution_process_count(self._h, GROUP[group], ctypes.byref(v)))
        return v.value

    def barrier(self, group="global"):
        check(lib().mlsl_distribution_barrier(self._h, GROUP[group]))

    def _dt(self, sbuf, rbuf, dtype):
        _, d1 = _as_ptr_dtype(sbuf)
        _, d2 = _as_ptr_dtype(rbuf)
        dt = dtype or d1 or d2
        if dt is None:
            raise ValueError("cannot infer dtype; pass dtype=")
        return dt

    def bcast(self, buf, count, root=0, dtype=None, group="global"):
        dt = self._dt(buf, None, dtype)
        p, _ = _as_ptr_dtype(buf)
        req = c_void_p()
        check(lib().mlsl_distribution_bcast(self._h, p, count, DTYPE[dt], root,
                                            GROUP[group], ctypes.byref(req)))
        return req

    def reduce(self, sbuf, rbuf, count, op="sum", root=0, dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_reduce(self._h, sp, rp, count, DTYPE[dt],
                                             REDOP[op], root, GROUP[group],
                                             ctypes.byref(req)))
        return req

    def all_reduce(self, sbuf, rbuf, count, op="sum", dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_all_reduce(self._h, sp, rp, count, DTYPE[dt],
                                                 REDOP[op], GROUP[group],
                                                 ctypes.byref(req)))
        return req

    def all_to_all(self, sbuf, send_count, rbuf, dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_all_to_all(self._h, sp, send_count, rp, DTYPE[dt],
                                                 GROUP[group], ctypes.byref(req)))
        return req

    def all_to_allv(self, sbuf, scnt, soff, rbuf, rcnt, roff, dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_all_to_allv(
            self._h, sp, _sizes(scnt), _sizes(soff), rp, _sizes(rcnt), _sizes(roff),
            DTYPE[dt], GROUP[group], ctypes.byref(req)))
        return req

    def gather(self, sbuf, send_count, rbuf, root=0, dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_gather(self._h, sp, send_count, rp, DTYPE[dt],
                                             root, GROUP[group], ctypes.byref(req)))
        return req

    def all_gather(self, sbuf, send_count, rbuf, dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_all_gather(self._h, sp, send_count, rp, DTYPE[dt],
                                                 GROUP[group], ctypes.byref(req)))
        return req

    def all_gatherv(self, sbuf, send_count, rbuf, rcnt, dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_all_gatherv(self._h, sp, send_count, rp,
                                                  _sizes(rcnt), DTYPE[dt], GROUP[group],
                                                  ctypes.byref(req)))
        return req

    def scatter(self, sbuf, rbuf, recv_count, root=0, dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_scatter(self._h, sp, rp, recv_count, DTYPE[dt],
                                              root, GROUP[group], ctypes.byref(req)))
        return req

    def send_recv_list(self, sbuf, rbuf, pairs, dtype=None, group="global"):
        """Neighbor exchange: pairs = [(peer, send_off, send_count,
        recv_off, recv_count), ...] in elements (CommOpSRList — declared but
        unimplemented in the reference; first-class here)."""
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        L = lib()
        if not hasattr(L, "_srl_declared"):
            P = ctypes.POINTER
            L.mlsl_distribution_send_recv_list.argtypes = [
                c_void_p, c_void_p, c_void_p, P(c_size_t), P(c_size_t),
                P(c_size_t), P(c_size_t), P(c_size_t), c_size_t, c_int, c_int,
                P(c_void_p)]
            L.mlsl_distribution_send_recv_list.restype = c_int
            L._srl_declared = True
        peers = _sizes([p[0] for p in pairs])
        soffs = _sizes([p[1] for p in pairs])
        scnts = _sizes([p[2] for p in pairs])
        roffs = _sizes([p[3] for p in pairs])
        rcnts = _sizes([p[4] for p in pairs])
        req = c_void_p()
        check(L.mlsl_distribution_send_recv_list(
            self._h, sp, rp, peers, soffs, scnts, roffs, rcnts, len(pairs),
            DTYPE[dt], GROUP[group], ctypes.byref(req)))
        return req

    def reduce_scatter(self, sbuf, rbuf, recv_count, op="sum", dtype=None, group="global"):
        dt = self._dt(sbuf, rbuf, dtype)
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        req = c_void_p()
        check(lib().mlsl_distribution_reduce_scatter(self._h, sp, rp, recv_count,
                                                     DTYPE[dt], REDOP[op], GROUP[group],
                                                     ctypes.byref(req)))
        return req

    def win_allocate(self, nbytes, group="global"):
        """Collective: allocate a one-sided RMA window of nbytes on every
        member of `group` (HBM in device mode). Returns a Win."""
        h = c_void_p()
        check(lib().mlsl_win_allocate(self._h, GROUP[group], nbytes,
                                      ctypes.byref(h)))
        return Win(h, nbytes)


class Win:
    """One-sided RMA window with fence-epoch semantics (MPI_Win_fence-like):
    put/get are nonblocking and complete at the next fence(); within an
    epoch, fence applies all puts first, then gets observe them. Created
    via Distribution.win_allocate."""

    def __init__(self, handle, nbytes):
        self._h = handle
        self.nbytes = nbytes

    def close(self):
        if self._h:
            check(lib().mlsl_win_free(self._h))
            self._h = None

    @property
    def base(self):
        """Local window memory address (device memory in device mode —
        use read()/write() or mlsl memcpy to access it from the host)."""
        p = c_void_p()
        n = c_size_t(0)
        check(lib().mlsl_win_buffer(self._h, ctypes.byref(p), ctypes.byref(n)))
        return p.value

    def put(self, src, nbytes, target, target_off=0):
        sp, _ = _as_ptr_dtype(src)
        check(lib().mlsl_win_put(self._h, sp, nbytes, target, target_off))

    def get(self, dst, nbytes, target, target_off=0):
        dp, _ = _as_ptr_dtype(dst)
        check(lib().mlsl_win_get(self._h, dp, nbytes, target, target_off))

    def fence(self):
        check(lib().mlsl_win_fence(self._h))

    def write(self, src, nbytes, off=0):
        """Coherently copy nbytes from src into the LOCAL window at off."""
        memcpy(self.base + off, src, nbytes)

    def read(self, dst, nbytes, off=0):
        """Coherently copy nbytes from the LOCAL window at off into dst."""
        memcpy(dst, self.base + off, nbytes)


class PersistentRequest:
    """Describe-once/run-many collective (persistent request): Setup happens
    at construction, start/wait/test are allocation-free."""

    def __init__(self, dist, kind, count, dtype="f32", op="sum", group="global",
                 quantized=False):
        import ctypes as _c
        L = lib()
        if not hasattr(L, "_persist_declared"):
            P = _c.POINTER
            L.mlsl_persistent_all_reduce.argtypes = [c_void_p, c_size_t, c_int, c_int,
                                                     c_int, c_int, P(c_void_p)]
            L.mlsl_persistent_reduce_scatter.argtypes = [c_void_p, c_size_t, c_int,
                                                         c_int, c_int, P(c_void_p)]
            L.mlsl_persistent_all_gather.argtypes = [c_void_p, c_size_t, c_int, c_int,
                                                     P(c_void_p)]
            L.mlsl_persistent_all_to_all.argtypes = [c_void_p, c_size_t, c_int, c_int,
                                                     P(c_void_p)]
            L.mlsl_request_start.argtypes = [c_void_p, c_void_p, c_void_p]
            L.mlsl_request_wait.argtypes = [c_void_p, P(c_void_p)]
            L.mlsl_request_test.argtypes = [c_void_p, P(c_int)]
            L.mlsl_request_destroy.argtypes = [c_void_p]
            for n in ("mlsl_persistent_all_reduce", "mlsl_persistent_reduce_scatter",
                      "mlsl_persistent_all_gather", "mlsl_persistent_all_to_all",
                      "mlsl_request_start", "mlsl_request_wait", "mlsl_request_test",
                      "mlsl_request_destroy"):
                getattr(L, n).restype = c_int
            L._persist_declared = True
        h = c_void_p()
        if kind == "all_reduce":
            check(L.mlsl_persistent_all_reduce(dist._h, count, DTYPE[dtype], REDOP[op],
                                               GROUP[group], 1 if quantized else 0,
                                               ctypes.byref(h)))
        elif kind == "reduce_scatter":
            check(L.mlsl_persistent_reduce_scatter(dist._h, count, DTYPE[dtype],
                                                   REDOP[op], GROUP[group],
                                                   ctypes.byref(h)))
        elif kind == "all_gather":
            check(L.mlsl_persistent_all_gather(dist._h, count, DTYPE[dtype],
                                               GROUP[group], ctypes.byref(h)))
        elif kind == "all_to_all":
            check(L.mlsl_persistent_all_to_all(dist._h, count, DTYPE[dtype],
                                               GROUP[group], ctypes.byref(h)))
        else:
            raise ValueError(kind)
        self._h = h

    def start(self, sbuf, rbuf):
        sp, _ = _as_ptr_dtype(sbuf)
        rp, _ = _as_ptr_dtype(rbuf)
        check(lib().mlsl_request_start(self._h, sp, rp))

    def wait(self):
        out = c_void_p()
        check(lib().mlsl_request_wait(self._h, ctypes.byref(out)))
        return out.value

    def test(self):
        done = c_int(0)
        check(lib().mlsl_request_test(self._h, ctypes.byref(done)))
        return bool(done.value)

    def destroy(self):
        if self._h:
            check(lib().mlsl_request_destroy(self._h))
            self._h = None


class CommBlockInfo:
    def __init__(self, handle):
        self._h = handle

    def _get(self, name):
        v = c_size_t(0)
        check(getattr(lib(), f"mlsl_comm_block_info_get_{name}")(self._h, ctypes.byref(v)))
        return v.value

    @property
    def mb_offset(self): return self._get("mb_offset")
    @property
    def mb_count(self): return self._get("mb_count")
    @property
    def fm_offset(self): return self._get("fm_offset")
    @property
    def fm_count(self): return self._get("fm_count")
    @property
    def fm_size(self): return self._get("fm_size")
    @property
    def buf_offset(self): return self._get("buf_offset")


class Activation:
    def __init__(self, handle):
        self._h = handle

    def _get(self, name):
        v = c_size_t(0)
        check(getattr(lib(), f"mlsl_activation_get_{name}")(self._h, ctypes.byref(v)))
        return v.value

    @property
    def global_fm_count(self): return self._get("global_fm_count")
    @property
    def global_fm_offset(self): return self._get("global_fm_offset")
    @property
    def local_fm_count(self): return self._get("local_fm_count")
    @property
    def fm_size(self): return self._get("fm_size")
    @property
    def comm_buf_size(self): return self._get("comm_buf_size")
    @property
    def pack_block_count(self): return self._get("pack_block_count")
    @property
    def unpack_block_count(self): return self._get("unpack_block_count")

    def pack_block(self, i):
        h = c_void_p()
        check(lib().mlsl_activation_get_pack_block(self._h, i, ctypes.byref(h)))
        return CommBlockInfo(h)

    def unpack_block(self, i):
        h = c_void_p()
        check(lib().mlsl_activation_get_unpack_block(self._h, i, ctypes.byref(h)))
        return CommBlockInfo(h)

    def start_comm(self, buf):
        p, _ = _as_ptr_dtype(buf)
        check(lib().mlsl_activation_start_comm(self._h, p))

    def wait_comm(self):
        out = c_void_p()
        check(lib().mlsl_activation_wait_comm(self._h, ctypes.byref(out)))
        return out.value


class ParameterSet:
    def __init__(self, handle):
        self._h = handle

    def _get(self, name):
        v = c_size_t(0)
        check(getattr(lib(), f"mlsl_parameter_set_get_{name}")(self._h, ctypes.byref(v)))
        return v.value

    @property
    def global_kernel_count(self): return self._get("global_kernel_count")
    @property
    def global_kernel_offset(self): return self._get("global_kernel_offset")
    @property
    def local_kernel_count(self): return self._get("local_kernel_count")
    @property
    def owned_kernel_count(self): return self._get("owned_kernel_count")
    @property
    def owned_kernel_offset(self): return self._get("owned_kernel_offset")
    @property
    def kernel_size(self): return self._get("kernel_size")

    @property
    def distributed_update(self):
        v = c_int(0)
        check(lib().mlsl_parameter_set_is_distributed_update(self._h, ctypes.byref(v)))
        return bool(v.value)

    def start_gradient_comm(self, buf):
        p, _ = _as_ptr_dtype(buf)
        check(lib().mlsl_parameter_set_start_gradient_comm(self._h, p))

    def wait_gradient_comm(self):
        out = c_void_p()
        check(lib().mlsl_parameter_set_wait_gradient_comm(self._h, ctypes.byref(out)))
        return out.value

    def test_gradient_comm(self):
        done = c_int(0)
        out = c_void_p()
        check(lib().mlsl_parameter_set_test_gradient_comm(self._h, ctypes.byref(done),
                                                          ctypes.byref(out)))
        return bool(done.value), out.value

    def start_increment_comm(self, buf):
        p, _ = _as_ptr_dtype(buf)
        check(lib().mlsl_parameter_set_start_increment_comm(self._h, p))

    def wait_increment_comm(self):
        out = c_void_p()
        check(lib().mlsl_parameter_set_wait_increment_comm(self._h, ctypes.byref(out)))
        return out.value


class Operation:
    def __init__(self, handle):
        self._h = handle

    def _get(self, name):
        v = c_size_t(0)
        check(getattr(lib(), f"mlsl_operation_get_{name}")(self._h, ctypes.byref(v)))
        return v.value

    @property
    def name(self):
        v = ctypes.c_char_p()
        check(lib().mlsl_operation_get_name(self._h, ctypes.byref(v)))
        return (v.value or b"").decode()

    @property
    def global_minibatch_size(self): return self._get("global_minibatch_size")
    @property
    def local_minibatch_size(self): return self._get("local_minibatch_size")
    @property
    def global_minibatch_offset(self): return self._get("global_minibatch_offset")
    @property
    def input_count(self): return self._get("input_count")
    @property
    def output_count(self): return self._get("output_count")
    @property
    def parameter_set_count(self): return self._get("parameter_set_count")

    def input(self, i):
        h = c_void_p()
        check(lib().mlsl_operation_get_input(self._h, i, ctypes.byref(h)))
        return Activation(h)

    def output(self, i):
        h = c_void_p()
        check(lib().mlsl_operation_get_output(self._h, i, ctypes.byref(h)))
        return Activation(h)

    def parameter_set(self, i):
        h = c_void_p()
        check(lib().mlsl_operation_get_parameter_set(self._h, i, ctypes.byref(h)))
        return ParameterSet(h)

    def set_prev(self, prev, act_idx, prev_out_idx):
        check(lib().mlsl_operation_set_prev(self._h, prev._h if prev else None,
                                            act_idx, prev_out_idx))

    def set_next(self, nxt, act_idx, next_in_idx):
        check(lib().mlsl_operation_set_next(self._h, nxt._h if nxt else None,
                                            act_idx, next_in_idx))


class OperationRegInfo:
    def __init__(self, handle, session):
        self._h = handle
        self._session = session

    def set_name(self, name):
        check(lib().mlsl_op_reg_info_set_name(self._h, name.encode()))

    def add_input(self, fm_count, fm_size, dtype="f32"):
        v = c_size_t(0)
        check(lib().mlsl_op_reg_info_add_input(self._h, fm_count, fm_size,
                                               DTYPE[dtype], ctypes.byref(v)))
        return v.value

    def add_output(self, fm_count, fm_size, dtype="f32"):
        v = c_size_t(0)
        check(lib().mlsl_op_reg_info_add_output(self._h, fm_count, fm_size,
                                                DTYPE[dtype], ctypes.byref(v)))
        return v.value

    def add_parameter_set(self, kernel_count, kernel_size, dtype="f32",
                          distributed_update=False, compression="none"):
        v = c_size_t(0)
        check(lib().mlsl_op_reg_info_add_parameter_set(
            self._h, kernel_count, kernel_size, DTYPE[dtype],
            1 if distributed_update else 0, COMPRESSION[compression], ctypes.byref(v)))
        return v.value

    def validate(self, dist=None):
        check(lib().mlsl_op_reg_info_validate(self._h, dist._h if dist else None))


class Statistics:
    def __init__(self, handle):
        self._h = handle

    def start(self): check(lib().mlsl_statistics_start(self._h))
    def stop(self): check(lib().mlsl_statistics_stop(self._h))
    def reset(self): check(lib().mlsl_statistics_reset(self._h))
    def print(self): check(lib().mlsl_statistics_print(self._h))

    @property
    def enabled(self):
        v = c_int(0)
        check(lib().mlsl_statistics_is_enabled(self._h, ctypes.byref(v)))
        return bool(v.value)

    def comm_cycles(self, op_idx):
        v = c_ull(0)
        check(lib().mlsl_statistics_get_comm_cycles(self._h, op_idx, ctypes.byref(v)))
        return v.value

    def compute_cycles(self, op_idx):
        v = c_ull(0)
        check(lib().mlsl_statistics_get_compute_cycles(self._h, op_idx, ctypes.byref(v)))
        return v.value

    def comm_size(self, op_idx):
        v = c_size_t(0)
        check(lib().mlsl_statistics_get_comm_size(self._h, op_idx, ctypes.byref(v)))
        return v.value

    def isolation_comm_cycles(self, op_idx):
        v = c_ull(0)
        check(lib().mlsl_statistics_get_isolation_comm_cycles(self._h, op_idx,
                                                              ctypes.byref(v)))
        return v.value

    @property
    def total_comm_size(self):
        v = c_size_t(0)
        check(lib().mlsl_statistics_get_total_comm_size(self._h, ctypes.byref(v)))
        return v.value

    @property
    def total_comm_cycles(self):
        v = c_ull(0)
        check(lib().mlsl_statistics_get_total_comm_cycles(self._h, ctypes.byref(v)))
        return v.value

    @property
    def total_compute_cycles(self):
        v = c_ull(0)
        check(lib().mlsl_statistics_get_total_compute_cycles(self._h, ctypes.byref(v)))
        return v.value

    def comm_device_ns(self, op_idx):
        v = c_ull(0)
        check(lib().mlsl_statistics_get_comm_device_ns(self._h, op_idx,
                                                       ctypes.byref(v)))
        return v.value

    @property
    def total_comm_device_ns(self):
        # hipEvent-measured GPU comm time (ns); 0 on the host path
        v = c_ull(0)
        check(lib().mlsl_statistics_get_total_comm_device_ns(self._h,
                                                             ctypes.byref(v)))
        return v.value


class Session:
    def __init__(self, phase="train"):
        h = c_void_p()
        check(lib().mlsl_session_create(0 if phase == "train" else 1, ctypes.byref(h)))
        self._h = h

    def close(self):
        if self._h:
            check(lib().mlsl_session_free(self._h))
            self._h = None

    def set_global_minibatch_size(self, mb):
        check(lib().mlsl_session_set_global_minibatch_size(self._h, mb))

    @property
    def global_minibatch_size(self):
        v = c_size_t(0)
        check(lib().mlsl_session_get_global_minibatch_size(self._h, ctypes.byref(v)))
        return v.value

    def create_op_reg_info(self, op_type="cc"):
        h = c_void_p()
        check(lib().mlsl_session_create_op_reg_info(self._h, OPTYPE[op_type],
                                                    ctypes.byref(h)))
        return OperationRegInfo(h, self)

    def delete_op_reg_info(self, info):
        check(lib().mlsl_session_delete_op_reg_info(self._h, info._h))

    def add_operation(self, info, dist):
        v = c_size_t(0)
        check(lib().mlsl_session_add_operation(self._h, info._h, dist._h,
                                               ctypes.byref(v)))
        return v.value

    @property
    def operation_count(self):
        v = c_size_t(0)
        check(lib().mlsl_session_get_operation_count(self._h, ctypes.byref(v)))
        return v.value

    def operation(self, i):
        h = c_void_p()
        check(lib().mlsl_session_get_operation(self._h, i, ctypes.byref(h)))
        return Operation(h)

    def remove_operations(self):
        check(lib().mlsl_session_remove_operations(self._h))

    def commit(self):
        check(lib().mlsl_session_commit(self._h))

    @property
    def stats(self):
        h = c_void_p()
        check(lib().mlsl_session_get_stats(self._h, ctypes.byref(h)))
        return Statistics(h)
