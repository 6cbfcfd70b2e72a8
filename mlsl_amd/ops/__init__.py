"""Standalone CDNA4 kernel ops (device-pointer API over the HIP kernels in
csrc/hip/kernels.hip): local reductions, int8 block quantization with error
feedback, and pack/unpack. Used by GPU numerics tests and by consumers that
want the fused kernels without the full Session machinery.

Buffers: torch CUDA tensors, raw integer device pointers, or anything with
``data_ptr()``.
"""
import ctypes

from .._lib import lib as _core_lib, check
from ..api import DTYPE, REDOP, _as_ptr_dtype

_declared = False


def _lib():
    global _declared
    L = _core_lib()
    if not _declared:
        c = ctypes
        L.mlsl_hip_device_count.argtypes = [c.POINTER(c.c_int)]
        L.mlsl_hip_synchronize.argtypes = []
        L.mlsl_hip_reduce.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t, c.c_int, c.c_int]
        L.mlsl_hip_reduce_nt.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t]
        L.mlsl_hip_copy.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t]
        L.mlsl_hip_reduce_nt2.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t]
        L.mlsl_hip_copy_variant.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t, c.c_int]
        L.mlsl_hip_quantize.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p, c.c_size_t,
                                        c.c_size_t, c.c_int, c.c_int]
        L.mlsl_hip_dequantize.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t,
                                          c.c_size_t, c.c_int]
        L.mlsl_hip_dequantize_nt.argtypes = L.mlsl_hip_dequantize.argtypes
        L.mlsl_hip_quantize_f32_nt.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p,
                                               c.c_size_t, c.c_size_t]
        L.mlsl_hip_quant_accum.argtypes = [c.c_void_p, c.c_void_p, c.c_size_t, c.c_size_t]
        ptypes = [c.c_void_p, c.c_void_p] + [c.c_size_t] * 8 + [c.c_int]
        L.mlsl_hip_pack.argtypes = ptypes
        L.mlsl_hip_unpack.argtypes = ptypes
        for n in ("mlsl_hip_device_count", "mlsl_hip_synchronize", "mlsl_hip_reduce",
                  "mlsl_hip_reduce_nt", "mlsl_hip_reduce_nt2", "mlsl_hip_copy", "mlsl_hip_copy_variant",
                  "mlsl_hip_quantize", "mlsl_hip_dequantize", "mlsl_hip_dequantize_nt", "mlsl_hip_quantize_f32_nt", "mlsl_hip_quant_accum",
                  "mlsl_hip_pack", "mlsl_hip_unpack"):
            getattr(L, n).restype = c.c_int
        _declared = True
    return L


def device_count():
    n = ctypes.c_int(0)
    _lib().mlsl_hip_device_count(ctypes.byref(n))
    return n.value


def synchronize():
    check(_lib().mlsl_hip_synchronize())


def reduce_(dst, src, count, dtype=None, op="sum"):
    """dst op= src on device (count elements)."""
    dp, d1 = _as_ptr_dtype(dst)
    sp, d2 = _as_ptr_dtype(src)
    dt = dtype or d1 or d2
    check(_lib().mlsl_hip_reduce(dp, sp, count, DTYPE[dt], REDOP[op]))


def reduce_nt(dst, src, count):
    """f32 sum with nontemporal loads/stores (benchmark variant)."""
    dp, _ = _as_ptr_dtype(dst)
    sp, _ = _as_ptr_dtype(src)
    check(_lib().mlsl_hip_reduce_nt(dp, sp, count))


def reduce_nt2(dst, src, count):
    dp, _ = _as_ptr_dtype(dst)
    sp, _ = _as_ptr_dtype(src)
    check(_lib().mlsl_hip_reduce_nt2(dp, sp, count))


def copy(dst, src, bytes_):
    """Streaming D2D copy (NT kernel for large aligned transfers)."""
    dp, _ = _as_ptr_dtype(dst)
    sp, _ = _as_ptr_dtype(src)
    check(_lib().mlsl_hip_copy(dp, sp, bytes_))


def copy_variant(dst, src, bytes_, nt):
    dp, _ = _as_ptr_dtype(dst)
    sp, _ = _as_ptr_dtype(src)
    check(_lib().mlsl_hip_copy_variant(dp, sp, bytes_, 1 if nt else 0))


def wire_bytes(count, block=256):
    """Bytes of the int8 wire format for `count` elements."""
    nblocks = (count + block - 1) // block
    return nblocks * (block + 8)


def quantize(inp, wire, count, err=None, block=256, dtype=None):
    ip, d1 = _as_ptr_dtype(inp)
    wp, _ = _as_ptr_dtype(wire)
    ep, _ = _as_ptr_dtype(err)
    dt = dtype or d1
    check(_lib().mlsl_hip_quantize(ip, ep, wp, count, block, DTYPE[dt],
                                   1 if err is not None else 0))


def dequantize(wire, out, count, block=256, dtype=None):
    wp, _ = _as_ptr_dtype(wire)
    op_, d1 = _as_ptr_dtype(out)
    dt = dtype or d1
    check(_lib().mlsl_hip_dequantize(wp, op_, count, block, DTYPE[dt]))


def quantize_f32_nt(inp, wire, count, err, block=256):
    ip, _ = _as_ptr_dtype(inp)
    wp, _ = _as_ptr_dtype(wire)
    ep, _ = _as_ptr_dtype(err)
    check(_lib().mlsl_hip_quantize_f32_nt(ip, ep, wp, count, block))


def dequantize_nt(wire, out, count, block=256, dtype=None):
    wp, _ = _as_ptr_dtype(wire)
    op_, d1 = _as_ptr_dtype(out)
    dt = dtype or d1
    check(_lib().mlsl_hip_dequantize_nt(wp, op_, count, block, DTYPE[dt]))


def quant_accum(acc_wire, wire, count, block=256):
    ap, _ = _as_ptr_dtype(acc_wire)
    wp, _ = _as_ptr_dtype(wire)
    check(_lib().mlsl_hip_quant_accum(ap, wp, count, block))


def pack(src, dst, *, mb_offset, mb_count, fm_offset, fm_count, fm_size,
         buf_offset, local_fm_count, local_mb_count, dtype):
    sp, _ = _as_ptr_dtype(src)
    dp, _ = _as_ptr_dtype(dst)
    check(_lib().mlsl_hip_pack(sp, dp, mb_offset, mb_count, fm_offset, fm_count,
                               fm_size, buf_offset, local_fm_count, local_mb_count,
                               DTYPE[dtype]))


def unpack(src, dst, *, mb_offset, mb_count, fm_offset, fm_count, fm_size,
           buf_offset, local_fm_count, local_mb_count, dtype):
    sp, _ = _as_ptr_dtype(src)
    dp, _ = _as_ptr_dtype(dst)
    check(_lib().mlsl_hip_unpack(sp, dp, mb_offset, mb_count, fm_offset, fm_count,
                                 fm_size, buf_offset, local_fm_count, local_mb_count,
                                 DTYPE[dtype]))
