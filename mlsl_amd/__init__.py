"""mlsl_amd — MI355X-native DL communication library.

A from-scratch rebuild of intel/MLSL's capabilities for AMD Instinct MI355X:
C++17 core, RCCL over xGMI, CDNA4 HIP kernels for fused reduce/pack/quant,
an async host progress engine, and this Python binding.

Quick start (one process per GPU, torchrun-style env):

    import mlsl_amd as mx
    mx.init()
    d = mx.Distribution(mx.world_size(), 1)
    req = d.all_reduce(grad, grad, grad.size)   # numpy / torch / raw ptr
    mx.wait(req)
    mx.finalize()
"""

from .api import (  # noqa: F401
    COMPRESSION, DTYPE, DTYPE_SIZE, GROUP, OPTYPE, REDOP,
    Activation, CommBlockInfo, Distribution, Operation, OperationRegInfo,
    ParameterSet, PersistentRequest, Session, Statistics, Win,
    alloc, configure, free, finalize, init, is_initialized, memcpy, rank,
    set_compute_stream,
    set_quant_params,
    test, version, wait, world_size,
)
from ._lib import MlslError  # noqa: F401

__version__ = "0.1.0"
