"""Migration shim: the reference Python binding's surface
(include/mlsl/mlsl.py — `MLSL` top object with CamelCase methods and
DataType/GroupType/ReductionType/OperationType enums) over the native
mlsl_amd API, so reference-era scripts port with minimal edits.

    from mlsl_amd.compat import MLSL, DataType, GroupType, ReductionType
    m = MLSL(); m.Init()
    dist = m.CreateDistribution(m.GetProcessCount(), 1)
    req = dist.AllReduce(buf, buf, n, DataType.FLOAT, ReductionType.SUM,
                         GroupType.DATA)
    m.Wait(req)
"""
from . import api as _api


class DataType:
    FLOAT = "f32"
    DOUBLE = "f64"
    BYTE = "u8"
    BFLOAT16 = "bf16"


class ReductionType:
    SUM = "sum"
    MIN = "min"
    MAX = "max"


class GroupType:
    DATA = "data"
    MODEL = "model"
    GLOBAL = "global"


class OperationType:
    CC = "cc"
    BIAS = "bias"
    ACT = "act"
    POOL = "pool"
    SPLIT = "split"
    CONCAT = "concat"
    BCAST = "bcast"
    REDUCE = "reduce"
    DATA = "data"
    EVAL = "eval"


class CompressionType:
    NONE = "none"
    QUANTIZATION = "int8"


class _DistributionShim:
    def __init__(self, d):
        self._d = d

    def GetProcessIdx(self, gt):
        return self._d.process_idx(gt)

    def GetProcessCount(self, gt):
        return self._d.process_count(gt)

    def Barrier(self, gt):
        self._d.barrier(gt)

    def Bcast(self, buf, count, dtype, root, gt):
        return self._d.bcast(buf, count, root=root, dtype=dtype, group=gt)

    def Reduce(self, sbuf, rbuf, count, dtype, op, root, gt):
        return self._d.reduce(sbuf, rbuf, count, op=op, root=root, dtype=dtype, group=gt)

    def AllReduce(self, sbuf, rbuf, count, dtype, op, gt):
        return self._d.all_reduce(sbuf, rbuf, count, op=op, dtype=dtype, group=gt)

    def AlltoAll(self, sbuf, send_count, rbuf, dtype, gt):
        return self._d.all_to_all(sbuf, send_count, rbuf, dtype=dtype, group=gt)

    def AlltoAllv(self, sbuf, scnt, soff, rbuf, rcnt, roff, dtype, gt):
        return self._d.all_to_allv(sbuf, scnt, soff, rbuf, rcnt, roff, dtype=dtype, group=gt)

    def Gather(self, sbuf, send_count, rbuf, dtype, root, gt):
        return self._d.gather(sbuf, send_count, rbuf, root=root, dtype=dtype, group=gt)

    def AllGather(self, sbuf, send_count, rbuf, dtype, gt):
        return self._d.all_gather(sbuf, send_count, rbuf, dtype=dtype, group=gt)

    def AllGatherv(self, sbuf, send_count, rbuf, rcnt, dtype, gt):
        return self._d.all_gatherv(sbuf, send_count, rbuf, rcnt, dtype=dtype, group=gt)

    def Scatter(self, sbuf, rbuf, recv_count, dtype, root, gt):
        return self._d.scatter(sbuf, rbuf, recv_count, root=root, dtype=dtype, group=gt)

    def ReduceScatter(self, sbuf, rbuf, recv_count, dtype, op, gt):
        return self._d.reduce_scatter(sbuf, rbuf, recv_count, op=op, dtype=dtype, group=gt)


class MLSL:
    """Reference-style top object (mlsl.py:556+)."""

    def Init(self):
        _api.init()

    def Finalize(self):
        _api.finalize()

    def IsInitialized(self):
        return _api.is_initialized()

    def GetProcessIdx(self):
        return _api.rank()

    def GetProcessCount(self):
        return _api.world_size()

    def GetVersion(self):
        maj, mnr = _api.version()
        return (maj << 16) | mnr

    def CreateDistribution(self, data_parts, model_parts):
        return _DistributionShim(_api.Distribution(data_parts, model_parts))

    def CreateDistributionWithColors(self, data_color, model_color):
        return _DistributionShim(_api.Distribution(colors=(data_color, model_color)))

    def DeleteDistribution(self, dist):
        dist._d.close()

    def CreateSession(self, phase="train"):
        return _api.Session(phase)

    def DeleteSession(self, session):
        session.close()

    def Wait(self, req):
        return _api.wait(req)

    def Test(self, req):
        return _api.test(req)

    def Alloc(self, size, alignment=64):
        return _api.alloc(size, alignment)

    def Free(self, ptr):
        _api.free(ptr)

    def SetQuantizationParams(self, block_elems=256):
        _api.set_quant_params(block_elems)
