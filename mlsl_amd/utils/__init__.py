"""Utility helpers: effective config, device/topology info, logging."""
import ctypes
import os

from .._lib import lib, check


def config():
    """Effective MLSL_* configuration (parsed the same way the core does)."""
    def geti(name, dflt):
        v = os.environ.get(name)
        return int(v) if v not in (None, "") else dflt

    return {
        "log_level": geti("MLSL_LOG_LEVEL", 0),
        "stats": bool(geti("MLSL_STATS", 0)),
        "progress": os.environ.get("MLSL_PROGRESS", "thread"),
        "transport": os.environ.get("MLSL_TRANSPORT", "auto"),
        "num_channels": geti("MLSL_NUM_CHANNELS", geti("MLSL_NUM_SERVERS", 1)),
        "large_msg_size_mb": geti("MLSL_LARGE_MSG_SIZE_MB", 128),
        "large_msg_chunks": geti("MLSL_LARGE_MSG_CHUNKS", 4),
        "allreduce_algo": os.environ.get("MLSL_ALLREDUCE_ALGO", "auto"),
        "msg_priority": bool(geti("MLSL_MSG_PRIORITY", 0)),
        "msg_priority_threshold": geti("MLSL_MSG_PRIORITY_THRESHOLD", 10000),
        "quant_block": geti("MLSL_QUANT_BLOCK", 256),
        "check_pointers": bool(geti("MLSL_CHECK_POINTERS", 0)),
        "timeout_sec": geti("MLSL_TIMEOUT", 300),
    }


def device_count():
    from .. import ops
    return ops.device_count()


def topology():
    """GPU/topology summary (device count; per-device name/memory via torch
    when available)."""
    info = {"device_count": device_count()}
    try:
        import torch
        if torch.cuda.is_available():
            props = torch.cuda.get_device_properties(0)
            info.update({
                "arch": getattr(props, "gcnArchName", "?"),
                "hbm_gb": round(props.total_memory / (1 << 30), 1),
                "cu_count": props.multi_processor_count,
            })
    except ImportError:
        pass
    return info
