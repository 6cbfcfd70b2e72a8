"""Parallelism helpers on top of the comm core.

- GradBucketer: bucketed non-blocking gradient allreduce overlapped with
  backward compute (the reference's Backward1/Backward2/Update split,
  tests/examples/mlsl_test/mlsl_test.cpp:464-528, generalized to arbitrary
  bucket lists — the driver's ResNet-50 overlap config).
- shard math helpers for distributed update (ZeRO-1 ancestor).
"""
from .bucketer import GradBucketer  # noqa: F401

from . import seqpar  # noqa: F401

try:  # torch is optional at import time
    from .ddp import DistributedData  # noqa: F401
    from .zero1 import ShardedOptimizer  # noqa: F401
except ImportError:  # pragma: no cover
    pass
