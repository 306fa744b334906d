"""tuplex_amd — MI355X-native execution backend for the Tuplex TransformStage hot
path, behind the reference's tuplex.Context / DataSet API (drop-in for that path).

Normal-case rows execute ONLY on the GPU (generated fused HIP kernels via the C-ABI
in include/tpx_abi.h); exception/fallback rows are replayed on the CPython
interpreter and merged in order, exactly as the reference does.
"""
from . import presolve as _presolve
_presolve.warm()  # forkserver before any HIP context
from .context import Context          # noqa: F401
from .dataset import DataSet          # noqa: F401

__version__ = "0.1.0"
