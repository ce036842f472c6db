"""scanner_amd — an MI355X-native distributed video-analysis dataflow
engine with the capabilities of Scanner (scanner-research/scanner,
SIGGRAPH 2018), built from scratch for ROCm/HIP/CDNA4.

See SURVEY.md at the repo root for the reference's structural analysis this
build follows, and README.md for the architecture."""

from .common import (CacheMode, ColumnType, DeviceType, FrameType,
                     PerfParams, ScannerException, SliceList,
                     default_machine_params)
from .client import Client
from .job import Job
from .op import Kernel, register_python_op
from .storage import NamedStream, NamedVideoStream
from . import types


def __getattr__(name):
    # `parallel` imports torch (~1.5 s cold): load it lazily so clients,
    # workers and kernel subprocesses that never touch the RCCL data plane
    # don't pay for it (PEP 562). importlib, not `from . import`: the
    # latter re-enters this __getattr__ through _handle_fromlist.
    if name == "parallel":
        import importlib
        return importlib.import_module(".parallel", __name__)
    raise AttributeError(f"module 'scanner_amd' has no attribute '{name}'")

__version__ = "0.1.0"

__all__ = [
    "CacheMode", "Client", "ColumnType", "DeviceType", "FrameType", "Job",
    "Kernel", "NamedStream", "NamedVideoStream", "PerfParams",
    "ScannerException", "default_machine_params", "parallel",
    "register_python_op", "types",
]
