"""Common client types (parity: python/scannerpy/common.py)."""
import enum
import math
import os


class DeviceType(enum.IntEnum):
    CPU = 0
    GPU = 1


class CacheMode(enum.IntEnum):
    """What to do when an output table already exists (parity:
    scannerpy CacheMode, client.py:1386-1432)."""
    Error = 0
    Ignore = 1     # skip streams whose outputs are committed (resume story)
    Overwrite = 2


class ColumnType(enum.IntEnum):
    Bytes = 0
    Video = 1


class FrameType(enum.IntEnum):
    U8 = 0
    U16 = 1
    F32 = 2
    F64 = 3


class ScannerException(Exception):
    pass


class SliceList(list):
    """Per-slice-group sampling args (parity: scannerpy SliceList) — pass as
    a per-stream arg to a streams op downstream of Slice to give each slice
    group its own sampling."""
    pass


class PerfParams:
    """Performance knobs (parity: scannerpy PerfParams, common.py:78-129).

    work_packet_size: rows per kernel-feed batch through the pipeline.
    io_packet_size: rows per task / storage item.
    pipeline_instances_per_node: parallel copies of the graph per worker
        (one per GPU for GPU graphs).
    cpu_pool / gpu_pool: pool allocator sizes in bytes (0 = no pool, direct
        system allocations).
    """

    def __init__(self, work_packet_size=16, io_packet_size=128,
                 cpu_pool=0, gpu_pool=0, pipeline_instances_per_node=None,
                 load_sparsity_threshold=8, queue_size_per_pipeline=4,
                 profiler_level=1, num_load_workers=0, span_cache=0):
        self.work_packet_size = int(work_packet_size)
        self.io_packet_size = int(io_packet_size)
        if self.work_packet_size < 1 or self.io_packet_size < 1:
            # 0 used to reach the engine and die as std::bad_alloc deep in
            # task partitioning; fail at construction with a real message
            raise ScannerException(
                "work_packet_size and io_packet_size must be >= 1 "
                f"(got {self.work_packet_size}, {self.io_packet_size})")
        if self.io_packet_size < self.work_packet_size:
            raise ScannerException(
                "io_packet_size must be >= work_packet_size "
                f"(got io={self.io_packet_size} < "
                f"work={self.work_packet_size})")
        self.cpu_pool = int(cpu_pool)
        self.gpu_pool = int(gpu_pool)
        self.pipeline_instances_per_node = pipeline_instances_per_node
        self.load_sparsity_threshold = int(load_sparsity_threshold)
        self.queue_size_per_pipeline = int(queue_size_per_pipeline)
        self.profiler_level = int(profiler_level)
        self.num_load_workers = int(num_load_workers)  # 0 = auto
        # HBM span-cache budget in bytes: compressed input GOP spans stay
        # resident in device memory across tasks/jobs, so each span is read
        # from storage and crosses PCIe once. 0 = auto (gpu_pool/3, max
        # 16 GB); 1 = disabled.
        self.span_cache = int(span_cache)

    @classmethod
    def manual(cls, work_packet_size, io_packet_size, **kw):
        return cls(work_packet_size=work_packet_size,
                   io_packet_size=io_packet_size, **kw)

    @classmethod
    def estimate(cls, total_rows=None, element_size=None, n_gpus=None,
                 **kw):
        """Heuristic auto-tuner (parity: PerfParams.estimate common.py:149,
        which probes GPUtil/psutil).

        Sizes packets so a task's working set stays within a fraction of
        memory, and — when a GPU is visible and the caller didn't choose —
        sizes the GPU pool from the device's actual free HBM so steady-state
        allocation never touches the driver (288 GB per MI355X: default to
        an 1/8 slab per pipeline-instance set, leaving room for 8 ranks per
        node).
        """
        element_size = element_size or (1920 * 1080 * 3)
        # keep one task's decoded frames under ~2 GB
        io = max(16, min(512, (2 << 30) // max(1, element_size)))
        work = max(4, min(64, io // 4))
        if total_rows is not None:
            io = min(io, max(1, int(math.ceil(total_rows / 4))))
            work = min(work, io)
        if "gpu_pool" not in kw:
            try:
                from . import _core
                if _core.have_gpu():
                    free = _core.gpu_free_memory(0)
                    kw["gpu_pool"] = min(free // 2, 32 << 30)
                    kw.setdefault("cpu_pool", 4 << 30)
            except Exception:
                pass
        return cls(work_packet_size=work, io_packet_size=io, **kw)

    def to_dict(self, n_instances=1):
        return {
            "io_packet_size": self.io_packet_size,
            "work_packet_size": self.work_packet_size,
            "pipeline_instances": n_instances,
            "cpu_pool_size": self.cpu_pool,
            "gpu_pool_size": self.gpu_pool,
            "sparsity_threshold": self.load_sparsity_threshold,
            "profiler_level": self.profiler_level,
            "load_workers": self.num_load_workers,
            "span_cache_size": self.span_cache,
        }


def default_machine_params():
    """Probe CPUs/GPUs (parity: default_machine_params database.cpp)."""
    from . import _core
    return {
        "num_cpus": os.cpu_count() or 1,
        "gpu_ids": list(range(_core.gpu_device_count())),
    }
