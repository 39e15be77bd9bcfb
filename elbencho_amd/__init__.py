"""elbencho_amd — an MI355X-native distributed storage benchmark.

A from-scratch rebuild of the capability surface of breuner/elbencho
(files, block devices, object storage; distributed service/master mode)
designed for AMD Instinct MI355X nodes:

  * C++ multithreaded I/O engine with io_uring async depth (``_core``),
  * hand-written gfx950 HIP kernels for GPU buffer fill and integrity
    verify, staged through pinned host buffers with hipMemcpyAsync on
    per-worker streams into 288 GB of HBM3E per GPU,
  * RCCL-over-xGMI phase barrier + stats all-reduce for multi-GPU
    lockstep runs (``elbencho_amd.parallel``), with elbencho's HTTP
    service/master control plane for multi-node compatibility.
"""

__version__ = "0.1.0"

VERSION = __version__
HTTP_PROTOCOL_VERSION = "1.0.0"  # master<->service wire compatibility


def load_core():
    """Import the native engine, building it in-tree if necessary.

    torch is imported first when present: PyTorch bundles its own HIP
    runtime, and loading _core's /opt/rocm runtime before torch's leaves
    hipGetDeviceCount reporting "no ROCm-capable device" once torch loads
    (two HSA runtimes in one process). torch-first keeps a single runtime.
    """
    try:
        import torch  # noqa: F401
    except ImportError:
        pass

    try:
        from elbencho_amd import _core  # type: ignore
        return _core
    except ImportError:
        from elbencho_amd.build import build

        build()
        from elbencho_amd import _core  # type: ignore

        return _core
