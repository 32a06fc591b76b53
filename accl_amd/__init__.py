"""accl_amd — MI355X-native collective offload engine (ACCL-class).

A brand-new implementation of the capabilities of Xilinx/ACCL (MPI-like
collectives with a device-resident engine) for AMD MI355X: the collective
microcode is a persistent HIP kernel per GPU, the transport is xGMI
peer-to-peer HBM, and a CPU emulator backend runs the same scheduler source
for hardware-free testing. See docs/DESIGN.md.
"""


def _load_core():
    # Load torch FIRST so its bundled HIP/ROCr libraries own the runtime and
    # _core's libamdhip64.so.7 / libhsa-runtime64.so.1 needs resolve to them.
    # With /opt/rocm's copies loaded first instead, two ROCr instances end up
    # in the process and the second one to initialize sees zero GPUs.
    try:
        import torch  # noqa: F401
    except Exception:
        pass
    try:
        from . import _core
        return _core
    except ImportError as e:
        raise ImportError(
            "accl_amd._core is not built. Run `python -m accl_amd.build` "
            "(requires hipcc / ROCm)."
        ) from e


_core = _load_core()

DataType = _core.DataType
ReduceFunction = _core.ReduceFunction
TAG_ANY = _core.TAG_ANY
GLOBAL_COMM = _core.GLOBAL_COMM
error_to_string = _core.error_to_string

from .accl import (ACCL, emu_job_name, generate_ranks,  # noqa: E402,F401
                   load_tuning)

__all__ = [
    "ACCL", "DataType", "ReduceFunction", "TAG_ANY", "GLOBAL_COMM",
    "error_to_string", "emu_job_name", "generate_ranks", "load_tuning",
]
