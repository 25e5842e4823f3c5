"""GPU runtime loader.

Loads the in-tree ``demodel_amd._hip`` extension (built by
demodel_amd/build.py, .so travels with the repo snapshot).  On a machine
WITH a GPU a missing/broken extension is a hard error — the engine must
never fall back to a silent CPU path there.  On CPU-only machines
``have_gpu()`` is False and callers use host landing targets.
"""

from __future__ import annotations

import os

_hip = None
_import_error: Exception | None = None

# The engine runs concurrent long kernels on many streams (per-shard
# decodes, per-worker verify/copy streams); ROCm multiplexes streams
# onto GPU_MAX_HW_QUEUES hardware queues (default 4) and kernels that
# share a queue SERIALIZE — measured: per-shard zstd decodes ran
# back-to-back instead of concurrently.  Must be set before the HIP
# runtime initializes (i.e. before torch loads).
os.environ.setdefault("GPU_MAX_HW_QUEUES", "8")

# torch MUST load before our extension: torch-rocm wheels bundle their own
# libamdhip64, and whichever HIP runtime initializes second in a process
# sees no devices (observed on MI355X: _hip-first broke torch.cuda,
# torch-first broke _hip when a different copy got mapped).  Importing
# torch first means the dynamic loader resolves our DT_NEEDED libamdhip64
# soname to torch's already-loaded copy — one runtime, shared state.
try:
    import torch  # noqa: F401
except Exception:
    pass

try:
    from demodel_amd import _hip as _hip  # type: ignore
except Exception as e:  # pragma: no cover - exercised only on broken builds
    _import_error = e


def _torch_has_gpu() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:
        return False


def have_gpu() -> bool:
    if _hip is None:
        if _torch_has_gpu():
            raise RuntimeError(
                "demodel_amd._hip failed to import on a GPU machine: "
                f"{_import_error!r}. Run python -m demodel_amd.build."
            )
        return False
    if _hip.device_count() > 0:
        return True
    # A GPU torch sees but our runtime doesn't is a broken install, not a
    # CPU box — never fall back silently (the HIP runtime can report
    # no-device transiently before first context init; poke torch first).
    if _torch_has_gpu():
        import torch

        torch.cuda.init()
        n, err = _hip.device_probe()
        if n > 0:
            return True
        env = {k: v for k, v in os.environ.items()
               if "VISIBLE" in k or k.startswith(("HSA", "ROCR", "HIP",
                                                  "GPU"))}
        raise RuntimeError(
            f"torch sees a GPU but demodel_amd._hip does not "
            f"(hipGetDeviceCount: {err}); refusing CPU fallback. "
            f"env={env} kfd={os.path.exists('/dev/kfd')}")
    return False


def hip():
    """The raw extension module; raises loudly if unavailable."""
    if _hip is None:
        raise RuntimeError(
            f"demodel_amd._hip is not available: {_import_error!r}. "
            "Build it with `python -m demodel_amd.build`."
        )
    return _hip
