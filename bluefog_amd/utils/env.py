# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Environment-variable configuration surface (reference analog:
docs/env_variable.rst). All knobs are read lazily so tests can monkeypatch.

Data-plane knobs:
- ``BLUEFOG_FUSION_THRESHOLD``: flat-bucket size in bytes for the optimizer
  wrappers (default 64 MiB — sized for xGMI links at ~153 GB/s, where a
  64 MiB transfer is ~0.4 ms and comfortably overlaps one backward stage;
  the reference default of 8 MiB is a PCIe/25GbE-era number).
- ``BLUEFOG_ALLOW_TORCH_FALLBACK``: if "1", GPU post-ops may silently fall
  back to torch elementwise chains when the HIP extension is missing.
  Default off: on a ROCm box the native kernels must load or ops raise.
- ``BLUEFOG_WIN_ON_GPU``: if "0", window buffers for GPU tensors are kept on
  CPU (debug aid; reference mpi_win_ops.cc:52-54). The reference's
  ``BLUEFOG_OPS_ON_CPU`` staging knob is intentionally not carried over:
  ops run where the tensor lives (docs/env_variables.md).
"""

import os


def _int_env(name: str, default: int) -> int:
    v = os.environ.get(name)
    return int(v) if v else default


def _bool_env(name: str, default: bool) -> bool:
    v = os.environ.get(name)
    if v is None:
        return default
    return v not in ("0", "false", "False", "")


def fusion_threshold_bytes() -> int:
    # BLUEFOG_FUSION_THRESHOLD is the reference's knob name
    # (operations.cc:474-487); BLUEFOG_BUCKET_BYTES is the alias used in
    # this framework's docs — either works
    v = os.environ.get("BLUEFOG_BUCKET_BYTES")
    if v:
        return int(v)
    return _int_env("BLUEFOG_FUSION_THRESHOLD", 64 * 1024 * 1024)


def allow_torch_fallback() -> bool:
    return _bool_env("BLUEFOG_ALLOW_TORCH_FALLBACK", False)


def win_on_gpu() -> bool:
    return _bool_env("BLUEFOG_WIN_ON_GPU", True)


def timeline_path() -> str:
    return os.environ.get("BLUEFOG_TIMELINE", "")
