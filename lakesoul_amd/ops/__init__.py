"""Native extension loading.

- ``cpp()``: host core (always required).
- ``hip()``: GPU kernels. On a machine with a GPU the HIP extension is
  REQUIRED — we fail loudly rather than fall back to an eager path, so a
  missing/broken native build can never silently pass GPU tests.
"""

from __future__ import annotations

_cpp = None
_hip = None


def cpp():
    global _cpp
    if _cpp is None:
        try:
            import torch  # noqa: F401  (loads libc10 et al. for the ext)
            from lakesoul_amd import _cpp as mod  # type: ignore
        except ImportError as e:
            raise ImportError(
                "lakesoul_amd._cpp native extension not built. "
                "Run `python setup.py build_ext --inplace` in the repo root."
            ) from e
        _cpp = mod
    return _cpp


def hip():
    global _hip
    if _hip is None:
        try:
            import torch  # noqa: F401  (loads libc10/libtorch_hip for the ext)
            from lakesoul_amd import _hip as mod  # type: ignore
        except ImportError as e:
            raise ImportError(
                "lakesoul_amd._hip HIP extension not built (gfx950). "
                "Run `python setup.py build_ext --inplace` in the repo root. "
                "GPU execution without the HIP kernels is not supported."
            ) from e
        _hip = mod
    return _hip


def hip_available() -> bool:
    try:
        hip()
        return True
    except ImportError:
        return False
