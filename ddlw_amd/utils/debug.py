"""Debug/sanitizer discipline.

The reference has no race detection; its only concurrency safety is rank-0
gating. Here:

- ``serialize_kernels()`` — run every HIP kernel synchronously
  (AMD_SERIALIZE_KERNEL=3, AMD_SERIALIZE_COPY=3): any kernel fault surfaces
  at its launch site instead of a later sync. Must be set before the first
  HIP call; tests can run under ``DDLW_DEBUG_SERIALIZE=1``.
- ``assert_stream_ordering()`` context — debug assertion that no ddlw op is
  launched while a capture is in progress on another stream.
"""
from __future__ import annotations

import os
from contextlib import contextmanager


def serialize_kernels(enable: bool = True) -> None:
    if enable:
        os.environ["AMD_SERIALIZE_KERNEL"] = "3"
        os.environ["AMD_SERIALIZE_COPY"] = "3"
        os.environ["HIP_LAUNCH_BLOCKING"] = "1"
    else:
        for k in ("AMD_SERIALIZE_KERNEL", "AMD_SERIALIZE_COPY", "HIP_LAUNCH_BLOCKING"):
            os.environ.pop(k, None)


def maybe_enable_from_env() -> None:
    if os.environ.get("DDLW_DEBUG_SERIALIZE", "0") == "1":
        serialize_kernels(True)


@contextmanager
def device_sync_each_op():
    """Context that hipDeviceSynchronizes after every ddlw kernel call —
    narrows a fault to the exact op (used by the kernel parity tests when
    chasing a miscompare)."""
    from ..ops import runtime

    lib = runtime.lib()
    prev = runtime.check

    def checked(status, name):
        prev(status, name)
        if lib is not None:
            if lib.ddlw_device_sync() != 0:
                raise RuntimeError(f"device fault after {name}: {lib.ddlw_last_error().decode()}")

    runtime.check = checked
    try:
        yield
    finally:
        runtime.check = prev
