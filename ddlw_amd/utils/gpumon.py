"""Per-GPU utilization/HBM/power sampler — the Ganglia-metrics equivalent
(reference prose ``Part 1 .../04_monitoring_and_optimization.py:25-29``).

Samples ``rocm-smi`` (or amd-smi) periodically on a background thread and
writes JSONL records; cheap enough to run alongside training.
"""
from __future__ import annotations

import json
import shutil
import subprocess
import threading
import time
from typing import Optional


class GpuMonitor:
    def __init__(self, out_path: str, interval_s: float = 2.0):
        self.out_path = out_path
        self.interval_s = interval_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._smi = shutil.which("rocm-smi")

    def _sample(self) -> Optional[dict]:
        if not self._smi:
            return None
        try:
            out = subprocess.run(
                [self._smi, "--showuse", "--showmemuse", "--showpower", "--json"],
                capture_output=True,
                text=True,
                timeout=10,
            )
            return {"t": time.time(), "smi": json.loads(out.stdout or "{}")}
        except Exception as e:
            return {"t": time.time(), "error": str(e)}

    def _loop(self) -> None:
        with open(self.out_path, "a") as f:
            while not self._stop.is_set():
                rec = self._sample()
                if rec:
                    f.write(json.dumps(rec) + "\n")
                    f.flush()
                self._stop.wait(self.interval_s)

    def __enter__(self) -> "GpuMonitor":
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def __exit__(self, *exc) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(5)
