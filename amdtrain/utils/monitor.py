"""Out-of-band GPU utilization poller (statistics.sh parity, AMD-native).

The reference polls ``nvidia-smi --query-gpu=timestamp,index,memory.total,
memory.used,memory.free,utilization.gpu,utilization.memory -lms 500`` into a
per-variant CSV (statistics.sh:1-4).  The MI355X equivalent polls
``amd-smi``/``rocm-smi`` (or the amdsmi Python bindings when importable) on a
background thread at the same 500 ms cadence into the same CSV columns.
"""

from __future__ import annotations

import csv
import datetime
import subprocess
import threading
from typing import List, Optional


def _query_rocm_smi() -> List[List[str]]:
    """One sample per GPU: [timestamp, index, mem_total, mem_used, mem_free, gpu%, mem%]."""
    ts = datetime.datetime.now().isoformat()
    try:
        out = subprocess.run(
            ["rocm-smi", "--showuse", "--showmemuse", "--showmeminfo", "vram",
             "--csv"],
            capture_output=True, text=True, timeout=5,
        ).stdout
    except (FileNotFoundError, subprocess.TimeoutExpired):
        return []
    rows = []
    reader = csv.DictReader([l for l in out.splitlines() if l.strip()])
    for r in reader:
        dev = r.get("device", "")
        if not dev.startswith("card"):
            continue
        total = r.get("VRAM Total Memory (B)", "")
        used = r.get("VRAM Total Used Memory (B)", "")
        free = ""
        try:
            free = str(int(total) - int(used))
        except ValueError:
            pass
        rows.append([
            ts, dev.replace("card", ""), total, used, free,
            r.get("GPU use (%)", ""), r.get("GFX Activity", r.get("Memory use (%)", "")),
        ])
    return rows


class GpuMonitor:
    """Background 500 ms rocm-smi poller writing a statistics.sh-style CSV."""

    HEADER = ["timestamp", "index", "memory.total", "memory.used",
              "memory.free", "utilization.gpu", "utilization.memory"]

    def __init__(self, csv_path: str, interval_s: float = 0.5):
        self.csv_path = csv_path
        self.interval_s = interval_s
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def _run(self) -> None:
        with open(self.csv_path, "w", newline="") as fh:
            writer = csv.writer(fh)
            writer.writerow(self.HEADER)
            while not self._stop.wait(self.interval_s):
                for row in _query_rocm_smi():
                    writer.writerow(row)
                fh.flush()

    def start(self) -> "GpuMonitor":
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None

    def __enter__(self) -> "GpuMonitor":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()
