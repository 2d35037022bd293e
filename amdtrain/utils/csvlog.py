"""Per-epoch wall-clock CSV logging.

Parity with the reference's epoch-time CSVs (dataparallel.py:188,207-213 and
distributed_slurm_main.py:209,229-235): one row per epoch with the epoch
index and elapsed seconds, appended to a file named after the launch style.
"""

from __future__ import annotations

import csv
import os
import time
from typing import Optional


class EpochTimer:
    """Context-style epoch timer that appends ``epoch,seconds`` rows to a CSV."""

    def __init__(self, path: Optional[str]):
        self.path = path
        self._t0 = 0.0
        if path:
            new = not os.path.exists(path)
            self._fh = open(path, "a", newline="")
            self._writer = csv.writer(self._fh)
            if new:
                self._writer.writerow(["epoch", "seconds"])
                self._fh.flush()
        else:
            self._fh = None
            self._writer = None

    def start(self) -> None:
        self._t0 = time.time()

    def stop(self, epoch: int) -> float:
        dt = time.time() - self._t0
        if self._writer is not None:
            self._writer.writerow([epoch, f"{dt:.3f}"])
            self._fh.flush()
        return dt

    def close(self) -> None:
        if self._fh is not None:
            self._fh.close()
            self._fh = None
