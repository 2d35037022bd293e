"""Progress meters with the reference's stdout format.

Behavioral parity with the reference's copy-pasted ``AverageMeter`` /
``ProgressMeter`` (reference: distributed.py:333-371) — value/avg/sum/count
tracking with a printf-style format string, and a tab-joined progress line
``Epoch: [e][ batch/total]\t<meter> <val> (<avg>) ...`` printed every
``--print-freq`` batches.
"""

from __future__ import annotations


class AverageMeter:
    """Tracks current value, running average, sum and count.

    Same observable surface as the reference meter (distributed.py:333-354):
    attributes ``val``, ``avg``, ``sum``, ``count``; ``str()`` renders
    ``"{name} {val<fmt>} ({avg<fmt>})"``.
    """

    def __init__(self, name: str, fmt: str = ":f"):
        self.name = name
        self.fmt = fmt
        self.reset()

    def reset(self) -> None:
        self.val = 0.0
        self.avg = 0.0
        self.sum = 0.0
        self.count = 0

    def update(self, val: float, n: int = 1) -> None:
        self.val = val
        self.sum += val * n
        self.count += n
        self.avg = self.sum / self.count if self.count else 0.0

    def __str__(self) -> str:
        fmtstr = "{name} {val" + self.fmt + "} ({avg" + self.fmt + "})"
        return fmtstr.format(name=self.name, val=self.val, avg=self.avg)


class ProgressMeter:
    """Batch-indexed progress line over a set of AverageMeters.

    Same stdout shape as the reference (distributed.py:357-371):
    ``prefix[ cur/total]`` followed by each meter, tab-joined.
    """

    def __init__(self, num_batches: int, meters, prefix: str = ""):
        self.batch_fmtstr = self._get_batch_fmtstr(num_batches)
        self.meters = list(meters)
        self.prefix = prefix

    def display(self, batch: int) -> str:
        entries = [self.prefix + self.batch_fmtstr.format(batch)]
        entries += [str(m) for m in self.meters]
        line = "\t".join(entries)
        print(line, flush=True)
        return line

    @staticmethod
    def _get_batch_fmtstr(num_batches: int) -> str:
        num_digits = len(str(num_batches // 1))
        fmt = "{:" + str(num_digits) + "d}"
        return "[" + fmt + "/" + fmt.format(num_batches) + "]"
