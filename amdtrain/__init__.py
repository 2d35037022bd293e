"""amdtrain — an MI355X-native single-node data-parallel ImageNet training framework.

Built from scratch on PyTorch-ROCm with hand-written CDNA4 (gfx950) HIP kernels
and RCCL collectives over xGMI.  Provides the same capabilities as the
``tczhangzhi/pytorch-distributed`` reference: five launch
styles (single-process scatter/gather DataParallel, launcher-style DDP,
spawn-style DDP, Apex-style mixed-precision DDP, Horovod-style
DistributedOptimizer), distributed evaluation, the reference CLI surface,
checkpoint dict schema, and meter/progress output.

Layout (mirrors SURVEY.md §1's logical layers):
  amdtrain.config    — shared argparse surface (L6)
  amdtrain.cli       — the six entrypoints (L5)
  amdtrain.comm      — process-group init + collectives over RCCL (L4)
  amdtrain.parallel  — gradient-sync engines: DDP reducer, Horovod-style
                       optimizer, Apex-style AMP, scatter/gather DP (L3)
  amdtrain.engine    — train/validate loops (L2)
  amdtrain.models    — native ResNet family (L1)
  amdtrain.ops       — hand-written gfx950 HIP kernels + CPU references (L1)
  amdtrain.data      — synthetic + ImageFolder pipelines, sampler, prefetcher (L0)
  amdtrain.utils     — meters, metrics, LR schedule, checkpointing (aux)
"""

__version__ = "0.2.0"

from . import utils  # noqa: F401
