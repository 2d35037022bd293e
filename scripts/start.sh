#!/bin/bash
# Canonical launch lines for the five styles (reference start.sh:1-5),
# MI355X edition. Use HIP_VISIBLE_DEVICES to pick GPUs.

# 1) spawn-style DDP (self-launching, tcp rendezvous)
# HIP_VISIBLE_DEVICES=0,1,2,3 python -m amdtrain.cli.multiprocessing_distributed -a resnet50 --synthetic

# 2) launcher-style DDP
# HIP_VISIBLE_DEVICES=0,1,2,3 python -m torch.distributed.run --nnodes=1 --nproc-per-node=4 --master-addr 127.0.0.1 -m amdtrain.cli.distributed -a resnet50 --synthetic

# 3) Apex-style AMP DDP
# HIP_VISIBLE_DEVICES=0,1,2,3 python -m torch.distributed.run --nnodes=1 --nproc-per-node=4 --master-addr 127.0.0.1 -m amdtrain.cli.apex_distributed -a resnet50 --synthetic

# 4) Horovod-style DistributedOptimizer (no MPI needed; self-spawns)
# HIP_VISIBLE_DEVICES=0,1,2,3 python -m amdtrain.cli.horovod_distributed -a resnet50 --synthetic
# 4b) ... or MPI-style, the reference's `horovodrun -np 4 -H localhost:4` lineage
# mpirun -np 4 python -m amdtrain.cli.horovod_distributed -a resnet50 --synthetic

# 5) Slurm multi-node
# srun -N2 --gres gpu:4 python -m amdtrain.cli.distributed_slurm_main --dist-file distfile -a resnet50 --synthetic

# 0) single-process scatter/gather DataParallel (documented-slow parity path)
# HIP_VISIBLE_DEVICES=0,1,2,3 python -m amdtrain.cli.dataparallel -a resnet50 --synthetic
