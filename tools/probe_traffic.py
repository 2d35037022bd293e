"""DO NOT RUN under rocprofv3 --pmc FETCH_SIZE,WRITE_SIZE: that counter
combination crashed rocprofv3 (signal 6) and hung its signal handler for
the full timeout on 2026-09-12 (cost 10 GPU-minutes).  Plain timing only.
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from amdtrain import _C
torch.manual_seed(0)
M, N, K = 1605632, 64, 64   # L1 c1 shape: A 205MB, B tiny, C 205MB
A = torch.randn(M, K, device="cuda").bfloat16()
B = torch.randn(N, K, device="cuda").bfloat16()
for _ in range(3):
    y = _C.gemm_bt(A, B, False)
torch.cuda.synchronize()
print("expected: A read 205MB + C write 205MB per call, 3 calls after 3 warmup")
