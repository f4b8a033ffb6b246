"""Run pairwise cosine similarity under rocprofv3 to show the hipBLASLt MFMA GEMM."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import metrics_amd as ma

x = torch.randn(4096, 1024, device="cuda", dtype=torch.bfloat16)
y = torch.randn(4096, 1024, device="cuda", dtype=torch.bfloat16)
for _ in range(5):
    ma.functional.pairwise_cosine_similarity(x, y)
torch.cuda.synchronize()
print("pairwise done")
