import os
import sys

import torch

sys.path.insert(0, "/root/repo")
from bert_pytorch_amd.ops import extension

dev = torch.device("cuda")
# the three routed production shapes (phase-1 K)
for (K, M, N) in [(12288, 3072, 1024), (12288, 4096, 1024), (12288, 1024, 1024)]:
    dy = torch.randn(K, M, device=dev, dtype=torch.bfloat16)
    x = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
    for _ in range(20):
        extension().wgrad_tn(dy, x)
torch.cuda.synchronize()
