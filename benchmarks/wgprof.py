import sys, os, torch
sys.path.insert(0, "/root/repo")
from bert_pytorch_amd.ops import extension
dev = torch.device("cuda")
K, M, N = 12288, 4096, 1024
dy = torch.randn(K, M, device=dev, dtype=torch.bfloat16)
x = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
for _ in range(30):
    out = extension().wgrad_tn(dy, x)
torch.cuda.synchronize()
