import sys

import torch

sys.path.insert(0, "/root/repo")
from bert_pytorch_amd.ops import extension

dev = torch.device("cuda")
P, V, K = 1280, 30528, 1024
h = (torch.randn(P, K, device=dev) * 0.5).bfloat16()
w = (torch.randn(V, K, device=dev) * 0.05).bfloat16()
b = torch.randn(V, device=dev).float()
labels = torch.randint(0, V, (P,), device=dev)
for _ in range(30):
    out = extension().mlm_head_fwd(h, w, b, labels, -1)
# library comparison point in the same trace
bb = b.bfloat16()
for _ in range(30):
    logits = torch.nn.functional.linear(h, w, bb)
torch.cuda.synchronize()
