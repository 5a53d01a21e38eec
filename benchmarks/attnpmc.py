import sys

import torch

sys.path.insert(0, "/root/repo")
from bert_pytorch_amd.ops import extension

dev = torch.device("cuda")
for (B, S, NH) in [(96, 128, 16), (16, 512, 16)]:
    H = NH * 64
    qkv = (torch.randn(B, S, 3 * H, device=dev) * 0.5).bfloat16()
    seqlens = torch.full((B,), S, device=dev, dtype=torch.int32)
    dout = torch.randn(B, S, H, device=dev).bfloat16()
    for _ in range(15):
        out, lse, mask = extension().attention_fwd(qkv, seqlens, NH, 0.1, 1, 2)
        extension().attention_bwd(dout, qkv, seqlens, out, lse, mask, NH,
                                  0.1, 1, 2)
torch.cuda.synchronize()
