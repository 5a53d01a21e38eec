"""A/B microbench: fused MLM-decoder GEMM+CE vs the unfused library path.

Times, at the production gathered-rows shape (P = B*max_pred, V=30528,
K=1024):
  forward:  ext.mlm_head_fwd (one kernel + fold)  vs  F.linear + ext.ce_fwd
  training: full autograd fwd+bwd of ops.mlm_decoder_loss vs the
            unfused scores path (F.linear + fused_cross_entropy)

Run on a GPU box: python benchmarks/mlm_bench.py
"""

import os
import sys
import time

os.environ["BPA_FUSED_MLM"] = "1"  # the in-repo kernel path is opt-in

sys.path.insert(
    0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
)

import torch
import torch.nn.functional as F

from bert_pytorch_amd import ops

DEV = "cuda:0"


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    ext = ops.extension()
    # 2048 = the phase-1 bench micro-batch (1920 = 96x20 preds, kernel-padded)
    for P, V, K in [(1280, 30528, 1024), (2048, 30528, 1024),
                    (2560, 30528, 1024)]:
        torch.manual_seed(0)
        h = (torch.randn(P, K, device=DEV) * 0.5).bfloat16()
        w = (torch.randn(V, K, device=DEV) * 0.05).bfloat16()
        b = torch.randn(V, device=DEV).float()
        labels = torch.randint(0, V, (P,), device=DEV)
        labels[::5] = -1
        flops = 2.0 * P * V * K

        t_fused = timeit(lambda: ext.mlm_head_fwd(h, w, b, labels, -1))

        bb = b.bfloat16()

        def unfused():
            logits = F.linear(h, w, bb)
            ext.ce_fwd(logits, labels, -1)

        t_unf = timeit(unfused)
        t_gemm = timeit(lambda: F.linear(h, w, bb))

        hg = h.clone().requires_grad_(True)
        wg = w.clone().requires_grad_(True)
        bg = b.clone().requires_grad_(True)

        def train_fused():
            loss = ops.mlm_decoder_loss(hg, wg, bg, labels)
            loss.backward()
            hg.grad = wg.grad = bg.grad = None

        def train_unfused():
            loss = ops.fused_cross_entropy(
                F.linear(hg, wg, bg.bfloat16()), labels, -1
            )
            loss.backward()
            hg.grad = wg.grad = bg.grad = None

        t_tf = timeit(train_fused, iters=30)
        t_tu = timeit(train_unfused, iters=30)

        print(
            f"P={P} V={V} K={K}: fwd fused {t_fused:.1f}us"
            f" ({flops / t_fused / 1e9:.0f} TF/s incl CE)"
            f" | unfused linear+ce {t_unf:.1f}us"
            f" (library GEMM alone {t_gemm:.1f}us ="
            f" {flops / t_gemm / 1e9:.0f} TF/s)"
            f" | train fused {t_tf:.1f}us vs unfused {t_tu:.1f}us"
        )


if __name__ == "__main__":
    main()
