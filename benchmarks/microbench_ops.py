#!/usr/bin/env python3
"""Per-op microbenchmarks on MI355X: time each HIP kernel at the
phase-1/phase-2 training shapes, report effective HBM bandwidth
against the measured copy ceiling (~6.3 TB/s) and against the eager
PyTorch composition.

Run (GPU box): python benchmarks/microbench_ops.py [--phase 1]
Prints one JSON line per op.
"""

from __future__ import annotations

import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bert_pytorch_amd import ops  # noqa: E402
from bert_pytorch_amd.ops import (  # noqa: E402
    fused_attention,
    fused_bias_dropout_residual_ln,
    fused_bias_gelu,
    fused_cross_entropy,
    fused_layer_norm,
)


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters * 1e3  # us


def report(name, us, bytes_moved, extra=None):
    rec = {
        "op": name,
        "us": round(us, 2),
        "GB": round(bytes_moved / 1e9, 4),
        "TB_s": round(bytes_moved / (us * 1e-6) / 1e12, 2),
    }
    if extra:
        rec.update(extra)
    print(json.dumps(rec), flush=True)
    return rec


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--phase", type=int, default=1, choices=[1, 2])
    args = p.parse_args()
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    b, s = (96, 128) if args.phase == 1 else (16, 512)
    rows = b * s
    H, FFN, NH, V = 1024, 4096, 16, 30528
    dt = torch.bfloat16
    e = 2  # bytes/elem

    # ---- ceiling: pure copy
    x_big = torch.randn(rows, FFN, device=dev, dtype=dt)
    y_big = torch.empty_like(x_big)
    us = timeit(lambda: y_big.copy_(x_big))
    report("copy_ceiling[rows,4096]", us, 2 * rows * FFN * e)

    # ---- bias-GELU fwd/bwd at FFN shape
    bias = torch.randn(FFN, device=dev, dtype=torch.float32)
    us = timeit(lambda: fused_bias_gelu(x_big, bias))
    report("bias_gelu_fwd[rows,4096]", us, 2 * rows * FFN * e)
    eag = lambda: torch.nn.functional.gelu(x_big.float() + bias).to(dt)  # noqa: E731
    us_e = timeit(eag, iters=20)
    report("bias_gelu_fwd_eager", us_e, 2 * rows * FFN * e)

    xg = x_big.clone().requires_grad_(True)
    yv = fused_bias_gelu(xg, bias)
    dy = torch.randn_like(yv)

    def bwd():
        ext = ops.extension()
        ext.bias_gelu_bwd(dy, x_big, bias)

    us = timeit(bwd)
    report("bias_gelu_bwd[rows,4096]", us, 3 * rows * FFN * e)

    # ---- bdrl fwd/bwd at H shape
    x = torch.randn(rows, H, device=dev, dtype=dt)
    res = torch.randn_like(x)
    g = torch.ones(H, device=dev, dtype=torch.float32)
    bt = torch.zeros(H, device=dev, dtype=torch.float32)
    hb = torch.randn(H, device=dev, dtype=torch.float32)
    ext = ops.extension()
    us = timeit(
        lambda: ext.bias_dropout_residual_ln_fwd(x, hb, res, g, bt, 0.1,
                                                 1e-12, 123, 0)
    )
    # x + res reads, z + y + mask writes
    report("bdrl_fwd[rows,1024]", us, rows * H * (4 * e + 1))
    yy, z, mask, mean, rstd = ext.bias_dropout_residual_ln_fwd(
        x, hb, res, g, bt, 0.1, 1e-12, 123, 0
    )
    dyt = torch.randn_like(yy)
    us = timeit(
        lambda: ext.bias_dropout_residual_ln_bwd(dyt, z, mask, g, mean, rstd,
                                                 0.1, True)
    )
    report("bdrl_bwd[rows,1024]", us, rows * H * (4 * e + 1))

    # ---- LayerNorm fwd/bwd
    w = torch.ones(H, device=dev, dtype=torch.float32)
    us = timeit(lambda: ext.ln_fwd(x, w, bt, 1e-12))
    report("ln_fwd[rows,1024]", us, 2 * rows * H * e)
    yv2, mean2, rstd2 = ext.ln_fwd(x, w, bt, 1e-12)
    us = timeit(lambda: ext.ln_bwd(dyt, x, w, mean2, rstd2))
    report("ln_bwd[rows,1024]", us, 3 * rows * H * e)

    # ---- attention fwd/bwd
    qkv = torch.randn(b, s, 3 * H, device=dev, dtype=dt)
    seqlens = torch.full((b,), s, device=dev, dtype=torch.int32)
    us = timeit(lambda: fused_attention(qkv, seqlens, NH, 0.0, True))
    flops = 4 * b * NH * s * s * 64  # 2 GEMMs of [s,64]x[64,s]
    report("attn_fwd", us, 0, {"TFLOP_s": round(flops / (us * 1e-6) / 1e12, 1)})
    out = fused_attention(qkv.requires_grad_(True), seqlens, NH, 0.0, True)
    do = torch.randn_like(out)
    us = timeit(lambda: torch.autograd.grad(out, qkv, do, retain_graph=True))
    report("attn_bwd", us, 0,
           {"TFLOP_s": round(2.5 * flops / (us * 1e-6) / 1e12, 1)})
    # production path: dropout p=0.1 (bit-packed stored mask: gen kernel
    # in fwd, mask reads in fwd + both bwd kernels)
    us = timeit(lambda: fused_attention(qkv.detach(), seqlens, NH, 0.1, True))
    report("attn_fwd_drop", us, 0,
           {"TFLOP_s": round(flops / (us * 1e-6) / 1e12, 1)})
    out = fused_attention(qkv, seqlens, NH, 0.1, True)
    us = timeit(lambda: torch.autograd.grad(out, qkv, do, retain_graph=True))
    report("attn_bwd_drop", us, 0,
           {"TFLOP_s": round(2.5 * flops / (us * 1e-6) / 1e12, 1)})

    # ---- cross entropy at vocab shape
    logits = torch.randn(rows, V, device=dev, dtype=dt)
    labels = torch.randint(-1, V, (rows,), device=dev)
    lf = logits.requires_grad_(True)
    us = timeit(lambda: fused_cross_entropy(lf, labels, -1), iters=20)
    report("ce_fwd[rows,30528]", us, rows * V * e)

    # ---- GEMM reference points (hipBLASLt through torch.matmul)
    a = torch.randn(rows, H, device=dev, dtype=dt)
    w1 = torch.randn(FFN, H, device=dev, dtype=dt)
    us = timeit(lambda: torch.nn.functional.linear(a, w1))
    report("gemm_ffn1", us, 0,
           {"TFLOP_s": round(2 * rows * H * FFN / (us * 1e-6) / 1e12, 1)})
    wv = torch.randn(V, H, device=dev, dtype=dt)
    us = timeit(lambda: torch.nn.functional.linear(a, wv))
    report("gemm_vocab", us, 0,
           {"TFLOP_s": round(2 * rows * H * V / (us * 1e-6) / 1e12, 1)})


if __name__ == "__main__":
    main()
