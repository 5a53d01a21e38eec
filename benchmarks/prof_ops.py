#!/usr/bin/env python3
"""Name the ops behind the `vectorized_elementwise_kernel` GPU-time
slice: run a few bench.py-shaped micro-steps under torch.profiler and
print the top ops by device time (op names + call counts), so the
"autocast casts vs grad-accumulation adds vs optimizer plumbing" split
is measured instead of guessed (VERDICT round-1, weak #3).

Run (GPU box): python benchmarks/prof_ops.py [--phase 1] [--steps 3]
"""

from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bert_pytorch_amd.config import BertConfig  # noqa: E402
from bert_pytorch_amd.models import (  # noqa: E402
    BertForPreTraining,
    BertPretrainingCriterion,
)
from bert_pytorch_amd.optim import FusedLAMB  # noqa: E402

PHASES = {
    1: dict(seq_len=128, local_batch=96, max_pred=20),
    2: dict(seq_len=512, local_batch=16, max_pred=80),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--phase", type=int, default=1, choices=[1, 2])
    ap.add_argument("--steps", type=int, default=3, help="accum windows")
    ap.add_argument("--accumulation", type=int, default=8)
    ap.add_argument("--pure_bf16", action="store_true")
    args = ap.parse_args()
    from bert_pytorch_amd.utils import tunable

    tunable.enable()
    phase = PHASES[args.phase]
    device = torch.device("cuda")
    cfg_path = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "config/bert_large_uncased_config.json")
    config = BertConfig.from_json_file(cfg_path)
    if config.vocab_size % 64:
        config.vocab_size += 64 - config.vocab_size % 64
    model = BertForPreTraining(config).to(device)
    if args.pure_bf16:
        model = model.to(torch.bfloat16)
    criterion = BertPretrainingCriterion(config.vocab_size)
    named = list(model.named_parameters())
    no_decay = ("bias", "LayerNorm", "qkv_bias")
    opt = FusedLAMB(
        [
            {"params": [p for n, p in named
                        if not any(d in n for d in no_decay)],
             "weight_decay": 0.01},
            {"params": [p for n, p in named if any(d in n for d in no_decay)],
             "weight_decay": 0.0},
        ],
        lr=1e-3,
        master_weights=args.pure_bf16,
    )
    bsz, seq, mp = phase["local_batch"], phase["seq_len"], phase["max_pred"]
    gen = torch.Generator(device=device).manual_seed(0)
    ids = torch.randint(1000, 30000, (bsz, seq), generator=gen, device=device)
    tt = torch.zeros_like(ids)
    mask = torch.ones_like(ids)
    labels = torch.full((bsz, seq), -1, dtype=torch.long, device=device)
    labels[:, :mp] = torch.randint(1000, 30000, (bsz, mp), generator=gen,
                                   device=device)
    nsp = torch.randint(0, 2, (bsz,), generator=gen, device=device)
    model.train()

    import contextlib

    def micro(i):
        with torch.autocast("cuda", dtype=torch.bfloat16) \
                if not args.pure_bf16 else contextlib.nullcontext():
            s, r, gl = model(ids, tt, mask, masked_lm_labels=labels,
                             max_predictions_per_seq=mp)
            loss = criterion(s, r, gl, nsp) / args.accumulation
        loss.backward()
        if (i + 1) % args.accumulation == 0:
            opt.step()
            opt.zero_grad(set_to_none=True)

    for i in range(2 * args.accumulation):  # warmup
        micro(i)
    torch.cuda.synchronize()

    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 record_shapes=False) as prof:
        for i in range(args.steps * args.accumulation):
            micro(i)
        torch.cuda.synchronize()
    print(prof.key_averages().table(
        sort_by="self_cuda_time_total", row_limit=30, top_level_events_only=False))


if __name__ == "__main__":
    main()
