#!/usr/bin/env python3
"""Flagship benchmark: BERT-Large pretraining step throughput on MI355X.

Measures the reference's headline metric (sequences/sec, BASELINE.json:
BERT-Large phase-1 seq128 / phase-2 seq512, bf16, LAMB) on synthetic
masked-LM data with random-init weights.

One "step" = one micro-batch forward+backward at the phase's local batch
size; the fused-LAMB optimizer step (+ DDP all-reduce when N>1) fires
every `--accumulation` micro-steps INSIDE the timed region. The named
configs reach global batch 65536/32768 via hundreds of accumulation
micro-steps per update; the bench default (accumulation=8) runs the
optimizer and gradient all-reduce 8-85x MORE often per sample than the
named config, so the reported seq/s is conservative w.r.t. it.

Launch: python bench.py --gpus N --steps K --warmup W
(N>1 comes via torch.distributed.run; RANK/LOCAL_RANK/WORLD_SIZE env.)
"""

from __future__ import annotations

import argparse
import contextlib
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from bert_pytorch_amd.config import BertConfig  # noqa: E402
from bert_pytorch_amd.models import (  # noqa: E402
    BertForPreTraining,
    BertPretrainingCriterion,
)
from bert_pytorch_amd.optim import FusedLAMB, PolyWarmUpScheduler  # noqa: E402
from bert_pytorch_amd.parallel import comm  # noqa: E402

PHASES = {
    # BERT-Large two-phase LAMB (BASELINE configs 2-3)
    1: dict(seq_len=128, local_batch=96, max_pred=20, lr=6e-3,
            named_global=65536),
    2: dict(seq_len=512, local_batch=16, max_pred=80, lr=4e-3,
            named_global=32768),
    # RoBERTa-Large single-phase fused-Adam (BASELINE config 4):
    # seq 512, local 16, global 8192, lr 4e-4 linear decay
    3: dict(seq_len=512, local_batch=16, max_pred=80, lr=4e-4,
            named_global=8192),
}


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=32)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--phase", type=int, default=1, choices=[1, 2, 3],
                   help="1/2 = BERT-Large seq128/seq512 LAMB; "
                        "3 = RoBERTa-Large seq512 fused-Adam (no NSP)")
    p.add_argument("--optimizer", type=str, default=None,
                   choices=["lamb", "adam"],
                   help="default: lamb (phases 1-2), adam (phase 3)")
    p.add_argument("--kfac", action="store_true",
                   help="K-FAC preconditioner before the optimizer step "
                        "(BASELINE config 5)")
    p.add_argument("--accumulation", type=int, default=8,
                   help="optimizer cadence in micro-steps; 0 = the named "
                        "config's full accumulation (update may not fire "
                        "within a short run)")
    p.add_argument("--local_batch", type=int, default=0)
    p.add_argument("--model_config", type=str,
                   default="config/bert_large_uncased_config.json")
    p.add_argument("--grad_compress", type=str, default=None,
                   choices=["bf16", "fp16"],
                   help="compressed-gradient all-reduce comm hook")
    p.add_argument("--window_autocast", action="store_true",
                   help="hold one autocast region per accumulation window "
                        "so weight-cast caching spans its micro-steps")
    p.add_argument("--pure_bf16", action="store_true",
                   help="bf16 model weights + fp32 LAMB masters instead of "
                        "fp32 weights + autocast (no per-microbatch weight "
                        "casts; bf16 gradient all-reduce)")
    p.add_argument("--fp32_weights", action="store_true",
                   help="fp32 weights + per-microbatch autocast casts (the "
                        "pre-round-2 default)")
    p.add_argument("--bf16_weights", action="store_true", default=True,
                   help="bf16 MATMUL/embedding weights only (LN params and "
                        "biases stay fp32) under bf16 autocast: the per-"
                        "microbatch big-weight casts become no-ops and their "
                        "grads accumulate in bf16 at half the bytes, while "
                        "the fused kernels keep their fp32-param call "
                        "pattern (avoids pure_bf16's per-call grad "
                        "downcasts). fp32 LAMB masters for the bf16 params.")
    p.add_argument("--seed", type=int, default=1234)
    return p.parse_args()


def make_batch(gen, device, bsz, seq, vocab, max_pred):
    lo = min(1000, vocab - 1)
    ids = torch.randint(lo, vocab, (bsz, seq), generator=gen, device=device)
    ids[:, 0] = 101
    tt = torch.zeros_like(ids)
    tt[:, seq // 2 :] = 1
    mask = torch.ones_like(ids)
    labels = torch.full((bsz, seq), -1, dtype=torch.long, device=device)
    pos = torch.rand(bsz, seq, generator=gen, device=device).argsort(dim=1)[:, :max_pred]
    vals = torch.randint(lo, vocab, (bsz, max_pred), generator=gen, device=device)
    labels.scatter_(1, pos, vals)
    ids.scatter_(1, pos, torch.full_like(vals, 103))
    nsp = torch.randint(0, 2, (bsz,), generator=gen, device=device)
    return ids, tt, mask, labels, nsp


def main():
    args = parse_args()
    if torch.cuda.is_available():
        from bert_pytorch_amd.utils import tunable
        tunable.enable()
    rank, local_rank, world = comm.init_distributed()
    if args.gpus > 1:
        assert world == args.gpus, f"WORLD_SIZE {world} != --gpus {args.gpus}"
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")
    torch.manual_seed(args.seed + rank)

    phase = PHASES[args.phase]
    seq = phase["seq_len"]
    bsz = args.local_batch or phase["local_batch"]
    accum = args.accumulation
    if accum <= 0:
        accum = max(1, -(-phase["named_global"] // (world * bsz)))

    opt_name = args.optimizer or ("adam" if args.phase == 3 else "lamb")
    if args.phase == 3 and args.model_config.endswith(
            "bert_large_uncased_config.json"):
        args.model_config = "config/roberta_large_cased_config.json"
    cfg_path = args.model_config
    if not os.path.exists(cfg_path):
        # resolve relative to this file so bench.py works from any cwd
        here = os.path.join(os.path.dirname(os.path.abspath(__file__)), cfg_path)
        if os.path.exists(here):
            cfg_path = here
    config = BertConfig.from_json_file(cfg_path)
    if config.vocab_size % 64:
        config.vocab_size += 64 - config.vocab_size % 64
    model = BertForPreTraining(config).to(device)
    if args.fp32_weights or args.pure_bf16:
        args.bf16_weights = False
    if args.pure_bf16:
        model = model.to(torch.bfloat16)
    elif args.bf16_weights and use_cuda:
        # CUDA-only mode (CPU smoke runs keep fp32: no autocast to
        # coordinate mixed dtypes there)
        for n, prm in model.named_parameters():
            if prm.dim() >= 2 and "LayerNorm" not in n:
                prm.data = prm.data.to(torch.bfloat16)
    criterion = BertPretrainingCriterion(config.vocab_size)
    model = comm.wrap_ddp(model, local_rank,
                          grad_compress=args.grad_compress)
    named = list(model.named_parameters())
    no_decay = ("bias", "LayerNorm", "qkv_bias")
    groups = [
        {"params": [p for n, p in named if not any(d in n for d in no_decay)],
         "weight_decay": 0.01},
        {"params": [p for n, p in named if any(d in n for d in no_decay)],
         "weight_decay": 0.0},
    ]
    if opt_name == "adam":
        from bert_pytorch_amd.optim import FusedAdam  # noqa: PLC0415

        optimizer = FusedAdam(
            groups, lr=phase["lr"],
            master_weights=args.pure_bf16 or args.bf16_weights,
        )
    else:
        optimizer = FusedLAMB(
            groups, lr=phase["lr"],
            master_weights=args.pure_bf16 or args.bf16_weights,
        )
    scheduler = PolyWarmUpScheduler(optimizer, warmup=0.2843, total_steps=7038)
    preconditioner = None
    if args.kfac:
        from bert_pytorch_amd.optim.kfac import KFAC  # noqa: PLC0415

        preconditioner = KFAC(
            model, optimizer, inv_update_interval=10,
            skip_layers=["BertLMPredictionHead", "embedding"],
        )
    gen = torch.Generator(device=device).manual_seed(args.seed + rank)
    # BERT uncased real vocab (30522) unless the model config is smaller
    vocab_unpadded = min(30522, config.vocab_size)

    ac_enabled = use_cuda and not args.pure_bf16

    class WindowAutocast:
        """Hold one autocast region open across an accumulation window
        so torch's weight-cast cache is reused for all its micro-steps
        (weights only change at the optimizer step, i.e. the window
        boundary). Saves the per-microbatch fp32->bf16 weight casts."""

        def __init__(self):
            self._ctx = None

        def cycle(self, micro_idx: int) -> None:
            if not (ac_enabled and args.window_autocast):
                return
            if micro_idx % accum == 0:
                self.close()
                self._ctx = torch.autocast(device.type, dtype=torch.bfloat16)
                self._ctx.__enter__()

        def close(self) -> None:
            if self._ctx is not None:
                self._ctx.__exit__(None, None, None)
                self._ctx = None

    wa = WindowAutocast()
    per_step_ac = ac_enabled and not args.window_autocast

    def one_step(micro_idx: int) -> None:
        batch = make_batch(gen, device, bsz, seq, vocab_unpadded, phase["max_pred"])
        ids, tt, mask, labels, nsp = batch
        sync = (micro_idx + 1) % accum == 0
        wa.cycle(micro_idx)
        with torch.autocast(device.type, dtype=torch.bfloat16,
                            enabled=per_step_ac) if per_step_ac or not ac_enabled \
                else contextlib.nullcontext():
            scores, rel, glabels = model(
                ids, tt, mask, masked_lm_labels=labels,
                max_predictions_per_seq=phase["max_pred"],
                compute_mlm_loss=True,
            )
            loss = criterion(scores, rel, glabels, nsp) / accum
        if sync or not isinstance(model, torch.nn.parallel.DistributedDataParallel):
            loss.backward()
        else:
            with model.no_sync():
                loss.backward()
        if sync:
            scheduler.step()
            if preconditioner is not None:
                preconditioner.step()
            optimizer.step()
            optimizer.zero_grad(set_to_none=True)

    model.train()
    for i in range(args.warmup):
        one_step(i)
    wa.close()

    comm.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        one_step(i)
    if use_cuda:
        torch.cuda.synchronize()
    wa.close()
    comm.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64)
    if world > 1:
        t = t.to(device) if use_cuda else t
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
    elapsed_max = float(t[0])

    value = world * bsz * args.steps / elapsed_max
    if rank == 0:
        result = {
            "metric": "sequences/sec",
            "value": round(value, 2),
            "unit": "seq/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed_max / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_cuda else "fp32(cpu)",
            "data": "synthetic",
            "config": {
                "model": f"{os.path.splitext(os.path.basename(args.model_config))[0]} "
                         f"({config.num_hidden_layers}L/{config.hidden_size}H/"
                         f"{config.num_attention_heads}h, vocab {config.vocab_size})",
                "phase": args.phase,
                "global_batch": world * bsz * accum,
                "named_global_batch": phase["named_global"],
                "seq_len": seq,
                "local_batch": bsz,
                "accumulation": accum,
                "optimizer": ("FusedAdam" if opt_name == "adam"
                              else "FusedLAMB") + " (HIP multi-tensor)"
                + (" + fp32 masters"
                   if args.pure_bf16 or args.bf16_weights else "")
                + (" + K-FAC" if args.kfac else ""),
                "weights": "bf16 (fp32 LAMB masters)" if args.pure_bf16
                else ("bf16 matmul weights + fp32 LN/bias (fp32 LAMB "
                      "masters)" if args.bf16_weights
                      else "fp32 + bf16 autocast"),
                "parallelism": f"dp{world}",
            },
        }
        print(json.dumps(result))
    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
