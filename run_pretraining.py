#!/usr/bin/env python3
"""BERT/RoBERTa pretraining on MI355X (gfx950).

API-compatible with the reference runner (/root/reference/
run_pretraining.py — call stack in SURVEY.md §3.1): same CLI flags, same
training-config JSON keys (CLI > JSON > defaults), same checkpoint
layout/dict schema, same CSV/TensorBoard metric names. The compute path
is MI355X-native: bf16/fp16 autocast + hand-written HIP kernels +
FusedLAMB (HIP multi-tensor) + RCCL DDP over xGMI.

Launch (single node, one process per GPU):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 run_pretraining.py \
        --config_file config/bert_pretraining_phase1_config.json \
        --input_dir DATA --output_dir OUT
"""

from __future__ import annotations

import argparse
import contextlib
import math
import os
import random
import time
from pathlib import Path

import numpy as np
import torch

from bert_pytorch_amd.config import BertConfig, merge_config_and_args
from bert_pytorch_amd.data import DistributedSampler, ShardedPretrainingDataset
from bert_pytorch_amd.models import BertForPreTraining, BertPretrainingCriterion
from bert_pytorch_amd.optim import FusedLAMB, PolyWarmUpScheduler
from bert_pytorch_amd.parallel import comm
from bert_pytorch_amd.utils import MetricLogger, checkpoint as ckpt_io
from bert_pytorch_amd.utils.profiling import StepTimer, torch_profile

MASK_TOKEN_DEFAULT = 103  # BERT [MASK]; overridden by --vocab_file when given


def parse_arguments(args=None) -> argparse.Namespace:
    parser = argparse.ArgumentParser(
        description="MI355X-native BERT pretraining"
    )
    parser.add_argument("--config_file", type=str, default=None,
                        help="training-config JSON; CLI flags override it")
    parser.add_argument("--model_config_file", type=str, default=None)
    parser.add_argument("--input_dir", type=str, default=None,
                        help="directory tree containing *.hdf5 shards")
    parser.add_argument("--output_dir", type=str, default=None)
    parser.add_argument("--log_prefix", type=str, default="pretraining_log")
    parser.add_argument("--local_batch_size", type=int, default=8)
    parser.add_argument("--global_batch_size", type=int, default=64)
    parser.add_argument("--max_steps", type=int, default=1000)
    parser.add_argument("--steps", type=int, default=0,
                        help="stop after this many optimizer steps this run "
                             "(0 = until max_steps)")
    parser.add_argument("--previous_phase_end_step", type=int, default=0)
    parser.add_argument("--max_predictions_per_seq", type=int, default=20)
    parser.add_argument("--masked_token_fraction", type=float, default=0.15)
    parser.add_argument("--learning_rate", type=float, default=6e-3)
    parser.add_argument("--warmup_proportion", type=float, default=0.2843)
    parser.add_argument("--lr_decay", type=str, default="poly",
                        choices=["poly", "linear", "cosine", "constant"])
    parser.add_argument("--optimizer", type=str, default="lamb",
                        choices=["lamb", "adam"],
                        help="fused LAMB (BERT two-phase) or fused Adam "
                             "(RoBERTa single-phase)")
    parser.add_argument("--bf16_weights", action="store_true",
                        help="bf16 matmul/embedding weights + fp32 LN/bias "
                             "under bf16 autocast, fp32 optimizer masters "
                             "for the bf16 params: the per-microbatch "
                             "big-weight autocast casts become no-ops and "
                             "their grads accumulate in bf16 (measured "
                             "+2-3%% seq/s over fp32 weights on MI355X)")
    parser.add_argument("--pure_bf16", action="store_true",
                        help="bf16 model weights + fp32 optimizer masters "
                             "instead of fp32 weights + autocast")
    parser.add_argument("--seed", type=int, default=42)
    parser.add_argument("--fp16", action="store_true",
                        help="fp16 autocast + dynamic GradScaler")
    parser.add_argument("--bf16", action="store_true",
                        help="bf16 autocast (MI355X-preferred; wins over --fp16)")
    parser.add_argument("--num_steps_per_checkpoint", type=int, default=200)
    parser.add_argument("--checkpoint_activations", action="store_true")
    parser.add_argument("--vocab_file", type=str, default=None)
    parser.add_argument("--mask_token_index", type=int, default=None)
    parser.add_argument("--vocab_pad_multiple", type=int, default=64,
                        help="pad vocab for MFMA tile alignment "
                             "(reference pads to 8: run_pretraining.py:237)")
    parser.add_argument("--num_workers", type=int, default=4)
    parser.add_argument("--grad_compress", type=str, default=None,
                        choices=["bf16", "fp16"],
                        help="compressed-gradient all-reduce comm hook")
    parser.add_argument("--kfac", action="store_true")
    parser.add_argument("--kfac_inv_interval", type=int, default=10)
    parser.add_argument("--kfac_factor_interval", type=int, default=1)
    parser.add_argument("--kfac_skip_layers", nargs="+",
                        default=["BertLMPredictionHead", "embedding"])
    parser.add_argument("--profile", type=int, default=0,
                        help="profile N optimizer steps with torch.profiler "
                             "(chrome trace under output_dir/profile) and stop")
    parser.add_argument("--timing_breakdown", action="store_true",
                        help="log per-step data/h2d/fwd/bwd/opt breakdown")
    parser.add_argument("--disable_progress_bar", action="store_true")
    parser.add_argument("--local_rank", type=int,
                        default=int(os.environ.get("LOCAL_RANK", 0)))
    ns = merge_config_and_args(parser, args)
    # bool-ish JSON values
    for key in ("fp16", "bf16", "bf16_weights", "kfac", "disable_progress_bar",
                "checkpoint_activations"):
        setattr(ns, key, bool(getattr(ns, key)))
    return ns


def setup_training(args):
    if torch.cuda.is_available():
        from bert_pytorch_amd.utils import tunable  # noqa: PLC0415

        tunable.enable()
    rank, local_rank, world_size = comm.init_distributed()
    args.local_rank = local_rank
    device = (
        torch.device("cuda", local_rank)
        if torch.cuda.is_available()
        else torch.device("cpu")
    )
    if args.output_dir:
        comm.mkdir_by_main_process(args.output_dir)
    log = MetricLogger(
        log_prefix=(
            os.path.join(args.output_dir, args.log_prefix)
            if args.output_dir
            else None
        )
        if comm.is_main_process()
        else None,
        tensorboard_dir=(
            os.path.join(args.output_dir, "tensorboard")
            if args.output_dir and comm.is_main_process()
            else None
        ),
        verbose=comm.is_main_process(),
    )

    # gradient accumulation math (reference: run_pretraining.py:218-228)
    local_acc = math.ceil(args.global_batch_size / world_size)
    args.accumulation_steps = math.ceil(local_acc / args.local_batch_size)
    effective = args.accumulation_steps * args.local_batch_size * world_size
    if effective != args.global_batch_size:
        log.info(
            "warning: global_batch_size %d not divisible by world %d x local %d"
            " -> effective global batch %d",
            args.global_batch_size, world_size, args.local_batch_size, effective,
        )
    return device, log


def prepare_model(args, device):
    config = BertConfig.from_json_file(args.model_config_file)
    # pad vocab so the MLM-decoder GEMM hits full MFMA tiles
    pad_to = args.vocab_pad_multiple
    if config.vocab_size % pad_to != 0:
        config.vocab_size += pad_to - config.vocab_size % pad_to
    model = BertForPreTraining(config)

    resume_step = 0
    resume_state = None
    if args.output_dir:
        latest = ckpt_io.find_latest(args.output_dir)
        if latest is not None:
            path, resume_step = latest
            resume_state = ckpt_io.load(path)
            model.load_state_dict(
                {
                    k.removeprefix("module."): v
                    for k, v in resume_state["model"].items()
                },
                strict=False,
            )
    model.to(device)
    if args.pure_bf16:
        model.to(torch.bfloat16)
    elif args.bf16_weights and device.type == "cuda":
        for n, prm in model.named_parameters():
            if prm.dim() >= 2 and "LayerNorm" not in n:
                prm.data = prm.data.to(torch.bfloat16)
    if args.checkpoint_activations:
        model.checkpoint_activations(True)
    global_steps = max(0, resume_step - args.previous_phase_end_step)
    model = comm.wrap_ddp(model, args.local_rank,
                          grad_compress=args.grad_compress)
    return model, config, resume_state, global_steps


def prepare_optimizers(args, model, resume_state):
    named = list(model.named_parameters())
    no_decay = ["bias", "LayerNorm.weight", "LayerNorm.bias", "qkv_bias"]
    decay_params = [p for n, p in named if not any(d in n for d in no_decay)]
    nodecay_params = [p for n, p in named if any(d in n for d in no_decay)]
    groups = [
        {"params": decay_params, "weight_decay": 0.01},
        {"params": nodecay_params, "weight_decay": 0.0},
    ]
    if args.optimizer == "adam":  # RoBERTa path (BASELINE config 4)
        from bert_pytorch_amd.optim import FusedAdam  # noqa: PLC0415

        optimizer = FusedAdam(
            groups, lr=args.learning_rate,
            master_weights=args.pure_bf16 or args.bf16_weights,
        )
    else:
        optimizer = FusedLAMB(
            groups, lr=args.learning_rate,
            master_weights=args.pure_bf16 or args.bf16_weights,
        )
    from bert_pytorch_amd.optim import (  # noqa: PLC0415
        ConstantWarmUpScheduler,
        CosineWarmUpScheduler,
        LinearWarmUpScheduler,
    )

    sched_cls = {
        "poly": PolyWarmUpScheduler,
        "linear": LinearWarmUpScheduler,
        "cosine": CosineWarmUpScheduler,
        "constant": ConstantWarmUpScheduler,
    }[args.lr_decay]
    scheduler = sched_cls(
        optimizer, warmup=args.warmup_proportion, total_steps=args.max_steps
    )
    scaler = torch.amp.GradScaler("cuda", enabled=args.fp16 and not args.bf16)

    preconditioner = None
    if args.kfac:
        from bert_pytorch_amd.optim.kfac import KFAC  # noqa: PLC0415

        preconditioner = KFAC(
            model,
            optimizer=optimizer,
            factor_update_interval=args.kfac_factor_interval,
            inv_update_interval=args.kfac_inv_interval,
            skip_layers=args.kfac_skip_layers,
        )

    if resume_state is not None:
        optimizer.load_state_dict(resume_state["optimizer"])
        # two-phase hand-off: the new phase's schedule/hyperparams replace
        # the loaded ones (reference: run_pretraining.py:298-309)
        if args.previous_phase_end_step > 0:
            resumed = max(
                0,
                _resume_step_of(resume_state) - args.previous_phase_end_step,
            )
            for group in optimizer.param_groups:
                group["step"] = resumed
                group["lr"] = args.learning_rate
                group["initial_lr"] = args.learning_rate
            for state in optimizer.state.values():
                state["step"] = resumed
            scheduler.base_lrs = [
                g["initial_lr"] for g in optimizer.param_groups
            ]
        if "scaler" in resume_state and scaler.is_enabled():
            scaler.load_state_dict(resume_state["scaler"])
        if "preconditioner" in resume_state and preconditioner is not None:
            preconditioner.load_state_dict(resume_state["preconditioner"])
    return optimizer, scheduler, scaler, preconditioner


def _resume_step_of(resume_state) -> int:
    for group in resume_state["optimizer"].get("param_groups", []):
        if "step" in group:
            return group["step"]
    return 0


def prepare_dataset(args, resume_state):
    files = sorted(str(p) for p in Path(args.input_dir).rglob("*.hdf5"))
    if not files:
        raise RuntimeError(f"no *.hdf5 shards under {args.input_dir}")

    mask_index = args.mask_token_index
    vocab_size = None
    if args.vocab_file and os.path.isfile(args.vocab_file):
        from bert_pytorch_amd.data.tokenization import get_wordpiece_tokenizer  # noqa: PLC0415

        tok = get_wordpiece_tokenizer(args.vocab_file)
        mask_index = tok.token_to_id("[MASK]")
        vocab_size = tok.vocab_size()
    if mask_index is None:
        mask_index = MASK_TOKEN_DEFAULT
    if vocab_size is None:
        # unpadded true vocab for random-token masking (random replacements
        # must stay < real vocab; reference: run_pretraining.py:370-388)
        model_cfg = BertConfig.from_json_file(args.model_config_file)
        vocab_size = model_cfg.vocab_size

    dataset = ShardedPretrainingDataset(
        files,
        mask_token_index=mask_index,
        max_pred_per_seq=args.max_predictions_per_seq,
        masked_lm_prob=args.masked_token_fraction,
        vocab_size=vocab_size,
        seed=args.seed,
    )
    sampler = DistributedSampler(
        dataset, comm.get_world_size(), rank=comm.get_rank(), seed=args.seed
    )
    if resume_state is not None and "sampler" in resume_state:
        sampler.load_state_dict(resume_state["sampler"])
    loader = torch.utils.data.DataLoader(
        dataset,
        sampler=sampler,
        batch_size=args.local_batch_size,
        num_workers=args.num_workers,
        pin_memory=torch.cuda.is_available(),
        worker_init_fn=comm.WorkerInitObj(args.seed + args.local_rank),
        drop_last=True,
        persistent_workers=args.num_workers > 0,
    )
    return loader, sampler


def forward_backward_pass(model, criterion, scaler, batch, args, sync_grads,
                          autocast_dtype, timer):
    input_ids, segment_ids, input_mask, mlm_labels, nsp_labels = batch
    enabled = autocast_dtype is not None
    with timer.phase("forward"), torch.autocast(
        device_type="cuda" if input_ids.is_cuda else "cpu",
        dtype=autocast_dtype or torch.bfloat16,
        enabled=enabled,
    ):
        scores, seq_rel, gathered_labels = model(
            input_ids, segment_ids, input_mask, masked_lm_labels=mlm_labels,
            max_predictions_per_seq=args.max_predictions_per_seq,
            compute_mlm_loss=True,
        )
        loss = criterion(scores, seq_rel, gathered_labels, nsp_labels)
        loss = loss / args.accumulation_steps
    with timer.phase("backward"):
        if sync_grads or not isinstance(
            model, torch.nn.parallel.DistributedDataParallel
        ):
            scaler.scale(loss).backward()
        else:
            with model.no_sync():
                scaler.scale(loss).backward()
    return loss.detach()


def take_optimizer_step(optimizer, scheduler, scaler, model, preconditioner):
    scheduler.step()
    if preconditioner is not None:
        if scaler.is_enabled():
            scaler.unscale_(optimizer)
        preconditioner.step()
    scaler.step(optimizer)
    scaler.update()
    optimizer.zero_grad(set_to_none=True)


def main(args) -> int:
    device, log = setup_training(args)
    random.seed(args.seed + args.local_rank)
    np.random.seed(args.seed + args.local_rank)
    torch.manual_seed(args.seed + args.local_rank)

    model, config, resume_state, global_steps = prepare_model(args, device)
    criterion = BertPretrainingCriterion(config.vocab_size)
    optimizer, scheduler, scaler, preconditioner = prepare_optimizers(
        args, model, resume_state
    )
    loader, sampler = prepare_dataset(args, resume_state)

    autocast_dtype = None
    if args.pure_bf16:
        autocast_dtype = None  # weights already bf16; no autocast casts
    elif args.bf16:
        autocast_dtype = torch.bfloat16
    elif args.fp16:
        autocast_dtype = torch.float16

    epoch = resume_state["epoch"] if resume_state else 0
    steps_this_run = args.steps if args.steps > 0 else args.max_steps
    start_steps = global_steps

    timer = StepTimer(enabled=args.timing_breakdown)
    profile_stack = contextlib.ExitStack()
    if args.profile > 0:
        steps_this_run = min(steps_this_run, args.profile)
        profile_stack.enter_context(
            torch_profile(
                os.path.join(args.output_dir or ".", "profile"),
                rank=comm.get_rank(),
            )
        )

    micro_step = 0
    accum_loss = 0.0
    window_samples = 0
    window_start = time.perf_counter()
    train_start = None
    done = False

    log.info(
        "pretraining: world=%d local_batch=%d accum=%d global_batch=%d "
        "steps(max/run)=%d/%d resume_step=%d",
        comm.get_world_size(), args.local_batch_size, args.accumulation_steps,
        args.global_batch_size, args.max_steps, steps_this_run, global_steps,
    )

    def timed_iter(data_loader):
        it = iter(data_loader)
        while True:
            with timer.phase("data"):
                try:
                    item = next(it)
                except StopIteration:
                    return
            yield item

    with profile_stack:
        while not done:
            sampler.set_epoch(epoch)
            for batch in timed_iter(loader):
                micro_step += 1
                with timer.phase("h2d"):
                    batch = [t.to(device, non_blocking=True) for t in batch]
                sync_grads = micro_step % args.accumulation_steps == 0
                loss = forward_backward_pass(
                    model, criterion, scaler, batch, args, sync_grads,
                    autocast_dtype, timer,
                )
                accum_loss += float(loss)
                window_samples += batch[0].shape[0]
                if not sync_grads:
                    continue

                with timer.phase("optimizer"):
                    take_optimizer_step(
                        optimizer, scheduler, scaler, model, preconditioner
                    )
                timer.step_end()
                global_steps += 1
                if train_start is None:
                    train_start = time.perf_counter()  # skip step-0 warmup cost

                if global_steps % 10 == 0 or global_steps <= 2:
                    now = time.perf_counter()
                    samples_per_second = (
                        window_samples * comm.get_world_size() / (now - window_start)
                    )
                    if comm.is_main_process():
                        log.log(
                            "train",
                            global_steps + args.previous_phase_end_step,
                            epoch=epoch,
                            average_loss=accum_loss,
                            step_loss=float(loss) * args.accumulation_steps,
                            learning_rate=optimizer.param_groups[0]["lr"],
                            samples_per_second=samples_per_second,
                        )
                    window_start = now
                    window_samples = 0
                accum_loss = 0.0

                if (
                    args.output_dir
                    and global_steps % args.num_steps_per_checkpoint == 0
                    and comm.is_main_process()
                ):
                    _save_checkpoint(
                        args, model, optimizer, sampler, scaler, preconditioner,
                        epoch, global_steps,
                    )
                if (
                    global_steps >= args.max_steps
                    or global_steps - start_steps >= steps_this_run
                ):
                    done = True
                    break
            epoch += 1

    if args.output_dir and comm.is_main_process():
        _save_checkpoint(
            args, model, optimizer, sampler, scaler, preconditioner, epoch,
            global_steps,
        )
    if args.timing_breakdown and comm.is_main_process():
        log.info(timer.format_summary())
    if train_start is not None and global_steps > start_steps + 1:
        elapsed = time.perf_counter() - train_start
        seq_per_sec = (
            args.global_batch_size * (global_steps - start_steps - 1) / elapsed
        )
        log.info("training_seq_per_sec = %.2f", seq_per_sec)
    log.close()
    return global_steps


def _save_checkpoint(args, model, optimizer, sampler, scaler, preconditioner,
                     epoch, global_steps):
    raw = model.module if hasattr(model, "module") else model
    state = {
        "model": raw.state_dict(),
        "optimizer": optimizer.state_dict(),
        "sampler": sampler.state_dict(),
        "epoch": epoch,
    }
    if scaler.is_enabled():
        state["scaler"] = scaler.state_dict()
    if preconditioner is not None:
        state["preconditioner"] = preconditioner.state_dict()
    ckpt_io.save(
        args.output_dir, global_steps + args.previous_phase_end_step, state
    )


if __name__ == "__main__":
    arguments = parse_arguments()
    for required in ("model_config_file", "input_dir"):
        if getattr(arguments, required) in (None, ""):
            raise SystemExit(f"--{required} is required (CLI or config JSON)")
    main(arguments)
