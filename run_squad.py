#!/usr/bin/env python3
"""SQuAD finetuning/inference on MI355X (gfx950).

API-compatible re-design of the reference run_squad.py (call stack:
SURVEY.md §3.3): same flags and artifacts (predictions.json,
nbest_predictions.json, EM/F1). The Apex-AMP-O2 + apex-DDP stack is
replaced MI355X-first: bf16 autocast (or fp32 + BertAdam), torch-DDP
over RCCL, multi-tensor HIP gradient clipping, and the in-repo C++
WordPiece tokenizer. The official evaluate-v1.1 metric runs in-repo
(no network for the script download).
"""

from __future__ import annotations

import argparse
import json
import os
import pickle
import random
import time

import numpy as np
import torch
from torch.utils.data import DataLoader, TensorDataset

from bert_pytorch_amd.config import BertConfig, merge_config_and_args
from bert_pytorch_amd.data import squad as squad_data
from bert_pytorch_amd.data.tokenization import get_wordpiece_tokenizer
from bert_pytorch_amd.models import BertForQuestionAnswering
from bert_pytorch_amd.optim import (
    BertAdam,
    FusedAdam,
    GradientClipper,
    LinearWarmUpScheduler,
)
from bert_pytorch_amd.parallel import comm
from bert_pytorch_amd.utils import MetricLogger


def parse_args(argv=None):
    parser = argparse.ArgumentParser(description="MI355X-native SQuAD runner")
    parser.add_argument("--config_file", type=str, default=None)
    parser.add_argument("--bert_model", type=str, default=None,
                        help="model config JSON (reference: model dir name)")
    parser.add_argument("--model_config_file", type=str, default=None)
    parser.add_argument("--init_checkpoint", type=str, default=None,
                        help="pretraining ckpt_*.pt (dict with 'model')")
    parser.add_argument("--vocab_file", type=str, required=False)
    parser.add_argument("--output_dir", type=str, default="squad_out")
    parser.add_argument("--train_file", type=str, default=None)
    parser.add_argument("--predict_file", type=str, default=None)
    parser.add_argument("--do_train", action="store_true")
    parser.add_argument("--do_predict", action="store_true")
    parser.add_argument("--do_eval", action="store_true")
    parser.add_argument("--train_batch_size", type=int, default=32)
    parser.add_argument("--predict_batch_size", type=int, default=32)
    parser.add_argument("--learning_rate", type=float, default=3e-5)
    parser.add_argument("--num_train_epochs", type=float, default=2.0)
    parser.add_argument("--max_steps", type=float, default=-1)
    parser.add_argument("--warmup_proportion", type=float, default=0.1)
    parser.add_argument("--max_seq_length", type=int, default=384)
    parser.add_argument("--doc_stride", type=int, default=128)
    parser.add_argument("--max_query_length", type=int, default=64)
    parser.add_argument("--n_best_size", type=int, default=20)
    parser.add_argument("--max_answer_length", type=int, default=30)
    parser.add_argument("--gradient_accumulation_steps", type=int, default=1)
    parser.add_argument("--seed", type=int, default=42)
    parser.add_argument("--fp16", action="store_true",
                        help="mixed precision (bf16 autocast on MI355X)")
    parser.add_argument("--bf16", action="store_true")
    parser.add_argument("--do_lower_case", action="store_true", default=True)
    parser.add_argument("--version_2_with_negative", action="store_true")
    parser.add_argument("--null_score_diff_threshold", type=float, default=0.0)
    parser.add_argument("--max_grad_norm", type=float, default=1.0)
    parser.add_argument("--local_rank", type=int,
                        default=int(os.environ.get("LOCAL_RANK", 0)))
    return merge_config_and_args(parser, argv)


def cached_features(args, examples, tokenizer, is_training, split):
    cache = os.path.join(
        args.output_dir,
        f"features_{split}_{args.max_seq_length}_{args.doc_stride}.pkl",
    )
    if os.path.exists(cache):
        with open(cache, "rb") as f:
            return pickle.load(f)
    features = squad_data.convert_examples_to_features(
        examples, tokenizer, args.max_seq_length, args.doc_stride,
        args.max_query_length, is_training,
    )
    if comm.is_main_process():
        os.makedirs(args.output_dir, exist_ok=True)
        with open(cache, "wb") as f:
            pickle.dump(features, f)
    return features


def features_to_dataset(features, is_training):
    t = lambda key, dtype=torch.long: torch.tensor(  # noqa: E731
        [getattr(f, key) for f in features], dtype=dtype
    )
    if is_training:
        return TensorDataset(
            t("input_ids"), t("input_mask"), t("segment_ids"),
            t("start_position"), t("end_position"),
        )
    return TensorDataset(
        t("input_ids"), t("input_mask"), t("segment_ids"),
        torch.arange(len(features)),
    )


def main(args=None):
    if args is None:
        args = parse_args()
    rank, local_rank, world = comm.init_distributed()
    if torch.cuda.is_available():
        from bert_pytorch_amd.utils import tunable  # noqa: PLC0415

        tunable.enable()
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")
    random.seed(args.seed)
    np.random.seed(args.seed)
    torch.manual_seed(args.seed)
    comm.mkdir_by_main_process(args.output_dir)
    log = MetricLogger(
        log_prefix=os.path.join(args.output_dir, "squad_log")
        if comm.is_main_process() else None,
        verbose=comm.is_main_process(),
    )

    cfg_path = args.model_config_file or args.bert_model
    config = BertConfig.from_json_file(cfg_path)
    vocab_file = args.vocab_file or getattr(config, "vocab_file", None)
    tokenizer = get_wordpiece_tokenizer(
        vocab_file, lowercase=getattr(config, "lowercase", args.do_lower_case)
    )

    model = BertForQuestionAnswering(config)
    if args.init_checkpoint:
        state = torch.load(args.init_checkpoint, map_location="cpu",
                           weights_only=False)
        if isinstance(state, dict) and "model" in state:
            state = state["model"]
        state = {k.removeprefix("module."): v for k, v in state.items()}
        missing, unexpected = model.load_state_dict(state, strict=False)
        log.info("loaded init checkpoint (missing=%d unexpected=%d)",
                 len(missing), len(unexpected))
    model.to(device)

    mixed = args.bf16 or args.fp16
    autocast_dtype = torch.bfloat16  # MI355X-preferred

    train_time = 0.0
    n_train_seqs = 0
    if args.do_train:
        examples = squad_data.read_squad_examples(
            args.train_file, is_training=True,
            version_2_with_negative=args.version_2_with_negative,
        )
        features = cached_features(args, examples, tokenizer, True, "train")
        dataset = features_to_dataset(features, True)
        sampler = (
            torch.utils.data.distributed.DistributedSampler(dataset)
            if world > 1 else torch.utils.data.RandomSampler(dataset)
        )
        loader = DataLoader(
            dataset, sampler=sampler,
            batch_size=args.train_batch_size // args.gradient_accumulation_steps,
            num_workers=2, drop_last=True,
        )
        steps_per_epoch = len(loader) // args.gradient_accumulation_steps
        num_steps = (
            int(args.max_steps) if args.max_steps > 0
            else int(steps_per_epoch * args.num_train_epochs)
        )

        no_decay = ("bias", "LayerNorm", "qkv_bias")
        groups = [
            {"params": [p for n, p in model.named_parameters()
                        if not any(d in n for d in no_decay)],
             "weight_decay": 0.01},
            {"params": [p for n, p in model.named_parameters()
                        if any(d in n for d in no_decay)],
             "weight_decay": 0.0},
        ]
        if mixed:
            optimizer = FusedAdam(groups, lr=args.learning_rate,
                                  bias_correction=False)
            scheduler = LinearWarmUpScheduler(
                optimizer, warmup=args.warmup_proportion, total_steps=num_steps
            )
        else:
            optimizer = BertAdam(
                groups, lr=args.learning_rate, warmup=args.warmup_proportion,
                t_total=num_steps,
            )
            scheduler = None
        clipper = GradientClipper(args.max_grad_norm)
        model_ddp = comm.wrap_ddp(model, local_rank)
        model_ddp.train()

        log.info("SQuAD train: %d examples, %d features, %d steps",
                 len(examples), len(features), num_steps)
        start = time.perf_counter()
        global_step = 0
        done = False
        for epoch in range(int(args.num_train_epochs) + 1):
            if done:
                break
            if world > 1:
                sampler.set_epoch(epoch)
            for step, batch in enumerate(loader):
                batch = [t.to(device, non_blocking=True) for t in batch]
                input_ids, input_mask, segment_ids, start_pos, end_pos = batch
                with torch.autocast(device.type, dtype=autocast_dtype,
                                    enabled=mixed and use_cuda):
                    start_logits, end_logits = model_ddp(
                        input_ids, segment_ids, input_mask
                    )
                    ignored = start_logits.size(1)
                    start_pos.clamp_(0, ignored)
                    end_pos.clamp_(0, ignored)
                    loss_fct = torch.nn.CrossEntropyLoss(ignore_index=ignored)
                    loss = (
                        loss_fct(start_logits.float(), start_pos)
                        + loss_fct(end_logits.float(), end_pos)
                    ) / 2
                    loss = loss / args.gradient_accumulation_steps
                loss.backward()
                if (step + 1) % args.gradient_accumulation_steps == 0:
                    clipper.step(model.parameters())
                    if scheduler is not None:
                        scheduler.step()
                    optimizer.step()
                    optimizer.zero_grad()
                    global_step += 1
                    n_train_seqs += args.train_batch_size * world
                    if global_step % 100 == 0 and comm.is_main_process():
                        log.log("train", global_step, loss=float(loss))
                    if global_step >= num_steps:
                        done = True
                        break
        train_time = time.perf_counter() - start
        if comm.is_main_process():
            torch.save(
                {"model": model.state_dict()},
                os.path.join(args.output_dir, "pytorch_model.bin"),
            )

    results = {}
    if args.do_predict and comm.is_main_process():
        examples = squad_data.read_squad_examples(
            args.predict_file, is_training=False,
            version_2_with_negative=args.version_2_with_negative,
        )
        features = cached_features(args, examples, tokenizer, False, "predict")
        dataset = features_to_dataset(features, False)
        loader = DataLoader(dataset, batch_size=args.predict_batch_size)
        model.eval()
        raw_results = []
        infer_start = time.perf_counter()
        with torch.no_grad():
            for input_ids, input_mask, segment_ids, idx in loader:
                input_ids = input_ids.to(device)
                input_mask = input_mask.to(device)
                segment_ids = segment_ids.to(device)
                with torch.autocast(device.type, dtype=autocast_dtype,
                                    enabled=mixed and use_cuda):
                    start_logits, end_logits = model(
                        input_ids, segment_ids, input_mask
                    )
                for i, feat_idx in enumerate(idx.tolist()):
                    raw_results.append(
                        squad_data.RawResult(
                            unique_id=features[feat_idx].unique_id,
                            start_logits=start_logits[i].float().tolist(),
                            end_logits=end_logits[i].float().tolist(),
                        )
                    )
        infer_time = time.perf_counter() - infer_start
        predictions, nbest = squad_data.get_answers(
            examples, features, raw_results,
            n_best_size=args.n_best_size,
            max_answer_length=args.max_answer_length,
            do_lower_case=getattr(config, "lowercase", True),
            version_2_with_negative=args.version_2_with_negative,
            null_score_diff_threshold=args.null_score_diff_threshold,
        )
        with open(os.path.join(args.output_dir, "predictions.json"), "w") as f:
            json.dump(predictions, f, indent=2)
        with open(os.path.join(args.output_dir, "nbest_predictions.json"), "w") as f:
            json.dump(nbest, f, indent=2)
        results["inference_sequences_per_second"] = len(features) / infer_time
        if args.do_eval:
            metrics = squad_data.evaluate_predictions(
                args.predict_file, predictions
            )
            results.update(metrics)
            log.info("exact_match=%.2f F1=%.2f",
                     metrics["exact_match"], metrics["f1"])

    if args.do_train and comm.is_main_process():
        results["e2e_train_time"] = train_time
        results["training_sequences_per_second"] = (
            n_train_seqs / train_time if train_time else 0.0
        )
    if comm.is_main_process():
        log.info("results: %s", json.dumps(results))
    log.close()
    return results


if __name__ == "__main__":
    main()
