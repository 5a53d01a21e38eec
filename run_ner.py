#!/usr/bin/env python3
"""NER (token classification) finetuning on MI355X (gfx950).

API-compatible re-design of the reference run_ner.py:
CoNLL data, BertForTokenClassification, FusedAdam (HIP multi-tensor)
with the warmup_exp_decay_exp LambdaLR schedule, macro-F1 evaluation.
"""

from __future__ import annotations

import argparse
import os
import random
import time

import numpy as np
import torch
from torch.utils.data import DataLoader

from bert_pytorch_amd.config import BertConfig, merge_config_and_args
from bert_pytorch_amd.data.ner_dataset import NERDataset
from bert_pytorch_amd.data.tokenization import get_wordpiece_tokenizer
from bert_pytorch_amd.models import BertForTokenClassification
from bert_pytorch_amd.optim import FusedAdam, warmup_exp_decay_exp
from bert_pytorch_amd.parallel import comm
from bert_pytorch_amd.utils import MetricLogger

PAD_LABEL_ID = -100


def parse_args(argv=None):
    parser = argparse.ArgumentParser(description="MI355X-native NER runner")
    parser.add_argument("--config_file", type=str, default=None)
    parser.add_argument("--model_config_file", type=str, required=False)
    parser.add_argument("--init_checkpoint", type=str, default=None)
    parser.add_argument("--vocab_file", type=str, default=None)
    parser.add_argument("--data_dir", type=str, required=False)
    parser.add_argument("--train_file", type=str, default="train.txt")
    parser.add_argument("--eval_file", type=str, default="valid.txt")
    parser.add_argument("--output_dir", type=str, default="ner_out")
    parser.add_argument("--max_seq_length", type=int, default=128)
    parser.add_argument("--batch_size", type=int, default=32)
    parser.add_argument("--learning_rate", type=float, default=5e-5)
    parser.add_argument("--epochs", type=int, default=3)
    parser.add_argument("--warmup", type=float, default=0.1)
    parser.add_argument("--lr_decay_rate", type=float, default=0.9)
    parser.add_argument("--lr_decay_steps", type=int, default=500)
    parser.add_argument("--seed", type=int, default=42)
    parser.add_argument("--fp16", action="store_true")
    parser.add_argument("--bf16", action="store_true")
    parser.add_argument("--do_train", action="store_true")
    parser.add_argument("--do_eval", action="store_true")
    parser.add_argument("--local_rank", type=int,
                        default=int(os.environ.get("LOCAL_RANK", 0)))
    return merge_config_and_args(parser, argv)


def compute_metrics(preds, labels, num_labels):
    """Macro F1 over label classes (reference: run_ner.py:127-142)."""
    f1s = []
    for cls in range(num_labels):
        tp = int(((preds == cls) & (labels == cls)).sum())
        fp = int(((preds == cls) & (labels != cls)).sum())
        fn = int(((preds != cls) & (labels == cls)).sum())
        if tp + fp == 0 or tp + fn == 0 or tp == 0:
            f1s.append(0.0)
            continue
        precision = tp / (tp + fp)
        recall = tp / (tp + fn)
        f1s.append(2 * precision * recall / (precision + recall))
    return float(np.mean(f1s))


def evaluate(model, loader, device, mixed, num_labels):
    model.eval()
    all_preds, all_labels = [], []
    with torch.no_grad():
        for ids, mask, labels in loader:
            ids, mask = ids.to(device), mask.to(device)
            with torch.autocast(device.type, dtype=torch.bfloat16,
                                enabled=mixed and device.type == "cuda"):
                logits = model(ids, None, mask)
            preds = logits.argmax(-1).cpu()
            sel = labels != PAD_LABEL_ID
            all_preds.append(preds[sel])
            all_labels.append(labels[sel])
    preds = torch.cat(all_preds).numpy()
    labels = torch.cat(all_labels).numpy()
    return compute_metrics(preds, labels, num_labels)


def main(args=None):
    if args is None:
        args = parse_args()
    rank, local_rank, world = comm.init_distributed()
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        from bert_pytorch_amd.utils import tunable  # noqa: PLC0415

        tunable.enable()
    device = torch.device("cuda", local_rank) if use_cuda else torch.device("cpu")
    random.seed(args.seed)
    np.random.seed(args.seed)
    torch.manual_seed(args.seed)
    comm.mkdir_by_main_process(args.output_dir)
    log = MetricLogger(
        log_prefix=os.path.join(args.output_dir, "ner_log")
        if comm.is_main_process() else None,
        verbose=comm.is_main_process(),
    )

    config = BertConfig.from_json_file(args.model_config_file)
    vocab_file = args.vocab_file or getattr(config, "vocab_file", None)
    tokenizer = get_wordpiece_tokenizer(
        vocab_file, lowercase=getattr(config, "lowercase", True)
    )
    train_set = NERDataset(
        os.path.join(args.data_dir, args.train_file), tokenizer,
        args.max_seq_length, pad_label_id=PAD_LABEL_ID,
    )
    eval_set = NERDataset(
        os.path.join(args.data_dir, args.eval_file), tokenizer,
        args.max_seq_length, labels=train_set.labels,
        pad_label_id=PAD_LABEL_ID,
    )
    model = BertForTokenClassification(config, num_labels=train_set.num_labels)
    if args.init_checkpoint:
        state = torch.load(args.init_checkpoint, map_location="cpu",
                           weights_only=False)
        if isinstance(state, dict) and "model" in state:
            state = state["model"]
        state = {k.removeprefix("module."): v for k, v in state.items()}
        model.load_state_dict(state, strict=False)
    model.to(device)
    mixed = args.bf16 or args.fp16

    train_loader = DataLoader(train_set, batch_size=args.batch_size,
                              shuffle=True, drop_last=True)
    eval_loader = DataLoader(eval_set, batch_size=args.batch_size)
    total_steps = max(1, len(train_loader) * args.epochs)

    optimizer = FusedAdam(model.parameters(), lr=args.learning_rate)
    lr_lambda = lambda step: warmup_exp_decay_exp(  # noqa: E731
        step, args.lr_decay_rate, args.lr_decay_steps, total_steps, args.warmup
    )

    if args.do_train:
        loss_fct = torch.nn.CrossEntropyLoss(ignore_index=PAD_LABEL_ID)
        start = time.perf_counter()
        global_step = 0
        for epoch in range(args.epochs):
            model.train()
            for ids, mask, labels in train_loader:
                ids, mask, labels = (
                    ids.to(device), mask.to(device), labels.to(device)
                )
                with torch.autocast(device.type, dtype=torch.bfloat16,
                                    enabled=mixed and use_cuda):
                    logits = model(ids, None, mask)
                    loss = loss_fct(
                        logits.view(-1, train_set.num_labels).float(),
                        labels.view(-1),
                    )
                optimizer.zero_grad()
                loss.backward()
                for group in optimizer.param_groups:
                    group["lr"] = args.learning_rate * lr_lambda(global_step)
                optimizer.step()
                global_step += 1
            f1 = evaluate(model, eval_loader, device, mixed, train_set.num_labels)
            log.log("eval", epoch, macro_f1=f1, loss=float(loss))
        train_time = time.perf_counter() - start
        log.info("e2e_train_time=%.1fs training_sequences_per_second=%.1f",
                 train_time, args.epochs * len(train_set) / train_time)
        if comm.is_main_process():
            torch.save({"model": model.state_dict()},
                       os.path.join(args.output_dir, "pytorch_model.bin"))
    if args.do_eval and not args.do_train:
        f1 = evaluate(model, eval_loader, device, mixed, train_set.num_labels)
        log.info("macro_f1=%.4f", f1)
    log.close()


if __name__ == "__main__":
    main()
