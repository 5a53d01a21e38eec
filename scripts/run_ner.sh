#!/bin/bash
# CoNLL NER finetune from a pretraining checkpoint
# (equivalent of the reference's scripts/run_ner.sh).
set -euo pipefail

CKPT="${1:-results/pretrain_ckpts/ckpt_8601.pt}"
DATA_DIR="${2:-data/conll}"
VOCAB="${3:-data/vocab/vocab.txt}"
OUT="${4:-results/ner}"
export HSA_ENABLE_IPC_MODE_LEGACY=0

python run_ner.py \
  --init_checkpoint "$CKPT" \
  --data_dir "$DATA_DIR" \
  --vocab_file "$VOCAB" \
  --model_config_file config/bert_large_uncased_config.json \
  --output_dir "$OUT" \
  --batch_size 32 --learning_rate 5e-5 --epochs 4 \
  --max_seq_length 128 --bf16 --do_train --do_eval
