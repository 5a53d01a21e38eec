#!/bin/bash
# Single-node SQuAD finetune from a pretraining checkpoint
# (equivalent of the reference's scripts/run_squad.sh).
set -euo pipefail

CKPT="${1:-results/pretrain_ckpts/ckpt_8601.pt}"
SQUAD_DIR="${2:-data/squad}"
VOCAB="${3:-data/vocab/vocab.txt}"
OUT="${4:-results/squad}"
NGPUS="${NGPUS:-8}"
export HSA_ENABLE_IPC_MODE_LEGACY=0

python -m torch.distributed.run --standalone --nproc-per-node "$NGPUS" \
  run_squad.py \
    --init_checkpoint "$CKPT" \
    --train_file "$SQUAD_DIR/train-v1.1.json" \
    --predict_file "$SQUAD_DIR/dev-v1.1.json" \
    --vocab_file "$VOCAB" \
    --model_config_file config/bert_large_uncased_config.json \
    --output_dir "$OUT" \
    --do_train --do_predict --do_eval \
    --train_batch_size 4 --predict_batch_size 8 \
    --learning_rate 3e-5 --num_train_epochs 2 \
    --max_seq_length 384 --doc_stride 128 --bf16
