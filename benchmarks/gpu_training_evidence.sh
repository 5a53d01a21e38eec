#!/bin/bash
# Training-correctness evidence on one MI355X with the real runner:
#  1. BERT-base 300-step convergence run (bf16, LAMB) -> CSV loss curve
#  2. BERT-Large phase-1 + K-FAC preconditioner (BASELINE config 5 path)
#  3. RoBERTa-style no-NSP + fused Adam + linear decay (config 4 path)
# Checkpoints/data stay in /tmp; only logs/CSVs are copied to $LOGDIR.
set -euo pipefail
LOGDIR="${1:-gpurun_out/evidence}"
WORK=/tmp/bpa_evidence
rm -rf "$WORK"; mkdir -p "$WORK/d128" "$WORK/d128nonsp" "$LOGDIR"

python - <<EOF
from bert_pytorch_amd.data import synth
synth.make_dataset("$WORK/d128", num_shards=2, samples_per_shard=4096,
                   seq_len=128, vocab_size=30522, seed=0)
synth.make_dataset("$WORK/d128nonsp", num_shards=1, samples_per_shard=2048,
                   seq_len=128, vocab_size=28996, seed=1, nsp=False)
EOF

# 1) convergence: BERT-base, 300 steps
python run_pretraining.py \
  --model_config_file config/bert_base_uncased_config.json \
  --input_dir "$WORK/d128" --output_dir "$WORK/conv" \
  --local_batch_size 64 --global_batch_size 128 --bf16 \
  --max_steps 300 --num_steps_per_checkpoint 300 \
  --learning_rate 2e-4 --warmup_proportion 0.1 \
  --num_workers 2 --disable_progress_bar \
  --log_prefix convergence
cp "$WORK"/conv/convergence_metrics.csv "$LOGDIR/convergence_bertbase.csv"
tail -3 "$WORK"/conv/convergence.txt > "$LOGDIR/convergence_tail.txt" || true

# 2) K-FAC on BERT-Large shapes (short)
python run_pretraining.py \
  --model_config_file config/bert_large_uncased_config.json \
  --input_dir "$WORK/d128" --output_dir "$WORK/kfac" \
  --local_batch_size 32 --global_batch_size 64 --bf16 --kfac \
  --max_steps 12 --num_steps_per_checkpoint 12 \
  --learning_rate 1e-4 --num_workers 2 --disable_progress_bar \
  --log_prefix kfac > "$LOGDIR/kfac_run.log" 2>&1
cp "$WORK"/kfac/kfac_metrics.csv "$LOGDIR/kfac_metrics.csv"

# 3) RoBERTa path: no NSP, fused Adam, linear decay
python run_pretraining.py \
  --model_config_file config/roberta_large_cased_config.json \
  --input_dir "$WORK/d128nonsp" --output_dir "$WORK/roberta" \
  --local_batch_size 16 --global_batch_size 32 --bf16 \
  --optimizer adam --lr_decay linear \
  --max_steps 12 --num_steps_per_checkpoint 12 \
  --learning_rate 1e-4 --num_workers 2 --disable_progress_bar \
  --log_prefix roberta > "$LOGDIR/roberta_run.log" 2>&1
cp "$WORK"/roberta/roberta_metrics.csv "$LOGDIR/roberta_metrics.csv"

echo "EVIDENCE OK"
python - <<EOF
import csv
rows = list(csv.DictReader(open("$LOGDIR/convergence_bertbase.csv")))
print("convergence: first loss", rows[0]["step_loss"], "last loss", rows[-1]["step_loss"])
EOF
