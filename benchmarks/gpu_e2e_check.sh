#!/bin/bash
# End-to-end GPU validation of the real pretraining runner: synthetic
# shards -> phase-1 steps (bf16, checkpoints) -> resume -> phase-2
# continuation (seq 512, previous_phase_end_step hand-off). Exercises
# dataloader H2D, fused kernels, FusedLAMB, checkpoint IO and the
# two-phase optimizer-state surgery on device.
set -euo pipefail
OUT="${1:-/tmp/bpa_e2e}"  # keep checkpoints OFF gpurun_out (64 MiB merge limit)
rm -rf "$OUT"; mkdir -p "$OUT/data128" "$OUT/data512"

python - <<EOF
from bert_pytorch_amd.data import synth
synth.make_dataset("$OUT/data128", num_shards=2, samples_per_shard=512,
                   seq_len=128, vocab_size=30522, seed=0)
synth.make_dataset("$OUT/data512", num_shards=1, samples_per_shard=128,
                   seq_len=512, vocab_size=30522, seed=1)
EOF

run() { python run_pretraining.py "$@"; }

# phase 1: 12 steps with a checkpoint every 4
run --model_config_file config/bert_large_uncased_config.json \
    --input_dir "$OUT/data128" --output_dir "$OUT/run" \
    --local_batch_size 32 --global_batch_size 64 --bf16 \
    --max_steps 12 --num_steps_per_checkpoint 4 \
    --learning_rate 1e-4 --num_workers 2 --disable_progress_bar \
    --timing_breakdown
test -f "$OUT/run/pretrain_ckpts/ckpt_12.pt" || { echo "MISSING ckpt_12"; exit 1; }

# resume (same phase): +4 steps
run --model_config_file config/bert_large_uncased_config.json \
    --input_dir "$OUT/data128" --output_dir "$OUT/run" \
    --local_batch_size 32 --global_batch_size 64 --bf16 \
    --max_steps 16 --steps 4 --num_steps_per_checkpoint 4 \
    --learning_rate 1e-4 --num_workers 2 --disable_progress_bar
test -f "$OUT/run/pretrain_ckpts/ckpt_16.pt" || { echo "MISSING ckpt_16"; exit 1; }

# phase 2: seq 512 continuation from the phase-1 checkpoint
run --model_config_file config/bert_large_uncased_config.json \
    --input_dir "$OUT/data512" --output_dir "$OUT/run" \
    --local_batch_size 8 --global_batch_size 16 --bf16 \
    --max_steps 4 --num_steps_per_checkpoint 4 \
    --previous_phase_end_step 16 --max_predictions_per_seq 80 \
    --learning_rate 5e-5 --num_workers 2 --disable_progress_bar
test -f "$OUT/run/pretrain_ckpts/ckpt_20.pt" || { echo "MISSING ckpt_20 (16+4)"; exit 1; }

echo "GPU E2E OK: phase1 + resume + phase2 hand-off"
