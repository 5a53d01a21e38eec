#!/bin/bash
# BASELINE config 1: BERT-base phase-1 seq128, 100 steps, world_size=1,
# CPU/gloo, synthetic HDF5 shard — the no-GPU plumbing run.
set -euo pipefail
ROOT="$(cd "$(dirname "$0")/.." && pwd)"
OUT="${1:-/tmp/bpa_cpu_plumbing}"
rm -rf "$OUT"; mkdir -p "$OUT/data"

python - <<EOF
from bert_pytorch_amd.data import synth
synth.make_dataset("$OUT/data", num_shards=2, samples_per_shard=2048,
                   seq_len=128, vocab_size=30522, seed=0)
EOF

MASTER_ADDR=127.0.0.1 python -m torch.distributed.run --standalone \
  --local-addr 127.0.0.1 --nproc-per-node 1 \
  "$ROOT/run_pretraining.py" \
    --model_config_file "$ROOT/config/bert_base_uncased_config.json" \
    --input_dir "$OUT/data" --output_dir "$OUT/run" \
    --local_batch_size 8 --global_batch_size 16 \
    --max_steps 100 --num_steps_per_checkpoint 50 \
    --learning_rate 1e-4 --num_workers 2 --disable_progress_bar
echo "CPU plumbing run complete; logs under $OUT/run"
