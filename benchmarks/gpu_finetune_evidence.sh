#!/bin/bash
# Finetune-runner evidence on one MI355X: BERT-Large SQuAD (seq 384) and
# CoNLL NER (seq 128) at the reference's operating points, random-init
# weights, synthetic data (no network on the box). Captures the
# reference's finetune headline metrics — training_sequences_per_second
# and inference_sequences_per_second (run_squad.py) — for BASELINE.md.
# Work dirs live under /tmp (gpurun's merge-back is capped at 64 MiB);
# only the logs go to gpurun_out/.
set -x
OUT=${OUT:-/tmp/bpa_finetune}
LOGS=${LOGS:-gpurun_out}
mkdir -p "$LOGS"
python benchmarks/gen_finetune_synth.py --out "$OUT" \
    --squad_train 8192 --squad_predict 4096 --ner_train 12000 --ner_valid 800

timeout 900 python run_squad.py \
    --model_config_file config/bert_large_uncased_config.json \
    --vocab_file "$OUT/vocab.txt" \
    --train_file "$OUT/squad_train.json" \
    --predict_file "$OUT/squad_predict.json" \
    --do_train --do_predict --do_eval --bf16 \
    --max_steps 250 --train_batch_size 32 --predict_batch_size 32 \
    --max_seq_length 384 --doc_stride 128 \
    --output_dir "$OUT/sq_out" > "$LOGS/squad_bertlarge_gpu.log" 2>&1
echo "SQUAD_RC=$?"
tail -2 "$LOGS/squad_bertlarge_gpu.log"

timeout 900 python run_ner.py \
    --model_config_file config/bert_large_uncased_config.json \
    --vocab_file "$OUT/vocab.txt" \
    --data_dir "$OUT" --train_file train.txt --eval_file valid.txt \
    --do_train --do_eval --bf16 --epochs 1 --batch_size 32 \
    --max_seq_length 128 \
    --output_dir "$OUT/ner_out" > "$LOGS/ner_bertlarge_gpu.log" 2>&1
echo "NER_RC=$?"
tail -2 "$LOGS/ner_bertlarge_gpu.log"
