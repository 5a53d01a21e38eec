#!/bin/bash
# End-to-end dataset build: download -> format -> vocab -> encode.
# Equivalent of the reference's scripts/create_datasets.sh (:58-142),
# driving the in-repo utils/ pipeline (no wikiextractor/nltk/HF-Rust
# dependencies; format.py parses wikiextractor output directly and the
# vocab trainer is the in-repo C++ BPE core).
#
# Usage:
#   scripts/create_datasets.sh --data-dir DATA [--model bert|roberta] \
#       [--download] [--format] [--vocab] [--encode]
set -euo pipefail

DATA_DIR=""
MODEL="bert"
DO_DOWNLOAD=0; DO_FORMAT=0; DO_VOCAB=0; DO_ENCODE=0
NPROC=$(nproc)

while [[ $# -gt 0 ]]; do
  case "$1" in
    --data-dir) DATA_DIR="$2"; shift 2 ;;
    --model)    MODEL="$2"; shift 2 ;;
    --download) DO_DOWNLOAD=1; shift ;;
    --format)   DO_FORMAT=1; shift ;;
    --vocab)    DO_VOCAB=1; shift ;;
    --encode)   DO_ENCODE=1; shift ;;
    *) echo "unknown arg: $1" >&2; exit 1 ;;
  esac
done
[[ -n "$DATA_DIR" ]] || { echo "--data-dir required" >&2; exit 1; }
ROOT="$(cd "$(dirname "$0")/.." && pwd)"
mkdir -p "$DATA_DIR"

if [[ $DO_DOWNLOAD -eq 1 ]]; then
  python "$ROOT/utils/download.py" --dataset wikicorpus_en \
      --output_dir "$DATA_DIR/raw/wiki"
  python "$ROOT/utils/download.py" --dataset squad \
      --output_dir "$DATA_DIR/raw/squad"
  echo "NOTE: run wikiextractor on $DATA_DIR/raw/wiki/*.xml before --format" >&2
fi

if [[ $DO_FORMAT -eq 1 ]]; then
  python "$ROOT/utils/format.py" --dataset wikicorpus \
      --input_glob "$DATA_DIR/raw/wiki/extracted/**/wiki_*" \
      --output_dir "$DATA_DIR/formatted" --shards 256 --processes "$NPROC"
fi

if [[ $DO_VOCAB -eq 1 ]]; then
  if [[ "$MODEL" == "roberta" ]]; then
    python "$ROOT/utils/build_vocab.py" --tokenizer bpe --vocab_size 50265 \
        --input_glob "$DATA_DIR/formatted/*.txt" --output_dir "$DATA_DIR/vocab"
  else
    python "$ROOT/utils/build_vocab.py" --tokenizer wordpiece --lowercase \
        --vocab_size 30522 \
        --input_glob "$DATA_DIR/formatted/*.txt" --output_dir "$DATA_DIR/vocab"
  fi
fi

if [[ $DO_ENCODE -eq 1 ]]; then
  if [[ "$MODEL" == "roberta" ]]; then
    # RoBERTa: seq512, no NSP (reference create_datasets.sh:121-124)
    python "$ROOT/utils/encode_data.py" --tokenizer bpe \
        --vocab_file "$DATA_DIR/vocab/vocab.json" \
        --merges_file "$DATA_DIR/vocab/merges.txt" \
        --input_dir "$DATA_DIR/formatted" \
        --output_dir "$DATA_DIR/hdf5/seq512_nsp0" \
        --max_seq_len 512 --nsp_probability 0 --processes "$NPROC"
  else
    # BERT: seq128 + seq512, NSP 0.5 (reference create_datasets.sh:133-140)
    python "$ROOT/utils/encode_data.py" --tokenizer wordpiece --lowercase \
        --vocab_file "$DATA_DIR/vocab/vocab.txt" \
        --input_dir "$DATA_DIR/formatted" \
        --output_dir "$DATA_DIR/hdf5/seq128_nsp5" \
        --max_seq_len 128 --nsp_probability 0.5 --processes "$NPROC"
    python "$ROOT/utils/encode_data.py" --tokenizer wordpiece --lowercase \
        --vocab_file "$DATA_DIR/vocab/vocab.txt" \
        --input_dir "$DATA_DIR/formatted" \
        --output_dir "$DATA_DIR/hdf5/seq512_nsp5" \
        --max_seq_len 512 --nsp_probability 0.5 --processes "$NPROC"
  fi
fi
echo "done."
