#!/bin/bash
# ASan+UBSan check of the C++ tokenizer core (CPU-only, no torch).
# The core includes <torch/extension.h> only for the extension build;
# it uses no torch symbols, so the sanitizer build strips the include
# and compiles it standalone.
set -euo pipefail
cd "$(dirname "$0")/.."
mkdir -p build/sanitize
sed 's|#include <torch/extension.h>||' csrc/tok/tokenizer.cpp \
  > build/sanitize/tokenizer_notorch.cpp
g++ -std=c++17 -g -O1 -fsanitize=address,undefined \
  -fno-sanitize-recover=all \
  build/sanitize/tokenizer_notorch.cpp csrc/tok/tok_sanitize_main.cpp \
  -o build/sanitize/tok_sanitize
./build/sanitize/tok_sanitize
