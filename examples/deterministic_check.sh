#!/bin/sh
# Greedy-decode determinism check (role of reference examples/macbeth.sh):
# two runs with temperature 0 must produce identical output.
set -e
MODEL=${1:?usage: deterministic_check.sh model.m tokenizer.t}
TOK=${2:?usage: deterministic_check.sh model.m tokenizer.t}
A=$(./dllama inference --model "$MODEL" --tokenizer "$TOK" \
      --prompt "To be, or not to be" --steps 32 --temperature 0 | head -n 1)
B=$(./dllama inference --model "$MODEL" --tokenizer "$TOK" \
      --prompt "To be, or not to be" --steps 32 --temperature 0 | head -n 1)
[ "$A" = "$B" ] && echo "✅ deterministic" || { echo "❌ outputs differ"; exit 1; }
