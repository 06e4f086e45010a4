#!/bin/sh
# Greedy-decode determinism check (role of reference examples/macbeth.sh):
# two runs with temperature 0 must produce identical generated text.
set -e
MODEL=${1:?usage: deterministic_check.sh model.m tokenizer.t}
TOK=${2:?usage: deterministic_check.sh model.m tokenizer.t}

gen() {
    # full output captured first (no mid-stream pipe close), then the
    # generation line = first line after the 💡 banner
    ./dllama inference --model "$MODEL" --tokenizer "$TOK" \
        --prompt "To be, or not to be" --steps 32 --temperature 0 \
        2>/dev/null | grep -v "^💡" | sed -n 1p
}

A=$(gen)
B=$(gen)
[ -n "$A" ] && [ "$A" = "$B" ] && echo "✅ deterministic: $A" \
    || { echo "❌ outputs differ"; exit 1; }
