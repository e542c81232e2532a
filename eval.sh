#!/bin/bash
# Batch-evaluate every saved checkpoint (parity with the reference's
# eval.sh loop): runs --phase=eval for each checkpoint in ./data/models
# and tees each run's stdout (the per-metric BLEU/METEOR/ROUGE/CIDEr
# lines) to <global_step>.txt next to the shell.
#
# Usage: ./eval.sh [extra main.py flags, e.g. --beam_size=3 --synthetic]
set -u
for file in ./data/models/*.npy; do
    [ -e "$file" ] || { echo "no checkpoints in ./data/models"; exit 1; }
    name="$(basename "$file" .npy)"
    echo "evaluating $file"
    python main.py --phase=eval --model_file="$file" "$@" | tee "${name}.txt"
done
