#!/bin/bash
# Batch-evaluate every saved checkpoint (parity with reference eval.sh:3-8):
# runs --phase=eval for each ./data/models/*.npy, teeing stdout to <step>.txt.
for file in ./data/models/*.npy; do
    name="$(basename "$file" .npy)"
    echo "evaluating $file"
    python main.py --phase=eval --model_file="$file" "$@" | tee "${name}.txt"
done
