#!/bin/bash
# Convert HF weights to a megatron_amd checkpoint (analog of
# examples/hf_to_megatron.sh).
python weights_conversion/hf_to_megatron.py llama2 --size 7 \
    --cache-dir ./hf/Llama-2-7b-hf \
    --out ./checkpoints/llama2-7b \
    "$@"
