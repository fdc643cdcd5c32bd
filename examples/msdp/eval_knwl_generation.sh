#!/bin/bash
# MSDP: token-level F1 of generated knowledge vs gold references
# (analog of examples/msdp/eval_knwl_generation.sh).
python tasks/main.py \
    --task MSDP-EVAL-F1 \
    --guess_file ./out/generated_knowledge.txt \
    --answer_file ./data/knowledge_reference.txt \
    "$@"
