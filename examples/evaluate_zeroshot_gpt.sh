#!/bin/bash
# Zero-shot LAMBADA / WikiText-103 evaluation of a GPT checkpoint
# (analog of examples/evaluate_zeroshot_gpt.sh). Use --task WIKITEXT103 with
# a wiki.test.tokens file for perplexity.
export HSA_ENABLE_IPC_MODE_LEGACY=0

python tasks/main.py \
    --task LAMBADA \
    --model_name gpt \
    --num_layers 24 --hidden_size 1024 --num_attention_heads 16 \
    --seq_length 1024 --max_position_embeddings 1024 \
    --micro_batch_size 8 --bf16 \
    --load ./checkpoints/gpt-345m \
    --valid_data ./data/lambada_test.jsonl --strict_lambada \
    --tokenizer_type GPT2BPETokenizer \
    --vocab_file gpt2-vocab.json --merge_file gpt2-merges.txt \
    "$@"
