#!/bin/bash
# Side-by-side numerical verification of a converted checkpoint vs the HF
# implementation (analog of examples/verify.sh).
export HSA_ENABLE_IPC_MODE_LEGACY=0

torchrun --nproc_per_node 1 --master_addr 127.0.0.1 verify_correctness.py \
    --model_name llama2 \
    --load ./checkpoints/llama2-7b \
    --num_layers 32 --hidden_size 4096 --num_attention_heads 32 \
    --seq_length 512 --max_position_embeddings 4096 --bf16 \
    --hf_cache_dir ./hf/Llama-2-7b-hf \
    --data_path ./data/my_corpus_text_document \
    --tokenizer_type SentencePieceTokenizer --vocab_file tokenizer.model \
    "$@"
