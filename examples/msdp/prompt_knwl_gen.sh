#!/bin/bash
# MSDP stage 1: prompt a large LM to generate grounding knowledge for each
# dialogue turn (analog of examples/msdp/prompt_knwl_gen.sh).
export HSA_ENABLE_IPC_MODE_LEGACY=0

python tasks/main.py \
    --task MSDP-PROMPT --prompt_type knowledge \
    --model_name llama2 \
    --num_layers 32 --hidden_size 4096 --num_attention_heads 32 \
    --seq_length 2048 --max_position_embeddings 4096 \
    --micro_batch_size 1 --bf16 --use_flash_attn \
    --load ./checkpoints/llama2-7b \
    --tokenizer_type SentencePieceTokenizer --vocab_file tokenizer.model \
    --sample_input_file ./data/wow_test.tsv \
    --sample_output_file ./out/generated_knowledge.txt \
    --prompt_file ./data/knowledge_prompts.jsonl \
    --num_prompt_examples 10 --out_seq_length 64 \
    "$@"
