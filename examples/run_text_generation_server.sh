#!/bin/bash
# REST text-generation server (PUT /api, same wire contract as the
# reference's Flask server; served by FastAPI/uvicorn here).
export HSA_ENABLE_IPC_MODE_LEGACY=0

torchrun --nproc_per_node 1 --master_addr 127.0.0.1 \
    tools/run_text_generation_server.py \
    --model_name llama2 \
    --num_layers 32 --hidden_size 4096 --num_attention_heads 32 \
    --seq_length 4096 --max_position_embeddings 4096 \
    --micro_batch_size 1 --bf16 --use_flash_attn \
    --load ./checkpoints/llama2-7b \
    --tokenizer_type SentencePieceTokenizer --vocab_file tokenizer.model \
    --port 5000 \
    "$@"

# query it:
#   curl -X PUT http://localhost:5000/api -H 'Content-Type: application/json' \
#        -d '{"prompts": ["The capital of France is"], "tokens_to_generate": 32}'
