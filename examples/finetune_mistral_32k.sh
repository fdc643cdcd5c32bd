#!/bin/bash
# Mistral-7B instruction tuning at seq 32k (RoPE-scaled, sliding window 4096),
# TP2 x DP4 (BASELINE config 5).
export HSA_ENABLE_IPC_MODE_LEGACY=0

torchrun --nproc_per_node 8 --master_addr 127.0.0.1 finetune.py \
    --model_name mistral \
    --tensor_model_parallel_size 2 \
    --sequence_parallel \
    --use_distributed_optimizer \
    --bf16 \
    --num_layers 32 --hidden_size 4096 --num_attention_heads 32 \
    --num_attention_heads_kv 8 --ffn_hidden_size 14336 \
    --seq_length 32768 --max_position_embeddings 32768 \
    --rope_scaling_factor 4.0 --sliding_window_size 4096 \
    --micro_batch_size 1 --global_batch_size 16 \
    --lr 1e-5 --lr_decay_style cosine --train_iters 2000 \
    --use_rms_norm --glu_activation swiglu --no_tie_embed_logits \
    --position_embedding_type rotary --no_use_bias \
    --model_type instruction --variable_seq_lengths \
    --data_path ./data/oasst \
    --tokenizer_type SentencePieceTokenizer --vocab_file ./tokenizer.model \
    --vocab_extra_ids_list "<|im_start|>,<|im_end|>" \
    "$@"
