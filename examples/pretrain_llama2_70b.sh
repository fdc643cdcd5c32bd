#!/bin/bash
# Llama-2-70B pretraining: TP4 x PP2 on one 8-GPU MI355X node (the BASELINE
# 70B config; 288 GB HBM3E per GPU holds the tp4/pp2 shards + fp32 optimizer
# state without DP sharding). Per-rank memory validated on real hardware via
# `bench.py --model llama2-70b-shard8`: 193 GiB peak at pipeline depth 2,
# 233 GiB at depth 4 (profiles/r02_progress.md) — full recompute below is
# therefore OPTIONAL at this sequence length; drop the two --recompute_*
# lines for ~25-30% more throughput if the batch plan keeps <=4 microbatches
# in flight per stage.
export HSA_ENABLE_IPC_MODE_LEGACY=0

torchrun --nproc_per_node 8 --master_addr 127.0.0.1 finetune.py \
    --model_name llama2 \
    --tensor_model_parallel_size 4 \
    --pipeline_model_parallel_size 2 \
    --sequence_parallel \
    --bf16 \
    --num_layers 80 --hidden_size 8192 --num_attention_heads 64 \
    --num_attention_heads_kv 8 --ffn_hidden_size 28672 \
    --seq_length 4096 --max_position_embeddings 4096 \
    --micro_batch_size 1 --global_batch_size 32 \
    --lr 1.5e-4 --min_lr 1.5e-5 --lr_decay_style cosine \
    --train_iters 10000 --clip_grad 1.0 \
    --use_rms_norm --glu_activation swiglu --no_tie_embed_logits \
    --position_embedding_type rotary --no_use_bias \
    --data_path ./data/my_corpus_text_document \
    --tokenizer_type SentencePieceTokenizer --vocab_file ./tokenizer.model \
    "$@"
