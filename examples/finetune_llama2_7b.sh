#!/bin/bash
# Llama-2-7B finetuning on one 8-GPU MI355X node (analog of the reference's
# examples/finetune.sh getting-started config, sized for 288 GB HBM3E).
export HSA_ENABLE_IPC_MODE_LEGACY=0

torchrun --nproc_per_node 8 --master_addr 127.0.0.1 finetune.py \
    --model_name llama2 \
    --tensor_model_parallel_size 2 \
    --pipeline_model_parallel_size 1 \
    --sequence_parallel \
    --use_distributed_optimizer \
    --bf16 \
    --num_layers 32 --hidden_size 4096 --num_attention_heads 32 \
    --ffn_hidden_size 11008 \
    --seq_length 4096 --max_position_embeddings 4096 \
    --micro_batch_size 4 --global_batch_size 64 \
    --lr 3e-4 --min_lr 3e-5 --lr_decay_style cosine \
    --lr_warmup_iters 100 --train_iters 5000 \
    --weight_decay 0.1 --clip_grad 1.0 \
    --use_rms_norm --glu_activation swiglu --no_tie_embed_logits \
    --position_embedding_type rotary --no_use_bias \
    --load ./checkpoints/llama2-7b --save ./checkpoints/llama2-7b-ft \
    --save_interval 500 --log_interval 10 --eval_interval 500 --eval_iters 10 \
    --data_path ./data/my_corpus_text_document \
    --tokenizer_type SentencePieceTokenizer --vocab_file ./tokenizer.model \
    "$@"
