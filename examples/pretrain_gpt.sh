#!/bin/bash
# GPT pretraining on a single MI355X (analog of examples/pretrain_gpt.sh).
export HSA_ENABLE_IPC_MODE_LEGACY=0

python finetune.py \
    --model_name gpt \
    --num_layers 24 --hidden_size 1024 --num_attention_heads 16 \
    --seq_length 1024 --max_position_embeddings 1024 \
    --micro_batch_size 8 --global_batch_size 64 \
    --lr 1.5e-4 --min_lr 1e-5 --lr_decay_style cosine \
    --lr_warmup_iters 1000 --train_iters 50000 \
    --weight_decay 0.01 --clip_grad 1.0 --bf16 --use_flash_attn \
    --save ./checkpoints/gpt-345m --save_interval 1000 \
    --log_interval 100 --eval_interval 1000 --eval_iters 10 \
    --data_path ./data/my_corpus_text_document \
    --tokenizer_type GPT2BPETokenizer \
    --vocab_file gpt2-vocab.json --merge_file gpt2-merges.txt \
    "$@"
