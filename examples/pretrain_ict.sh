#!/bin/bash
# Inverse-Cloze-Task retriever pretraining (analog of examples/pretrain_ict.sh).
export HSA_ENABLE_IPC_MODE_LEGACY=0

python pretrain_ict.py \
    --num_layers 12 --hidden_size 768 --num_attention_heads 12 \
    --seq_length 256 --max_position_embeddings 512 \
    --micro_batch_size 32 --global_batch_size 32 \
    --lr 1e-4 --lr_decay_style linear --train_iters 100000 \
    --weight_decay 0.01 --clip_grad 1.0 --bf16 \
    --save ./checkpoints/ict --save_interval 2000 \
    --data_path ./data/wiki_text_sentence \
    --tokenizer_type BertWordPieceLowerCase --vocab_file bert-vocab.txt \
    "$@"
