#!/bin/bash
# BERT pretraining (MLM + NSP) on one 8-GPU MI355X node
# (analog of examples/pretrain_bert_distributed.sh).
export HSA_ENABLE_IPC_MODE_LEGACY=0

torchrun --nproc_per_node 8 --master_addr 127.0.0.1 pretrain_bert.py \
    --num_layers 24 --hidden_size 1024 --num_attention_heads 16 \
    --seq_length 512 --max_position_embeddings 512 \
    --micro_batch_size 8 --global_batch_size 256 \
    --lr 1e-4 --min_lr 1e-5 --lr_decay_style linear \
    --lr_warmup_fraction 0.01 --train_iters 100000 \
    --weight_decay 0.01 --clip_grad 1.0 --bf16 \
    --save ./checkpoints/bert-large --save_interval 2000 \
    --log_interval 100 --eval_interval 1000 --eval_iters 10 \
    --data_path ./data/bert_corpus_text_sentence \
    --tokenizer_type BertWordPieceLowerCase --vocab_file bert-vocab.txt \
    "$@"
