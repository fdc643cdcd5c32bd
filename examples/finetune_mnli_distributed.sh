#!/bin/bash
# MNLI classification finetuning of a pretrained BERT
# (analog of examples/finetune_mnli_distributed.sh).
export HSA_ENABLE_IPC_MODE_LEGACY=0

torchrun --nproc_per_node 8 --master_addr 127.0.0.1 tasks/main.py \
    --task MNLI \
    --num_layers 24 --hidden_size 1024 --num_attention_heads 16 \
    --seq_length 512 --max_position_embeddings 512 \
    --micro_batch_size 8 --global_batch_size 64 \
    --lr 5e-5 --lr_decay_style linear --lr_warmup_fraction 0.065 \
    --epochs 3 --weight_decay 1e-2 --clip_grad 1.0 --bf16 \
    --pretrained_checkpoint ./checkpoints/bert-large \
    --save ./checkpoints/bert-mnli --save_interval 500000 \
    --log_interval 10 --eval_interval 100 --eval_iters 50 \
    --train_data ./data/MNLI/train.tsv \
    --valid_data ./data/MNLI/dev_matched.tsv ./data/MNLI/dev_mismatched.tsv \
    --tokenizer_type BertWordPieceLowerCase --vocab_file bert-vocab.txt \
    "$@"
