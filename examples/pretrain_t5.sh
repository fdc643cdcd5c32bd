#!/bin/bash
# T5 span-corruption pretraining on a single MI355X
# (analog of examples/pretrain_t5.sh).
export HSA_ENABLE_IPC_MODE_LEGACY=0

python pretrain_t5.py \
    --num_layers 12 --hidden_size 768 --num_attention_heads 12 \
    --kv_channels 64 --ffn_hidden_size 3072 \
    --encoder_seq_length 512 --decoder_seq_length 128 \
    --seq_length 512 --max_position_embeddings 512 \
    --micro_batch_size 16 --global_batch_size 16 \
    --lr 1e-4 --lr_decay_style linear --lr_warmup_fraction 0.01 \
    --train_iters 100000 --weight_decay 0.01 --clip_grad 1.0 --bf16 \
    --vocab_extra_ids 100 \
    --save ./checkpoints/t5-base --save_interval 2000 \
    --log_interval 100 --eval_interval 1000 --eval_iters 10 \
    --data_path ./data/t5_corpus_text_sentence \
    --tokenizer_type SentencePieceTokenizer --vocab_file spiece.model \
    "$@"
