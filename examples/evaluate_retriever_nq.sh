#!/bin/bash
# Retriever top-k accuracy on Natural Questions: embeds the evidence corpus
# and retrieves with dense GPU MIPS (analog of examples/evaluate_retriever_nq.sh).
export HSA_ENABLE_IPC_MODE_LEGACY=0

python tasks/main.py \
    --task RETRIEVER-EVAL \
    --num_layers 12 --hidden_size 768 --num_attention_heads 12 \
    --seq_length 256 --max_position_embeddings 512 \
    --micro_batch_size 128 --bf16 \
    --load ./checkpoints/ict \
    --evidence_data_path ./data/wikipedia_evidence.tsv \
    --qa_data_dev ./data/nq_dev.jsonl \
    --tokenizer_type BertWordPieceLowerCase --vocab_file bert-vocab.txt \
    "$@"
