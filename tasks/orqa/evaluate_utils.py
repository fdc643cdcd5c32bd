"""ORQA retrieval evaluation (reference tasks/orqa/evaluate_utils.py):
embed the evidence corpus with the context tower, embed NQ questions with
the query tower, retrieve top-k passages by MIPS (dense GPU matmul search —
megatron_amd.inference.realm_index; no faiss in this stack, 288 GB HBM
holds the embedding matrix), report top-k answer-hit accuracies."""

from __future__ import annotations

import numpy as np
import torch

from megatron_amd.checkpointing import load_checkpoint
from megatron_amd.config import get_config
from megatron_amd.inference.realm_index import BlockData, FaissMIPSIndex
from megatron_amd.models import ModelType
from megatron_amd.models.biencoder_model import BiEncoderModel
from megatron_amd.training import get_model
from megatron_amd.utils import print_rank_0

from tasks.orqa.evidence import get_open_retrieval_wiki_dataset
from tasks.orqa.unsupervised.nq import (
    get_nq_dataset,
    get_one_epoch_nq_dataloader,
    process_nq_batch,
)
from tasks.orqa.unsupervised.qa_utils import calculate_matches


class ORQAEvaluator:
    def __init__(self, model=None):
        cfg = self.cfg = get_config()
        self.evidence_dataset = get_open_retrieval_wiki_dataset()

        if model is None:
            def provider(pre_process=True, post_process=True):
                return BiEncoderModel(
                    cfg,
                    shared_query_context_model=(
                        cfg.biencoder_shared_query_context_model
                    ),
                )

            model = get_model(provider, ModelType.encoder_or_decoder,
                              wrap_with_ddp=False, cfg=cfg)
            if cfg.load is not None:
                load_checkpoint(model, None, None, cfg)
            model = model[0]
        self.model = model
        self.model.eval()

        self.mips_index = self._build_index()

    @torch.no_grad()
    def _build_index(self):
        """Embed every evidence passage with the context tower into a
        BlockData, then hand it to the MIPS index."""
        cfg = self.cfg
        device = "cuda" if torch.cuda.is_available() else "cpu"
        block_data = BlockData()
        loader = torch.utils.data.DataLoader(
            self.evidence_dataset, batch_size=cfg.micro_batch_size,
            shuffle=False, num_workers=cfg.num_workers,
        )
        for batch in loader:
            tokens = batch["context"].long().to(device)
            mask = batch["context_pad_mask"].long().to(device)
            types = batch["context_types"].long().to(device)
            embeds = self.model.embed_context(tokens, mask)
            block_data.add_block_data(
                batch["row_id"].numpy(),
                embeds.float().cpu().numpy(),
            )
        index = FaissMIPSIndex(
            embed_size=next(iter(block_data.embed_data.values())).shape[-1],
            embed_data=block_data,
            use_gpu=torch.cuda.is_available(),
        )
        return index

    @torch.no_grad()
    def generate_query_vectors(self, qa_data, split):
        dataset = get_nq_dataset(qa_data, split)
        loader = get_one_epoch_nq_dataloader(dataset)
        query_vectors, references = [], []
        for batch in loader:
            tokens, mask, types, reference = process_nq_batch(batch)
            embeds = self.model.embed_query(tokens, mask)
            query_vectors.extend(embeds.float().cpu().numpy())
            references.extend(reference)
        return np.array(query_vectors), references

    def evaluate(self, qa_data, split, top_k=None):
        cfg = self.cfg
        top_k = top_k or max(cfg.report_topk_accuracies or [20])
        query_vectors, references = self.generate_query_vectors(qa_data,
                                                                split)
        scores, doc_ids = self.mips_index.search_mips_index(
            torch.from_numpy(query_vectors), top_k
        )
        closest = [(list(ids), list(s)) for ids, s in zip(doc_ids, scores)]
        stats = calculate_matches(
            self.evidence_dataset.id2text, references, closest,
            match_type=getattr(cfg, "match", "string"),
        )
        n = len(references)
        print_rank_0(f"Validation results on {split}:")
        for k in (cfg.report_topk_accuracies or [1, 5, 20]):
            if k <= len(stats.top_k_hits):
                print_rank_0(
                    f"  top-{k}: {100.0 * stats.top_k_hits[k - 1] / n:.2f}"
                )
        return stats
