"""Index builder: embed every block with the biencoder context tower and
store embeddings (reference megatron/indexer.py, condensed)."""

from __future__ import annotations

import torch

from .realm_index import BlockData


class IndexBuilder:
    def __init__(self, model, dataset, batch_size=128,
                 embedding_path="block_embeds.pkl"):
        self.model = model
        self.dataset = dataset
        self.batch_size = batch_size
        self.block_data = BlockData(embedding_path)

    @torch.no_grad()
    def build_and_save_index(self):
        self.model.eval()
        loader = torch.utils.data.DataLoader(
            self.dataset, batch_size=self.batch_size
        )
        for batch in loader:
            tokens = batch["context_tokens"]
            mask = batch["context_mask"]
            ids = batch.get("block_id",
                            torch.arange(tokens.shape[0]))
            if torch.cuda.is_available():
                tokens, mask = tokens.cuda(), mask.cuda()
            embeds = self.model.embed_context(tokens, mask)
            self.block_data.add_block_data(
                ids.tolist(), embeds.float().cpu().numpy()
            )
        rank = (torch.distributed.get_rank()
                if torch.distributed.is_initialized() else 0)
        self.block_data.save_shard(rank)
        return self.block_data
