"""Dense retrieval index over block embeddings (reference
megatron/data/realm_index.py, condensed): build, save, load and
exact-inner-product search — the MI355X path does brute-force matmul search
on-GPU (288 GB HBM holds hundreds of millions of 128-d embeddings; no faiss
in this environment)."""

from __future__ import annotations

import os
import pickle

import numpy as np
import torch


class BlockData:
    """block id -> embedding store."""

    def __init__(self, embedding_path=None, load_from_path=False):
        self.embed_data = {}
        self.embedding_path = embedding_path
        if load_from_path and embedding_path and os.path.exists(embedding_path):
            self.load_from_file()

    def add_block_data(self, block_indices, block_embeds):
        for idx, embed in zip(block_indices, block_embeds):
            self.embed_data[int(idx)] = np.asarray(embed, dtype=np.float16)

    def save_shard(self, rank=0):
        path = f"{self.embedding_path}.rank{rank}"
        with open(path, "wb") as f:
            pickle.dump(self.embed_data, f)

    def merge_shards_and_save(self, paths):
        for p in paths:
            with open(p, "rb") as f:
                self.embed_data.update(pickle.load(f))
        with open(self.embedding_path, "wb") as f:
            pickle.dump(self.embed_data, f)

    def load_from_file(self):
        with open(self.embedding_path, "rb") as f:
            self.embed_data = pickle.load(f)

    def clear(self):
        self.embed_data = {}


class FaissMIPSIndex:
    """Exact max-inner-product search; name kept for reference parity, the
    backend is a dense GPU matmul."""

    def __init__(self, embed_size, embed_data=None, use_gpu=True):
        self.embed_size = embed_size
        self.ids = None
        self.matrix = None
        self.use_gpu = use_gpu and torch.cuda.is_available()
        if embed_data is not None:
            self.add_block_embed_data(embed_data)

    def add_block_embed_data(self, block_data: BlockData):
        ids = sorted(block_data.embed_data.keys())
        mat = np.stack([block_data.embed_data[i] for i in ids]).astype(
            np.float32
        )
        self.ids = np.asarray(ids)
        self.matrix = torch.from_numpy(mat)
        if self.use_gpu:
            self.matrix = self.matrix.cuda()

    def search_mips_index(self, query_embeds, top_k, reconstruct=False):
        q = torch.as_tensor(query_embeds, dtype=torch.float32)
        if self.use_gpu:
            q = q.cuda()
        scores = q @ self.matrix.t()
        top = torch.topk(scores, k=min(top_k, scores.shape[1]), dim=1)
        idx = self.ids[top.indices.cpu().numpy()]
        if reconstruct:
            return top.values.cpu().numpy(), self.matrix[
                top.indices.cpu()
            ].cpu().numpy(), idx
        return top.values.cpu().numpy(), idx
