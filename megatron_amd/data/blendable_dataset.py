"""Weighted multi-dataset mixing (reference megatron/data/blendable_dataset.py).

The reference computes blending indices in C++ (helpers.cpp:20-80); here the
same greedy weight-balancing loop runs vectorized-enough in numpy at build
time (one pass over num_samples)."""

from __future__ import annotations

import time

import numpy as np
import torch

from ..utils import print_rank_0


def _build_blending_indices(dataset_index, dataset_sample_index, weights,
                            num_datasets, size, verbose):
    """Greedy assignment: at each step pick the dataset whose current sampled
    fraction is furthest below its weight (same algorithm as
    helpers.cpp build_blending_indices)."""
    current_samples = np.zeros(num_datasets, dtype=np.int64)
    for sample_idx in range(size):
        errors = weights * (sample_idx + 1) - current_samples
        max_error_index = np.argmax(errors)
        dataset_index[sample_idx] = max_error_index
        dataset_sample_index[sample_idx] = current_samples[max_error_index]
        current_samples[max_error_index] += 1


class BlendableDataset(torch.utils.data.Dataset):
    def __init__(self, datasets, weights):
        self.datasets = datasets
        num_datasets = len(datasets)
        assert num_datasets == len(weights)

        self.size = 0
        for dataset in self.datasets:
            self.size += len(dataset)

        weights = np.array(weights, dtype=np.float64)
        sum_weights = np.sum(weights)
        assert sum_weights > 0.0
        weights /= sum_weights

        start_time = time.time()
        assert num_datasets < 255
        self.dataset_index = np.zeros(self.size, dtype=np.uint8)
        self.dataset_sample_index = np.zeros(self.size, dtype=np.int64)
        verbose = (torch.distributed.get_rank() == 0
                   if torch.distributed.is_initialized() else True)
        from ..ops import ext as _ext

        mod = _ext.load(required=False)
        if mod is not None and hasattr(mod, "build_blending_indices"):
            di = torch.from_numpy(self.dataset_index)
            dsi = torch.from_numpy(self.dataset_sample_index)
            mod.build_blending_indices(
                di, dsi, torch.from_numpy(weights), num_datasets, self.size,
                verbose,
            )
        else:
            _build_blending_indices(
                self.dataset_index, self.dataset_sample_index, weights,
                num_datasets, self.size, verbose,
            )
        print_rank_0(
            f"> elapsed time for building blendable dataset indices: "
            f"{time.time() - start_time:.2f} (sec)"
        )

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        dataset_idx = self.dataset_index[idx]
        sample_idx = self.dataset_sample_index[idx]
        return self.datasets[dataset_idx][sample_idx]
