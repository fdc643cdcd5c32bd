"""GPT pretraining dataset: epoch-spanning sample extraction over concatenated
documents, with cached doc/sample/shuffle index mappings.

Reference: megatron/data/gpt_dataset.py:20-513. sample_idx is built by the
native C++ helper (ops/csrc/data_helpers.cpp, the analog of the reference's
helpers.cpp build_sample_idx) when the extension is built, with a vectorized
numpy path (cumsum + searchsorted, same output) as the no-extension fallback.
"""

from __future__ import annotations

import os
import time

import numpy as np
import torch

from .. import parallel as mpu
from ..utils import print_rank_0
from .blendable_dataset import BlendableDataset
from .indexed_dataset import make_dataset as make_indexed_dataset


def get_datasets_weights_and_num_samples(data_prefix, train_valid_test_num_samples):
    """(reference dataset_utils.py:47-92)"""
    assert len(data_prefix) % 2 == 0 or len(data_prefix) == 1
    if len(data_prefix) == 1:
        return [data_prefix[0]], [1.0], [train_valid_test_num_samples]
    num_datasets = len(data_prefix) // 2
    weights = [0] * num_datasets
    prefixes = [0] * num_datasets
    for i in range(num_datasets):
        weights[i] = float(data_prefix[2 * i])
        prefixes[i] = (data_prefix[2 * i + 1]).strip()
    weight_sum = sum(weights)
    assert weight_sum > 0.0
    weights = [weight / weight_sum for weight in weights]
    datasets_train_valid_test_num_samples = []
    for weight in weights:
        datasets_train_valid_test_num_samples.append(
            [
                int(np.ceil(val * weight * 1.005))
                for val in train_valid_test_num_samples
            ]
        )
    return prefixes, weights, datasets_train_valid_test_num_samples


def get_train_valid_test_split_(splits_string, size):
    """(reference dataset_utils.py:94-125)"""
    splits = []
    if splits_string.find(",") != -1:
        splits = [float(s) for s in splits_string.split(",")]
    elif splits_string.find("/") != -1:
        splits = [float(s) for s in splits_string.split("/")]
    else:
        splits = [float(splits_string)]
    while len(splits) < 3:
        splits.append(0.0)
    splits = splits[:3]
    splits_sum = sum(splits)
    assert splits_sum > 0.0
    splits = [split / splits_sum for split in splits]
    splits_index = [0]
    for index, split in enumerate(splits):
        splits_index.append(splits_index[index] + int(round(split * float(size))))
    diff = splits_index[-1] - size
    for index in range(1, len(splits_index)):
        splits_index[index] -= diff
    assert len(splits_index) == 4
    assert splits_index[-1] == size
    return splits_index


def build_train_valid_test_datasets(data_prefix, data_impl, splits_string,
                                    train_valid_test_num_samples, seq_length,
                                    seed, skip_warmup,
                                    train_data_prefix=None,
                                    valid_data_prefix=None,
                                    test_data_prefix=None):
    """(reference gpt_dataset.py:20-124). When per-split prefixes are given
    (--train_data_path / --valid_data_path / --test_data_path) each split is
    built over its own corpus and the split string is ignored."""
    if train_data_prefix or valid_data_prefix or test_data_prefix:
        def one_split(prefix_list, n):
            if not prefix_list:
                return None
            if len(prefix_list) == 1:
                ds, _, _ = _build_train_valid_test_datasets(
                    prefix_list[0], data_impl, "1,0,0", [n, 0, 0],
                    seq_length, seed, skip_warmup,
                )
                return ds
            prefixes, weights, nums = get_datasets_weights_and_num_samples(
                prefix_list, [n, 0, 0]
            )
            parts = [
                _build_train_valid_test_datasets(
                    pref, data_impl, "1,0,0", [nm[0], 0, 0], seq_length,
                    seed, skip_warmup,
                )[0]
                for pref, nm in zip(prefixes, nums)
            ]
            return BlendableDataset(parts, weights)

        return (
            one_split(train_data_prefix, train_valid_test_num_samples[0]),
            one_split(valid_data_prefix, train_valid_test_num_samples[1]),
            one_split(test_data_prefix, train_valid_test_num_samples[2]),
        )

    if len(data_prefix) == 1:
        return _build_train_valid_test_datasets(
            data_prefix[0], data_impl, splits_string,
            train_valid_test_num_samples, seq_length, seed, skip_warmup,
        )

    prefixes, weights, datasets_train_valid_test_num_samples = (
        get_datasets_weights_and_num_samples(
            data_prefix, train_valid_test_num_samples
        )
    )
    train_datasets, valid_datasets, test_datasets = [], [], []
    for i in range(len(prefixes)):
        train_ds, valid_ds, test_ds = _build_train_valid_test_datasets(
            prefixes[i], data_impl, splits_string,
            datasets_train_valid_test_num_samples[i], seq_length, seed,
            skip_warmup,
        )
        if train_ds:
            train_datasets.append(train_ds)
        if valid_ds:
            valid_datasets.append(valid_ds)
        if test_ds:
            test_datasets.append(test_ds)

    blending_train_dataset = None
    if train_datasets:
        blending_train_dataset = BlendableDataset(train_datasets, weights)
    blending_valid_dataset = None
    if valid_datasets:
        blending_valid_dataset = BlendableDataset(valid_datasets, weights)
    blending_test_dataset = None
    if test_datasets:
        blending_test_dataset = BlendableDataset(test_datasets, weights)
    return (blending_train_dataset, blending_valid_dataset,
            blending_test_dataset)


def _build_train_valid_test_datasets(data_prefix, data_impl, splits_string,
                                     train_valid_test_num_samples, seq_length,
                                     seed, skip_warmup):
    indexed_dataset = get_indexed_dataset_(data_prefix, data_impl, skip_warmup)
    total_num_of_documents = indexed_dataset.sizes.shape[0]
    splits = get_train_valid_test_split_(splits_string, total_num_of_documents)

    print_rank_0(" > dataset split:")

    def print_split_stats(name, index):
        print_rank_0(f"    {name}:")
        print_rank_0(
            f"     document indices in [{splits[index]}, {splits[index + 1]}) "
            f"total of {splits[index + 1] - splits[index]} documents"
        )

    print_split_stats("train", 0)
    print_split_stats("validation", 1)
    print_split_stats("test", 2)

    def build_dataset(index, name):
        dataset = None
        if splits[index + 1] > splits[index]:
            documents = np.arange(
                start=splits[index], stop=splits[index + 1], step=1,
                dtype=np.int32,
            )
            dataset = GPTDataset(
                name, data_prefix, documents, indexed_dataset,
                train_valid_test_num_samples[index], seq_length, seed,
            )
        return dataset

    train_dataset = build_dataset(0, "train")
    valid_dataset = build_dataset(1, "valid")
    test_dataset = build_dataset(2, "test")
    return train_dataset, valid_dataset, test_dataset


def get_indexed_dataset_(data_prefix, data_impl, skip_warmup):
    print_rank_0(" > building dataset index ...")
    start_time = time.time()
    indexed_dataset = make_indexed_dataset(data_prefix, data_impl, skip_warmup)
    print_rank_0(
        f" > finished creating indexed dataset in "
        f"{time.time() - start_time:4f} seconds"
    )
    print_rank_0(f"    number of documents: {indexed_dataset.sizes.shape[0]}")
    return indexed_dataset


class GPTDataset(torch.utils.data.Dataset):
    """(reference gpt_dataset.py:221-269)"""

    def __init__(self, name, data_prefix, documents, indexed_dataset,
                 num_samples, seq_length, seed):
        self.name = name
        self.indexed_dataset = indexed_dataset
        assert np.min(documents) >= 0
        assert np.max(documents) < indexed_dataset.sizes.shape[0]

        self.doc_idx, self.sample_idx, self.shuffle_idx = _build_index_mappings(
            self.name, data_prefix, documents, self.indexed_dataset.sizes,
            num_samples, seq_length, seed,
        )
        # the sample mapping yields seq_length+1 tokens; finetune.get_batch
        # splits them into input/label views
        self.seq_length = seq_length

    def __len__(self):
        return self.sample_idx.shape[0] - 1

    def __getitem__(self, idx):
        idx = self.shuffle_idx[idx]
        doc_index_f = self.sample_idx[idx][0]
        doc_index_l = self.sample_idx[idx + 1][0]
        offset_f = self.sample_idx[idx][1]
        offset_l = self.sample_idx[idx + 1][1]
        if doc_index_f == doc_index_l:
            sample = self.indexed_dataset.get(
                self.doc_idx[doc_index_f], offset=offset_f,
                length=offset_l - offset_f + 1,
            )
        else:
            sample_list = [
                self.indexed_dataset.get(self.doc_idx[doc_index_f],
                                         offset=offset_f)
            ]
            for i in range(doc_index_f + 1, doc_index_l):
                sample_list.append(self.indexed_dataset.get(self.doc_idx[i]))
            sample_list.append(
                self.indexed_dataset.get(
                    self.doc_idx[doc_index_l], length=offset_l + 1
                )
            )
            sample = np.concatenate(sample_list)
        return {"text": np.array(sample, dtype=np.int64)}


def _num_tokens(documents, sizes):
    return np.sum(sizes[documents])


def _num_epochs(tokens_per_epoch, seq_length, num_samples):
    num_epochs = 0
    total_tokens = 0
    while True:
        num_epochs += 1
        total_tokens += tokens_per_epoch
        if ((total_tokens - 1) // seq_length) >= num_samples:
            return num_epochs


def _build_doc_idx(documents, num_epochs, np_rng, separate_last_epoch):
    """(reference gpt_dataset.py:409-427)"""
    if not separate_last_epoch or num_epochs == 1:
        doc_idx = np.mgrid[0:num_epochs, 0:len(documents)][1]
        doc_idx[:] = documents
        doc_idx = doc_idx.reshape(-1)
        doc_idx = doc_idx.astype(np.int32)
        np_rng.shuffle(doc_idx)
        return doc_idx
    doc_idx_first = _build_doc_idx(documents, num_epochs - 1, np_rng, False)
    doc_idx_last = _build_doc_idx(documents, 1, np_rng, False)
    return np.concatenate((doc_idx_first, doc_idx_last))


def _build_sample_idx(sizes, doc_idx, seq_length, num_samples):
    """Vectorized equivalent of helpers.cpp build_sample_idx
    (reference helpers.cpp:99-162): sample i begins at flat token position
    i*seq_length within the epoch-concatenated shuffled documents."""
    doc_sizes = sizes[doc_idx].astype(np.int64)
    doc_ends = np.cumsum(doc_sizes)  # exclusive end position of each doc
    boundaries = np.arange(num_samples + 1, dtype=np.int64) * seq_length
    # doc containing each boundary token
    doc_pos = np.searchsorted(doc_ends, boundaries, side="right")
    doc_starts = doc_ends - doc_sizes
    offsets = boundaries - doc_starts[np.minimum(doc_pos, len(doc_sizes) - 1)]
    sample_idx = np.empty((num_samples + 1, 2), dtype=np.int64)
    sample_idx[:, 0] = doc_pos
    sample_idx[:, 1] = offsets
    return sample_idx


def _build_shuffle_idx(num_samples, total_size, np_rng):
    """(reference gpt_dataset.py:487-513)"""
    dtype_ = np.uint32
    if total_size >= (np.iinfo(np.uint32).max - 1):
        dtype_ = np.int64
    shuffle_idx_first = np.arange(
        start=0, stop=num_samples, step=1, dtype=dtype_
    )
    np_rng.shuffle(shuffle_idx_first)
    if num_samples == total_size:
        return shuffle_idx_first
    shuffle_idx_last = np.arange(
        start=num_samples, stop=total_size, step=1, dtype=dtype_
    )
    np_rng.shuffle(shuffle_idx_last)
    return np.concatenate((shuffle_idx_first, shuffle_idx_last))


def _build_index_mappings(name, data_prefix, documents, sizes, num_samples,
                          seq_length, seed):
    """Build (and cache as .npy) the doc/sample/shuffle indices
    (reference gpt_dataset.py:272-406)."""
    tokens_per_epoch = _num_tokens(documents, sizes)
    num_epochs = _num_epochs(tokens_per_epoch, seq_length, num_samples)
    np_rng = np.random.RandomState(seed=seed)

    _filename = data_prefix
    _filename += f"_{name}_indexmap"
    _filename += f"_{num_samples}ns"
    _filename += f"_{seq_length}sl"
    _filename += f"_{seed}s"
    doc_idx_filename = _filename + "_doc_idx.npy"
    sample_idx_filename = _filename + "_sample_idx.npy"
    shuffle_idx_filename = _filename + "_shuffle_idx.npy"

    build_on_this_rank = (
        not torch.distributed.is_initialized()
        or torch.distributed.get_rank() == 0
    )
    if build_on_this_rank and (
        not os.path.isfile(doc_idx_filename)
        or not os.path.isfile(sample_idx_filename)
        or not os.path.isfile(shuffle_idx_filename)
    ):
        print_rank_0(
            " > WARNING: could not find index map files, building on rank 0 ..."
        )
        if num_epochs == 1:
            separate_last_epoch = False
        else:
            num_samples_from_epochs_minus_one = (
                (num_epochs - 1) * tokens_per_epoch - 1
            ) // seq_length
            last_epoch_num_samples = (
                num_samples - num_samples_from_epochs_minus_one
            )
            assert last_epoch_num_samples >= 0
            num_samples_per_epoch = (tokens_per_epoch - 1) // seq_length
            assert last_epoch_num_samples <= (num_samples_per_epoch + 1)
            separate_last_epoch = (
                last_epoch_num_samples < int(0.80 * num_samples_per_epoch)
            )

        start_time = time.time()
        doc_idx = _build_doc_idx(documents, num_epochs, np_rng,
                                 separate_last_epoch)
        np.save(doc_idx_filename, doc_idx, allow_pickle=True)

        num_samples_ = (num_epochs * tokens_per_epoch - 1) // seq_length
        from ..ops import ext as _ext

        _mod = _ext.load(required=False)
        if _mod is not None and hasattr(_mod, "build_sample_idx"):
            sample_idx = _mod.build_sample_idx(
                torch.from_numpy(np.ascontiguousarray(sizes, dtype=np.int32)),
                torch.from_numpy(np.ascontiguousarray(doc_idx,
                                                      dtype=np.int32)),
                seq_length, num_samples_,
            ).numpy()
        else:
            sample_idx = _build_sample_idx(sizes, doc_idx, seq_length,
                                           num_samples_)
        np.save(sample_idx_filename, sample_idx, allow_pickle=True)

        if separate_last_epoch:
            num_samples_shuffle = num_samples_from_epochs_minus_one
        else:
            num_samples_shuffle = sample_idx.shape[0] - 1
        shuffle_idx = _build_shuffle_idx(
            num_samples_shuffle, sample_idx.shape[0] - 1, np_rng
        )
        np.save(shuffle_idx_filename, shuffle_idx, allow_pickle=True)
        print_rank_0(
            f" > elapsed time to build and save index mapping files: "
            f"{time.time() - start_time} (seconds)"
        )

    # barrier via all-reduce so other ranks wait for the files
    # (reference gpt_dataset.py:378-386)
    if torch.distributed.is_initialized():
        counts = torch.tensor(
            [1], dtype=torch.long,
            device="cuda" if torch.cuda.is_available() else "cpu",
        )
        torch.distributed.all_reduce(counts,
                                     group=mpu.get_data_parallel_group())
        torch.distributed.all_reduce(
            counts, group=mpu.get_pipeline_model_parallel_group()
        )
        assert counts[0].item() == (
            torch.distributed.get_world_size()
            // torch.distributed.get_world_size(
                group=mpu.get_tensor_model_parallel_group()
            )
        )

    doc_idx = np.load(doc_idx_filename, allow_pickle=True, mmap_mode="r")
    sample_idx = np.load(sample_idx_filename, allow_pickle=True, mmap_mode="r")
    shuffle_idx = np.load(shuffle_idx_filename, allow_pickle=True, mmap_mode="r")
    print_rank_0(
        f"    total number of samples: {sample_idx.shape[0]}"
    )
    print_rank_0(f"    total number of epochs: {num_epochs}")
    return doc_idx, sample_idx, shuffle_idx
