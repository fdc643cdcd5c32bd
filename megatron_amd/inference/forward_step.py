"""Inference forward pass with KV-cache and pipeline-aware micro-batching
(reference megatron/text_generation/forward_step.py:17-204)."""

from __future__ import annotations

import torch

from .. import parallel as mpu
from ..config import get_config
from .communication import (
    recv_from_prev_pipeline_rank_,
    send_to_next_pipeline_rank,
)


class InferenceParams:
    """(reference forward_step.py:17-41)"""

    def __init__(self, max_batch_size, max_sequence_len):
        self.max_sequence_len = max_sequence_len
        self.max_batch_size = max_batch_size
        self.sequence_len_offset = 0
        self.batch_size_offset = 0
        self.key_value_memory_dict = {}

    def swap_key_value_dict(self, batch_idx):
        if len(self.key_value_memory_dict) == 0:
            raise ValueError("should not swap when dict in empty")
        for layer_number in self.key_value_memory_dict.keys():
            inference_key_memory, inference_value_memory = (
                self.key_value_memory_dict[layer_number]
            )
            assert len(batch_idx) == inference_key_memory.shape[1]
            # swap IN-PLACE so cache pointers stay valid (a captured decode
            # graph holds them)
            inference_key_memory.copy_(inference_key_memory[:, batch_idx])
            inference_value_memory.copy_(
                inference_value_memory[:, batch_idx]
            )


class ForwardStep:
    """Forward step wrapper handling pipelining (reference :44-204).

    MI355X addition: single-token decode steps are captured as a HIP graph
    (torch.cuda.CUDAGraph is hipGraph on ROCm) and replayed — the decode
    inner loop is launch-bound (~1k tiny kernels per token at 7B), so one
    graph launch replaces the whole per-token launch storm. Enabled when
    `cfg.use_hip_graph_decode`, TP=PP=1, on GPU; the prompt step and any
    variable-shape step stay eager."""

    def __init__(self, model, max_batch_size, max_sequence_len):
        assert not isinstance(model, list)
        self.model = model
        self.inference_params = InferenceParams(max_batch_size,
                                                max_sequence_len)
        self.pipeline_size_larger_than_one = (
            mpu.get_pipeline_model_parallel_world_size() > 1
        )
        cfg = get_config()
        self.pipelining_batch_x_seqlen = (
            cfg.inference_batch_times_seqlen_threshold
        )
        self._graph = None
        self._graph_batch = None
        self._use_graph_decode = (
            getattr(cfg, "use_hip_graph_decode", False)
            and torch.cuda.is_available()
            and not self.pipeline_size_larger_than_one
            and mpu.get_tensor_model_parallel_world_size() == 1
        )

    def __call__(self, tokens, position_ids, attention_mask):
        if self.pipeline_size_larger_than_one:
            current_batch_x_seqlen = tokens.size(0) * tokens.size(1)
            if current_batch_x_seqlen >= self.pipelining_batch_x_seqlen:
                micro_batch_size = max(
                    1, self.pipelining_batch_x_seqlen // tokens.size(1)
                )
                return self._with_pipelining_forward_step(
                    tokens, position_ids, attention_mask, micro_batch_size
                )
        if (
            self._use_graph_decode
            and tokens.size(1) == 1
            and self.inference_params.key_value_memory_dict
        ):
            return self._graph_decode_step(tokens, position_ids)
        return self._no_pipelining_forward_step(
            tokens, position_ids, attention_mask
        )

    def _init_graph_state(self, batch_size):
        ip = self.inference_params
        device = torch.cuda.current_device()
        ip.use_graph = True
        ip.graph_pos = torch.zeros(1, dtype=torch.long, device=device)
        ip.graph_arange = torch.arange(ip.max_sequence_len, device=device)
        self._s_tokens = torch.zeros(batch_size, 1, dtype=torch.long,
                                     device=device)
        self._s_pos = torch.zeros(batch_size, 1, dtype=torch.long,
                                  device=device)
        self._graph_batch = batch_size

    def _graph_decode_step(self, tokens, position_ids):
        ip = self.inference_params
        batch_size = tokens.size(0)
        if self._graph_batch is None:
            self._init_graph_state(batch_size)
        assert batch_size == self._graph_batch

        # stage this step's inputs into the static buffers the graph reads
        self._s_tokens.copy_(tokens)
        self._s_pos.copy_(position_ids)
        ip.graph_pos.fill_(ip.sequence_len_offset)

        if self._graph is None:
            # warm-up eager run through the static path (allocator, blas
            # workspaces), then capture the second run
            ip.use_graph = True
            self.model(self._s_tokens, self._s_pos, None,
                       inference_params=ip)
            torch.cuda.synchronize()
            self._graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self._graph):
                self._s_logits = self.model(self._s_tokens, self._s_pos,
                                            None, inference_params=ip)
            # capture only RECORDS the work — replay to actually execute
            # this step (the warm-up already wrote the same KV slot, and the
            # replay rewrites it identically)
            self._graph.replay()
        else:
            self._graph.replay()
        ip.sequence_len_offset += 1
        return self._s_logits

    def _forward(self, tokens, position_ids, attention_mask):
        return self.model(
            tokens, position_ids, attention_mask,
            inference_params=self.inference_params,
        )

    def _no_pipelining_forward_step(self, tokens, position_ids,
                                    attention_mask, recv_buffer=None):
        output_tensor = _forward_step_helper(
            self.model, tokens, position_ids, attention_mask,
            self.inference_params, recv_buffer=recv_buffer,
        )
        self.inference_params.sequence_len_offset += tokens.size(1)
        logits = None
        if mpu.is_pipeline_last_stage():
            logits = output_tensor
        return logits

    def _with_pipelining_forward_step(self, tokens, position_ids,
                                      attention_mask, micro_batch_size):
        sequence_length = tokens.size(1)
        batch_size = tokens.size(0)
        num_micro_batches, last_chunk = divmod(batch_size, micro_batch_size)
        if last_chunk > 0:
            num_micro_batches += 1

        logits = None
        if mpu.is_pipeline_last_stage():
            cfg = get_config()
            logits = torch.empty(
                (batch_size, sequence_length, cfg.padded_vocab_size),
                dtype=torch.float32,
                device=torch.cuda.current_device()
                if torch.cuda.is_available() else "cpu",
            )

        for micro_batch_index in range(num_micro_batches):
            start = micro_batch_index * micro_batch_size
            end = min(start + micro_batch_size, batch_size)
            tokens2use = tokens[start:end, ...]
            position_ids2use = position_ids[start:end, ...]
            self.inference_params.batch_size_offset = start
            output = _forward_step_helper(
                self.model, tokens2use, position_ids2use, attention_mask,
                self.inference_params,
            )
            if mpu.is_pipeline_last_stage():
                logits[start:end, ...] = output

        self.inference_params.batch_size_offset = 0
        self.inference_params.sequence_len_offset += sequence_length
        return logits


def _get_recv_buffer_dtype(cfg):
    return cfg.params_dtype


def _forward_step_helper(model, tokens, position_ids, attention_mask,
                         inference_params, recv_buffer=None):
    """Single forward with pipeline send/recv (reference :106-142)."""
    cfg = get_config()
    batch_size = tokens.size(0)
    sequence_length = tokens.size(1)
    if not mpu.is_pipeline_first_stage():
        if recv_buffer is None:
            recv_buffer = torch.empty(
                (sequence_length, batch_size, cfg.hidden_size),
                dtype=_get_recv_buffer_dtype(cfg),
                device=torch.cuda.current_device()
                if torch.cuda.is_available() else "cpu",
            )
        recv_from_prev_pipeline_rank_(recv_buffer)
        model.set_input_tensor(recv_buffer)

    output_tensor = model(
        tokens, position_ids, attention_mask,
        inference_params=inference_params,
    )

    if not mpu.is_pipeline_last_stage():
        send_to_next_pipeline_rank(output_tensor)
    return output_tensor
