"""Transformer block: parallel MLP / attention / layer / stack.

Reference: megatron/model/transformer.py (ParallelMLP :77-141, CoreAttention
:144-277, ParallelAttention :280-560, ParallelTransformerLayer :612-846,
ParallelTransformer :927-1282). Hidden states are [s, b, h] throughout.

MI355X mapping:
 - QKV / MLP GEMMs -> hipBLASLt via torch.matmul inside the TP linear layers.
 - attention -> CDNA4 flash-attention kernel (ops/csrc/flash_attn.hip):
   LDS-staged K/V tiles, MFMA bf16, online softmax, causal + sliding window,
   GQA without K/V head expansion. The fallback CoreAttention path keeps the fused
   scale+mask+softmax HIP kernel.
 - norms / glu / rope / bias-dropout-add -> fused HIP elementwise kernels.
"""

from __future__ import annotations

import math
from contextlib import nullcontext

import torch
import torch.nn.functional as F

from .. import parallel as mpu
from ..ops import functional as ops_f
from .enums import AttnMaskType, AttnType, LayerType
from .module import MegatronModule
from .norms import get_norm
from .rope import apply_rotary_emb, precompute_freqs


def _args_device():
    return torch.cuda.current_device() if torch.cuda.is_available() else "cpu"


class ParallelMLP(MegatronModule):
    """h -> ffn (x2 for GLU) -> act -> h (reference transformer.py:77-141)."""

    def __init__(self, cfg, init_method, output_layer_init_method,
                 world_size=None):
        super().__init__()
        self.cfg = cfg
        self.glu = cfg.glu_activation
        ffn_mult = 2 if self.glu else 1

        self.dense_h_to_4h = mpu.ColumnParallelLinear(
            cfg.hidden_size, cfg.ffn_hidden_size * ffn_mult,
            bias=cfg.use_bias, gather_output=False, init_method=init_method,
            skip_bias_add=True, params_dtype=cfg.params_dtype,
            fp8=getattr(cfg, "fp8", False),
            use_cpu_initialization=cfg.use_cpu_initialization,
            perform_initialization=cfg.perform_initialization,
            gradient_accumulation_fusion=cfg.gradient_accumulation_fusion,
            sequence_parallel_enabled=cfg.sequence_parallel,
            async_tensor_model_parallel_allreduce=(
                not cfg.no_async_tensor_model_parallel_allreduce
                and not cfg.sequence_parallel
            ),
            world_size=world_size,
        )
        self.bias_gelu_fusion = cfg.bias_gelu_fusion and not self.glu and cfg.use_bias
        self.activation_func = F.gelu

        self.dense_4h_to_h = mpu.RowParallelLinear(
            cfg.ffn_hidden_size, cfg.hidden_size,
            bias=cfg.use_bias, input_is_parallel=True,
            init_method=output_layer_init_method, skip_bias_add=True,
            params_dtype=cfg.params_dtype, fp8=getattr(cfg, "fp8", False),
            use_cpu_initialization=cfg.use_cpu_initialization,
            perform_initialization=cfg.perform_initialization,
            gradient_accumulation_fusion=cfg.gradient_accumulation_fusion,
            sequence_parallel_enabled=cfg.sequence_parallel,
            world_size=world_size,
        )

    def forward(self, hidden_states):
        intermediate, bias = self.dense_h_to_4h(hidden_states)
        if self.glu:
            if bias is not None:
                intermediate = intermediate + bias
            intermediate = ops_f.glu_activation(intermediate, self.glu)
        elif self.bias_gelu_fusion:
            intermediate = F.gelu(intermediate + bias)
        else:
            if bias is not None:
                intermediate = intermediate + bias
            intermediate = self.activation_func(intermediate)
        output, output_bias = self.dense_4h_to_h(intermediate)
        return output, output_bias


class CoreAttention(MegatronModule):
    """Unfused attention: QK^T -> fused scaled-masked-softmax -> PV
    (reference transformer.py:144-277). Used when flash attention is off or
    shapes are unsupported."""

    def __init__(self, cfg, layer_number, attn_mask_type=AttnMaskType.causal):
        super().__init__()
        self.fp16 = cfg.fp16
        self.bf16 = cfg.bf16
        self.attention_softmax_in_fp32 = cfg.attention_softmax_in_fp32
        self.layer_number = max(1, layer_number)
        self.attn_mask_type = attn_mask_type
        self.sequence_parallel = cfg.sequence_parallel

        projection_size = cfg.kv_channels * cfg.num_attention_heads
        world_size = mpu.get_tensor_model_parallel_world_size()
        self.hidden_size_per_partition = projection_size // world_size
        self.hidden_size_per_attention_head = cfg.kv_channels
        self.num_attention_heads_per_partition = (
            cfg.num_attention_heads // world_size
        )

        coeff = None
        self.norm_factor = math.sqrt(self.hidden_size_per_attention_head)
        if cfg.apply_query_key_layer_scaling:
            coeff = self.layer_number
            self.norm_factor *= coeff
        self.coeff = coeff
        self.attention_dropout = torch.nn.Dropout(cfg.attention_dropout)

    def forward(self, query_layer, key_layer, value_layer, attention_mask):
        # [s, b, np, hn]
        sq, b, np_, hn = query_layer.shape
        sk = key_layer.shape[0]

        # [b*np, sq, sk]
        q = query_layer.reshape(sq, b * np_, hn).transpose(0, 1)
        k = key_layer.reshape(sk, b * np_, hn).transpose(0, 1)

        matmul_result = torch.empty(
            b * np_, sq, sk, dtype=q.dtype, device=q.device
        )
        matmul_result = torch.baddbmm(
            matmul_result, q, k.transpose(1, 2),
            beta=0.0, alpha=(1.0 / self.norm_factor),
        )
        attention_scores = matmul_result.view(b, np_, sq, sk)

        scale = self.coeff if self.coeff is not None else 1.0
        causal = self.attn_mask_type == AttnMaskType.causal and (
            attention_mask is None or sq > 1
        )
        probs = ops_f.scaled_masked_softmax(
            attention_scores,
            attention_mask if not causal else None,
            scale, causal,
        )

        if not self.sequence_parallel:
            with mpu.get_cuda_rng_tracker().fork():
                probs = self.attention_dropout(probs)
        else:
            probs = self.attention_dropout(probs)

        v = value_layer.reshape(sk, b * np_, hn).transpose(0, 1)
        context = torch.bmm(probs.view(b * np_, sq, sk), v)
        context = context.view(b, np_, sq, hn).permute(2, 0, 1, 3)
        return context.reshape(sq, b, np_ * hn)


class FlashSelfAttention(torch.nn.Module):
    """CDNA4 flash-attention wrapper; q/k/v arrive [s, b, n, h] and are fed to
    the kernel as [b, s, n, h] (reference transformer.py:345-404 wrapping
    flash_attn_func)."""

    def __init__(self, causal=True, softmax_scale=None, attention_dropout=0.0,
                 window_size=None):
        super().__init__()
        self.causal = causal
        self.softmax_scale = softmax_scale
        self.dropout_p = attention_dropout
        self.window_size = window_size

    def forward(self, q, k, v):
        # [s,b,n,h] -> [b,s,n,h] as VIEWS: the FA kernels are stride-aware
        # over batch/seq, so no transpose copies are materialized and the
        # kernel writes its output straight into an sbhd buffer (the final
        # [s,b,n*h] reshape below is then also a view)
        q, k, v = (x.transpose(0, 1) for x in (q, k, v))
        out = ops_f.flash_attention(
            q, k, v, causal=self.causal, softmax_scale=self.softmax_scale,
            window_size=self.window_size, dropout_p=self.dropout_p,
            training=self.training,
        )
        # [b,s,n,h] -> [s,b,n*h]
        b, s, n, h = out.shape
        return out.transpose(0, 1).reshape(s, b, n * h)


class ParallelAttention(MegatronModule):
    """Self-attention with fused QKV column-parallel projection, GQA/MQA, RoPE
    and KV-cache (reference transformer.py:280-560).

    QKV weight layout matches the reference checkpoint format: per KV group
    [q_1..q_nq, k, v] each of kv_channels rows (transformer.py:450-465), so
    HF<->Megatron conversion (permute_qkv) carries over unchanged."""

    def __init__(self, cfg, init_method, output_layer_init_method, layer_number,
                 attention_type=AttnType.self_attn,
                 attn_mask_type=AttnMaskType.causal, world_size=None):
        super().__init__()
        self.cfg = cfg
        self.layer_number = max(1, layer_number)
        self.attention_type = attention_type
        self.attn_mask_type = attn_mask_type
        self.params_dtype = cfg.params_dtype
        self.sequence_parallel = cfg.sequence_parallel

        self.num_attention_heads = cfg.num_attention_heads
        self.num_attention_heads_kv = cfg.num_attention_heads_kv
        projection_size = cfg.kv_channels * cfg.num_attention_heads

        if world_size is None:
            world_size = mpu.get_tensor_model_parallel_world_size()
        self.hidden_size_per_attention_head = cfg.kv_channels
        self.num_attention_heads_per_partition = mpu.divide(
            cfg.num_attention_heads, world_size
        )
        self.num_attention_heads_kv_per_partition = mpu.divide(
            cfg.num_attention_heads_kv, world_size
        )
        self.n_rep = (
            self.num_attention_heads_per_partition
            // self.num_attention_heads_kv_per_partition
        )

        linear_kwargs = dict(
            bias=cfg.use_bias, gather_output=False, init_method=init_method,
            params_dtype=cfg.params_dtype, fp8=getattr(cfg, "fp8", False),
            use_cpu_initialization=cfg.use_cpu_initialization,
            perform_initialization=cfg.perform_initialization,
            gradient_accumulation_fusion=cfg.gradient_accumulation_fusion,
            sequence_parallel_enabled=cfg.sequence_parallel,
            async_tensor_model_parallel_allreduce=(
                not cfg.no_async_tensor_model_parallel_allreduce
                and not cfg.sequence_parallel
            ),
            world_size=world_size,
        )
        if attention_type == AttnType.self_attn:
            qkv_out = cfg.kv_channels * (
                cfg.num_attention_heads + 2 * cfg.num_attention_heads_kv
            )
            self.query_key_value = mpu.ColumnParallelLinear(
                cfg.hidden_size, qkv_out, **linear_kwargs
            )
        else:
            # cross attention (T5 decoder): separate q and fused kv
            assert cfg.num_attention_heads == cfg.num_attention_heads_kv, (
                "cross attention does not support GQA"
            )
            self.query = mpu.ColumnParallelLinear(
                cfg.hidden_size, projection_size, **linear_kwargs
            )
            self.key_value = mpu.ColumnParallelLinear(
                cfg.hidden_size, 2 * projection_size, **linear_kwargs
            )

        self.use_flash_attn = cfg.use_flash_attn
        if self.use_flash_attn:
            self.flash_attention = FlashSelfAttention(
                causal=(attn_mask_type == AttnMaskType.causal),
                attention_dropout=cfg.attention_dropout,
                window_size=cfg.sliding_window_size,
            )
        self.core_attention = CoreAttention(cfg, self.layer_number, attn_mask_type)

        self.dense = mpu.RowParallelLinear(
            projection_size, cfg.hidden_size,
            bias=cfg.use_bias, input_is_parallel=True,
            init_method=output_layer_init_method, skip_bias_add=True,
            params_dtype=cfg.params_dtype, fp8=getattr(cfg, "fp8", False),
            use_cpu_initialization=cfg.use_cpu_initialization,
            perform_initialization=cfg.perform_initialization,
            gradient_accumulation_fusion=cfg.gradient_accumulation_fusion,
            sequence_parallel_enabled=cfg.sequence_parallel,
            world_size=world_size,
        )

        if cfg.position_embedding_type == "rotary":
            cos, sin = precompute_freqs(
                self.hidden_size_per_attention_head,
                cfg.max_position_embeddings,
                theta=cfg.rope_theta,
                scaling_factor=cfg.rope_scaling_factor,
            )
            self.register_buffer("rope_cos", cos, persistent=False)
            self.register_buffer("rope_sin", sin, persistent=False)
        else:
            self.rope_cos = None
            self.rope_sin = None

    def _allocate_kv_cache(self, inference_max_seq_len, batch_size, device, dtype):
        return torch.empty(
            inference_max_seq_len, batch_size,
            self.num_attention_heads_kv_per_partition,
            self.hidden_size_per_attention_head,
            dtype=dtype, device=device,
        )

    def forward(self, hidden_states, attention_mask, position_ids=None,
                inference_params=None, encoder_output=None):
        # hidden_states [s, b, h]
        sq, b = hidden_states.shape[0], hidden_states.shape[1]
        np_ = self.num_attention_heads_per_partition
        nkv = self.num_attention_heads_kv_per_partition
        hn = self.hidden_size_per_attention_head

        if self.attention_type == AttnType.self_attn:
            mixed, _ = self.query_key_value(hidden_states)
            if self.sequence_parallel:
                sq = mixed.shape[0]
            # per-group layout [nq+2, hn] within each kv group
            ngroups = nkv
            nq_per_group = np_ // nkv
            mixed = mixed.view(sq, b, ngroups, (nq_per_group + 2) * hn)
            # training fast path: split + RoPE as ONE fused autograd node —
            # q/k read as strided views, rotated by the stride-aware RoPE
            # kernel, and the backward assembles d_mixed in a single buffer
            # (no per-slice zeros+copy+add chains)
            if (
                mixed.is_cuda
                and self.rope_cos is not None
                and inference_params is None
            ):
                cos = self.rope_cos.to(device=mixed.device,
                                       dtype=torch.float32)
                sin = self.rope_sin.to(device=mixed.device,
                                       dtype=torch.float32)
                query, key, value = ops_f.fused_qkv_split_rope(
                    mixed, cos[:sq], sin[:sq], np_, nkv, hn
                )
                return self._attend(query, key, value, attention_mask,
                                    inference_params)
            # k/v (and q when nq_per_group == 1) stay uniformly-strided VIEWS
            # into the fused projection output: the FA and RoPE kernels are
            # stride-aware, so no slice-materializing copies are needed. A
            # GQA query (heads non-uniform across groups) still reshapes.
            if nq_per_group == 1:
                query = mixed[..., :hn]
            else:
                query = mixed[..., : nq_per_group * hn].reshape(
                    sq, b, np_, hn
                )
            key = mixed[..., nq_per_group * hn : (nq_per_group + 1) * hn]
            value = mixed[..., (nq_per_group + 1) * hn :]
        else:
            # cross attention: q from decoder states, kv from encoder output
            q_out, _ = self.query(hidden_states)
            kv_out, _ = self.key_value(encoder_output)
            if self.sequence_parallel:
                sq = q_out.shape[0]
            sk = kv_out.shape[0]
            query = q_out.reshape(sq, b, np_, hn)
            kv = kv_out.view(sk, b, np_, 2 * hn)
            key = kv[..., :hn].contiguous()
            value = kv[..., hn:].contiguous()

        # fused decode step: {rope + cache-append}, {attention} — three
        # kernels total (with the dense gemv) instead of the ~20-op eager
        # chain; all positions device-held so the step is graph-capturable
        if (
            self.rope_cos is not None
            and inference_params is not None
            and getattr(inference_params, "use_graph", False)
            and sq == 1
            and self.layer_number in inference_params.key_value_memory_dict
            and self.attention_type == AttnType.self_attn
            and query.is_cuda
            and query.dtype == torch.bfloat16
            and hn in (64, 128)
            and not torch.is_grad_enabled()
        ):
            from ..ops import ext as _oext

            _m = _oext.load(required=True)
            ip = inference_params
            k_cache, v_cache = ip.key_value_memory_dict[self.layer_number]
            cos = self.rope_cos.to(device=query.device, dtype=torch.float32)
            sin = self.rope_sin.to(device=query.device, dtype=torch.float32)
            q_rot = _m.decode_rope_append(
                query[0], key[0], value[0], cos, sin, ip.graph_pos,
                k_cache, v_cache,
            )
            win = -1
            if self.use_flash_attn and self.flash_attention.window_size:
                win = int(self.flash_attention.window_size)
            ctx_row = _m.decode_attn(q_rot, k_cache, v_cache, ip.graph_pos,
                                     1.0 / math.sqrt(hn), win)
            return self.dense(ctx_row.unsqueeze(0))

        # rotary embedding
        if self.rope_cos is not None:
            # tables stay fp32 regardless of module dtype conversion
            cos = self.rope_cos.to(device=query.device, dtype=torch.float32)
            sin = self.rope_sin.to(device=query.device, dtype=torch.float32)
            if inference_params is not None:
                offset = inference_params.sequence_len_offset
                if position_ids is None:
                    pos = torch.arange(
                        offset, offset + sq, device=query.device
                    ).unsqueeze(0).expand(b, sq)
                else:
                    pos = position_ids
                query, key = apply_rotary_emb(query, key, cos, sin, pos)
            else:
                query, key = apply_rotary_emb(query, key, cos, sin, None)

        # hipGraph-capturable decode: every op below is shape-static (the
        # current length lives in device tensors), so ForwardStep can capture
        # one decode step as a HIP graph and replay it per token — the decode
        # loop is launch-bound, exactly what the graph removes.
        if (
            inference_params is not None
            and getattr(inference_params, "use_graph", False)
            and sq == 1
            and self.layer_number in inference_params.key_value_memory_dict
        ):
            return self._static_decode(query, key, value, inference_params)

        # KV cache (reference transformer.py:412-419, 492-505)
        if inference_params is not None:
            if self.layer_number not in inference_params.key_value_memory_dict:
                inference_params.key_value_memory_dict[self.layer_number] = (
                    self._allocate_kv_cache(
                        inference_params.max_sequence_len, b,
                        key.device, key.dtype,
                    ),
                    self._allocate_kv_cache(
                        inference_params.max_sequence_len, b,
                        value.device, value.dtype,
                    ),
                )
            k_cache, v_cache = inference_params.key_value_memory_dict[
                self.layer_number
            ]
            start = inference_params.sequence_len_offset
            k_cache[start : start + sq] = key
            v_cache[start : start + sq] = value
            key = k_cache[: start + sq]
            value = v_cache[: start + sq]

        return self._attend(query, key, value, attention_mask,
                            inference_params)

    def _static_decode(self, query, key, value, inference_params):
        """One decode step with every shape fixed: KV written by tensor
        index, attention over the FULL cache with a length mask read from a
        device tensor — capturable as a HIP graph (see
        inference/forward_step.py)."""
        ip = inference_params
        k_cache, v_cache = ip.key_value_memory_dict[self.layer_number]
        # [1,b,nkv,h] written at the device-held position
        k_cache.index_copy_(0, ip.graph_pos, key)
        v_cache.index_copy_(0, ip.graph_pos, value)

        b = query.shape[1]
        n, hn = query.shape[2], query.shape[3]

        if query.is_cuda and hn in (64, 128) and query.dtype == torch.bfloat16:
            # fused decode attention (ops/csrc/decode_attn.hip): one kernel
            # replaces the fp32-cast + bmm + mask + softmax + bmm chain
            from ..ops import ext as _oext

            ctx_row = _oext.load(required=True).decode_attn(
                query[0].contiguous(), k_cache, v_cache, ip.graph_pos,
                1.0 / math.sqrt(hn),
            )
            output, bias = self.dense(ctx_row.unsqueeze(0))
            return output, bias
        nkv = k_cache.shape[2]
        L = k_cache.shape[0]
        q = query.permute(1, 2, 0, 3).reshape(b * n, 1, hn)  # [b*n,1,h]
        k = k_cache.permute(1, 2, 0, 3)  # [b,nkv,L,h]
        v = v_cache.permute(1, 2, 0, 3)
        if n != nkv:
            rep = n // nkv
            k = k.repeat_interleave(rep, dim=1)
            v = v.repeat_interleave(rep, dim=1)
        k = k.reshape(b * n, L, hn)
        v = v.reshape(b * n, L, hn)

        scale = 1.0 / math.sqrt(hn)
        scores = torch.bmm(q.float(), k.float().transpose(1, 2)) * scale
        invalid = ip.graph_arange > ip.graph_pos  # [L] device-side length
        if self.use_flash_attn and self.flash_attention.window_size:
            # sliding window (HF/Mistral: keep w keys incl. current)
            w = int(self.flash_attention.window_size)
            invalid = invalid | (ip.graph_arange < ip.graph_pos - w + 1)
        scores = scores.masked_fill(invalid.view(1, 1, L), float("-inf"))
        probs = torch.softmax(scores, dim=-1).to(v.dtype)
        ctx = torch.bmm(probs, v)  # [b*n,1,h]
        ctx = ctx.view(b, n, hn).view(b, n * hn).unsqueeze(0)  # [1,b,n*h]
        output, bias = self.dense(ctx)
        return output, bias

    def _attend(self, query, key, value, attention_mask, inference_params):
        sq = query.shape[0]
        use_flash = (
            self.use_flash_attn
            and self.attention_type == AttnType.self_attn
            and (inference_params is None or sq > 1)
        )
        if use_flash:
            context = self.flash_attention(query, key, value)
        else:
            # expand kv heads for the unfused path
            if self.n_rep > 1:
                key = key.repeat_interleave(self.n_rep, dim=2)
                value = value.repeat_interleave(self.n_rep, dim=2)
            context = self.core_attention(query, key, value, attention_mask)

        output, bias = self.dense(context)
        return output, bias


class ParallelTransformerLayer(MegatronModule):
    """One transformer layer: pre/post-LN, attention, MLP, fused residual adds
    (reference transformer.py:612-846). Supports Falcon parallel-attention and
    parallel-layernorm variants."""

    def __init__(self, cfg, init_method, output_layer_init_method, layer_number,
                 layer_type=LayerType.encoder,
                 self_attn_mask_type=AttnMaskType.causal, world_size=None):
        super().__init__()
        self.cfg = cfg
        self.layer_number = layer_number
        self.layer_type = layer_type
        self.apply_residual_connection_post_layernorm = (
            cfg.apply_residual_connection_post_layernorm
        )
        self.use_post_ln = cfg.use_post_ln
        self.parallel_attn = cfg.parallel_attn
        self.parallel_layernorm = cfg.parallel_layernorm
        self.fp32_residual_connection = cfg.fp32_residual_connection
        self.hidden_dropout = cfg.hidden_dropout
        self.bias_dropout_fusion = cfg.bias_dropout_fusion

        self.input_layernorm = get_norm(cfg)
        self.self_attention = ParallelAttention(
            cfg, init_method, output_layer_init_method, layer_number,
            attn_mask_type=self_attn_mask_type, world_size=world_size,
        )
        if self.parallel_layernorm:
            self.mlp_layernorm = get_norm(cfg)
        if not self.parallel_attn:
            self.post_attention_layernorm = get_norm(cfg)
        if self.layer_type == LayerType.decoder:
            # T5-style cross attention over the encoder output
            self.inter_attention = ParallelAttention(
                cfg, init_method, output_layer_init_method, layer_number,
                attention_type=AttnType.cross_attn,
                attn_mask_type=AttnMaskType.padding, world_size=world_size,
            )
            self.post_inter_attention_layernorm = get_norm(cfg)
        self.mlp = ParallelMLP(cfg, init_method, output_layer_init_method,
                               world_size=world_size)

    def forward(self, hidden_states, attention_mask, position_ids=None,
                inference_params=None, encoder_output=None,
                enc_dec_attn_mask=None):
        # [s, b, h]. Where hidden_states feeds both a norm and the residual
        # stream, forward_res returns a pass-through second output so the
        # residual gradient is folded into the norm backward in-kernel
        # (no autograd fan-in add — see RMSNormResFunction).
        fuse_res = not self.apply_residual_connection_post_layernorm
        if self.parallel_layernorm:
            # Falcon-40B: chain both norms' pass-throughs so each tensor has
            # a single extra consumer
            mlp_ln_out, h_mid = self.mlp_layernorm.forward_res(hidden_states)
            if fuse_res:
                ln_out, residual = self.input_layernorm.forward_res(h_mid)
            else:
                ln_out = self.input_layernorm(h_mid)
                residual = hidden_states
        elif fuse_res:
            ln_out, residual = self.input_layernorm.forward_res(hidden_states)
        else:
            residual = hidden_states
            ln_out = self.input_layernorm(hidden_states)

        attn_out, attn_bias = self.self_attention(
            ln_out, attention_mask, position_ids=position_ids,
            inference_params=inference_params,
        )

        if self.apply_residual_connection_post_layernorm:
            residual = ln_out

        if self.parallel_attn:
            # Falcon: MLP input = same LN output; single residual add at end
            if self.parallel_layernorm:
                mlp_in = mlp_ln_out
            else:
                mlp_in = ln_out
            mlp_out, mlp_bias = self.mlp(mlp_in)
            out = attn_out + mlp_out
            bias = None
            if attn_bias is not None:
                bias = attn_bias + (mlp_bias if mlp_bias is not None else 0)
            out = ops_f.bias_dropout_add(
                out, bias, residual, self.hidden_dropout, self.training
            )
            return out

        attn_res = ops_f.bias_dropout_add(
            attn_out, attn_bias, residual, self.hidden_dropout, self.training
        )

        if self.layer_type == LayerType.decoder:
            ln_cross, attn_res_b = (
                self.post_attention_layernorm.forward_res(attn_res)
            )
            cross_out, cross_bias = self.inter_attention(
                ln_cross, enc_dec_attn_mask, encoder_output=encoder_output
            )
            attn_res = ops_f.bias_dropout_add(
                cross_out, cross_bias, attn_res_b, self.hidden_dropout,
                self.training,
            )
            if fuse_res:
                ln2_out, residual2 = (
                    self.post_inter_attention_layernorm.forward_res(attn_res)
                )
            else:
                ln2_out = self.post_inter_attention_layernorm(attn_res)
                residual2 = ln2_out
        elif fuse_res:
            ln2_out, residual2 = (
                self.post_attention_layernorm.forward_res(attn_res)
            )
        else:
            ln2_out = self.post_attention_layernorm(attn_res)
            residual2 = ln2_out

        mlp_out, mlp_bias = self.mlp(ln2_out)
        out = ops_f.bias_dropout_add(
            mlp_out, mlp_bias, residual2, self.hidden_dropout, self.training
        )
        if self.use_post_ln:
            out = self.input_layernorm(out)
        return out


class ParallelTransformer(MegatronModule):
    """Stack of layers with activation recompute and pipeline input plumbing
    (reference transformer.py:927-1282)."""

    def __init__(self, cfg, init_method, output_layer_init_method,
                 layer_type=LayerType.encoder,
                 self_attn_mask_type=AttnMaskType.causal,
                 pre_process=True, post_process=True, world_size=None):
        super().__init__()
        self.cfg = cfg
        self.pre_process = pre_process
        self.post_process = post_process
        self.input_tensor = None
        self.recompute_granularity = cfg.recompute_granularity
        self.recompute_method = cfg.recompute_method
        self.recompute_num_layers = cfg.recompute_num_layers
        self.distribute_saved_activations = (
            cfg.distribute_saved_activations and not cfg.sequence_parallel
        )
        self.sequence_parallel = cfg.sequence_parallel

        # layer partition across pipeline stages (reference
        # transformer.py:875-924 _get_num_layers)
        pp = cfg.pipeline_model_parallel_size
        vpp = cfg.virtual_pipeline_model_parallel_size
        assert cfg.num_layers % pp == 0, "num_layers must divide pp size"
        if vpp is not None:
            assert cfg.num_layers % (pp * vpp) == 0
            self.num_layers = cfg.num_layers // (pp * vpp)
        else:
            self.num_layers = cfg.num_layers // pp

        if mpu.model_parallel_is_initialized():
            pp_rank = mpu.get_pipeline_model_parallel_rank()
        else:
            pp_rank = 0
        if vpp is not None:
            vpp_rank = mpu.get_virtual_pipeline_model_parallel_rank() or 0
            offset = vpp_rank * (cfg.num_layers // vpp) + pp_rank * self.num_layers
        else:
            offset = pp_rank * self.num_layers

        self.layers = torch.nn.ModuleList(
            [
                ParallelTransformerLayer(
                    cfg, init_method, output_layer_init_method,
                    layer_number=i + 1 + offset, layer_type=layer_type,
                    self_attn_mask_type=self_attn_mask_type,
                    world_size=world_size,
                )
                for i in range(self.num_layers)
            ]
        )

        if self.post_process:
            self.final_layernorm = get_norm(cfg)

    def _get_layer(self, i):
        return self.layers[i]

    def set_input_tensor(self, input_tensor):
        self.input_tensor = input_tensor

    def _checkpointed_forward(self, hidden_states, attention_mask,
                              position_ids, encoder_output=None,
                              enc_dec_attn_mask=None):
        def custom(start, end):
            def custom_forward(*args):
                x = args[0]
                for index in range(start, end):
                    layer = self._get_layer(index)
                    x = layer(x, attention_mask, position_ids=position_ids,
                              encoder_output=encoder_output,
                              enc_dec_attn_mask=enc_dec_attn_mask)
                return x

            return custom_forward

        if self.recompute_method == "uniform" or self.recompute_method is None:
            chunk = self.recompute_num_layers
            l = 0
            while l < self.num_layers:
                hidden_states = mpu.checkpoint(
                    custom(l, min(l + chunk, self.num_layers)),
                    self.distribute_saved_activations, hidden_states,
                )
                l += chunk
        elif self.recompute_method == "block":
            for l in range(self.num_layers):
                if l < self.recompute_num_layers:
                    hidden_states = mpu.checkpoint(
                        custom(l, l + 1),
                        self.distribute_saved_activations, hidden_states,
                    )
                else:
                    hidden_states = custom(l, l + 1)(hidden_states)
        else:
            raise ValueError(self.recompute_method)
        return hidden_states

    def forward(self, hidden_states, attention_mask, position_ids=None,
                inference_params=None, encoder_output=None,
                enc_dec_attn_mask=None):
        if not self.pre_process:
            hidden_states = self.input_tensor

        if self.sequence_parallel:
            rng_context = mpu.get_cuda_rng_tracker().fork()
        else:
            rng_context = nullcontext()

        with rng_context:
            if self.recompute_granularity == "full" and self.training:
                hidden_states = self._checkpointed_forward(
                    hidden_states, attention_mask, position_ids,
                    encoder_output, enc_dec_attn_mask,
                )
            else:
                for index in range(self.num_layers):
                    layer = self._get_layer(index)
                    hidden_states = layer(
                        hidden_states, attention_mask,
                        position_ids=position_ids,
                        inference_params=inference_params,
                        encoder_output=encoder_output,
                        enc_dec_attn_mask=enc_dec_attn_mask,
                    )

        if self.post_process:
            hidden_states = self.final_layernorm(hidden_states)
        return hidden_states
