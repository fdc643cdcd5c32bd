"""Checkpoint resharding: change TP/PP partitioning of a saved checkpoint.

Reference: tools/checkpoint_util.py:6-156 with the loader/saver plugin
protocol (checkpoint_loader_megatron.py / checkpoint_saver_megatron.py)
connected by an mp.Queue. The message protocol is preserved: the loader
reconstructs full (unsharded) tensors TP-merging shard-by-shard and streams
named messages ("embeddings", "transformer layer N", "final layernorm",
"lm head") to the saver, which re-splits them for the target tp/pp and
writes megatron checkpoint directories.

  python tools/checkpoint_util.py --model_type llama2 \
      --load_dir ckpt_in --save_dir ckpt_out \
      --target_tensor_parallel_size 2 --target_pipeline_parallel_size 2
"""

from __future__ import annotations

import argparse
import os
import sys

import torch

sys.path.append(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from megatron_amd.checkpointing import (  # noqa: E402
    get_checkpoint_tracker_filename,
)


def _load_shards(load_dir):
    tracker = get_checkpoint_tracker_filename(load_dir)
    with open(tracker) as f:
        meta = f.read().strip()
    release = meta == "release"
    iteration = 0 if release else int(meta)
    sub = "release" if release else f"iter_{iteration:07d}"

    base = os.path.join(load_dir, sub)
    shard_dirs = sorted(
        d for d in os.listdir(base) if d.startswith("mp_rank_")
    )
    # organize by (tp, pp)
    shards = {}
    for d in shard_dirs:
        parts = d.split("_")
        tp = int(parts[2])
        pp = int(parts[3]) if len(parts) > 3 else 0
        path = os.path.join(base, d, "model_optim_rng.pt")
        if not os.path.exists(path):
            path = os.path.join(base, d, "model_rng.pt")
        shards[(tp, pp)] = torch.load(path, map_location="cpu",
                                      weights_only=False)
    tp_size = max(k[0] for k in shards) + 1
    pp_size = max(k[1] for k in shards) + 1
    return shards, tp_size, pp_size, iteration, release


# TP merge/split rules per parameter kind. The fused QKV is sharded along
# dim 0 in units of whole KV groups, so plain dim-0 concat/split is correct
# as long as n_kv_heads % tp == 0 (enforced by the model).
_DIM0_KEYS = (
    "word_embeddings.weight", "lm_head",
    "query_key_value.weight", "query_key_value.bias",
    "dense_h_to_4h.weight", "dense_h_to_4h.bias",
)
_DIM1_KEYS = ("dense.weight", "dense_4h_to_h.weight")
_REPLICATED = ("layernorm", "norm.weight", "norm.bias", ".bias",
               "position_embeddings")


def _merge_kind(key):
    for k in _DIM1_KEYS:
        if key.endswith(k):
            return 1
    for k in _DIM0_KEYS:
        if key.endswith(k):
            return 0
    return None  # replicated


def _glu_aware_cat(key, tensors, ffn_hidden_size, glu):
    """dense_h_to_4h packs [up; gate] per TP shard — merging shards must
    interleave halves, not plain-concat (reference
    checkpoint_loader_megatron.py handles this via args.glu_activation)."""
    if glu and key.endswith("dense_h_to_4h.weight"):
        ups, gates = [], []
        for t in tensors:
            up, gate = torch.chunk(t, 2, dim=0)
            ups.append(up)
            gates.append(gate)
        return torch.cat(ups + gates, dim=0)
    return torch.cat(tensors, dim=0)


def _glu_aware_split(key, tensor, tp, glu):
    if glu and key.endswith("dense_h_to_4h.weight"):
        up, gate = torch.chunk(tensor, 2, dim=0)
        ups = torch.chunk(up, tp, dim=0)
        gates = torch.chunk(gate, tp, dim=0)
        return [torch.cat([u, g], dim=0) for u, g in zip(ups, gates)]
    return list(torch.chunk(tensor, tp, dim=0))


def merge_full_state(shards, tp_size, pp_size, num_layers, glu):
    """Loader side: reconstruct the full unsharded model dict."""
    full = {}
    layer_offset = 0
    for pp in range(pp_size):
        models = []
        for tp in range(tp_size):
            m = shards[(tp, pp)]["model"]
            if "language_model" in m:
                m = m["language_model"]
            models.append(m)
        keys = models[0].keys()
        n_local_layers = len(
            {k.split(".")[2] for k in keys if k.startswith("encoder.layers.")}
        )
        for key in keys:
            out_key = key
            if key.startswith("encoder.layers."):
                parts = key.split(".")
                local_idx = int(parts[2])
                parts[2] = str(local_idx + layer_offset)
                out_key = ".".join(parts)
            kind = _merge_kind(key)
            tensors = [m[key] for m in models]
            if kind == 0:
                full[out_key] = _glu_aware_cat(key, tensors, None, glu)
            elif kind == 1:
                full[out_key] = torch.cat(tensors, dim=1)
            else:
                full[out_key] = tensors[0]
        layer_offset += n_local_layers
    return full


def split_full_state(full, tp, pp, num_layers, glu):
    """Saver side: re-split for the target tp/pp."""
    assert num_layers % pp == 0
    layers_per_stage = num_layers // pp
    out = {}
    for tpr in range(tp):
        for ppr in range(pp):
            out[(tpr, ppr)] = {}

    for key, tensor in full.items():
        kind = _merge_kind(key)
        if key.startswith("encoder.layers."):
            parts = key.split(".")
            gidx = int(parts[2])
            ppr = gidx // layers_per_stage
            parts[2] = str(gidx % layers_per_stage)
            local_key = ".".join(parts)
            targets = [ppr]
        elif key.startswith("embedding.") or key == "lm_head":
            # embedding first stage; lm_head / final LN last stage
            targets = [0] if key.startswith("embedding.") else [pp - 1]
            local_key = key
        elif key.startswith("encoder.final_layernorm"):
            targets = [pp - 1]
            local_key = key
        else:
            targets = list(range(pp))
            local_key = key

        if kind == 0:
            pieces = _glu_aware_split(key, tensor, tp, glu)
        elif kind == 1:
            pieces = list(torch.chunk(tensor, tp, dim=1))
        else:
            pieces = [tensor] * tp
        for ppr in targets:
            for tpr in range(tp):
                out[(tpr, ppr)][local_key] = pieces[tpr]
    return out


# ---------------------------------------------------------------------------
# Streaming (bounded-memory) reshard. The reference bounds memory with a
# loader subprocess + mp.Queue streaming one named tensor group at a time
# (tools/checkpoint_util.py:13-86); here a single process achieves the same
# bound: source shard files are memory-mapped (clean, OS-evictable pages)
# and opened one source-PP stage at a time, and each TARGET rank's state is
# assembled and written before the next target starts — peak anonymous RSS
# is ~one target shard (model/(tp*pp)) + transient merged tensors, never the
# full unsharded model.


class _ShardCache:
    """mmap-backed loader of source shard payloads, one PP stage resident at
    a time. `max_loaded` is exposed for the bounded-memory test."""

    def __init__(self, paths, tp_size):
        self.paths = paths
        self.tp_size = tp_size
        self._stage = None
        self._stage_models = None
        self.max_loaded = 0
        self.load_calls = 0

    def _load(self, path):
        self.load_calls += 1
        try:
            return torch.load(path, map_location="cpu", weights_only=False,
                              mmap=True)
        except Exception:
            return torch.load(path, map_location="cpu", weights_only=False)

    def meta(self):
        """args/iteration payload from shard (0,0) — loaded and dropped."""
        payload = self._load(self.paths[(0, 0)])
        meta = {"args": payload.get("args"),
                "iteration": payload.get("iteration", 0)}
        del payload
        return meta

    def stage_models(self, pp):
        if self._stage != pp:
            self._stage_models = None
            models = []
            for tp in range(self.tp_size):
                payload = self._load(self.paths[(tp, pp)])
                m = payload["model"]
                if "language_model" in m:
                    m = m["language_model"]
                models.append(m)
            self._stage = pp
            self._stage_models = models
            self.max_loaded = max(self.max_loaded, len(models))
        return self._stage_models


def _shard_paths(load_dir):
    tracker = get_checkpoint_tracker_filename(load_dir)
    with open(tracker) as f:
        meta = f.read().strip()
    release = meta == "release"
    iteration = 0 if release else int(meta)
    sub = "release" if release else f"iter_{iteration:07d}"
    base = os.path.join(load_dir, sub)
    paths = {}
    for d in sorted(os.listdir(base)):
        if not d.startswith("mp_rank_"):
            continue
        parts = d.split("_")
        tp = int(parts[2])
        pp = int(parts[3]) if len(parts) > 3 else 0
        path = os.path.join(base, d, "model_optim_rng.pt")
        if not os.path.exists(path):
            path = os.path.join(base, d, "model_rng.pt")
        paths[(tp, pp)] = path
    tp_size = max(k[0] for k in paths) + 1
    pp_size = max(k[1] for k in paths) + 1
    return paths, tp_size, pp_size, iteration, release


def _merge_key(models, key, glu):
    kind = _merge_kind(key)
    tensors = [m[key] for m in models]
    if kind == 0:
        return _glu_aware_cat(key, tensors, None, glu)
    if kind == 1:
        return torch.cat(tensors, dim=1)
    return tensors[0]


def stream_reshard(load_dir, save_dir, tp, pp, glu, progress=print):
    """Bounded-memory reshard loop: one target rank's dict in flight."""
    paths, tp_src, pp_src, iteration, release = _shard_paths(load_dir)
    cache = _ShardCache(paths, tp_src)
    meta = cache.meta()
    margs = meta["args"]
    num_layers = getattr(margs, "num_layers")
    assert num_layers % pp == 0
    layers_per_tgt = num_layers // pp

    # source stage layer spans (count local layers per source stage lazily)
    src_spans = []
    offset = 0
    for sp in range(pp_src):
        models = cache.stage_models(sp)
        n_local = len({k.split(".")[2] for k in models[0].keys()
                       if k.startswith("encoder.layers.")})
        src_spans.append((offset, offset + n_local))
        offset += n_local
    assert offset == num_layers, (offset, num_layers)

    sub = "release" if release else f"iter_{iteration:07d}"
    if margs is not None:
        margs.tensor_model_parallel_size = tp
        margs.pipeline_model_parallel_size = pp

    def split_piece(key, merged, tpr):
        kind = _merge_kind(key)
        if kind == 0:
            return _glu_aware_split(key, merged, tp, glu)[tpr]
        if kind == 1:
            return torch.chunk(merged, tp, dim=1)[tpr]
        return merged

    # iterate targets outer pp (sequential source stages), inner tp
    for ppr in range(pp):
        lo, hi = ppr * layers_per_tgt, (ppr + 1) * layers_per_tgt
        for tpr in range(tp):
            sd = {}
            # non-layer keys (fetch the source stage only when this target
            # actually needs it — the cache holds ONE stage at a time)
            if ppr == 0:
                first_models = cache.stage_models(0)
                for key in list(first_models[0].keys()):
                    if key.startswith("embedding."):
                        sd[key] = split_piece(key, _merge_key(first_models,
                                                              key, glu), tpr)
            if ppr == pp - 1:
                last_models = cache.stage_models(pp_src - 1)
                for key in list(last_models[0].keys()):
                    if (key.startswith("encoder.final_layernorm")
                            or key == "lm_head"):
                        sd[key] = split_piece(key, _merge_key(last_models,
                                                              key, glu), tpr)
            # layers in [lo, hi) from their source stages
            for sp, (slo, shi) in enumerate(src_spans):
                if shi <= lo or slo >= hi:
                    continue
                models = cache.stage_models(sp)
                for key in list(models[0].keys()):
                    if not key.startswith("encoder.layers."):
                        continue
                    parts = key.split(".")
                    gidx = int(parts[2]) + slo
                    if not (lo <= gidx < hi):
                        continue
                    parts[2] = str(gidx - lo)
                    local_key = ".".join(parts)
                    sd[local_key] = split_piece(key, _merge_key(models, key,
                                                                glu), tpr)
            if pp == 1:
                d = os.path.join(save_dir, sub, f"mp_rank_{tpr:02d}")
            else:
                d = os.path.join(save_dir, sub, f"mp_rank_{tpr:02d}_{ppr:03d}")
            os.makedirs(d, exist_ok=True)
            state = {
                "args": margs,
                "checkpoint_version": 3.0,
                "iteration": iteration,
                "model": {"language_model": sd},
            }
            torch.save(state, os.path.join(d, "model_optim_rng.pt"))
            progress(f"  wrote tp{tpr} pp{ppr} ({len(sd)} tensors)")
            del sd, state
    with open(get_checkpoint_tracker_filename(save_dir), "w") as f:
        f.write("release" if release else str(iteration))
    return cache


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_type", default="llama2",
                        choices=["gpt", "llama", "llama2", "codellama",
                                 "falcon", "mistral"])
    parser.add_argument("--load_dir", required=True)
    parser.add_argument("--save_dir", required=True)
    parser.add_argument("--target_tensor_parallel_size", type=int, default=1)
    parser.add_argument("--target_pipeline_parallel_size", type=int,
                        default=1)
    args = parser.parse_args()

    paths, tp_size, pp_size, _, _ = _shard_paths(args.load_dir)
    glu = args.model_type in ("llama", "llama2", "codellama", "mistral")
    # honor the saved args' glu flag when present
    probe = torch.load(paths[(0, 0)], map_location="cpu", weights_only=False)
    margs = probe.get("args")
    if margs is not None and getattr(margs, "glu_activation", None):
        glu = True
    del probe

    print(f"streaming reshard from tp={tp_size} pp={pp_size} "
          f"-> tp={args.target_tensor_parallel_size} "
          f"pp={args.target_pipeline_parallel_size}")
    stream_reshard(args.load_dir, args.save_dir,
                   args.target_tensor_parallel_size,
                   args.target_pipeline_parallel_size, glu)
    print(f"saved resharded checkpoint to {args.save_dir}")


if __name__ == "__main__":
    main()
