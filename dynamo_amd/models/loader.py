"""Local HF-checkpoint loading: safetensors + config.json -> our modules.

Reference parity: lib/llm/src/local_model (the reference points engines at
a local HF snapshot; hub.rs fetching is out of scope here - no network -
but a user's on-disk checkpoint must load). Maps HF Llama/Qwen2/Mixtral
tensor names onto this build's fused/TP-sharded layouts:

  q_proj/k_proj/v_proj (+bias)   -> Attention.wqkv/bqkv rows (q | k | v),
                                    q sharded by rank, k/v by the rank's
                                    kv-head index (replicated when
                                    tp > num_kv_heads)
  o_proj                         -> Attention.wo columns (rank slice)
  gate_proj/up_proj              -> SwiGLUMLP.w_gate_up rows (gate | up),
                                    each I-sharded by rank
  down_proj                      -> SwiGLUMLP.w_down columns
  block_sparse_moe.gate          -> MoEMLP.router
  experts.e.{w1,w3,w2}           -> MoEMLP.w_gate_up / w_down (expert-
                                    sliced under EP, I-sliced under MoE-TP)
  *_layernorm / model.norm       -> *_norm_w
  embed_tokens / lm_head         -> embed / lm_head (tied falls back)

Round-trip tested (export our random-init weights under HF names, load
into a fresh model, outputs must be bit-identical) including TP shard
slicing - see tests/test_weight_loading.py.
"""
from __future__ import annotations

import json
import os
from typing import Dict, Iterator, Tuple

import torch

from dynamo_amd.engine.config import ModelConfig


# ---------------------------------------------------------------------------
def config_from_hf(path: str) -> ModelConfig:
    """Build a ModelConfig from an HF checkpoint dir's config.json."""
    with open(os.path.join(path, "config.json")) as f:
        hf = json.load(f)
    archs = [a.lower() for a in hf.get("architectures", [])]
    if any("mixtral" in a for a in archs):
        arch = "mixtral"
    elif any("qwen2" in a for a in archs):
        arch = "qwen2"
    else:
        arch = "llama"
    heads = hf["num_attention_heads"]
    return ModelConfig(
        name=os.path.basename(os.path.normpath(path)),
        arch="llama" if arch == "qwen2" else arch,
        hidden_size=hf["hidden_size"],
        intermediate_size=hf["intermediate_size"],
        num_layers=hf["num_hidden_layers"],
        num_q_heads=heads,
        num_kv_heads=hf.get("num_key_value_heads", heads),
        head_dim=hf.get("head_dim", hf["hidden_size"] // heads),
        vocab_size=hf["vocab_size"],
        max_position=hf.get("max_position_embeddings", 8192),
        rope_theta=float(hf.get("rope_theta", 10000.0)),
        rope_scaling=hf.get("rope_scaling"),
        rms_eps=float(hf.get("rms_norm_eps", 1e-5)),
        tie_embeddings=bool(hf.get("tie_word_embeddings", False)),
        num_experts=hf.get("num_local_experts", 0),
        num_experts_per_tok=hf.get("num_experts_per_tok", 2),
        attn_bias=(arch == "qwen2"),
    )


def _iter_safetensors(path: str) -> Iterator[Tuple[str, torch.Tensor]]:
    from safetensors import safe_open
    files = sorted(f for f in os.listdir(path) if f.endswith(".safetensors"))
    if not files:
        raise FileNotFoundError(f"no *.safetensors under {path}")
    for fn in files:
        with safe_open(os.path.join(path, fn), framework="pt") as f:
            for key in f.keys():
                yield key, f.get_tensor(key)


def _shard(full: torch.Tensor, dim: int, n: int, idx: int) -> torch.Tensor:
    size = full.shape[dim] // n
    return full.narrow(dim, idx * size, size)


# ---------------------------------------------------------------------------
def load_weights(model, path: str, strict: bool = True) -> int:
    """Load an HF checkpoint dir into a built model (in place). Returns the
    number of checkpoint tensors consumed."""
    cfg = model.cfg
    tp = model.tp
    hd = cfg.head_dim
    kv_idx = (tp.rank if tp.size <= cfg.num_kv_heads
              else (tp.rank * cfg.num_kv_heads) // tp.size)
    consumed = 0
    seen = set()

    def put(dst: torch.Tensor, src: torch.Tensor):
        if dst.shape != src.shape:
            raise ValueError(f"shape mismatch {tuple(dst.shape)} vs "
                             f"{tuple(src.shape)}")
        with torch.no_grad():
            dst.copy_(src.to(dst.dtype))

    def attn_rows(layer, kind: str, full: torch.Tensor, bias: bool):
        at = layer.attn
        dst = at.bqkv if bias else at.wqkv
        if dst is None:
            raise ValueError("checkpoint has qkv bias but attn_bias=False")
        q_rows = at.hq * hd
        k_rows = at.hkv * hd
        if kind == "q":
            put(dst[:q_rows],
                _shard(full, 0, tp.size, tp.rank) if tp.size > 1 else full)
        else:
            # kv_idx picks this rank's kv-head window; covers both the
            # sharded (tp <= kv heads) and replicated (tp > kv heads) cases
            shard = full.narrow(0, kv_idx * k_rows, k_rows)
            off = q_rows if kind == "k" else q_rows + k_rows
            put(dst[off:off + k_rows], shard)

    for name, t in _iter_safetensors(path):
        seen.add(name)
        consumed += 1
        if name == "model.embed_tokens.weight":
            put(model.embed, t)
            if cfg.tie_embeddings:
                pass  # lm_head IS embed
            continue
        if name == "model.norm.weight":
            put(model.final_norm_w, t)
            continue
        if name == "lm_head.weight":
            if not cfg.tie_embeddings:
                put(model.lm_head, t)
            continue
        if not name.startswith("model.layers."):
            consumed -= 1
            seen.discard(name)
            if strict:
                raise ValueError(f"unmapped tensor {name}")
            continue
        rest = name[len("model.layers."):]
        li, _, rest = rest.partition(".")
        layer = model.layers[int(li)]
        is_bias = rest.endswith(".bias")
        if rest == "input_layernorm.weight":
            put(layer.input_norm_w, t)
        elif rest == "post_attention_layernorm.weight":
            put(layer.post_norm_w, t)
        elif rest.startswith("self_attn.q_proj."):
            attn_rows(layer, "q", t, is_bias)
        elif rest.startswith("self_attn.k_proj."):
            attn_rows(layer, "k", t, is_bias)
        elif rest.startswith("self_attn.v_proj."):
            attn_rows(layer, "v", t, is_bias)
        elif rest == "self_attn.o_proj.weight":
            sh = _shard(t, 1, tp.size, tp.rank) if tp.size > 1 else t
            put(layer.attn.wo, sh)
        elif rest == "mlp.gate_proj.weight":
            mlp = layer.mlp
            sh = _shard(t, 0, tp.size, tp.rank) if tp.size > 1 else t
            put(mlp.w_gate_up[:mlp.I], sh)
        elif rest == "mlp.up_proj.weight":
            mlp = layer.mlp
            sh = _shard(t, 0, tp.size, tp.rank) if tp.size > 1 else t
            put(mlp.w_gate_up[mlp.I:], sh)
        elif rest == "mlp.down_proj.weight":
            mlp = layer.mlp
            sh = _shard(t, 1, tp.size, tp.rank) if tp.size > 1 else t
            put(mlp.w_down, sh)
        elif rest == "block_sparse_moe.gate.weight":
            put(layer.moe.router, t)
        elif rest.startswith("block_sparse_moe.experts."):
            e_s, _, w = rest[len("block_sparse_moe.experts."):].partition(".")
            e = int(e_s)
            moe = layer.moe
            if moe.ep:
                if not (moe.e0 <= e < moe.e0 + moe.El):
                    continue  # another rank's expert
                el = e - moe.e0
                if w == "w1.weight":
                    put(moe.w_gate_up[el, :moe.I], t)
                elif w == "w3.weight":
                    put(moe.w_gate_up[el, moe.I:], t)
                elif w == "w2.weight":
                    put(moe.w_down[el], t)
            else:
                if w == "w1.weight":
                    sh = (_shard(t, 0, tp.size, tp.rank)
                          if tp.size > 1 else t)
                    put(moe.w_gate_up[e, :moe.I], sh)
                elif w == "w3.weight":
                    sh = (_shard(t, 0, tp.size, tp.rank)
                          if tp.size > 1 else t)
                    put(moe.w_gate_up[e, moe.I:], sh)
                elif w == "w2.weight":
                    sh = (_shard(t, 1, tp.size, tp.rank)
                          if tp.size > 1 else t)
                    put(moe.w_down[e], sh)
        else:
            consumed -= 1
            if strict:
                raise ValueError(f"unmapped layer tensor {name}")
    return consumed


# ---------------------------------------------------------------------------
def export_hf(model, path: str):
    """Inverse mapping (tp=1 only): write this model's weights as an
    HF-named safetensors checkpoint + config.json. Used by the round-trip
    test and as a conversion utility."""
    from safetensors.torch import save_file
    cfg = model.cfg
    assert model.tp.size == 1, "export is tp=1 only"
    hd = cfg.head_dim
    out: Dict[str, torch.Tensor] = {}
    out["model.embed_tokens.weight"] = model.embed
    out["model.norm.weight"] = model.final_norm_w
    if not cfg.tie_embeddings:
        out["lm_head.weight"] = model.lm_head
    for i, layer in enumerate(model.layers):
        p = f"model.layers.{i}."
        out[p + "input_layernorm.weight"] = layer.input_norm_w
        out[p + "post_attention_layernorm.weight"] = layer.post_norm_w
        at = layer.attn
        q_rows, k_rows = at.hq * hd, at.hkv * hd
        out[p + "self_attn.q_proj.weight"] = at.wqkv[:q_rows]
        out[p + "self_attn.k_proj.weight"] = at.wqkv[q_rows:q_rows + k_rows]
        out[p + "self_attn.v_proj.weight"] = at.wqkv[q_rows + k_rows:]
        if at.bqkv is not None:
            out[p + "self_attn.q_proj.bias"] = at.bqkv[:q_rows]
            out[p + "self_attn.k_proj.bias"] = at.bqkv[q_rows:q_rows + k_rows]
            out[p + "self_attn.v_proj.bias"] = at.bqkv[q_rows + k_rows:]
        out[p + "self_attn.o_proj.weight"] = at.wo
        if hasattr(layer, "mlp"):
            out[p + "mlp.gate_proj.weight"] = layer.mlp.w_gate_up[:layer.mlp.I]
            out[p + "mlp.up_proj.weight"] = layer.mlp.w_gate_up[layer.mlp.I:]
            out[p + "mlp.down_proj.weight"] = layer.mlp.w_down
        else:
            moe = layer.moe
            out[p + "block_sparse_moe.gate.weight"] = moe.router
            for e in range(moe.E):
                ep = p + f"block_sparse_moe.experts.{e}."
                out[ep + "w1.weight"] = moe.w_gate_up[e, :moe.I]
                out[ep + "w3.weight"] = moe.w_gate_up[e, moe.I:]
                out[ep + "w2.weight"] = moe.w_down[e]
    os.makedirs(path, exist_ok=True)
    save_file({k: v.detach().contiguous().cpu() for k, v in out.items()},
              os.path.join(path, "model.safetensors"))
    hf = {
        "architectures": [{"mixtral": "MixtralForCausalLM"}.get(
            cfg.arch, "LlamaForCausalLM")],
        "hidden_size": cfg.hidden_size,
        "intermediate_size": cfg.intermediate_size,
        "num_hidden_layers": cfg.num_layers,
        "num_attention_heads": cfg.num_q_heads,
        "num_key_value_heads": cfg.num_kv_heads,
        "head_dim": cfg.head_dim,
        "vocab_size": cfg.vocab_size,
        "max_position_embeddings": cfg.max_position,
        "rope_theta": cfg.rope_theta,
        "rms_norm_eps": cfg.rms_eps,
        "tie_word_embeddings": cfg.tie_embeddings,
    }
    if cfg.num_experts:
        hf["num_local_experts"] = cfg.num_experts
        hf["num_experts_per_tok"] = cfg.num_experts_per_tok
    if cfg.attn_bias:
        hf["architectures"] = ["Qwen2ForCausalLM"]
    with open(os.path.join(path, "config.json"), "w") as f:
        json.dump(hf, f, indent=1)
