"""Model / engine configuration.

ModelConfig carries the architecture hyperparameters (random-init synthetic
weights — no network access for checkpoints); EngineConfig carries serving
knobs. Mirrors the role of the reference's ModelDeploymentCard +
ModelRuntimeConfig (ai-dynamo/dynamo lib/llm/src/model_card.rs:834,
local_model/runtime_config.rs:183) as the worker->frontend contract.
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Optional


@dataclass
class ModelConfig:
    name: str = "llama-3-8b"
    arch: str = "llama"            # llama | opt | mixtral | qwen2
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_q_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    vocab_size: int = 128256
    max_position: int = 131072
    rope_theta: float = 500000.0
    rope_scaling: Optional[dict] = None
    rms_eps: float = 1e-5
    tie_embeddings: bool = False
    # MoE (mixtral)
    num_experts: int = 0
    num_experts_per_tok: int = 2
    moe_ep: bool = False   # expert parallelism over the TP group (vs MoE-TP)
    # OPT-style extras
    activation: str = "silu"       # silu | gelu
    norm: str = "rmsnorm"          # rmsnorm | layernorm
    learned_pos_emb: bool = False
    # qwen2-style QKV bias
    attn_bias: bool = False
    # local HF checkpoint dir to load real weights from ("" = random init)
    weights_path: str = ""


    def to_dict(self):
        return dataclasses.asdict(self)

    @staticmethod
    def from_dict(d: dict) -> "ModelConfig":
        return ModelConfig(**d)


PRESETS = {
    "opt-125m": ModelConfig(
        name="opt-125m", arch="opt", hidden_size=768, intermediate_size=3072,
        num_layers=12, num_q_heads=12, num_kv_heads=12, head_dim=64,
        vocab_size=50272, max_position=2048, activation="gelu",
        norm="layernorm", learned_pos_emb=True, tie_embeddings=True),
    "llama-3-8b": ModelConfig(
        name="llama-3-8b", arch="llama", hidden_size=4096,
        intermediate_size=14336, num_layers=32, num_q_heads=32, num_kv_heads=8,
        head_dim=128, vocab_size=128256, rope_theta=500000.0),
    "llama-3-70b": ModelConfig(
        name="llama-3-70b", arch="llama", hidden_size=8192,
        intermediate_size=28672, num_layers=80, num_q_heads=64, num_kv_heads=8,
        head_dim=128, vocab_size=128256, rope_theta=500000.0),
    "qwen2-7b": ModelConfig(
        name="qwen2-7b", arch="qwen2", hidden_size=3584,
        intermediate_size=18944, num_layers=28, num_q_heads=28,
        num_kv_heads=4, head_dim=128, vocab_size=152064,
        max_position=131072, rope_theta=1000000.0, attn_bias=True),
    "mixtral-8x7b": ModelConfig(
        name="mixtral-8x7b", arch="mixtral", hidden_size=4096,
        intermediate_size=14336, num_layers=32, num_q_heads=32, num_kv_heads=8,
        head_dim=128, vocab_size=32000, rope_theta=1000000.0, num_experts=8,
        num_experts_per_tok=2),
    # tiny configs for CPU tests
    "tiny-llama": ModelConfig(
        name="tiny-llama", arch="llama", hidden_size=256, intermediate_size=512,
        num_layers=2, num_q_heads=8, num_kv_heads=2, head_dim=32,
        vocab_size=512, max_position=4096, rope_theta=10000.0),
    # tiny but head_dim=128 so the native GPU kernels apply
    "tiny-llama-gpu": ModelConfig(
        name="tiny-llama-gpu", arch="llama", hidden_size=512,
        intermediate_size=1024, num_layers=2, num_q_heads=4, num_kv_heads=2,
        head_dim=128, vocab_size=1024, max_position=8192, rope_theta=10000.0),
    "tiny-mixtral-gpu": ModelConfig(
        name="tiny-mixtral-gpu", arch="mixtral", hidden_size=512,
        intermediate_size=1024, num_layers=2, num_q_heads=4, num_kv_heads=2,
        head_dim=128, vocab_size=1024, max_position=8192, rope_theta=10000.0,
        num_experts=4, num_experts_per_tok=2),
    "tiny-opt": ModelConfig(
        name="tiny-opt", arch="opt", hidden_size=128, intermediate_size=256,
        num_layers=2, num_q_heads=4, num_kv_heads=4, head_dim=32,
        vocab_size=512, max_position=2048, activation="gelu",
        norm="layernorm", learned_pos_emb=True, tie_embeddings=True),
    "tiny-qwen": ModelConfig(
        name="tiny-qwen", arch="qwen2", hidden_size=224,
        intermediate_size=448, num_layers=2, num_q_heads=7, num_kv_heads=1,
        head_dim=32, vocab_size=512, max_position=2048, rope_theta=10000.0,
        attn_bias=True),
    "tiny-qwen-gpu": ModelConfig(
        name="tiny-qwen-gpu", arch="qwen2", hidden_size=896,
        intermediate_size=1024, num_layers=2, num_q_heads=7, num_kv_heads=1,
        head_dim=128, vocab_size=1024, max_position=8192,
        rope_theta=10000.0, attn_bias=True),
    "tiny-mixtral": ModelConfig(
        name="tiny-mixtral", arch="mixtral", hidden_size=128,
        intermediate_size=256, num_layers=2, num_q_heads=4, num_kv_heads=2,
        head_dim=32, vocab_size=512, max_position=2048, rope_theta=10000.0,
        num_experts=4, num_experts_per_tok=2),
}


@dataclass
class EngineConfig:
    model: ModelConfig = field(default_factory=ModelConfig)
    device: str = "cuda:0"
    dtype: str = "bfloat16"
    page_size: int = 64
    max_num_seqs: int = 64
    max_batched_tokens: int = 8192      # chunked-prefill budget per step
    max_model_len: int = 16384
    kv_pool_pages: int = 0              # 0 = size from gpu_mem_fraction
    gpu_mem_fraction: float = 0.90
    enable_prefix_caching: bool = True
    enable_hip_graphs: bool = True
    host_cache_pages: int = 0           # KVBM G2 tier size (0 = disabled)
    # G2 eviction policy: "lru", or "tinylfu" = LRU victim + TinyLFU
    # admission filter (reference kvbm-logical tinylfu.rs parity)
    host_cache_policy: str = "lru"
    disk_cache_pages: int = 0           # KVBM G3 tier size (0 = disabled)
    disk_cache_path: str = ""           # G3 backing file (required if G3 on)
    object_cache_dir: str = ""          # G4 shared object store (disabled="")
    kv_events: bool = True              # emit stored/removed block events
    # KV cache storage dtype: "auto" matches the compute dtype; "fp8"
    # stores OCP e4m3 (halves decode-attention HBM traffic; GPU-only,
    # requires GQA group in 2/4/8/16 and head_dim 128 - reference parity
    # with its engines' --kv-cache-dtype fp8)
    kv_cache_dtype: str = "auto"
    # waiting-queue admission policy (reference parity: kv-router
    # scheduling/policy.rs SchedulingPolicy FCFS/LCFS/WSPT):
    #   fcfs = arrival order; lcfs = newest first; wspt = shortest
    #   remaining prompt first (weighted shortest processing time)
    queue_policy: str = "fcfs"          # fcfs | lcfs | wspt
    block_salt: int = 0
    # disaggregation
    worker_type: str = "aggregated"     # aggregated | prefill | decode
    # CPU-only: back the KV pool with a /dev/shm file so a peer PROCESS can
    # map it by path (the CPU stand-in for the GPU pool's hipIpc handle;
    # lets cross-process disagg pulls run in CPU/gloo tests)
    cpu_shm_pool: bool = False
    # tensor parallelism (process group wired by the worker)
    tp_size: int = 1
    tp_rank: int = 0
    # V-cache page layout: "auto" stores V pages d-major ([P, Hkv, hd, ps])
    # on GPU whenever the native kernels support it - the decode PV
    # A-fragment then reads as contiguous token runs, deleting the V^T
    # staging transpose (decode sweep: G8 3911 -> 4476, G4 3951 -> 4987
    # GB/s). "never" keeps token-major pages everywhere.
    kv_v_layout: str = "auto"           # auto | never

    @property
    def torch_dtype(self):
        import torch
        return {"bfloat16": torch.bfloat16, "float32": torch.float32}[self.dtype]

    @property
    def v_transposed(self) -> bool:
        """Resolved V-page layout: d-major needs the swapped decode kernel
        (head_dim 128, page_size % 32 == 0, GQA group 2..16) and the
        prefill32/fallback staging (page_size % 8 == 0)."""
        if self.kv_v_layout == "never":
            return False
        m = self.model
        g = (m.num_q_heads // m.num_kv_heads
             if m.num_kv_heads and m.num_q_heads % m.num_kv_heads == 0 else 0)
        return (self.device.startswith("cuda") and m.head_dim == 128
                and self.page_size % 32 == 0 and 2 <= g <= 16)

    @property
    def kv_torch_dtype(self):
        import torch
        if self.kv_cache_dtype == "fp8":
            if not self.device.startswith("cuda"):
                raise ValueError("fp8 KV cache is GPU-only")
            return torch.float8_e4m3fn
        return self.torch_dtype
