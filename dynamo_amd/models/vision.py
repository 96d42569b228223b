"""Vision encoder for multimodal serving (random-init synthetic weights).

The encoder side of the reference's E/PD disaggregation
(ai-dynamo/dynamo docs/.../encoder-disaggregation.md; encode workers
components/src/dynamo/vllm/multimodal_handlers/encode_worker_handler.py):
a ViT-style patch encoder producing per-patch embeddings projected to the
language model's hidden size. Dense non-causal attention over a few
hundred patch tokens per image — bf16 GEMMs (hipBLASLt) + our
rmsnorm/gelu kernels; this is the encode WORKER's model, not the LLM
serving hot path (the reference likewise runs its vision towers inside
the engine process, not hand-fused).
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from dynamo_amd import ops


@dataclass
class VisionConfig:
    name: str = "vit-base"
    image_size: int = 224
    patch_size: int = 16
    hidden_size: int = 768
    intermediate_size: int = 3072
    num_layers: int = 12
    num_heads: int = 12
    out_hidden_size: int = 4096     # LLM hidden size to project into
    rms_eps: float = 1e-5

    @property
    def num_patches(self) -> int:
        return (self.image_size // self.patch_size) ** 2


VISION_PRESETS = {
    "vit-base": VisionConfig(),
    "tiny-vit": VisionConfig(name="tiny-vit", image_size=32, patch_size=8,
                             hidden_size=64, intermediate_size=128,
                             num_layers=2, num_heads=4, out_hidden_size=256),
}


class VisionEncoder(torch.nn.Module):
    def __init__(self, cfg: VisionConfig, device, dtype=torch.float32,
                 seed: int = 0):
        super().__init__()
        self.cfg = cfg
        self.device = torch.device(device)
        self.dtype = dtype
        g = torch.Generator(device="cpu").manual_seed(seed)

        def w(*shape, std=0.02):
            t = torch.empty(shape).normal_(0, std, generator=g)
            return torch.nn.Parameter(t.to(self.device, dtype),
                                      requires_grad=False)

        D = cfg.hidden_size
        P = cfg.patch_size
        self.patch_w = w(D, 3 * P * P)
        self.pos = w(cfg.num_patches, D, std=0.01)
        self.blocks = []
        for i in range(cfg.num_layers):
            blk = {
                "ln1": w(D, std=0.0), "ln2": w(D, std=0.0),
                "qkv": w(3 * D, D), "o": w(D, D),
                "fc1": w(cfg.intermediate_size, D),
                "fc2": w(D, cfg.intermediate_size),
            }
            with torch.no_grad():
                blk["ln1"].fill_(1.0)
                blk["ln2"].fill_(1.0)
            self.blocks.append(blk)
            for k, v in blk.items():
                self.register_parameter(f"b{i}_{k}", v)
        self.out_ln = w(D, std=0.0)
        with torch.no_grad():
            self.out_ln.fill_(1.0)
        self.proj = w(cfg.out_hidden_size, D)

    @torch.no_grad()
    def forward(self, pixels: torch.Tensor) -> torch.Tensor:
        """pixels [3, H, W] in [0,1] -> [num_patches, out_hidden]."""
        cfg = self.cfg
        P = cfg.patch_size
        x = pixels.to(self.device, self.dtype)
        if x.shape[-2:] != (cfg.image_size, cfg.image_size):
            x = torch.nn.functional.interpolate(
                x.unsqueeze(0).float(), size=(cfg.image_size, cfg.image_size),
                mode="bilinear", align_corners=False)[0].to(self.dtype)
        # patchify: [3,H,W] -> [np, 3*P*P]
        x = x.unfold(1, P, P).unfold(2, P, P)        # [3, gh, gw, P, P]
        x = x.permute(1, 2, 0, 3, 4).reshape(cfg.num_patches, 3 * P * P)
        h = torch.nn.functional.linear(x, self.patch_w) + self.pos
        nh = cfg.num_heads
        hd = cfg.hidden_size // nh
        scale = hd ** -0.5
        for blk in self.blocks:
            y = ops.rmsnorm(h, blk["ln1"], cfg.rms_eps)
            qkv = torch.nn.functional.linear(y, blk["qkv"])
            q, k, v = qkv.chunk(3, dim=-1)
            q = q.view(-1, nh, hd).transpose(0, 1)
            k = k.view(-1, nh, hd).transpose(0, 1)
            v = v.view(-1, nh, hd).transpose(0, 1)
            att = torch.softmax((q @ k.transpose(-1, -2)).float() * scale,
                                dim=-1).to(self.dtype)
            y = (att @ v).transpose(0, 1).reshape(-1, cfg.hidden_size)
            h = h + torch.nn.functional.linear(y, blk["o"])
            y = ops.rmsnorm(h, blk["ln2"], cfg.rms_eps)
            y = ops.gelu(torch.nn.functional.linear(y, blk["fc1"]))
            h = h + torch.nn.functional.linear(y, blk["fc2"])
        h = ops.rmsnorm(h, self.out_ln, cfg.rms_eps)
        return torch.nn.functional.linear(h, self.proj)


def decode_image(data: bytes) -> torch.Tensor:
    """Decode PNG/JPEG bytes -> [3, H, W] float in [0,1] (media decode of
    the reference preprocessor, preprocessor.rs:2248)."""
    import io

    import numpy as np
    from PIL import Image
    img = Image.open(io.BytesIO(data)).convert("RGB")
    arr = np.asarray(img, dtype=np.float32) / 255.0
    return torch.from_numpy(arr).permute(2, 0, 1).contiguous()
