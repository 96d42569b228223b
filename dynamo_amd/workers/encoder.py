"""Encode worker: serves the vision-encoder side of E/PD disaggregation.

Reference parity: dedicated vision-encode workers whose embeddings are
transferred to P/D workers (docs/.../encoder-disaggregation.md;
encode_worker_handler.py). MI355X-native stance: embeddings return inline
(fp16 b64) through the request plane — single-node scale makes the
RDMA-handle indirection of the reference unnecessary.

CLI: python -m dynamo_amd.workers.encoder --vision tiny-vit --model <llm>
"""
from __future__ import annotations

import argparse
import asyncio
import base64
import binascii
import logging
import signal
from typing import Optional

import numpy as np
import torch

from dynamo_amd.models.vision import VISION_PRESETS, VisionEncoder, decode_image
from dynamo_amd.runtime import DistributedRuntime

log = logging.getLogger("dynamo_amd.encoder")


class EncoderService:
    def __init__(self, encoder: VisionEncoder, runtime: DistributedRuntime,
                 namespace: str = "dynamo", component: str = "encoder",
                 model_name: str = ""):
        self.encoder = encoder
        self.runtime = runtime
        self.comp = runtime.namespace(namespace).component(component)
        self.model_name = model_name
        self.count = 0

    async def start(self):
        self.comp.serve_endpoint("encode", self.encode)
        await self.comp.register(
            model_card=None,
            metadata={"worker_type": "encoder",
                      "model": self.model_name,
                      "vision": self.encoder.cfg.name,
                      "out_hidden": self.encoder.cfg.out_hidden_size,
                      "tokens_per_image": self.encoder.cfg.num_patches})
        return self

    async def stop(self):
        self.comp.deregister()

    @property
    def instance_id(self):
        return self.comp.instance_id

    def _encode_one(self, img: dict) -> torch.Tensor:
        if "b64_image" in img:       # PNG/JPEG bytes
            pixels = decode_image(base64.b64decode(img["b64_image"]))
        elif "b64_pixels" in img:    # raw float tensor {b64, shape}
            arr = np.frombuffer(base64.b64decode(img["b64_pixels"]),
                                dtype=np.dtype(img.get("dtype", "float32")))
            pixels = torch.from_numpy(arr.reshape(img["shape"]).copy())
        else:
            raise ValueError("image needs b64_image or b64_pixels")
        return self.encoder.forward(pixels)

    async def encode(self, payload: dict, ctx):
        self.count += 1
        out = []
        for img in payload.get("images", []):
            emb = await asyncio.to_thread(self._encode_one, img)
            e16 = emb.to(torch.float16).cpu().numpy()
            out.append({"b64": base64.b64encode(e16.tobytes()).decode(),
                        "shape": list(e16.shape), "dtype": "float16"})
        yield {"embeddings": out}


def main():
    p = argparse.ArgumentParser("dynamo_amd.workers.encoder")
    p.add_argument("--vision", default="vit-base")
    p.add_argument("--model", default="", help="LLM model this encoder feeds")
    p.add_argument("--out-hidden", type=int, default=0,
                   help="override projection width (match the LLM hidden)")
    p.add_argument("--discovery", default="memory")
    p.add_argument("--namespace", default="dynamo")
    p.add_argument("--device", default=None)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--host", default="127.0.0.1")
    args = p.parse_args()
    logging.basicConfig(level=logging.INFO)

    import dataclasses
    cfg = VISION_PRESETS[args.vision]
    if args.out_hidden:
        cfg = dataclasses.replace(cfg, out_hidden_size=args.out_hidden)
    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.startswith("cuda") else torch.float32
    enc = VisionEncoder(cfg, device, dtype, seed=args.seed)

    async def run():
        rt = DistributedRuntime(args.discovery, host=args.host)
        svc = EncoderService(enc, rt, namespace=args.namespace,
                             model_name=args.model)
        await svc.start()
        print(f"ENCODER_READY {svc.instance_id} {rt.server.address}",
              flush=True)
        stop = asyncio.Event()
        loop = asyncio.get_running_loop()
        for sig in (signal.SIGINT, signal.SIGTERM):
            loop.add_signal_handler(sig, stop.set)
        await stop.wait()
        await svc.stop()
        await rt.shutdown()

    asyncio.run(run())


if __name__ == "__main__":
    main()
