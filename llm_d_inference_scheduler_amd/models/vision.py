"""Vision encoder for the encode role (E/PD, E/P/D).

The reference's encode stage fans multimodal items out to encoder pods and
hands embeddings to prefill via vLLM's EC_Connector (outside its repo,
SURVEY.md §2.11). Here the encoder is in-process on the encode-role GPU:
a ViT-shaped tower (patch projection + pre-norm transformer blocks +
projection into the language hidden size), random-init like every other
model (no checkpoints without egress). "Images" are synthesized
deterministically from the item URL (no network), so prefill results are
reproducible across ranks and runs, and the URL doubles as the dedupe key
(connector_epd_shared_storage.go:125-208 dedupes by URL the same way).
"""
import math
from typing import Dict, Optional

import torch

from .. import _router_core as rc
from .configs import ModelConfig


class VisionEncoder:
    def __init__(self, config: ModelConfig, device,
                 dtype: torch.dtype = torch.bfloat16, seed: int = 0,
                 cache_items: int = 256):
        assert config.vision_hidden > 0, "model has no vision tower"
        self.cfg = config
        self.device = torch.device(device)
        self.dtype = dtype
        vh = config.vision_hidden
        self.n_patches = config.vision_patches
        self.patch_dim = 3 * 14 * 14  # ViT-L/14 patch pixels
        if self.device.type == "cuda":
            torch.cuda.manual_seed_all(seed + 77)
            gen = None
        else:
            gen = torch.Generator(device="cpu").manual_seed(seed + 77)

        def w(*shape, std=0.02):
            t = torch.empty(*shape, dtype=torch.float32,
                            device=self.device if gen is None else "cpu")
            t.normal_(0.0, std, generator=gen)
            return t.to(self.dtype).to(self.device)

        self.patch_proj = w(self.patch_dim, vh)
        self.pos_embed = w(self.n_patches, vh, std=0.01)
        out_std = 0.02 / math.sqrt(2 * max(1, config.vision_layers))
        self.blocks = []
        for _ in range(config.vision_layers):
            self.blocks.append({
                "wqkv": w(vh, 3 * vh), "wo": w(vh, vh, std=out_std),
                "w1": w(vh, 4 * vh), "w2": w(4 * vh, vh, std=out_std),
            })
        self.project = w(vh, config.hidden_size)
        self._cache: Dict[str, torch.Tensor] = {}
        self._cache_items = cache_items
        self.n_heads = max(1, vh // 64)

    def synth_image(self, url: str) -> torch.Tensor:
        """Deterministic synthetic patches for a URL (no egress)."""
        seed = rc.xxh64(url.encode("utf-8"), 0) & 0x7FFFFFFF
        gen = torch.Generator(device="cpu").manual_seed(seed)
        img = torch.randn(self.n_patches, self.patch_dim, generator=gen)
        return img.to(self.dtype).to(self.device)

    def encode_url(self, url: str) -> torch.Tensor:
        """[n_patches, text_hidden] embedding; URL-deduped cache."""
        hit = self._cache.get(url)
        if hit is not None:
            return hit
        emb = self.forward(self.synth_image(url))
        if len(self._cache) >= self._cache_items:
            self._cache.pop(next(iter(self._cache)))
        self._cache[url] = emb
        return emb

    def forward(self, patches: torch.Tensor) -> torch.Tensor:
        vh = self.cfg.vision_hidden
        h = patches @ self.patch_proj + self.pos_embed
        nh, hd = self.n_heads, vh // self.n_heads
        scale = hd ** -0.5
        for blk in self.blocks:
            x = _layernorm(h)
            qkv = x @ blk["wqkv"]
            q, k, v = qkv.split(vh, dim=-1)
            P = q.shape[0]
            q = q.view(P, nh, hd).permute(1, 0, 2).float()
            k = k.view(P, nh, hd).permute(1, 0, 2).float()
            v = v.view(P, nh, hd).permute(1, 0, 2).float()
            attn = torch.softmax(q @ k.transpose(1, 2) * scale, dim=-1) @ v
            attn = attn.permute(1, 0, 2).reshape(P, vh).to(h.dtype)
            h = h + attn @ blk["wo"]
            x = _layernorm(h)
            h = h + torch.nn.functional.gelu(x @ blk["w1"]) @ blk["w2"]
        return _layernorm(h) @ self.project


def _layernorm(x: torch.Tensor) -> torch.Tensor:
    xf = x.float()
    y = (xf - xf.mean(-1, keepdim=True)) / \
        (xf.var(-1, keepdim=True, unbiased=False) + 1e-5).sqrt()
    return y.to(x.dtype)
