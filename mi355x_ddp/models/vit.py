"""ViT-L/32, defined locally (torchvision is not installed here).

The reference profile stage imports `vit_l_32` alongside `resnet50`
(reference multigpu_profile.py:14) but leaves it commented out at :24;
this module completes that capability so the profile entrypoint can run
either workload (`MI355X_PROFILE_MODEL=vit`). Architecture matches
torchvision's `vit_l_32` (ViT-Large, 32x32 patches, 224 input: 24 layers,
hidden 1024, MLP 4096, 16 heads, learned position embeddings, pre-norm
encoder blocks, class token head) so the parameter census and DDP
gradient payload (~305 M params, ~1.2 GB fp32 grads) are the real thing.

Attention/LayerNorm/GELU run through PyTorch-ROCm (MIOpen/rocBLAS — the
same scoping as ResNet's convs); the classifier head
uses the hand-written MFMA linear.
"""

from __future__ import annotations

import torch
from torch import nn

from .toy import HipLinear


class EncoderBlock(nn.Module):
    def __init__(self, hidden: int, heads: int, mlp: int):
        super().__init__()
        self.ln_1 = nn.LayerNorm(hidden, eps=1e-6)
        self.self_attention = nn.MultiheadAttention(hidden, heads,
                                                    batch_first=True)
        self.ln_2 = nn.LayerNorm(hidden, eps=1e-6)
        self.mlp = nn.Sequential(
            nn.Linear(hidden, mlp), nn.GELU(), nn.Linear(mlp, hidden))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = self.ln_1(x)
        y, _ = self.self_attention(y, y, y, need_weights=False)
        x = x + y
        return x + self.mlp(self.ln_2(x))


class VisionTransformer(nn.Module):
    def __init__(self, image_size: int = 224, patch: int = 32,
                 layers: int = 24, hidden: int = 1024, heads: int = 16,
                 mlp: int = 4096, num_classes: int = 1000):
        super().__init__()
        self.patch = patch
        self.conv_proj = nn.Conv2d(3, hidden, kernel_size=patch, stride=patch)
        n_tokens = (image_size // patch) ** 2 + 1
        self.class_token = nn.Parameter(torch.zeros(1, 1, hidden))
        self.pos_embedding = nn.Parameter(
            torch.empty(1, n_tokens, hidden).normal_(std=0.02))
        self.encoder = nn.ModuleList(
            EncoderBlock(hidden, heads, mlp) for _ in range(layers))
        self.ln = nn.LayerNorm(hidden, eps=1e-6)
        self.head = HipLinear(hidden, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.conv_proj(x)                       # [B, H, gh, gw]
        x = x.flatten(2).transpose(1, 2)            # [B, tokens, H]
        cls = self.class_token.expand(x.shape[0], -1, -1)
        x = torch.cat([cls, x], dim=1) + self.pos_embedding
        for blk in self.encoder:
            x = blk(x)
        return self.head(self.ln(x[:, 0]))


def vit_l_32(num_classes: int = 1000) -> VisionTransformer:
    return VisionTransformer(num_classes=num_classes)


def vit_tiny(num_classes: int = 10) -> VisionTransformer:
    """Small configuration for tests."""
    return VisionTransformer(image_size=64, patch=16, layers=2, hidden=64,
                             heads=4, mlp=128, num_classes=num_classes)
