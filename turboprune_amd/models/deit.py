"""DeiT / ViT family with masked layers (timm-compatible module naming).

The reference registers timm factories ``local_deit_{tiny,small,base}_
patch16_224`` (+ distilled and 384 variants) over timm's VisionTransformer
(reference: utils/deit.py:69-253), but its Custom/timm path is
latent-broken. This is a working native implementation:

- module names follow timm (``cls_token``, ``pos_embed``,
  ``patch_embed.proj``, ``blocks.N.attn.qkv`` ...) so checkpoints line up;
- every Linear is a ``Conv1dMask`` (the reference's replacement map,
  custom_models.py:219) and the patch-embed conv is ``ConvMask``;
- attention itself uses torch SDPA on ROCm; the masked qkv/proj/mlp
  projections are the MFMA masked-GEMM hot path.
"""

from __future__ import annotations

from functools import partial

import torch
import torch.nn as nn

from turboprune_amd.ops.mask_layers import Conv1dMask, ConvMask
from turboprune_amd.ops.norm_act import FusedGELU, FusedLayerNorm


class PatchEmbed(nn.Module):
    """Patchify + project. The stride-P PxP conv is mathematically a GEMM
    over non-overlapping patches; on GPU it runs on the masked MFMA GEMM
    (MIOpen's only kernel for this 3-channel odd shape is a ~60 ms naive
    fallback on some boxes). The parameter stays a ConvMask for
    state-dict compatibility."""

    def __init__(self, img_size=224, patch_size=16, in_chans=3, embed_dim=768):
        super().__init__()
        self.img_size = (img_size, img_size)
        self.patch_size = (patch_size, patch_size)
        self.num_patches = (img_size // patch_size) ** 2
        self.proj = ConvMask(in_channels=in_chans, out_channels=embed_dim,
                             kernel_size=patch_size, stride=patch_size,
                             bias=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, C, H, W = x.shape
        P = self.patch_size[0]
        if x.is_cuda and H % P == 0 and W % P == 0:
            from turboprune_amd.ops import functional as TF
            E = self.proj.out_channels
            # patches ordered (row-in-patch, col-in-patch, channel) to
            # match a channels_last-reshaped weight
            xp = (x.permute(0, 2, 3, 1)
                   .reshape(B, H // P, P, W // P, P, C)
                   .permute(0, 1, 3, 2, 4, 5)
                   .reshape(B, (H // P) * (W // P), P * P * C))
            w2 = self.proj.weight.permute(0, 2, 3, 1).reshape(E, -1)
            m2 = self.proj.mask.permute(0, 2, 3, 1).reshape(E, -1)
            cache = self.proj._fresh_cache()
            c2 = cache.permute(0, 2, 3, 1).reshape(E, -1) \
                if cache is not None else None
            return TF.masked_linear(xp, w2, m2, self.proj.bias, c2,
                                    self.proj.compute_dtype)
        x = self.proj(x)                      # B, C, H/ps, W/ps
        return x.flatten(2).transpose(1, 2)   # B, N, C


class Attention(nn.Module):
    def __init__(self, dim, num_heads=8, qkv_bias=True, attn_drop=0.0,
                 proj_drop=0.0):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = dim // num_heads
        self.scale = self.head_dim ** -0.5
        self.qkv = Conv1dMask(dim, dim * 3, bias=qkv_bias)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = Conv1dMask(dim, dim, bias=True)
        self.proj_drop = nn.Dropout(proj_drop)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, N, C = x.shape
        qkv = self.qkv(x).reshape(B, N, 3, self.num_heads, self.head_dim)
        qkv = qkv.permute(2, 0, 3, 1, 4)      # 3, B, heads, N, hd
        q, k, v = qkv[0], qkv[1], qkv[2]
        drop = self.attn_drop.p if self.training else 0.0
        if drop == 0.0:
            from turboprune_amd.ops.attention import sdpa
            x = sdpa(q, k, v)  # torch SDPA, or fused HIP fwd when opted in
        else:
            x = torch.nn.functional.scaled_dot_product_attention(
                q, k, v, dropout_p=drop)
        x = x.transpose(1, 2).reshape(B, N, C)
        return self.proj_drop(self.proj(x))


class Mlp(nn.Module):
    def __init__(self, in_features, hidden_features, drop=0.0):
        super().__init__()
        self.fc1 = Conv1dMask(in_features, hidden_features, bias=True)
        self.act = FusedGELU()
        self.fc2 = Conv1dMask(hidden_features, in_features, bias=True)
        self.drop = nn.Dropout(drop)

    def forward(self, x):
        return self.drop(self.fc2(self.drop(self.act(self.fc1(x)))))


class Block(nn.Module):
    def __init__(self, dim, num_heads, mlp_ratio=4.0, qkv_bias=True,
                 drop=0.0, attn_drop=0.0, norm_layer=nn.LayerNorm):
        super().__init__()
        self.norm1 = norm_layer(dim)
        self.attn = Attention(dim, num_heads, qkv_bias, attn_drop, drop)
        self.norm2 = norm_layer(dim)
        self.mlp = Mlp(dim, int(dim * mlp_ratio), drop)

    def forward(self, x):
        x = x + self.attn(self.norm1(x))
        x = x + self.mlp(self.norm2(x))
        return x


class VisionTransformer(nn.Module):
    def __init__(self, img_size=224, patch_size=16, in_chans=3,
                 num_classes=1000, embed_dim=768, depth=12, num_heads=12,
                 mlp_ratio=4.0, qkv_bias=True, drop_rate=0.0,
                 attn_drop_rate=0.0, norm_layer=None):
        super().__init__()
        norm_layer = norm_layer or partial(FusedLayerNorm, eps=1e-6)
        self.num_classes = num_classes
        self.embed_dim = embed_dim
        self.patch_embed = PatchEmbed(img_size, patch_size, in_chans, embed_dim)
        num_patches = self.patch_embed.num_patches

        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim))
        self.pos_embed = nn.Parameter(torch.zeros(1, num_patches + 1, embed_dim))
        self.pos_drop = nn.Dropout(drop_rate)
        self.blocks = nn.ModuleList([
            Block(embed_dim, num_heads, mlp_ratio, qkv_bias, drop_rate,
                  attn_drop_rate, norm_layer)
            for _ in range(depth)])
        self.norm = norm_layer(embed_dim)
        self.head = Conv1dMask(embed_dim, num_classes, bias=True) \
            if num_classes > 0 else nn.Identity()

        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)
        self.apply(self._init_weights)

    @staticmethod
    def _init_weights(m):
        if isinstance(m, (nn.Linear, nn.Conv1d)):
            nn.init.trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.constant_(m.bias, 0)
        elif isinstance(m, nn.LayerNorm):
            nn.init.constant_(m.bias, 0)
            nn.init.constant_(m.weight, 1.0)

    def forward_features(self, x):
        B = x.shape[0]
        x = self.patch_embed(x)
        cls = self.cls_token.expand(B, -1, -1)
        x = torch.cat((cls, x), dim=1)
        x = self.pos_drop(x + self.pos_embed)
        for blk in self.blocks:
            x = blk(x)
        x = self.norm(x)
        return x[:, 0]

    def forward(self, x):
        return self.head(self.forward_features(x))


class DistilledVisionTransformer(VisionTransformer):
    """DeiT distilled variant: extra dist token + head, averaged heads at
    inference (reference: utils/deit.py:21-66)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)
        self.dist_token = nn.Parameter(torch.zeros(1, 1, self.embed_dim))
        num_patches = self.patch_embed.num_patches
        self.pos_embed = nn.Parameter(
            torch.zeros(1, num_patches + 2, self.embed_dim))
        self.head_dist = Conv1dMask(self.embed_dim, self.num_classes,
                                    bias=True)
        nn.init.trunc_normal_(self.dist_token, std=0.02)
        nn.init.trunc_normal_(self.pos_embed, std=0.02)

    def forward_features(self, x):
        B = x.shape[0]
        x = self.patch_embed(x)
        cls = self.cls_token.expand(B, -1, -1)
        dist = self.dist_token.expand(B, -1, -1)
        x = torch.cat((cls, dist, x), dim=1)
        x = self.pos_drop(x + self.pos_embed)
        for blk in self.blocks:
            x = blk(x)
        x = self.norm(x)
        return x[:, 0], x[:, 1]

    def forward(self, x):
        feat, feat_dist = self.forward_features(x)
        out, out_dist = self.head(feat), self.head_dist(feat_dist)
        if self.training:
            return out, out_dist
        return (out + out_dist) / 2


def _deit(embed_dim, depth, num_heads, img_size=224, distilled=False,
          num_classes=1000, **kwargs):
    cls = DistilledVisionTransformer if distilled else VisionTransformer
    return cls(img_size=img_size, patch_size=16, embed_dim=embed_dim,
               depth=depth, num_heads=num_heads, mlp_ratio=4.0,
               qkv_bias=True, num_classes=num_classes,
               norm_layer=partial(FusedLayerNorm, eps=1e-6), **kwargs)


# factory names mirror the reference's timm registrations (utils/deit.py:69-253)
def local_deit_tiny_patch16_224(num_classes=1000, **kw):
    return _deit(192, 12, 3, num_classes=num_classes, **kw)


def local_deit_small_patch16_224(num_classes=1000, **kw):
    return _deit(384, 12, 6, num_classes=num_classes, **kw)


def local_deit_base_patch16_224(num_classes=1000, **kw):
    return _deit(768, 12, 12, num_classes=num_classes, **kw)


def local_deit_tiny_distilled_patch16_224(num_classes=1000, **kw):
    return _deit(192, 12, 3, distilled=True, num_classes=num_classes, **kw)


def local_deit_small_distilled_patch16_224(num_classes=1000, **kw):
    return _deit(384, 12, 6, distilled=True, num_classes=num_classes, **kw)


def local_deit_base_distilled_patch16_224(num_classes=1000, **kw):
    return _deit(768, 12, 12, distilled=True, num_classes=num_classes, **kw)


def local_deit_base_patch16_384(num_classes=1000, **kw):
    return _deit(768, 12, 12, img_size=384, num_classes=num_classes, **kw)


def local_deit_base_distilled_patch16_384(num_classes=1000, **kw):
    return _deit(768, 12, 12, img_size=384, distilled=True,
                 num_classes=num_classes, **kw)
