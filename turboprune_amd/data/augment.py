"""Batch GPU augmentation primitives (K11 in SURVEY §2.4).

These are the per-epoch GPU augmentations of the airbench-style CIFAR
loader (reference: utils/dataset.py:38-98,192-217) and the post-decode
stage of the ImageNet pipeline: normalize, alternating-epoch flip,
random-translate crop, cutout.

Dispatch: on ROCm tensors these route to the fused HIP kernels in the
extension (one pass over the batch, bf16/f32 out); the torch
implementations below are the CPU path and the numerics oracle.
"""

from __future__ import annotations

from typing import Optional

import torch

from turboprune_amd.ops import _backend


def normalize_u8(images_u8: torch.Tensor, mean: torch.Tensor,
                 std: torch.Tensor, out_dtype: torch.dtype = torch.float32,
                 flip: Optional[torch.Tensor] = None) -> torch.Tensor:
    """uint8 NCHW -> (x/255 - mean)/std in out_dtype, optionally flipping
    rows of the batch horizontally (flip: bool tensor [N])."""
    if _backend.use_native(images_u8) and hasattr(_backend.extension(),
                                                  "normalize_u8"):
        return _backend.extension().normalize_u8(
            images_u8, mean.to(images_u8.device), std.to(images_u8.device),
            flip if flip is not None else torch.Tensor(), out_dtype)
    x = images_u8.to(out_dtype).div_(255.0)
    mean = mean.to(x.device, x.dtype).view(1, -1, 1, 1)
    std = std.to(x.device, x.dtype).view(1, -1, 1, 1)
    x = (x - mean) / std
    if flip is not None:
        x[flip] = torch.flip(x[flip], dims=[-1])
    return x


def batch_flip_lr(images: torch.Tensor,
                  flip_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Horizontally flip selected rows (all rows if flip_mask is None)."""
    if flip_mask is None:
        return torch.flip(images, dims=[-1])
    out = images.clone()
    out[flip_mask] = torch.flip(images[flip_mask], dims=[-1])
    return out


def batch_crop_translate(padded: torch.Tensor, crop_size: int,
                         shifts: torch.Tensor) -> torch.Tensor:
    """Random-translate crop from reflect-padded images.

    padded: [N, C, H+2p, W+2p]; shifts: int64 [N, 2] in [0, 2p]; returns
    [N, C, crop, crop] (reference: utils/dataset.py:43-69)."""
    if _backend.use_native(padded) and padded.dtype == torch.float32:
        return _backend.extension().crop_translate(
            padded.contiguous(), crop_size, shifts)
    n, c, hp, wp = padded.shape
    out = torch.empty(n, c, crop_size, crop_size, dtype=padded.dtype,
                      device=padded.device)
    # gather rows then cols via advanced indexing, vectorized over batch
    ar = torch.arange(crop_size, device=padded.device)
    rows = shifts[:, 0].view(-1, 1) + ar.view(1, -1)      # [N, crop]
    cols = shifts[:, 1].view(-1, 1) + ar.view(1, -1)      # [N, crop]
    bi = torch.arange(n, device=padded.device).view(-1, 1, 1)
    out = padded[bi, :, rows.view(n, crop_size, 1),
                 cols.view(n, 1, crop_size)]               # [N, crop, crop, C]
    return out.permute(0, 3, 1, 2).contiguous()


def make_cutout_mask(n: int, h: int, w: int, size: int,
                     centers: torch.Tensor,
                     device: torch.device) -> torch.Tensor:
    """Boolean [N, H, W] mask of cutout squares; centers int64 [N, 2]
    (reference: utils/dataset.py:72-98)."""
    ar_h = torch.arange(h, device=device).view(1, -1, 1)
    ar_w = torch.arange(w, device=device).view(1, 1, -1)
    cy = centers[:, 0].view(-1, 1, 1)
    cx = centers[:, 1].view(-1, 1, 1)
    half = size // 2
    return ((ar_h >= cy - half) & (ar_h <= cy + half) &
            (ar_w >= cx - half) & (ar_w <= cx + half))


def batch_cutout(images: torch.Tensor, size: int,
                 generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Zero a random size×size square per image, in place."""
    if size <= 0:
        return images
    n, _, h, w = images.shape
    centers = torch.stack([
        torch.randint(0, h, (n,), device=images.device, generator=generator),
        torch.randint(0, w, (n,), device=images.device, generator=generator),
    ], dim=1)
    if _backend.use_native(images) and images.dtype == torch.float32 \
            and images.is_contiguous():
        _backend.extension().cutout_(images, centers, size)
        return images
    mask = make_cutout_mask(n, h, w, size, centers, images.device)
    images.masked_fill_(mask.unsqueeze(1), 0.0)
    return images


def sample_rrc_boxes(n: int, h: int, w: int,
                     scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3),
                     generator: Optional[torch.Generator] = None,
                     device: Optional[torch.device] = None) -> torch.Tensor:
    """Sample RandomResizedCrop boxes (top, left, height, width) with
    torchvision's area/ratio semantics (10 tries then center fallback)."""
    import math
    boxes = torch.empty(n, 4, dtype=torch.int64)
    area = h * w
    log_r = (math.log(ratio[0]), math.log(ratio[1]))
    for i in range(n):
        ok = False
        for _ in range(10):
            target = area * (torch.empty(1).uniform_(*scale,
                                                     generator=generator)
                             .item())
            ar = math.exp(torch.empty(1).uniform_(*log_r,
                                                  generator=generator)
                          .item())
            cw = int(round(math.sqrt(target * ar)))
            ch = int(round(math.sqrt(target / ar)))
            if 0 < cw <= w and 0 < ch <= h:
                top = int(torch.randint(0, h - ch + 1, (1,),
                                        generator=generator).item())
                left = int(torch.randint(0, w - cw + 1, (1,),
                                         generator=generator).item())
                boxes[i] = torch.tensor([top, left, ch, cw])
                ok = True
                break
        if not ok:
            side = min(h, w)
            boxes[i] = torch.tensor([(h - side) // 2, (w - side) // 2,
                                     side, side])
    return boxes.to(device) if device is not None else boxes


def random_resized_crop(images_u8: torch.Tensor, boxes: torch.Tensor,
                        mean: torch.Tensor, std: torch.Tensor,
                        out_size: int = 224,
                        flip: Optional[torch.Tensor] = None,
                        out_dtype: torch.dtype = torch.float32
                        ) -> torch.Tensor:
    """Fused bilinear crop-resize + flip + normalize (the
    FFCV train-pipeline equivalent, reference dataset.py:385-392)."""
    if _backend.use_native(images_u8):
        return _backend.extension().random_resized_crop(
            images_u8, boxes.to(images_u8.device),
            flip if flip is not None else torch.Tensor(),
            mean.to(images_u8.device), std.to(images_u8.device),
            out_size, out_dtype)
    # torch oracle: per-image interpolate of the crop
    outs = []
    x = images_u8.float()
    for i in range(images_u8.shape[0]):
        t, l, ch, cw = boxes[i].tolist()
        crop = x[i:i + 1, :, t:t + ch, l:l + cw]
        out = torch.nn.functional.interpolate(
            crop, size=(out_size, out_size), mode="bilinear",
            align_corners=False)
        outs.append(out)
    out = torch.cat(outs, 0).div_(255.0)
    out = (out - mean.view(1, -1, 1, 1)) / std.view(1, -1, 1, 1)
    if flip is not None:
        out[flip] = torch.flip(out[flip], dims=[-1])
    return out.to(out_dtype)
