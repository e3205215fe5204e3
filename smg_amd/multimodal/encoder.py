"""EPD encode-rank role: the vision tower that runs on a dedicated fleet in
the reference (grpc_servicer/.../encoder_servicer.py + mm_rdma NIXL
transport).  MI355X-native version: pixels arrive over the xGMI plane
(comm/plane.py PIX_SEND/PIX_RECV), the encoder runs on the encode rank's
GPU, and the [E, d_model] embeddings ship rank-to-rank (EMB_SEND/EMB_RECV)
straight into the decode engine's prefill — no host staging, no sidecar
transport.

ToyVisionEncoder is a deterministic random-init patch encoder standing in
for a real tower (the bench models are random-init too): resize to
`image_size`, 16x16 patches, one linear projection.  EncodeWorker is the
plane-protocol wrapper (accept_pixels / export_embed) an encode rank runs.
"""
from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.nn.functional as F


class ToyVisionEncoder:
    def __init__(self, d_model: int, image_size: int = 224, patch: int = 16,
                 seed: int = 1234, device: str = "cpu", dtype=torch.float32):
        self.d_model = d_model
        self.image_size = image_size
        self.patch = patch
        self.device = torch.device(device)
        self.dtype = dtype
        g = torch.Generator().manual_seed(seed)
        in_dim = 3 * patch * patch
        w = torch.randn(in_dim, d_model, generator=g) * (in_dim ** -0.5)
        self.proj = w.to(self.device, dtype)
        self.n_embed = (image_size // patch) ** 2

    @torch.no_grad()
    def encode(self, pixels_u8: torch.Tensor) -> torch.Tensor:
        """[3, H, W] uint8 -> [n_embed, d_model]."""
        x = pixels_u8.to(self.device, torch.float32).unsqueeze(0) / 255.0
        x = F.interpolate(x, size=(self.image_size, self.image_size),
                          mode="bilinear", align_corners=False)
        p = self.patch
        n = self.image_size // p
        # [1,3,H,W] -> [n*n, 3*p*p] patch rows
        x = x.reshape(3, n, p, n, p).permute(1, 3, 0, 2, 4).reshape(n * n, 3 * p * p)
        return (x.to(self.dtype) @ self.proj)


class EncodeWorker:
    """The encode rank's plane endpoint: PIX_RECV -> encode -> parked
    embedding -> EMB_SEND (instruction pairs scheduled by the gateway, like
    the PD KV handoff)."""

    def __init__(self, encoder: ToyVisionEncoder):
        self.encoder = encoder
        self.device = encoder.device
        self.dtype = encoder.dtype
        self._embeds: Dict = {}
        self._pixels: Dict = {}

    # plane protocol ---------------------------------------------------------
    def accept_pixels(self, rid, pixels: torch.Tensor) -> None:
        self._embeds[rid] = self.encoder.encode(pixels)

    def export_embed(self, rid) -> torch.Tensor:
        return self._embeds.pop(rid)

    def export_pixels(self, rid) -> torch.Tensor:
        """Gateway-side use: outgoing pixel stash (PIX_SEND)."""
        return self._pixels.pop(rid)

    def embed_len(self) -> int:
        return self.encoder.n_embed


class PixelSource:
    """Gateway-side stash for outgoing pixels (export_pixels protocol)."""

    def __init__(self):
        self._pixels: Dict = {}

    def put(self, rid, pixels: torch.Tensor) -> None:
        self._pixels[rid] = pixels

    def export_pixels(self, rid) -> torch.Tensor:
        return self._pixels.pop(rid)
