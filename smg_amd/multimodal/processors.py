"""Per-model vision processors (reference: crates/multimodal/src/vision/
processors/ — qwen_vl_base.rs (smart resize), llava.rs, phi3_vision.rs, ...;
registry/).

Each processor maps a decoded RGB image to model-ready pixel tensors using
the gfx950 resize+normalize kernel (Pillow-exact bicubic, csrc/image.hip).
Sizing strategies:
  * fixed:        square target (CLIP/LLaVA-style, e.g. 336x336)
  * smart_resize: snap H/W to multiples of `factor` within [min,max] pixels,
                  preserving aspect ratio (Qwen-VL family)
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np

CLIP_MEAN = [0.48145466, 0.4578275, 0.40821073]
CLIP_STD = [0.26862954, 0.26130258, 0.27577711]


@dataclass
class ProcessorConfig:
    name: str = "clip"
    strategy: str = "fixed"  # fixed | smart_resize
    size: int = 336
    factor: int = 28
    min_pixels: int = 56 * 56
    max_pixels: int = 14 * 14 * 4 * 1280
    mean: List[float] = field(default_factory=lambda: list(CLIP_MEAN))
    std: List[float] = field(default_factory=lambda: list(CLIP_STD))
    patch_size: int = 14
    merge_size: int = 2


def smart_resize(h: int, w: int, factor: int, min_pixels: int, max_pixels: int) -> Tuple[int, int]:
    """Qwen-VL smart resize (public algorithm): round H,W to multiples of
    `factor`, rescale into the pixel budget, keep aspect ratio."""
    if h < factor or w < factor:
        scale = factor / min(h, w)
        h, w = max(factor, int(h * scale)), max(factor, int(w * scale))
    h_bar = max(factor, round(h / factor) * factor)
    w_bar = max(factor, round(w / factor) * factor)
    if h_bar * w_bar > max_pixels:
        beta = math.sqrt((h * w) / max_pixels)
        h_bar = max(factor, math.floor(h / beta / factor) * factor)
        w_bar = max(factor, math.floor(w / beta / factor) * factor)
    elif h_bar * w_bar < min_pixels:
        beta = math.sqrt(min_pixels / (h * w))
        h_bar = math.ceil(h * beta / factor) * factor
        w_bar = math.ceil(w * beta / factor) * factor
    return h_bar, w_bar


class ImageProcessor:
    def __init__(self, config: Optional[ProcessorConfig] = None, use_gpu: bool = True):
        self.config = config or ProcessorConfig()
        try:
            import torch  # noqa: F401 — HIP runtime ordering

            from .. import _core

            self._kernel = _core.ImageProcessor(use_gpu=use_gpu)
        except ImportError:
            self._kernel = None

    @property
    def on_gpu(self) -> bool:
        return self._kernel is not None and self._kernel.on_gpu()

    def target_size(self, h: int, w: int) -> Tuple[int, int]:
        cfg = self.config
        if cfg.strategy == "fixed":
            return cfg.size, cfg.size
        return smart_resize(h, w, cfg.factor, cfg.min_pixels, cfg.max_pixels)

    def process(self, image: np.ndarray) -> Dict:
        """u8 HWC RGB -> {pixel_values f32 CHW, grid (th, tw) for patch models}."""
        h, w = image.shape[:2]
        th, tw = self.target_size(h, w)
        if self._kernel is not None:
            _, f32 = self._kernel.resize_normalize(
                image, tw, th, mean=self.config.mean, std=self.config.std, want_u8=False, want_f32=True
            )
        else:  # PIL fallback (identical semantics by construction)
            from PIL import Image

            resized = np.asarray(Image.fromarray(image).resize((tw, th), Image.BICUBIC))
            f32 = resized.astype(np.float32) / 255.0
            f32 = (f32 - np.array(self.config.mean, dtype=np.float32)) / np.array(self.config.std, dtype=np.float32)
            f32 = f32.transpose(2, 0, 1).copy()
        out = {"pixel_values": f32, "height": th, "width": tw}
        if self.config.strategy == "smart_resize":
            out["grid_thw"] = (1, th // self.config.patch_size, tw // self.config.patch_size)
        return out


# ---- registry (reference vision registry/) --------------------------------
_CONFIGS: Dict[str, ProcessorConfig] = {
    "clip": ProcessorConfig("clip", "fixed", size=336),
    "llava": ProcessorConfig("llava", "fixed", size=336),
    "phi3_vision": ProcessorConfig("phi3_vision", "fixed", size=336),
    "phi4_vision": ProcessorConfig("phi4_vision", "fixed", size=448),
    "llama4_vision": ProcessorConfig("llama4_vision", "fixed", size=336),
    "pixtral": ProcessorConfig("pixtral", "smart_resize", factor=16, max_pixels=1024 * 1024),
    "qwen2_vl": ProcessorConfig("qwen2_vl", "smart_resize", factor=28),
    "qwen3_vl": ProcessorConfig("qwen3_vl", "smart_resize", factor=28),
    "kimi_k3_vision": ProcessorConfig("kimi_k3_vision", "smart_resize", factor=28),
    "inkling_vision": ProcessorConfig("inkling_vision", "fixed", size=384),
}

_MODEL_PATTERNS = [
    ("qwen3-vl", "qwen3_vl"),
    ("qwen2-vl", "qwen2_vl"),
    ("qwen2.5-vl", "qwen2_vl"),
    ("llava", "llava"),
    ("pixtral", "pixtral"),
    ("phi-3", "phi3_vision"),
    ("phi-4", "phi4_vision"),
    ("llama-4", "llama4_vision"),
    ("kimi", "kimi_k3_vision"),
]


def processor_for_model(model_id: Optional[str], use_gpu: bool = True) -> ImageProcessor:
    cfg = _CONFIGS["clip"]
    if model_id:
        low = model_id.lower()
        for pat, name in _MODEL_PATTERNS:
            if pat in low:
                cfg = _CONFIGS[name]
                break
    return ImageProcessor(cfg, use_gpu=use_gpu)


def register_processor(name: str, cfg: ProcessorConfig) -> None:
    _CONFIGS[name] = cfg
