"""Per-model vision processors (reference: crates/multimodal/src/vision/
processors/ — qwen_vl_base.rs, qwen2_vl.rs, qwen3_vl.rs, llava.rs,
phi3_vision.rs, phi4_vision.rs, pixtral.rs, llama4_vision.rs, registry/).

Each processor maps a decoded RGB image (u8 HWC) to model-ready pixel
tensors using the gfx950 resize+normalize kernel (Pillow-exact bicubic,
csrc/image.hip) with a PIL fallback of identical semantics, plus the
model-specific geometry the reference implements per model:

  * CLIP/fixed:  square resize (llava-hf default also center-crops)
  * LLaVA:       aspect modes `square` (shortest-edge resize + center crop)
                 and `pad` (expand to square with mean-color fill) — llava.rs
  * Qwen-VL:     smart_resize to factor multiples within a pixel budget,
                 temporal patchify to [n_patches, C*t*p*p] + grid_thw +
                 merge-aware token count — qwen_vl_base.rs:186,640
  * Pixtral:     fit inside longest_edge, ceil-snap to patch multiples,
                 dynamic size, tokens = (h/p)*(w/p) — pixtral.rs:107,259
  * Phi-4:       dynamic-HD crop grid (closest aspect ratio under max crops),
                 pad-resize + attention mask, global view + 448 tiles,
                 mask-aware token count — phi4_vision.rs:144-443
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np

CLIP_MEAN = [0.48145466, 0.4578275, 0.40821073]
CLIP_STD = [0.26862954, 0.26130258, 0.27577711]
HALF_MEAN = [0.5, 0.5, 0.5]
HALF_STD = [0.5, 0.5, 0.5]


@dataclass
class ProcessorConfig:
    name: str = "clip"
    strategy: str = "fixed"  # fixed | square | pad | smart_resize | longest_edge | dynamic_hd
    size: int = 336
    factor: int = 28
    min_pixels: int = 56 * 56
    max_pixels: int = 14 * 14 * 4 * 1280
    mean: List[float] = field(default_factory=lambda: list(CLIP_MEAN))
    std: List[float] = field(default_factory=lambda: list(CLIP_STD))
    patch_size: int = 14
    merge_size: int = 2
    temporal_patch_size: int = 2
    longest_edge: int = 1024       # pixtral
    base_resolution: int = 448     # phi4 tile size
    dynamic_hd: int = 36           # phi4 max crops


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
    """Base processor: resize+normalize through the HIP kernel; subclasses
    override geometry (`target_size`), layout (`preprocess`) and
    `num_tokens`."""

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

    # ---- primitives --------------------------------------------------------
    def _resize(self, image: np.ndarray, tw: int, th: int) -> np.ndarray:
        """u8 HWC -> u8 HWC at (th, tw), Pillow-exact bicubic."""
        if image.shape[0] == th and image.shape[1] == tw:
            return image
        if self._kernel is not None:
            u8, _ = self._kernel.resize_normalize(
                image, tw, th, mean=[0, 0, 0], std=[1, 1, 1], want_u8=True, want_f32=False)
            return u8
        from PIL import Image

        return np.asarray(Image.fromarray(image).resize((tw, th), Image.BICUBIC))

    def _resize_normalize(self, image: np.ndarray, tw: int, th: int) -> np.ndarray:
        """u8 HWC -> normalized f32 CHW at (th, tw)."""
        if self._kernel is not None:
            _, f32 = self._kernel.resize_normalize(
                image, tw, th, mean=self.config.mean, std=self.config.std,
                want_u8=False, want_f32=True)
            return f32
        resized = self._resize(image, tw, th)
        return self._normalize(resized)

    def _normalize(self, u8_hwc: np.ndarray) -> np.ndarray:
        f32 = u8_hwc.astype(np.float32) / 255.0
        f32 = (f32 - np.array(self.config.mean, dtype=np.float32)) / np.array(
            self.config.std, dtype=np.float32)
        return f32.transpose(2, 0, 1).copy()

    # ---- per-model surface -------------------------------------------------
    def target_size(self, h: int, w: int) -> Tuple[int, int]:
        return self.config.size, self.config.size

    def num_tokens(self, h: int, w: int) -> int:
        p = self.config.patch_size
        return (self.config.size // p) ** 2

    def process(self, image: np.ndarray) -> Dict:
        """u8 HWC RGB -> {pixel_values f32 CHW, ...}."""
        h, w = image.shape[:2]
        th, tw = self.target_size(h, w)
        f32 = self._resize_normalize(image, tw, th)
        return {"pixel_values": f32, "height": th, "width": tw,
                "num_tokens": self.num_tokens(h, w)}


class LlavaProcessor(ImageProcessor):
    """llava.rs: `square` mode (llava-hf/*) resizes the shortest edge to
    `size` then center-crops size x size; `pad` mode (liuhaotian/llava-*)
    expands to a square with mean-color padding first, then resizes."""

    def process(self, image: np.ndarray) -> Dict:
        cfg = self.config
        h, w = image.shape[:2]
        s = cfg.size
        if cfg.strategy == "pad":
            side = max(h, w)
            fill = np.array([round(m * 255) for m in cfg.mean], dtype=np.uint8)
            canvas = np.broadcast_to(fill, (side, side, 3)).copy()
            top, left = (side - h) // 2, (side - w) // 2
            canvas[top: top + h, left: left + w] = image
            f32 = self._resize_normalize(canvas, s, s)
        else:  # square: shortest edge -> size, center crop
            if h <= w:
                th, tw = s, max(s, round(w * s / h))
            else:
                th, tw = max(s, round(h * s / w)), s
            resized = self._resize(image, tw, th)
            top, left = (th - s) // 2, (tw - s) // 2
            crop = resized[top: top + s, left: left + s]
            f32 = self._normalize(crop)
        return {"pixel_values": f32, "height": s, "width": s,
                "num_tokens": self.num_tokens(h, w)}


class QwenVLProcessor(ImageProcessor):
    """qwen_vl_base.rs: smart_resize + temporal patchify.  Output layout per
    image/video: patches [grid_t*grid_h*grid_w, C*temporal*patch*patch]
    (merge-ordered rows, qwen_vl_base.rs:648) + grid_thw; tokens =
    t*h*w / merge^2 (qwen_vl_base.rs:640)."""

    def target_size(self, h: int, w: int) -> Tuple[int, int]:
        c = self.config
        return smart_resize(h, w, c.factor, c.min_pixels, c.max_pixels)

    def num_tokens(self, h: int, w: int) -> int:
        c = self.config
        th, tw = self.target_size(h, w)
        gh, gw = th // c.patch_size, tw // c.patch_size
        return (1 * gh * gw) // (c.merge_size ** 2)

    @staticmethod
    def patchify(frames_chw: np.ndarray, patch: int, merge: int, temporal: int) -> np.ndarray:
        """[T, C, H, W] f32 -> [grid_t*gh*gw, C*temporal*patch*patch] in the
        HF/reference merge-aware order.  T must be a multiple of `temporal`
        (pad by repeating the last frame before calling)."""
        T, C, H, W = frames_chw.shape
        gt = T // temporal
        gh, gw = H // patch, W // patch
        x = frames_chw.reshape(gt, temporal, C, gh // merge, merge, patch, gw // merge, merge, patch)
        # -> [gt, gh/m, gw/m, m, m, C, temporal, patch, patch]
        x = x.transpose(0, 3, 6, 4, 7, 2, 1, 5, 8)
        return x.reshape(gt * gh * gw, C * temporal * patch * patch).copy()

    def process(self, image: np.ndarray) -> Dict:
        c = self.config
        h, w = image.shape[:2]
        th, tw = self.target_size(h, w)
        f32 = self._resize_normalize(image, tw, th)  # [C, th, tw]
        frames = np.repeat(f32[None], c.temporal_patch_size, axis=0)  # single image: repeat
        patches = self.patchify(frames, c.patch_size, c.merge_size, c.temporal_patch_size)
        gh, gw = th // c.patch_size, tw // c.patch_size
        return {
            "pixel_values": patches,
            "grid_thw": (1, gh, gw),
            "height": th, "width": tw,
            "num_tokens": (1 * gh * gw) // (c.merge_size ** 2),
        }

    def process_video(self, frames: List[np.ndarray], sample_fps: float = 1.0) -> Dict:
        """Frame list (u8 HWC, pre-sampled) -> stacked temporal patches.
        All frames are resized to the first frame's smart-resize target."""
        c = self.config
        h, w = frames[0].shape[:2]
        th, tw = self.target_size(h, w)
        chw = np.stack([self._resize_normalize(f, tw, th) for f in frames])
        # pad to a temporal multiple by repeating the last frame
        t = c.temporal_patch_size
        if len(frames) % t:
            pad = t - len(frames) % t
            chw = np.concatenate([chw, np.repeat(chw[-1:], pad, axis=0)])
        patches = self.patchify(chw, c.patch_size, c.merge_size, t)
        gt = chw.shape[0] // t
        gh, gw = th // c.patch_size, tw // c.patch_size
        return {
            "pixel_values": patches,
            "grid_thw": (gt, gh, gw),
            "height": th, "width": tw,
            "num_tokens": (gt * gh * gw) // (c.merge_size ** 2),
            "second_per_grid": t / max(sample_fps, 1e-6),
        }


class PixtralProcessor(ImageProcessor):
    """pixtral.rs:107 get_resize_output_size — fit inside longest_edge (only
    downscale), then ceil-snap both dims to patch multiples; dynamic output
    size; tokens = (h/p)*(w/p) (pixtral.rs:259)."""

    def target_size(self, h: int, w: int) -> Tuple[int, int]:
        c = self.config
        ratio = max(h / c.longest_edge, w / c.longest_edge)
        if ratio > 1.0:
            h = math.floor(h / ratio)
            w = math.floor(w / ratio)
        p = c.patch_size
        gh = (max(h, 1) - 1) // p + 1
        gw = (max(w, 1) - 1) // p + 1
        return gh * p, gw * p

    def num_tokens(self, h: int, w: int) -> int:
        th, tw = self.target_size(h, w)
        p = self.config.patch_size
        return (th // p) * (tw // p)

    def process(self, image: np.ndarray) -> Dict:
        h, w = image.shape[:2]
        th, tw = self.target_size(h, w)
        f32 = self._resize_normalize(image, tw, th)
        p = self.config.patch_size
        return {"pixel_values": f32, "height": th, "width": tw,
                "image_sizes": (th, tw),
                "num_tokens": (th // p) * (tw // p)}


class Phi4VisionProcessor(ImageProcessor):
    """phi4_vision.rs dynamic-HD: pick an (w_crops, h_crops) grid — the
    natural ceil grid when under `dynamic_hd`, else the closest-aspect-ratio
    factor pair (phi4_vision.rs:152-201) — aspect-preserving resize into the
    grid with bottom/right padding + attention mask, then a global
    base_resolution view + the 448x448 tiles; tokens from the 2x-downsampled
    mask: 256 + 1 + mask_sum + mask_col0_sum + 16 (phi4_vision.rs:374)."""

    @staticmethod
    def _target_ratios(min_num: int, max_num: int) -> List[Tuple[int, int]]:
        out = set()
        for n in range(min_num, max_num + 1):
            for i in range(1, int(math.isqrt(n)) + 1):
                if n % i == 0:
                    out.add((i, n // i))
                    out.add((n // i, i))
        return sorted(out, key=lambda p: p[0] * p[1])

    def _closest_ratio(self, aspect: float, ratios, w: int, h: int) -> Tuple[int, int]:
        best, best_diff = (1, 1), float("inf")
        base_area = self.config.base_resolution ** 2
        area = w * h
        for wr, hr in ratios:
            diff = abs(aspect - wr / hr)
            if diff < best_diff:
                best_diff, best = diff, (wr, hr)
            elif abs(diff - best_diff) < 1e-6 and area > 0.5 * base_area * wr * hr:
                best = (wr, hr)
        return best

    def crop_grid(self, h: int, w: int) -> Tuple[int, int]:
        """(w_crops, h_crops) for an image (phi4_vision.rs dynamic_preprocess)."""
        base = self.config.base_resolution
        wc, hc = math.ceil(w / base), math.ceil(h / base)
        if wc * hc > self.config.dynamic_hd:
            ratios = self._target_ratios(1, self.config.dynamic_hd)
            wc, hc = self._closest_ratio(w / h, ratios, w, h)
        return wc, hc

    def process(self, image: np.ndarray) -> Dict:
        c = self.config
        base = c.base_resolution
        h, w = image.shape[:2]
        wc, hc = self.crop_grid(h, w)
        tw, th = base * wc, base * hc
        # aspect-preserving fit into (th, tw), pad bottom/right
        scale = min(tw / w, th / h)
        rw, rh = max(1, int(w * scale)), max(1, int(h * scale))
        resized = self._resize(image, rw, rh)
        canvas = np.zeros((th, tw, 3), dtype=np.uint8)
        canvas[:rh, :rw] = resized
        hd = self._normalize(canvas)  # [C, th, tw]
        # attention mask at patch granularity over the padded HD image
        p = c.patch_size
        mask = np.zeros((th // p, tw // p), dtype=np.uint32)
        mask[: math.ceil(rh / p), : math.ceil(rw / p)] = 1
        # global view
        global_view = self._resize_normalize(canvas, base, base)
        # tiles [hc*wc, C, base, base]
        tiles = (
            hd.reshape(3, hc, base, wc, base)
            .transpose(1, 3, 0, 2, 4)
            .reshape(hc * wc, 3, base, base)
        )
        pixel_values = np.concatenate([global_view[None], tiles])
        # token count from the 2x-downsampled mask (phi4_vision.rs:374)
        m2 = mask[::2, ::2]
        num_tokens = 256 + 1 + int(m2.sum()) + int(m2[:, 0].sum()) + 16
        return {
            "pixel_values": pixel_values,
            "attention_mask": mask,
            "height": th, "width": tw,
            "crops": (hc, wc),
            "num_tokens": num_tokens,
        }

    def num_tokens(self, h: int, w: int) -> int:
        return int(self.process(np.zeros((h, w, 3), dtype=np.uint8))["num_tokens"])


# ---- registry (reference vision registry/) --------------------------------
_CLASSES = {
    "fixed": ImageProcessor,
    "square": LlavaProcessor,
    "pad": LlavaProcessor,
    "smart_resize": QwenVLProcessor,
    "longest_edge": PixtralProcessor,
    "dynamic_hd": Phi4VisionProcessor,
}

_CONFIGS: Dict[str, ProcessorConfig] = {
    "clip": ProcessorConfig("clip", "fixed", size=336),
    "llava": ProcessorConfig("llava", "square", size=336),
    "llava_pad": ProcessorConfig("llava_pad", "pad", size=336),
    "phi3_vision": ProcessorConfig("phi3_vision", "dynamic_hd", base_resolution=336,
                                   dynamic_hd=16, mean=list(HALF_MEAN), std=list(HALF_STD)),
    "phi4_vision": ProcessorConfig("phi4_vision", "dynamic_hd", base_resolution=448,
                                   dynamic_hd=36, mean=list(HALF_MEAN), std=list(HALF_STD)),
    "llama4_vision": ProcessorConfig("llama4_vision", "fixed", size=336),
    "pixtral": ProcessorConfig("pixtral", "longest_edge", factor=16, patch_size=16,
                               longest_edge=1024),
    "qwen2_vl": ProcessorConfig("qwen2_vl", "smart_resize", factor=28),
    "qwen3_vl": ProcessorConfig("qwen3_vl", "smart_resize", factor=28),
    "kimi_k3_vision": ProcessorConfig("kimi_k3_vision", "smart_resize", factor=28),
    "inkling_vision": ProcessorConfig("inkling_vision", "fixed", size=384),
}

_MODEL_PATTERNS = [
    ("qwen3-vl", "qwen3_vl"),
    ("qwen2-vl", "qwen2_vl"),
    ("qwen2.5-vl", "qwen2_vl"),
    ("liuhaotian/llava", "llava_pad"),
    ("llava", "llava"),
    ("pixtral", "pixtral"),
    ("phi-3", "phi3_vision"),
    ("phi-4", "phi4_vision"),
    ("llama-4", "llama4_vision"),
    ("kimi", "kimi_k3_vision"),
]


def make_processor(cfg: ProcessorConfig, use_gpu: bool = True) -> ImageProcessor:
    return _CLASSES.get(cfg.strategy, ImageProcessor)(cfg, use_gpu=use_gpu)


def processor_for_model(model_id: Optional[str], use_gpu: bool = True) -> ImageProcessor:
    cfg = _CONFIGS["clip"]
    if model_id:
        low = model_id.lower()
        for pat, name in _MODEL_PATTERNS:
            if pat in low:
                cfg = _CONFIGS[name]
                break
    return make_processor(cfg, use_gpu=use_gpu)


def register_processor(name: str, cfg: ProcessorConfig) -> None:
    _CONFIGS[name] = cfg
