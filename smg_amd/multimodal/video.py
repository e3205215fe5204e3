"""Video frame decoding + sampling (reference: crates/multimodal/src/video/ —
OpenCV-backed frame capture with `opencv_buffer_capture.cpp` and fps/uniform
frame sampling).

This image has no OpenCV/ffmpeg, so decode is pure-Python via PIL's animated
formats (GIF / animated WebP / APNG / multi-frame TIFF) plus raw ``.npy``
frame stacks — the sampling and per-frame kernel-resize pipeline is the same
one a container decoder would feed.  Per-frame resize+normalize runs on the
gfx950 image kernel through the model's processor
(QwenVLProcessor.process_video)."""
from __future__ import annotations

import io
from dataclasses import dataclass
from typing import List, Optional, Tuple

import numpy as np

from .media import MediaError


@dataclass
class VideoSampleConfig:
    """Frame sampling policy (reference video/mod.rs sampling knobs)."""

    num_frames: Optional[int] = None  # uniform-sample exactly N frames
    sample_fps: float = 1.0           # else sample at this rate
    max_frames: int = 64
    min_frames: int = 2


def decode_video_frames(data: bytes) -> Tuple[List[np.ndarray], float]:
    """bytes -> (frames u8 HWC RGB, native_fps).

    Accepts PIL animated images (GIF/WebP/APNG/TIFF) and numpy ``.npy``
    stacks of shape [T, H, W, 3] u8 (the raw-frame escape hatch the tests and
    offline pipelines use)."""
    if data[:6] == b"\x93NUMPY":
        arr = np.load(io.BytesIO(data), allow_pickle=False)
        if arr.ndim != 4 or arr.shape[-1] != 3:
            raise MediaError(f"npy video must be [T,H,W,3] u8, got {arr.shape}")
        return [f for f in arr.astype(np.uint8)], 24.0
    from PIL import Image, ImageSequence

    try:
        img = Image.open(io.BytesIO(data))
    except Exception as e:
        raise MediaError(f"video decode failed: {e}")
    n_frames = getattr(img, "n_frames", 1)
    if n_frames <= 1:
        raise MediaError("not an animated image / video")
    frames = []
    durations_ms = []
    for frame in ImageSequence.Iterator(img):
        frames.append(np.asarray(frame.convert("RGB")))
        durations_ms.append(frame.info.get("duration", 42) or 42)
    fps = 1000.0 / (sum(durations_ms) / len(durations_ms))
    return frames, fps


def sample_frames(
    frames: List[np.ndarray], native_fps: float, cfg: Optional[VideoSampleConfig] = None
) -> Tuple[List[np.ndarray], float]:
    """Uniform / fps-based frame sampling -> (sampled frames, effective fps)."""
    cfg = cfg or VideoSampleConfig()
    total = len(frames)
    if total == 0:
        raise MediaError("no frames to sample")
    if cfg.num_frames is not None:
        n = max(cfg.min_frames, min(cfg.num_frames, total, cfg.max_frames))
    else:
        duration_s = total / max(native_fps, 1e-6)
        n = int(round(duration_s * cfg.sample_fps))
        n = max(cfg.min_frames, min(n if n > 0 else cfg.min_frames, total, cfg.max_frames))
    idx = np.linspace(0, total - 1, n).round().astype(int)
    eff_fps = native_fps * n / total
    return [frames[i] for i in idx], eff_fps


def process_video(
    data: bytes,
    processor,
    cfg: Optional[VideoSampleConfig] = None,
):
    """bytes -> model-ready video tensors via the model's processor.
    Processors exposing process_video (Qwen-VL family) get temporal patches;
    others get per-frame pixel_values stacked on a new leading axis."""
    frames, native_fps = decode_video_frames(data)
    sampled, eff_fps = sample_frames(frames, native_fps, cfg)
    if hasattr(processor, "process_video"):
        return processor.process_video(sampled, sample_fps=eff_fps)
    outs = [processor.process(f) for f in sampled]
    return {
        "pixel_values": np.stack([o["pixel_values"] for o in outs]),
        "num_frames": len(sampled),
        "num_tokens": sum(o["num_tokens"] for o in outs),
        "height": outs[0]["height"],
        "width": outs[0]["width"],
    }
