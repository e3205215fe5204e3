"""Audio decode + transforms (reference: crates/multimodal/src/audio/ —
decode.rs WAV/PCM decode + mono mixdown, transforms.rs band-limited sinc
resample matching torchaudio defaults (width 6, rolloff 0.99, Hann window)
and Slaney mel filter bank, processors/qwen3_audio.rs whisper-style log-mel
(n_fft 400, hop 160, reflect center pad, peak-8 floor, (x+4)/4 scaling)).

All numpy (vectorized polyphase kernels instead of the reference's scalar
loops); the output feature tensor feeds the audio tower on the GPU engine.
"""
from __future__ import annotations

import math
import struct
from typing import Tuple

import numpy as np


class AudioError(ValueError):
    pass


# ---- decode (decode.rs; WAV/PCM only — no codec libs in this image) ---------
def decode_wav(data: bytes) -> Tuple[np.ndarray, int]:
    """RIFF/WAVE -> (mono float32 in [-1, 1], sample_rate).  Supports PCM
    8/16/24/32-bit int and IEEE float32/64, any channel count (mixed down)."""
    if len(data) < 44 or data[:4] != b"RIFF" or data[8:12] != b"WAVE":
        raise AudioError("not a RIFF/WAVE file")
    pos = 12
    fmt = None
    raw = None
    while pos + 8 <= len(data):
        cid, size = data[pos:pos + 4], struct.unpack("<I", data[pos + 4:pos + 8])[0]
        body = data[pos + 8:pos + 8 + size]
        if cid == b"fmt ":
            fmt = struct.unpack("<HHIIHH", body[:16])
        elif cid == b"data":
            raw = body
        pos += 8 + size + (size & 1)
    if fmt is None or raw is None:
        raise AudioError("missing fmt/data chunk")
    audio_format, channels, sample_rate, _, _, bits = fmt
    if audio_format == 0xFFFE and len(data) >= pos:  # WAVE_FORMAT_EXTENSIBLE
        audio_format = 1
    if audio_format == 1:  # PCM int
        if bits == 8:
            x = (np.frombuffer(raw, dtype=np.uint8).astype(np.float32) - 128.0) / 128.0
        elif bits == 16:
            x = np.frombuffer(raw, dtype="<i2").astype(np.float32) / 32768.0
        elif bits == 24:
            b = np.frombuffer(raw, dtype=np.uint8).reshape(-1, 3)
            x = (
                (b[:, 0].astype(np.int32))
                | (b[:, 1].astype(np.int32) << 8)
                | (b[:, 2].astype(np.int8).astype(np.int32) << 16)
            ).astype(np.float32) / 8388608.0
        elif bits == 32:
            x = np.frombuffer(raw, dtype="<i4").astype(np.float32) / 2147483648.0
        else:
            raise AudioError(f"unsupported PCM bit depth {bits}")
    elif audio_format == 3:  # IEEE float
        x = np.frombuffer(raw, dtype="<f4" if bits == 32 else "<f8").astype(np.float32)
    else:
        raise AudioError(f"unsupported WAV format code {audio_format}")
    if channels > 1:
        x = x[: len(x) // channels * channels].reshape(-1, channels).mean(axis=1)
    return np.ascontiguousarray(x, dtype=np.float32), sample_rate


# ---- resample (transforms.rs:68-151 — torchaudio-default sinc) -------------
def bandlimited_resample(
    samples: np.ndarray,
    src_rate: int,
    dst_rate: int,
    lowpass_filter_width: float = 6.0,
    rolloff: float = 0.99,
) -> np.ndarray:
    if src_rate <= 0 or dst_rate <= 0:
        raise AudioError("audio resampling rates must be positive")
    samples = np.asarray(samples, dtype=np.float32)
    if samples.size == 0 or src_rate == dst_rate:
        return samples.copy()
    g = math.gcd(src_rate, dst_rate)
    orig, new = src_rate // g, dst_rate // g
    base_freq = min(orig, new) * rolloff
    width = int(math.ceil(lowpass_filter_width * orig / base_freq))
    kernel_len = 2 * width + orig
    # kernel[phase, k]: windowed sinc at t = ((k-width)/orig - phase/new)*base
    k = np.arange(kernel_len, dtype=np.float64)
    phase = np.arange(new, dtype=np.float64)[:, None]
    t = ((k[None, :] - width) / orig - phase / new) * base_freq
    t = np.clip(t, -lowpass_filter_width, lowpass_filter_width)
    window = np.cos(t * np.pi / lowpass_filter_width / 2.0) ** 2
    kernels = (np.sinc(t) * window * (base_freq / orig)).astype(np.float32)

    target_len = -((-samples.size * new) // orig)  # ceil
    n_blocks = -((-samples.size) // orig)
    # pad so every block's kernel window is in-bounds (zeros left of width,
    # zeros beyond the tail — matches the reference's bounds checks)
    padded = np.concatenate(
        [np.zeros(width, np.float32), samples, np.zeros(n_blocks * orig + width + kernel_len, np.float32)]
    )
    # frames[block, k] = padded[block*orig + k]
    frames = np.lib.stride_tricks.as_strided(
        padded,
        shape=(n_blocks, kernel_len),
        strides=(padded.strides[0] * orig, padded.strides[0]),
    )
    # out[block, phase] = frames[block] . kernels[phase]
    out = frames @ kernels.T
    return out.reshape(-1)[:target_len].astype(np.float32)


# ---- mel filter bank (transforms.rs:13-64, Slaney-normalized) ---------------
def _hz_to_mel(f):
    f = np.asarray(f, dtype=np.float64)
    f_sp = 200.0 / 3
    min_log_hz = 1000.0
    min_log_mel = min_log_hz / f_sp
    logstep = np.log(6.4) / 27.0
    return np.where(f >= min_log_hz, min_log_mel + np.log(np.maximum(f, 1e-30) / min_log_hz) / logstep, f / f_sp)


def _mel_to_hz(m):
    m = np.asarray(m, dtype=np.float64)
    f_sp = 200.0 / 3
    min_log_hz = 1000.0
    min_log_mel = min_log_hz / f_sp
    logstep = np.log(6.4) / 27.0
    return np.where(m >= min_log_mel, min_log_hz * np.exp(logstep * (m - min_log_mel)), m * f_sp)


def mel_basis(sample_rate: int, n_fft: int, n_mels: int) -> np.ndarray:
    """(n_mels, n_fft//2+1) Slaney-normalized triangular filters."""
    fft_bins = n_fft // 2 + 1
    fft_freqs = np.arange(fft_bins, dtype=np.float64) * sample_rate / n_fft
    mel_min, mel_max = _hz_to_mel(0.0), _hz_to_mel(sample_rate / 2.0)
    mel_edges = _mel_to_hz(mel_min + (mel_max - mel_min) * np.arange(n_mels + 2) / (n_mels + 1))
    widths = np.diff(mel_edges)
    lower = (fft_freqs[None, :] - mel_edges[:-2, None]) / widths[:-1, None]
    upper = (mel_edges[2:, None] - fft_freqs[None, :]) / widths[1:, None]
    enorm = 2.0 / (mel_edges[2:] - mel_edges[:-2])
    weights = np.maximum(0.0, np.minimum(lower, upper)) * enorm[:, None]
    return weights.astype(np.float32)


# ---- whisper-style log-mel (qwen3_audio.rs:340-393) -------------------------
def log_mel_spectrogram(
    samples: np.ndarray,
    sample_rate: int = 16_000,
    n_fft: int = 400,
    hop_length: int = 160,
    n_mels: int = 128,
) -> np.ndarray:
    """(n_mels, n_frames) features: reflect center pad, Hann window, power
    spectrum, Slaney mel, log10 clamped at 1e-10, floored at peak-8,
    scaled (x+4)/4."""
    samples = np.asarray(samples, dtype=np.float32)
    frame_count = max(1, len(samples) // hop_length)
    pad = n_fft // 2
    if len(samples) >= 2:
        padded = np.concatenate([samples[1:pad + 1][::-1], samples, samples[-2:-pad - 2:-1]])
    else:
        padded = np.full(len(samples) + 2 * pad, samples[0] if len(samples) else 0.0, np.float32)
    need = (frame_count - 1) * hop_length + n_fft
    if len(padded) < need:
        padded = np.concatenate([padded, np.zeros(need - len(padded), np.float32)])
    window = np.hanning(n_fft + 1)[:-1].astype(np.float32)  # periodic Hann
    idx = np.arange(frame_count)[:, None] * hop_length + np.arange(n_fft)[None, :]
    frames = padded[idx] * window[None, :]
    spec = np.fft.rfft(frames, n=n_fft, axis=1)
    power = (spec.real ** 2 + spec.imag ** 2).astype(np.float32)  # (frames, bins)
    mel = mel_basis(sample_rate, n_fft, n_mels) @ power.T  # (mels, frames)
    log_spec = np.log10(np.maximum(mel, 1e-10))
    log_spec = np.maximum(log_spec, log_spec.max() - 8.0)
    return ((log_spec + 4.0) / 4.0).astype(np.float32)


def preprocess_audio(
    data: bytes,
    target_rate: int = 16_000,
    n_mels: int = 128,
    max_samples: int | None = None,
) -> dict:
    """WAV bytes -> log-mel features dict (the audio analogue of
    ImageProcessor.process): {"features": (n_mels, frames) f32,
    "feature_length": frames, "sample_rate": target}."""
    samples, rate = decode_wav(data)
    if rate != target_rate:
        samples = bandlimited_resample(samples, rate, target_rate)
    if max_samples is not None:
        samples = samples[:max_samples]
    feats = log_mel_spectrogram(samples, sample_rate=target_rate, n_mels=n_mels)
    return {"features": feats, "feature_length": feats.shape[1], "sample_rate": target_rate}
