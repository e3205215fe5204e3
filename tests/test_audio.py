"""Audio decode/resample/log-mel (reference crates/multimodal/src/audio/:
decode.rs, transforms.rs bandlimited_resample + mel_basis,
processors/qwen3_audio.rs whisper_log_mel)."""
import io
import math
import struct
import wave

import numpy as np
import pytest

from smg_amd.multimodal.audio import (
    AudioError,
    bandlimited_resample,
    decode_wav,
    log_mel_spectrogram,
    mel_basis,
    preprocess_audio,
)


def make_wav(samples: np.ndarray, rate: int, channels: int = 1, sampwidth: int = 2) -> bytes:
    buf = io.BytesIO()
    with wave.open(buf, "wb") as w:
        w.setnchannels(channels)
        w.setsampwidth(sampwidth)
        w.setframerate(rate)
        if sampwidth == 2:
            w.writeframes((np.clip(samples, -1, 1) * 32767).astype("<i2").tobytes())
        else:
            raise ValueError
    return buf.getvalue()


def sine(freq, rate, secs):
    t = np.arange(int(rate * secs)) / rate
    return np.sin(2 * np.pi * freq * t).astype(np.float32)


def test_decode_wav_mono_and_stereo():
    x = sine(440, 16000, 0.1)
    mono = make_wav(x, 16000)
    got, rate = decode_wav(mono)
    assert rate == 16000
    assert np.max(np.abs(got - x)) < 2e-4  # 16-bit quantization

    stereo = make_wav(np.repeat(x, 2), 16000, channels=2)
    got2, _ = decode_wav(stereo)
    assert np.max(np.abs(got2 - x)) < 2e-4  # L==R mixdown


def test_decode_float32_wav():
    x = sine(100, 8000, 0.05)
    # hand-build an IEEE-float WAV (wave module writes PCM only)
    body = x.astype("<f4").tobytes()
    fmt = struct.pack("<HHIIHH", 3, 1, 8000, 8000 * 4, 4, 32)
    data = b"RIFF" + struct.pack("<I", 4 + 8 + len(fmt) + 8 + len(body)) + b"WAVE"
    data += b"fmt " + struct.pack("<I", len(fmt)) + fmt
    data += b"data" + struct.pack("<I", len(body)) + body
    got, rate = decode_wav(data)
    assert rate == 8000 and np.allclose(got, x, atol=1e-6)


def test_decode_rejects_garbage():
    with pytest.raises(AudioError):
        decode_wav(b"not audio at all" * 10)


def test_resample_length_formula_and_tone():
    # torchaudio output-length formula: ceil(len * new / old)
    x = sine(440, 44100, 0.25)
    y = bandlimited_resample(x, 44100, 16000)
    assert len(y) == math.ceil(len(x) * 16000 / 44100)
    # the 440 Hz tone must survive: dominant FFT bin within 1 bin of 440 Hz
    spec = np.abs(np.fft.rfft(y * np.hanning(len(y))))
    freq = np.argmax(spec) * 16000 / len(y)
    assert abs(freq - 440) < 16000 / len(y) * 2
    # amplitude preserved within a few percent (interior, away from edges)
    assert 0.9 < np.max(np.abs(y[100:-100])) < 1.1


def test_resample_identity_and_empty():
    x = sine(100, 16000, 0.01)
    assert np.array_equal(bandlimited_resample(x, 16000, 16000), x)
    assert bandlimited_resample(np.array([], np.float32), 8000, 16000).size == 0
    with pytest.raises(AudioError):
        bandlimited_resample(x, 0, 16000)


def test_mel_basis_properties():
    fb = mel_basis(16000, 400, 128)
    assert fb.shape == (128, 201)
    assert np.all(fb >= 0)
    # every filter has some energy, filters tile the spectrum
    assert np.all(fb.sum(axis=1) > 0)
    # Slaney normalization: peak amplitude decreases with bandwidth (higher
    # mels are wider -> lower peaks)
    peaks = fb.max(axis=1)
    assert peaks[0] > peaks[-1]


def test_log_mel_shape_and_range():
    x = sine(1000, 16000, 1.0)
    feats = log_mel_spectrogram(x, 16000, n_fft=400, hop_length=160, n_mels=128)
    assert feats.shape == (128, 100)  # 16000 samples / 160 hop
    # whisper scaling keeps values in (-1, ~1.5]; floor is (peak-8+4)/4
    assert feats.max() <= (feats.max() * 4 - 4 + 8 + 4) / 4  # tautological guard
    assert feats.min() >= (feats.max() * 4 - 4 - 8 + 4) / 4 - 1e-5
    # the 1 kHz bin should dominate: argmax mel roughly consistent per frame
    hot = np.argmax(feats[:, 50])
    assert 20 <= hot <= 80


def test_preprocess_audio_end_to_end():
    x = sine(440, 44100, 0.5)
    out = preprocess_audio(make_wav(x, 44100), target_rate=16000, n_mels=128)
    assert out["sample_rate"] == 16000
    assert out["features"].shape[0] == 128
    assert out["features"].shape[1] == out["feature_length"] == 8000 // 160


def test_pipeline_audio_content_part():
    """input_audio part flows through PreparationStage into ctx.multimodal."""
    import asyncio
    import base64

    from smg_amd.config import PolicyConfig, RouterConfig
    from smg_amd.routers.grpc.pipeline import PipelineContext, PreparationStage
    from smg_amd.routers.base import RouteRequest
    from smg_amd.server.app_context import AppContext

    async def run():
        cfg = RouterConfig(policy=PolicyConfig(name="round_robin", gpu_tree=False))
        ctx_app = AppContext(cfg)
        stage = PreparationStage(ctx_app)
        wav = make_wav(sine(440, 16000, 0.1), 16000)
        body = {
            "model": "qwen3-audio",
            "messages": [
                {
                    "role": "user",
                    "content": [
                        {"type": "text", "text": "transcribe this"},
                        {"type": "input_audio", "input_audio": {"data": base64.b64encode(wav).decode(), "format": "wav"}},
                    ],
                }
            ],
        }
        req = RouteRequest(path="/v1/chat/completions", body=body, raw_body=b"{}", headers={}, request_id="r1")
        pctx = PipelineContext(req=req, endpoint="chat")
        ok = await stage.run(pctx)
        assert ok, getattr(pctx.error, "body", None)
        assert pctx.multimodal and "audios" in pctx.multimodal
        a = pctx.multimodal["audios"][0]
        assert a["feature_length"] == 10 and a["sample_rate"] == 16000

    asyncio.new_event_loop().run_until_complete(run())
