"""Multimodal preprocessing tests (reference: crates/multimodal processors +
grpc/multimodal pipeline)."""
import base64
import io
import json

import numpy as np
import pytest

from smg_amd.multimodal.media import MediaError, decode_image, fetch_image_bytes
from smg_amd.multimodal.processors import (
    ImageProcessor,
    ProcessorConfig,
    processor_for_model,
    smart_resize,
)
from smg_amd.multimodal.transport import decode_tensor, encode_tensor, release_tensor


def png_data_url(w=40, h=30, color=(200, 30, 60)):
    from PIL import Image

    buf = io.BytesIO()
    Image.new("RGB", (w, h), color).save(buf, format="PNG")
    return "data:image/png;base64," + base64.b64encode(buf.getvalue()).decode()


class TestMedia:
    def test_data_url_roundtrip(self, runner):
        async def run():
            data = await fetch_image_bytes(png_data_url())
            arr = decode_image(data)
            assert arr.shape == (30, 40, 3)

        runner(run())

    def test_bad_base64(self, runner):
        async def run():
            with pytest.raises(MediaError):
                await fetch_image_bytes("data:image/png;base64,!!!notb64")

        runner(run())


class TestSmartResize:
    def test_multiples_of_factor(self):
        h, w = smart_resize(1000, 700, 28, 56 * 56, 14 * 14 * 4 * 1280)
        assert h % 28 == 0 and w % 28 == 0
        assert abs((h / w) - (1000 / 700)) < 0.15

    def test_max_pixel_budget(self):
        h, w = smart_resize(8000, 8000, 28, 56 * 56, 1280 * 28 * 28)
        assert h * w <= 1280 * 28 * 28

    def test_min_pixel_floor(self):
        h, w = smart_resize(10, 10, 28, 56 * 56, 14 * 14 * 4 * 1280)
        assert h * w >= 56 * 56


class TestProcessors:
    def test_fixed_clip(self):
        proc = ImageProcessor(ProcessorConfig("clip", "fixed", size=336), use_gpu=False)
        rng = np.random.default_rng(0)
        out = proc.process(rng.integers(0, 256, (480, 640, 3), dtype=np.uint8))
        assert out["pixel_values"].shape == (3, 336, 336)
        assert out["pixel_values"].dtype == np.float32

    def test_qwen_grid(self):
        proc = processor_for_model("Qwen2-VL-7B", use_gpu=False)
        rng = np.random.default_rng(1)
        out = proc.process(rng.integers(0, 256, (280, 420, 3), dtype=np.uint8))
        t, th, tw = out["grid_thw"]
        assert th == out["height"] // 14 and tw == out["width"] // 14

    def test_registry_default(self):
        assert processor_for_model(None, use_gpu=False).config.name == "clip"
        assert processor_for_model("llava-1.6", use_gpu=False).config.name == "llava"


class TestTransport:
    def test_inline(self):
        arr = np.arange(24, dtype=np.float32).reshape(2, 3, 4)
        desc = encode_tensor(arr, "inline")
        assert desc["kind"] == "inline"
        assert np.array_equal(decode_tensor(desc), arr)

    def test_shm_threshold(self):
        small = np.zeros(16, dtype=np.float32)
        assert encode_tensor(small, "shm")["kind"] == "inline"  # below 64 KiB
        big = np.random.default_rng(0).random((256, 256), dtype=np.float32)
        desc = encode_tensor(big, "shm")
        assert desc["kind"] == "shm"
        out = decode_tensor(desc)
        assert np.array_equal(out, big)
        release_tensor(desc)


def test_grpc_chat_with_image(runner):
    """Image part flows through the pipeline into the engine request."""
    from tests.test_grpc_mode import setup, teardown, _req

    async def run():
        ctx, router, servers = await setup(n_workers=1)
        try:
            body = {
                "model": "mock-model",
                "max_tokens": 2,
                "messages": [
                    {
                        "role": "user",
                        "content": [
                            {"type": "text", "text": "what is this?"},
                            {"type": "image_url", "image_url": {"url": png_data_url(64, 64)}},
                        ],
                    }
                ],
            }
            resp = await router.route(_req("/v1/chat/completions", body))
            assert resp.status == 200, resp.body
            data = json.loads(resp.body)
            assert data["choices"][0]["message"]["content"]
        finally:
            await teardown(router, servers)

    runner(run())
