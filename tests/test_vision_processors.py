"""Per-model vision processors + video path (reference:
crates/multimodal/src/vision/processors/{llava,qwen2_vl,pixtral,phi4_vision}.rs
and video/).  Geometry and layout checks from the published algorithms, plus
an HF-parity differential against transformers' CLIPImageProcessor (the only
HF image processor importable without torchvision in this image)."""
import io

import numpy as np
import pytest

from smg_amd.multimodal.processors import (
    ImageProcessor,
    LlavaProcessor,
    Phi4VisionProcessor,
    PixtralProcessor,
    ProcessorConfig,
    QwenVLProcessor,
    processor_for_model,
)
from smg_amd.multimodal.video import (
    VideoSampleConfig,
    decode_video_frames,
    process_video,
    sample_frames,
)


def img(h, w, seed=0):
    return np.random.default_rng(seed).integers(0, 256, (h, w, 3), dtype=np.uint8)


class TestLlava:
    def test_square_mode_geometry(self):
        proc = LlavaProcessor(ProcessorConfig("llava", "square", size=336), use_gpu=False)
        out = proc.process(img(480, 640))
        assert out["pixel_values"].shape == (3, 336, 336)
        assert out["num_tokens"] == (336 // 14) ** 2  # 576

    def test_pad_mode_mean_fill(self):
        proc = LlavaProcessor(ProcessorConfig("llava_pad", "pad", size=336), use_gpu=False)
        # a tall thin all-black image: the padded columns must be mean-colored
        image = np.zeros((300, 100, 3), dtype=np.uint8)
        out = proc.process(image)
        assert out["pixel_values"].shape == (3, 336, 336)
        # mean-color padding normalizes to ~0 (value == mean)
        left_pad = out["pixel_values"][:, 168, 10]
        assert np.all(np.abs(left_pad) < 0.1)
        # the image region normalizes to the black value (-mean/std), clearly negative
        center = out["pixel_values"][:, 168, 168]
        assert np.all(center < -1.0)

    def test_hf_clip_parity(self):
        """Differential vs transformers CLIPImageProcessor: shortest-edge
        resize + center crop + normalize must match bit-for-bit-ish."""
        tfi = pytest.importorskip("transformers.models.clip.image_processing_clip")
        hf = tfi.CLIPImageProcessor(
            do_resize=True, size={"shortest_edge": 336}, do_center_crop=True,
            crop_size={"height": 336, "width": 336}, do_rescale=True, do_normalize=True,
            do_convert_rgb=False,
        )
        from PIL import Image

        image = img(400, 520, seed=3)
        try:
            ref = hf(Image.fromarray(image), return_tensors="np")["pixel_values"][0]
        except NameError:
            pytest.skip("transformers image backend requires torchvision in this build")
        proc = LlavaProcessor(ProcessorConfig("llava", "square", size=336), use_gpu=False)
        got = proc.process(image)["pixel_values"]
        assert got.shape == ref.shape
        assert np.abs(got - ref).max() < 1e-3, np.abs(got - ref).max()

    def test_registry_pad_variant(self):
        assert processor_for_model("liuhaotian/llava-v1.5", use_gpu=False).config.strategy == "pad"
        assert processor_for_model("llava-hf/llava-1.5", use_gpu=False).config.strategy == "square"


class TestQwenVL:
    def test_patchify_matches_naive_reference(self):
        c = ProcessorConfig("qwen2_vl", "smart_resize", factor=28)
        frames = np.random.default_rng(0).normal(size=(2, 3, 56, 84)).astype(np.float32)
        patches = QwenVLProcessor.patchify(frames, patch=14, merge=2, temporal=2)
        gh, gw = 4, 6
        assert patches.shape == (1 * gh * gw, 3 * 2 * 14 * 14)
        # naive per-patch extraction in the merge-aware order
        row = 0
        for bh in range(gh // 2):
            for bw in range(gw // 2):
                for mh in range(2):
                    for mw in range(2):
                        ph, pw = bh * 2 + mh, bw * 2 + mw
                        ref = frames[:, :, ph * 14:(ph + 1) * 14, pw * 14:(pw + 1) * 14]
                        # row layout: C, temporal, p, p
                        ref = ref.transpose(1, 0, 2, 3).reshape(-1)
                        assert np.array_equal(patches[row], ref), (row, ph, pw)
                        row += 1

    def test_image_tokens_and_grid(self):
        proc = QwenVLProcessor(ProcessorConfig("qwen2_vl", "smart_resize", factor=28), use_gpu=False)
        out = proc.process(img(280, 420))
        t, gh, gw = out["grid_thw"]
        assert t == 1 and gh == out["height"] // 14 and gw == out["width"] // 14
        assert out["num_tokens"] == (gh * gw) // 4
        assert out["pixel_values"].shape == (gh * gw, 3 * 2 * 14 * 14)

    def test_video_temporal_grid(self):
        proc = QwenVLProcessor(ProcessorConfig("qwen2_vl", "smart_resize", factor=28), use_gpu=False)
        frames = [img(112, 112, seed=i) for i in range(5)]  # odd count -> pad to 6
        out = proc.process_video(frames, sample_fps=2.0)
        t, gh, gw = out["grid_thw"]
        assert t == 3  # ceil(5/2) temporal groups
        assert out["pixel_values"].shape == (t * gh * gw, 3 * 2 * 14 * 14)
        assert out["num_tokens"] == (t * gh * gw) // 4


class TestPixtral:
    def test_downscale_and_snap(self):
        proc = PixtralProcessor(
            ProcessorConfig("pixtral", "longest_edge", patch_size=16, longest_edge=1024),
            use_gpu=False)
        th, tw = proc.target_size(2000, 1000)
        assert th == 1024 and tw == 512
        assert proc.num_tokens(2000, 1000) == (1024 // 16) * (512 // 16)

    def test_no_upscale_snaps_up_to_patch(self):
        proc = PixtralProcessor(
            ProcessorConfig("pixtral", "longest_edge", patch_size=16, longest_edge=1024),
            use_gpu=False)
        # small image is NOT upscaled beyond patch snapping (ceil division)
        th, tw = proc.target_size(100, 60)
        assert th == 112 and tw == 64  # ceil(100/16)=7 -> 112, ceil(60/16)=4 -> 64
        out = proc.process(img(100, 60))
        assert out["pixel_values"].shape == (3, 112, 64)
        assert out["image_sizes"] == (112, 64)


class TestPhi4:
    def cfg(self, **kw):
        base = dict(strategy="dynamic_hd", base_resolution=448, dynamic_hd=36,
                    mean=[0.5] * 3, std=[0.5] * 3)
        base.update(kw)
        return ProcessorConfig("phi4_vision", **base)

    def test_natural_grid_under_cap(self):
        proc = Phi4VisionProcessor(self.cfg(), use_gpu=False)
        assert proc.crop_grid(448, 448) == (1, 1)
        assert proc.crop_grid(500, 1000) == (3, 2)  # (wc=ceil(1000/448)=3, hc=2)

    def test_ratio_search_over_cap(self):
        proc = Phi4VisionProcessor(self.cfg(dynamic_hd=4), use_gpu=False)
        wc, hc = proc.crop_grid(448, 448 * 10)  # 10:1 aspect, cap 4 crops
        assert wc * hc <= 4
        assert wc >= hc  # wide image picks a wide grid

    def test_tiles_and_global(self):
        proc = Phi4VisionProcessor(self.cfg(), use_gpu=False)
        out = proc.process(img(500, 900))
        hc, wc = out["crops"]
        # global view + hc*wc tiles, each 448x448
        assert out["pixel_values"].shape == (1 + hc * wc, 3, 448, 448)
        assert out["attention_mask"].shape == (out["height"] // 14, out["width"] // 14)
        # padded region masked out
        assert out["attention_mask"].min() == 0 or out["attention_mask"].all()
        assert out["num_tokens"] > 256

    def test_tiles_reassemble_hd_image(self):
        proc = Phi4VisionProcessor(self.cfg(), use_gpu=False)
        image = img(448, 896, seed=7)  # exactly 1x2 crops, no padding
        out = proc.process(image)
        hc, wc = out["crops"]
        assert (hc, wc) == (1, 2)
        tiles = out["pixel_values"][1:]
        # stitch tiles back: [hc, wc, C, 448, 448] -> [C, hc*448, wc*448]
        stitched = tiles.reshape(hc, wc, 3, 448, 448).transpose(2, 0, 3, 1, 4).reshape(
            3, hc * 448, wc * 448)
        # un-normalize and compare to the source (mean/std 0.5 -> x*0.5+0.5)
        restored = ((stitched * 0.5 + 0.5) * 255).round().clip(0, 255).astype(np.uint8)
        assert np.abs(restored.transpose(1, 2, 0).astype(int) - image.astype(int)).max() <= 1


class TestVideo:
    def make_gif(self, n=8, size=64):
        from PIL import Image

        frames = [Image.fromarray(img(size, size, seed=i)) for i in range(n)]
        buf = io.BytesIO()
        frames[0].save(buf, format="GIF", save_all=True, append_images=frames[1:],
                       duration=100, loop=0)
        return buf.getvalue()

    def test_decode_gif(self):
        frames, fps = decode_video_frames(self.make_gif(6))
        assert len(frames) == 6
        assert frames[0].shape == (64, 64, 3)
        assert 5 <= fps <= 20  # 100ms/frame -> 10fps

    def test_npy_stack(self):
        arr = np.stack([img(32, 48, seed=i) for i in range(4)])
        buf = io.BytesIO()
        np.save(buf, arr)
        frames, fps = decode_video_frames(buf.getvalue())
        assert len(frames) == 4 and frames[0].shape == (32, 48, 3)

    def test_uniform_sampling(self):
        frames = [img(16, 16, seed=i) for i in range(30)]
        sampled, eff = sample_frames(frames, 30.0, VideoSampleConfig(num_frames=6))
        assert len(sampled) == 6
        sampled, eff = sample_frames(frames, 30.0, VideoSampleConfig(sample_fps=2.0))
        assert len(sampled) == 2  # 1s of video at 2 fps
        sampled, _ = sample_frames(frames, 30.0, VideoSampleConfig(num_frames=100))
        assert len(sampled) == 30  # capped at available

    def test_end_to_end_qwen_video(self):
        proc = QwenVLProcessor(ProcessorConfig("qwen2_vl", "smart_resize", factor=28), use_gpu=False)
        out = process_video(self.make_gif(8, size=112), proc, VideoSampleConfig(num_frames=4))
        t, gh, gw = out["grid_thw"]
        assert t == 2  # 4 frames / temporal 2
        assert out["pixel_values"].shape[0] == t * gh * gw

    def test_end_to_end_fixed_processor_video(self):
        proc = ImageProcessor(ProcessorConfig("clip", "fixed", size=224), use_gpu=False)
        out = process_video(self.make_gif(6), proc, VideoSampleConfig(num_frames=3))
        assert out["pixel_values"].shape == (3, 3, 224, 224)
        assert out["num_frames"] == 3


@pytest.mark.gpu
class TestProcessorsGpu:
    def test_llava_square_gpu_matches_pil(self):
        """Per-model processors on the gfx950 kernel path must equal the PIL
        fallback (the kernel is bit-identical PIL BICUBIC)."""
        cfg = ProcessorConfig("llava", "square", size=336)
        gpu = LlavaProcessor(cfg, use_gpu=True)
        assert gpu.on_gpu
        cpu = LlavaProcessor(cfg, use_gpu=False)
        image = img(400, 520, seed=5)
        a = gpu.process(image)["pixel_values"]
        b = cpu.process(image)["pixel_values"]
        assert np.abs(a - b).max() < 1e-5

    def test_qwen_patches_gpu_matches_pil(self):
        cfg = ProcessorConfig("qwen2_vl", "smart_resize", factor=28)
        gpu = QwenVLProcessor(cfg, use_gpu=True)
        assert gpu.on_gpu
        cpu = QwenVLProcessor(cfg, use_gpu=False)
        image = img(280, 420, seed=6)
        a = gpu.process(image)
        b = cpu.process(image)
        assert a["grid_thw"] == b["grid_thw"]
        assert np.abs(a["pixel_values"] - b["pixel_values"]).max() < 1e-5
