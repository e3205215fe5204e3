"""3-rank gloo EPD-over-the-plane helper (launched by test_epd_plane.py):

  rank 0 (gateway)  -- pixels --> rank 1 (encode) -- embeddings --> rank 2 (decode)

The decode rank also computes the single-engine reference (same seeds,
direct mm_embed submit) and asserts the disaggregated stream matches
exactly.  Prints EPD_OK on success.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from smg_amd.comm.plane import EMB_RECV, EMB_SEND, PIX_RECV, PIX_SEND, execute_transfers
from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig
from smg_amd.multimodal.encoder import EncodeWorker, PixelSource, ToyVisionEncoder

RID = 77
H = W = 48
IMAGE = 64
PATCH = 16


def main():
    rank = int(os.environ["RANK"])
    dist.init_process_group("gloo")
    cfg = TorchEngineConfig.tiny()
    enc = ToyVisionEncoder(cfg.d_model, image_size=IMAGE, patch=PATCH, seed=5)
    E = enc.n_embed
    g = torch.Generator().manual_seed(9)
    pixels = (torch.rand(3, H, W, generator=g) * 255).to(torch.uint8)
    prompt = list(range(20))

    if rank == 0:
        src = PixelSource()
        src.put(RID, pixels)
        execute_transfers(src, [(RID, 1, PIX_SEND, 3, H, W)])
        dist.barrier()
        dist.barrier()
        ok = torch.zeros(1)
        dist.broadcast(ok, src=2)
        print("EPD_OK" if int(ok[0]) == 1 else "EPD_FAIL", flush=True)
    elif rank == 1:
        worker = EncodeWorker(enc)
        execute_transfers(worker, [(RID, 0, PIX_RECV, 3, H, W)])
        dist.barrier()  # pixels landed + encoded
        execute_transfers(worker, [(RID, 2, EMB_SEND, E, 0, 0)])
        dist.barrier()
        ok = torch.zeros(1)
        dist.broadcast(ok, src=2)
    else:
        eng = TorchEngine(cfg, device="cpu")
        dist.barrier()
        execute_transfers(eng, [(RID, 1, EMB_RECV, E, 0, 0)])
        dist.barrier()  # embeddings landed
        eng.submit(prompt, 6, rid=RID)  # pairs with the pending embedding
        while not eng.finished(RID):
            eng.step()
        got = eng.collect(RID)
        # single-engine reference: same weights, direct mm_embed
        ref_eng = TorchEngine(cfg, device="cpu")
        emb = enc.encode(pixels)
        r = ref_eng.submit(prompt, 6, rid="ref", mm_embed=emb)
        while not ref_eng.finished(r):
            ref_eng.step()
        ref = ref_eng.collect(r)
        ok = torch.tensor([1.0 if got == ref and len(got) == 6 else 0.0])
        dist.broadcast(ok, src=2)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
