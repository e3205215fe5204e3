"""3-rank gloo EPD SERVING helper (launched by test_epd_plane.py):
TickGateway on rank 0 (decode pool = ranks 0/2), EncodeWorker on rank 1,
TorchEngine worker on rank 2.  A pixel-carrying request and plain text
requests serve through the same tick loop; the multimodal stream must
match the single-engine mm_embed reference.  Prints EPD_GW_OK.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from smg_amd.comm.plane import GatewayPlane, PlaneConfig, WorkerPlane
from smg_amd.config import PolicyConfig
from smg_amd.engine.torch_engine import TorchEngine, TorchEngineConfig
from smg_amd.multimodal.encoder import EncodeWorker, ToyVisionEncoder
from smg_amd.policies import create_policy
from smg_amd.routers.rccl_router import TickGateway, epd_rank_roles, run_worker_loop
from smg_amd.workers.worker import Worker


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo")
    cfg = TorchEngineConfig.tiny()
    pcfg = PlaneConfig(max_prompt=64, device="cpu")
    g = torch.Generator().manual_seed(21)
    pixels = (torch.rand(3, 40, 40, generator=g) * 255).to(torch.uint8)
    prompt_mm = list(range(18))
    if rank == 0:
        eng = TorchEngine(cfg, device="cpu")
        plane = GatewayPlane(pcfg, [1, 2])
        workers = [Worker(f"rccl://rank-{r}", rccl_rank=r) for r in range(world)]
        policy = create_policy(PolicyConfig(name="round_robin"))
        seen = {}
        gw = TickGateway(workers, policy, plane=plane, local_engine=eng,
                         pd_roles=epd_rank_roles(world),
                         on_event=lambda rid, tok, fl: seen.setdefault(rid, []).append(tok))
        assert gw.pipeline, "EPD (no prefill role) must keep the pipelined tick"
        gw.submit(list(range(10)), 4, rid=1)
        gw.submit(list(range(12)), 4, rid=2)
        gw.submit(prompt_mm, 5, rid=3, pixels=pixels)
        for _ in range(200):
            gw.tick()
            if gw.completed_total >= 3:
                break
        gw._drain_pipeline()
        gw.stop_workers()
        # reference: same weights + same toy encoder, single engine
        ref_eng = TorchEngine(cfg, device="cpu")
        enc = ToyVisionEncoder(cfg.d_model, image_size=64, patch=16)
        r = ref_eng.submit(prompt_mm, 5, mm_embed=enc.encode(pixels))
        while not ref_eng.finished(r):
            ref_eng.step()
        ref = ref_eng.collect(r)
        ok = gw.completed_total >= 3 and seen.get(3) == ref
        print("EPD_GW_OK" if ok else f"EPD_GW_FAIL got={seen.get(3)} ref={ref} "
              f"done={gw.completed_total}", flush=True)
    elif rank == 1:
        worker = EncodeWorker(ToyVisionEncoder(cfg.d_model, image_size=64, patch=16))
        plane = WorkerPlane(pcfg)
        run_worker_loop(worker, plane, role="encode")
    else:
        eng = TorchEngine(cfg, device="cpu")
        plane = WorkerPlane(pcfg)
        run_worker_loop(eng, plane, role="regular")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
