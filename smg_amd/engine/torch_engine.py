"""GPU worker engine: a continuous-batching transformer decode engine on
PyTorch-ROCm (bf16, random-init weights).

Role: the MI355X stand-in for the real inference engines the gateway fronts
(vLLM/SGLang in the reference).  The bench runs one TorchEngine
per GPU rank behind the gateway's RCCL data plane, so the measured req/s is a
full gateway+engine system on real hardware.  Tests use the CPU mock engine
(mock/engine.py); this class is the GPU path and fails loudly without one.

Design: slot-based continuous batching with a preallocated KV cache
[slots, layers, 2, heads, max_seq, head_dim] in bf16, chunked prefill
(one chunk per step alongside the batched single-token decode), greedy
sampling.  The decode forward is shape-stable per active-slot-count, so HIP
graphs can capture it (enable with graphs=True).
"""
from __future__ import annotations

import math
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

import numpy as np
import torch
import torch.nn.functional as F


@dataclass
class TorchEngineConfig:
    vocab_size: int = 32768
    n_layers: int = 16
    d_model: int = 2048
    n_heads: int = 16
    # GQA (Llama3/Qwen-style): n_kv_heads < n_heads shares each KV head
    # across a group of query heads — the decode-attention kernel streams
    # each K/V row once per GROUP, cutting its HBM bytes by n_heads/n_kv_heads.
    # None = MHA (n_kv_heads == n_heads).
    n_kv_heads: Optional[int] = None
    ffn_mult: float = 2.6875  # 5504/2048, llama-style
    max_slots: int = 64
    max_seq: int = 2048
    prefill_chunk: int = 2048
    # batched-prefill shaping: how many equal-remaining requests share one
    # forward, and the total-token budget multiple that caps the group
    prefill_group: int = 32
    prefill_budget_mult: float = 2.0
    dtype: str = "bfloat16"
    seed: int = 1234
    # prefix KV cache (vLLM/SGLang-style): shared prompt prefixes skip
    # recompute — the engine-side half of cache-aware routing
    prefix_cache_slots: int = 16
    prefix_cache_page: int = 64
    prefix_cache_max: int = 1024  # max cached prefix tokens
    # opt-in fp8 (OCP e4m3) KV cache: halves the HBM bytes decode attention
    # streams; compute stays bf16 (only the cache storage is quantized).
    # Requires the gfx950 fused kernels.  Default OFF — the bench headline
    # is measured with the bf16 cache.
    kv_fp8: bool = False

    @classmethod
    def tiny(cls) -> "TorchEngineConfig":
        return cls(vocab_size=2048, n_layers=2, d_model=256, n_heads=4, max_slots=8, max_seq=256, prefill_chunk=256)

    @classmethod
    def bench_1b(cls) -> "TorchEngineConfig":
        """The flagship bench model: ~1.1B params bf16 (≈2.3 GB weights)."""
        return cls()

    @classmethod
    def bench_1b_gqa(cls) -> "TorchEngineConfig":
        """GQA flagship variant: 16 q heads / 4 kv heads (group 4, the
        Llama3-8B ratio), with d_ffn widened 5504->6528 so total parameter
        count matches bench_1b (the q/k/v projection shrinks by 2*(D-KD)*D
        per layer; 3*D*Δf recovers it)."""
        return cls(n_kv_heads=4, ffn_mult=6528 / 2048)

    @property
    def head_dim(self) -> int:
        return self.d_model // self.n_heads

    @property
    def kv_heads(self) -> int:
        return self.n_kv_heads if self.n_kv_heads else self.n_heads

    @property
    def kv_dim(self) -> int:
        return self.kv_heads * self.head_dim

    @property
    def gqa_group(self) -> int:
        return self.n_heads // self.kv_heads

    @property
    def d_ffn(self) -> int:
        return int(self.d_model * self.ffn_mult) // 64 * 64


class _Layer:
    def __init__(self, cfg: TorchEngineConfig, device, dtype, gen):
        d, f = cfg.d_model, cfg.d_ffn
        k = 1.0 / math.sqrt(d)

        def w(*shape):
            return (torch.rand(*shape, generator=gen, device=device, dtype=torch.float32) * 2 - 1).to(dtype) * k

        self.wqkv = w(d, d + 2 * cfg.kv_dim)  # q(D) | k(KD) | v(KD)
        self.wo = w(d, d)
        self.w13 = w(d, 2 * f)  # gate+up fused: one GEMM instead of two
        self.w2 = w(f, d)
        self.ln1 = torch.ones(d, device=device, dtype=dtype)
        self.ln2 = torch.ones(d, device=device, dtype=dtype)


class _StepHandle:
    """Deferred-read result of TorchEngine.step_launch: a CUDA event gating a
    pinned slab of sampled token values, plus the (container, key, slab_idx)
    patch list and the event rows awaiting those values."""

    __slots__ = ("ev", "slab", "n", "fills", "events", "produced", "dec_batches")

    def __init__(self):
        self.ev = None
        self.slab: Optional[torch.Tensor] = None
        self.n = 0
        self.fills: List[tuple] = []
        self.events: List[list] = []
        self.produced = 0
        # decode bursts as arrays: (reqs, act_np, done_mask, slab_base) — the
        # per-token tuple building happens in step_finish, overlapped with the
        # NEXT tick's GPU work in the pipelined gateway
        self.dec_batches: List[tuple] = []


def _rms(x, weight):
    # fused on ROCm (single kernel); computes in fp32 internally
    return F.rms_norm(x, (x.shape[-1],), weight=weight, eps=1e-5)


class _Request:
    __slots__ = ("rid", "tokens", "max_new", "slot", "prefilled", "generated", "done",
                 "prefill_only", "pc_keys", "mm_embed", "await_embed")

    def __init__(self, rid, tokens, max_new, prefill_only=False):
        self.rid = rid
        self.tokens = tokens
        self.max_new = max_new
        self.slot = -1
        self.prefilled = 0
        self.generated: List[int] = []
        self.done = False
        self.prefill_only = prefill_only
        self.pc_keys = None  # prefix-cache chain keys, precomputed at submit
        self.mm_embed = None  # [E, d_model] EPD vision embeddings (seq prefix)
        self.await_embed = False  # EPD: hold prefill until accept_embed pairs


class TorchEngine:
    _flash_gqa = False
    _sdpa_gqa = False
    _attn_split = 1
    _hip_attn_split = None
    _use_mfma = False

    def __init__(self, cfg: Optional[TorchEngineConfig] = None, device: str = "cuda:0", graphs: bool = False):
        if device.startswith("cuda") and not torch.cuda.is_available():
            raise RuntimeError("TorchEngine requires a GPU (use mock/engine.py on CPU)")
        self.cfg = cfg or TorchEngineConfig()
        self.device = torch.device(device)
        # cpu device is for logic tests only; compute dtype stays bf16 on GPU
        self.dtype = getattr(torch, self.cfg.dtype) if device.startswith("cuda") else torch.float32
        gen = torch.Generator(device=self.device).manual_seed(self.cfg.seed)
        c = self.cfg
        with torch.device(self.device):
            self.embed = (torch.rand(c.vocab_size, c.d_model, generator=gen, device=self.device,
                                     dtype=torch.float32) * 2 - 1).to(self.dtype) / math.sqrt(c.d_model)
            self.layers = [_Layer(c, self.device, self.dtype, gen) for _ in range(c.n_layers)]
            self.ln_f = torch.ones(c.d_model, device=self.device, dtype=self.dtype)
            # KV cache: [layers, 2, slots, heads, max_seq, head_dim]
            self.kv_dtype = (
                torch.float8_e4m3fn if (c.kv_fp8 and self.device.type == "cuda") else self.dtype
            )
            self.kv = torch.zeros(
                c.n_layers, 2, c.max_slots, c.kv_heads, c.max_seq, c.head_dim,
                device=self.device, dtype=self.kv_dtype,
            )
            # rotary tables (complex form: one complex mul applies the rotation)
            inv = 1.0 / (10000.0 ** (torch.arange(0, c.head_dim, 2, device=self.device).float() / c.head_dim))
            t = torch.arange(c.max_seq, device=self.device).float()
            freqs = torch.outer(t, inv)
            self.freqs_cis = torch.polar(torch.ones_like(freqs), freqs)  # complex64 [T, D/2]
        self.seq_len = torch.zeros(c.max_slots, dtype=torch.long, device=self.device)
        self._seq_len_host = np.zeros(c.max_slots, dtype=np.int64)
        self._gen_cnt = np.zeros(c.max_slots, dtype=np.int64)  # len(generated) per slot
        self._maxnew_arr = np.zeros(c.max_slots, dtype=np.int64)
        self._last_tok = torch.zeros(c.max_slots, dtype=torch.long, device=self.device)
        self._arange_slots = torch.arange(c.max_slots, device=self.device)
        self._free_slots = list(range(c.max_slots - 1, -1, -1))
        self.waiting: List[_Request] = []
        self.running: Dict[int, _Request] = {}  # slot -> request
        self._requests: Dict[str, _Request] = {}
        self._rid_counter = 0
        self._parked: Dict[str, tuple] = {}  # rid -> (slot, req, first_tok) awaiting export_kv
        self._pending_embeds: Dict = {}  # rid -> [E, d_model] EPD embeddings (accept_embed)
        self.total_generated = 0
        self._step_events: List[tuple] = []  # (rid, token, done) since last drain
        self._slabs: List[Optional[torch.Tensor]] = [None] * 4  # pinned D2H slabs
        self._slab_i = 0
        self._pf_stages: List[Optional[torch.Tensor]] = [None] * 3  # pinned H2D staging
        self._pf_stage_i = 0
        self._pfg_cache: Dict[tuple, tuple] = {}  # (B,L,start0,fresh) -> captured graph
        self._pfg_bad: set = set()  # shapes whose capture failed (stay eager)
        self._pfg_seen: Dict[tuple, int] = {}  # shape repeat counts (capture gate)
        self.pfg_hits = 0
        self.pfg_eager = 0
        # host-time breakdown of step_launch (pf_book/dec_book include pf/decode)
        self.launch_t = {"restore": 0.0, "pf": 0.0, "pf_book": 0.0,
                         "decode": 0.0, "dec_book": 0.0}
        # small-transfer staging: torch.tensor(list, device=cuda) is a
        # BLOCKING H2D — the host waits for the copy, which is stream-ordered
        # behind the whole queued tick, so each one drains the pipeline
        self._h2d_bufs: List[Optional[tuple]] = [None] * 12
        self._h2d_i = 0
        self.graphs = graphs
        self._graph_cache: Dict[int, tuple] = {}
        # prefix KV cache arena: same [L, 2, slot, H, T, D] layout as self.kv
        # so a hit is one strided device copy
        pc = self.cfg
        if pc.prefix_cache_slots > 0:
            self._pc_arena = torch.zeros(
                pc.n_layers, 2, pc.prefix_cache_slots, pc.kv_heads,
                min(pc.prefix_cache_max, pc.max_seq), pc.head_dim,
                device=self.device, dtype=self.kv.dtype,
            )
        else:
            self._pc_arena = None
        self._pc_keys: Dict[int, tuple] = {}  # chain-key -> (slot, plen)
        self._pc_lru: List[int] = list(range(pc.prefix_cache_slots))  # front = LRU victim
        self._pc_slot_keys: Dict[int, List[int]] = {}  # slot -> its chain keys
        self._pc_slot_tokens: Dict[int, List[int]] = {}  # slot -> cached prefix (hit verification)
        self.prefix_cache_hits = 0
        self.prefix_cache_miss = 0
        # fused gfx950 decode-attention kernel (csrc/attn_decode.hip): reads
        # only kv[slot][:pos+1] per slot instead of sdpa's rectangular window
        self._hip_attn = None
        self._hip_fused = None
        self._hip_silu_mul = None
        if self.device.type == "cuda" and self.dtype == torch.bfloat16 and c.head_dim <= 128:
            import os as _os

            try:
                from .. import _core

                if hasattr(_core, "attn_decode"):
                    # v9 (attn_decode2): v_dot2c_f32_bf16 K phase, +14% at bench
                    # shapes; its launcher routes fp8 back to the v7 kernel
                    # (fp8 is convert-bound).  SMG_ATTN_V1=1 pins v7.
                    self._hip_attn = _core.attn_decode
                    if hasattr(_core, "attn_decode2") and not _os.environ.get("SMG_ATTN_V1"):
                        self._hip_attn = _core.attn_decode2
                    self._pos_i32 = torch.zeros(c.max_slots, dtype=torch.int32, device=self.device)
                    self._attn_out = torch.zeros(
                        c.max_slots, c.n_heads, c.head_dim, device=self.device, dtype=self.dtype
                    )
                    # v8 T-split (flash-decoding): split each KV window over
                    # n_split waves + LSE merge when slots*kv_heads alone
                    # truly under-fills the chip.  Measured on MI355X: at
                    # >=2048 waves the plain kernel wins (splitting cost
                    # 670->565 req/s at 520 slots x 4 kv heads); below that
                    # (small fleets / low concurrency) the extra parallelism
                    # pays.  SMG_ATTN_SPLIT overrides.
                    waves = c.max_slots * c.kv_heads
                    split = 1
                    while split < 8 and waves * split < 2048:
                        split *= 2
                    split = int(_os.environ.get("SMG_ATTN_SPLIT", split) or split)
                    self._attn_split = max(1, min(16, split))
                    if self._attn_split > 1 and hasattr(_core, "attn_decode_split"):
                        self._hip_attn_split = _core.attn_decode_split
                        G = c.gqa_group
                        self._attn_part = torch.zeros(
                            c.max_slots, c.kv_heads, self._attn_split, G, c.head_dim,
                            device=self.device, dtype=torch.float32)
                        self._attn_ml = torch.zeros(
                            c.max_slots, c.kv_heads, self._attn_split, G, 2,
                            device=self.device, dtype=torch.float32)
                    else:
                        self._attn_split = 1
                        self._hip_attn_split = None
                # fused rms_norm + MFMA GEMM (csrc/rms_gemm.hip): the decode
                # QKV/W13 projections as hand-written 16x16x32 bf16 MFMA
                # tiles with the rms fold (g into the weight, 1/rms as an
                # output row scale).  Measured on MI355X (scripts/mfma_bench):
                # 154-451 TF vs hipBLASLt's 259-702 on the decode shapes —
                # the library GEMM still wins, so this path is OPT-IN
                # (SMG_MFMA=1); kept in-tree as the measured MFMA baseline
                # the next tiling iteration starts from.
                self._use_mfma = False
                if (hasattr(_core, "rms_gemm") and c.d_model % 512 == 0
                        and (c.d_model + 2 * c.kv_dim) % 128 == 0
                        and (2 * c.d_ffn) % 128 == 0
                        and _os.environ.get("SMG_MFMA", "0") == "1"):
                    self._hip_rms_gemm = _core.rms_gemm
                    self._hip_row_invrms = _core.row_invrms
                    self._use_mfma = True
                    self._invrms = torch.zeros(c.max_slots, dtype=torch.float32, device=self.device)
                    self._qkv_buf = torch.zeros(
                        c.max_slots, c.d_model + 2 * c.kv_dim, device=self.device, dtype=self.dtype)
                    self._gu_buf = torch.zeros(
                        c.max_slots, 2 * c.d_ffn, device=self.device, dtype=self.dtype)
                    for layer in self.layers:
                        layer.wqkv_tg = (
                            (layer.wqkv.float() * layer.ln1.float().unsqueeze(1))
                            .t().contiguous().to(self.dtype))
                        layer.w13_tg = (
                            (layer.w13.float() * layer.ln2.float().unsqueeze(1))
                            .t().contiguous().to(self.dtype))
                # fused rope+KV-store+q-pack and silu*mul (csrc/fused_decode.hip):
                # removes ~10 elementwise launches per layer from the decode loop
                if hasattr(_core, "rope_kv_store"):
                    self._hip_fused = _core.rope_kv_store
                    self._hip_silu_mul = _core.silu_mul
                    self._hip_rope_prefill = getattr(_core, "rope_prefill", None)
                    self._hip_lse_merge = getattr(_core, "lse_merge", None)
                    self._q_buf = torch.zeros(
                        c.max_slots, c.d_model, device=self.device, dtype=self.dtype
                    )
                    self._smul_buf = torch.zeros(
                        c.max_slots, c.d_ffn, device=self.device, dtype=self.dtype
                    )
            except ImportError as exc:
                # on a GPU box the gfx950 kernels ARE the engine; a silent
                # torch-eager fallback would hide a broken build.  CPU logic
                # tests never reach here (device=cpu); set SMG_ALLOW_EAGER=1
                # to debug eager on a GPU explicitly.
                if not _os.environ.get("SMG_ALLOW_EAGER"):
                    raise RuntimeError(
                        "smg_amd._core (gfx950 kernels) failed to import on a CUDA/HIP "
                        "device — run `python -m smg_amd.csrc.build` (or set "
                        "SMG_ALLOW_EAGER=1 to force the torch-eager fallback)"
                    ) from exc
        self._flash_lse = self._probe_flash_lse() if self.device.type == "cuda" else None
        if c.kv_fp8 and (self._hip_fused is None or self._hip_attn is None or
                         getattr(self, "_hip_rope_prefill", None) is None):
            raise RuntimeError(
                "kv_fp8=True requires the gfx950 fused kernels (smg_amd._core with"
                " rope_kv_store/rope_prefill/attn_decode) on a CUDA/HIP device"
            )

    # ---- API -------------------------------------------------------------
    def submit(self, tokens: List[int], max_new_tokens: int, rid: Optional[str] = None,
               prefill_only: bool = False, mm_embed=None, await_embed: bool = False) -> str:
        """`prefill_only` is the PD prefill leg: the request prefills, samples
        its first token (emitted with the PREFILLED flag, value 4) and PARKS —
        the slot stays allocated until export_kv() hands its KV off to the
        decode rank (PD over the rccl plane; reference PD delegates this
        transfer to engine-side Mooncake/NIXL, here the engine is ours and
        the handoff is an xGMI p2p send).

        `mm_embed` is the EPD embedding leg: an [E, d_model] tensor of
        vision-tower embeddings that occupy sequence positions 0..E-1 in
        front of the prompt tokens (the reference ships these engine-side
        via Mooncake after the encode fleet runs; here they arrive over the
        xGMI plane — comm/plane.py EMB transfer directions)."""
        if rid is None:
            self._rid_counter += 1
            rid = f"req-{self._rid_counter}"
        c = self.cfg
        # prompts stay numpy int64 end-to-end: vectorized clamp here, zero-copy
        # page-slices in the batched prefill, vectorized chain-key hashing in
        # the prefix cache (the per-token Python list work was milliseconds per
        # serving tick at 32 admissions x ~1.7K tokens)
        toks = np.asarray(tokens, dtype=np.int64) % c.vocab_size
        n_emb = 0
        if mm_embed is not None:
            mm_embed = torch.as_tensor(mm_embed).to(self.device, self.dtype)
            if mm_embed.dim() != 2 or mm_embed.shape[1] != c.d_model:
                raise ValueError(f"mm_embed must be [E, {c.d_model}]")
            n_emb = mm_embed.shape[0]
        if len(toks) + n_emb > c.max_seq - 2:
            keep = max(1, c.max_seq - 2 - n_emb)
            toks = toks[-keep:]  # keep the prompt tail
        max_new_tokens = max(1, min(max_new_tokens, c.max_seq - 1 - len(toks) - n_emb))
        req = _Request(rid, toks, max_new_tokens, prefill_only=prefill_only)
        if mm_embed is None:
            # EPD decode leg: embeddings that arrived over the plane
            # (accept_embed) pair with the request by rid
            mm_embed = self._pending_embeds.pop(rid, None)
        req.mm_embed = mm_embed
        # hold prefill until the embedding transfer lands (pipelined gateway:
        # the local EMB_RECV may flush after this submit)
        req.await_embed = bool(await_embed) and mm_embed is None
        if mm_embed is None and not req.await_embed and self._pc_arena is not None:
            # hash here: in the pipelined gateway, submit() runs in the
            # routing phase (overlapped with the GPU tick) while the prefix
            # lookup runs in the launch critical path
            page = c.prefix_cache_page
            pmax = min(((len(toks) - 1) // page) * page, c.prefix_cache_max, c.max_seq - 1)
            if pmax >= page:
                from ..kvindex.chain_keys import chain_keys

                req.pc_keys = chain_keys(toks[:pmax], page)
        self._requests[rid] = req
        self.waiting.append(req)
        return rid

    # ---- PD disaggregation: KV handoff -----------------------------------
    PREFILLED = 4  # event flag: prefill leg done, KV parked for export
    PLEN_INFO = 8  # event flag: token field carries the prefilled length

    def export_kv(self, rid: str):
        """Parked prefill -> (kv tensor [L, 2, KVH, plen, hd], plen,
        first_token); frees the slot."""
        entry = self._parked.pop(rid, None)
        if entry is None:
            raise KeyError(f"no parked prefill for {rid}")
        slot, req, first_tok = entry
        plen = req.prefilled
        t = self.kv[:, :, slot, :, :plen].clone()
        self._free_slots.append(slot)
        self._seq_len_host[slot] = 0
        self.seq_len[slot] = 0
        self._requests.pop(rid, None)
        return t, plen, first_tok

    def import_kv(self, rid, kv_tensor, plen: int, first_tok: int, max_new: int) -> bool:
        """Decode-leg intake: place the transferred KV into a free slot and
        continue decoding from the prefill-sampled first token."""
        if not self._free_slots:
            return False
        c = self.cfg
        slot = self._free_slots.pop()
        self.kv[:, :, slot, :, :plen] = kv_tensor.to(self.kv.dtype)
        self._seq_len_host[slot] = plen
        self.seq_len[slot] = plen
        self._last_tok[slot] = int(first_tok)
        req = _Request(rid, [0] * plen, max_new)  # token VALUES unused post-prefill
        req.prefilled = plen
        req.slot = slot
        req.generated = [int(first_tok)]
        self._gen_cnt[slot] = 1
        self._maxnew_arr[slot] = max_new
        self._requests[rid] = req
        self.running[slot] = req
        return True

    def kv_transfer_shape(self, plen: int):
        c = self.cfg
        return (c.n_layers, 2, c.kv_heads, plen, c.head_dim)

    # ---- EPD: vision embeddings over the plane ----------------------------
    def accept_embed(self, rid, emb: torch.Tensor) -> None:
        """EMB_RECV landing: [E, d_model] embeddings stash until the matching
        request submit()s (plane transfers execute before submits each tick)."""
        if len(self._pending_embeds) > 4096:  # orphaned rids (aborted requests)
            self._pending_embeds.clear()
        self._pending_embeds[rid] = emb.to(self.device, self.dtype)
        # a request that arrived first picks the embedding up here instead
        req = self._requests.get(rid)
        if req is not None and req.prefilled == 0 and req.mm_embed is None:
            req.mm_embed = self._pending_embeds.pop(rid)
            req.await_embed = False

    def finished(self, rid: str) -> bool:
        r = self._requests.get(rid)
        return r is None or r.done

    def collect(self, rid: str) -> List[int]:
        r = self._requests.pop(rid, None)
        return r.generated if r else []

    def n_active(self) -> int:
        return len(self.running) + len(self.waiting)

    def drain_events(self) -> List[tuple]:
        """(rid, token, done) tuples produced since the last drain — the
        event stream the RCCL plane ships back to the gateway each tick."""
        out, self._step_events = self._step_events, []
        return out

    def load_snapshot(self) -> Dict:
        c = self.cfg
        used = int(sum(self._seq_len_host[s] for s in self.running))
        return {
            "num_queue_tokens": sum(len(r.tokens) for r in self.waiting),
            "num_inflight_tokens": used,
            "num_running_reqs": len(self.running),
            "num_queue_reqs": len(self.waiting),
            "token_usage": used / (c.max_slots * c.max_seq),
            "gen_throughput": None,
        }

    # ---- engine step -----------------------------------------------------
    def step_launch(self, decode_burst: int = 1) -> "_StepHandle":
        """Launch one engine iteration WITHOUT reading results back.

        All admission/completion control flow here is host-deterministic —
        "done" is a token-COUNT condition and sampling is argmax payload — so
        the sampled token VALUES are only event payload.  They are copied
        device→pinned-slab asynchronously and read in step_finish().  The
        pipelined TickGateway resolves tick N-1's handle while tick N's
        kernels run, which keeps the GPU fed through the gateway's
        routing/event CPU phase (measured: 56% GPU-busy in the timed bench
        region before this split, with a ~5 ms CPU gap every tick)."""
        c = self.cfg
        lt = self.launch_t
        _t = time.perf_counter
        t0 = _t()
        # admission
        while self.waiting and self._free_slots:
            req = self.waiting.pop(0)
            req.slot = self._free_slots.pop()
            self.running[req.slot] = req
            self._seq_len_host[req.slot] = 0
            self._gen_cnt[req.slot] = len(req.generated)
            self._maxnew_arr[req.slot] = req.max_new

        h = _StepHandle()
        samples: List[tuple] = []  # (device tensor, slab base offset)
        off = 0
        # prefill: prefix-cache restore for fresh slots, then BATCHED chunked
        # prefill — equal-remaining requests share one forward (the common
        # case: cache-hit suffixes of identical length).  The arena copies
        # are grouped by prefix length into ONE indexed device copy per
        # length: per-slot 6-D slice assigns were ~150 us of host dispatch
        # each, x32 hits/tick in the bench.
        restore: Dict[int, list] = {}
        for slot, req in self.running.items():
            if (req.prefilled == 0 and self._pc_arena is not None
                    and req.mm_embed is None and not req.await_embed):
                hit = self._prefix_lookup(req.tokens, keys=req.pc_keys)
                if hit is not None:
                    pslot, plen = hit
                    restore.setdefault(plen, []).append((slot, pslot, int(req.tokens[plen - 1])))
                    self._seq_len_host[slot] = plen
                    req.prefilled = plen
                    self.prefix_cache_hits += 1
                else:
                    self.prefix_cache_miss += 1
        for plen, group in restore.items():
            if len(group) == 1:
                slot, pslot, last = group[0]
                self.kv[:, :, slot, :, :plen] = self._pc_arena[:, :, pslot, :, :plen]
                self.seq_len[slot] = plen
                self._last_tok[slot] = last
            else:
                meta = self._h2d_i64(np.asarray(group, dtype=np.int64).ravel()).view(-1, 3)
                slots_t, pslots_t, lasts_t = meta[:, 0], meta[:, 1], meta[:, 2]
                self.kv[:, :, slots_t, :, :plen] = self._pc_arena[:, :, pslots_t, :, :plen]
                self.seq_len[slots_t] = plen
                self._last_tok[slots_t] = lasts_t
        t1 = _t()
        lt["restore"] += t1 - t0
        # EPD embedding-prefixed requests prefill one at a time (exception
        # path: [E, D] vision embeddings occupy positions 0..E-1, so they
        # can't share the token-batched forward)
        emb_pending = [(s, r) for s, r in self.running.items()
                       if r.prefilled == 0 and r.mm_embed is not None]
        for slot, req in emb_pending[:4]:
            nxt_e = self._prefill_embedded(slot, req)
            samples.append((nxt_e, off))
            req.prefilled = len(req.tokens)
            req.generated.append(-1)
            h.fills.append((req.generated, 0, off))
            self._gen_cnt[slot] = 1
            h.produced += 1
            self.total_generated += 1
            if len(req.generated) >= req.max_new or self._seq_len_host[slot] >= c.max_seq - 2:
                req.done = True
                del self.running[slot]
                self._free_slots.append(slot)
                self._seq_len_host[slot] = 0
                self.seq_len[slot] = 0
            row = [req.rid, None, 1 if req.done else 0]
            h.fills.append((row, 1, off))
            h.events.append(row)
            off += 1
        pending = [(s, r) for s, r in self.running.items()
                   if r.prefilled < len(r.tokens) and r.mm_embed is None
                   and not r.await_embed]
        if pending:
            pending.sort(key=lambda sr: len(sr[1].tokens) - sr[1].prefilled)
            min_rem = len(pending[0][1].tokens) - pending[0][1].prefilled
            L = min(min_rem, c.prefill_chunk)
            group = [sr for sr in pending if len(sr[1].tokens) - sr[1].prefilled >= L][: c.prefill_group]
            # keep the total under the budget multiple
            while len(group) > 1 and len(group) * L > c.prefill_budget_mult * c.prefill_chunk:
                group.pop()
            items = []
            for slot, req in group:
                items.append((slot, req.prefilled, req.tokens[req.prefilled: req.prefilled + L]))
            t2 = _t()
            nxt_pf = self._prefill_batch(items)  # device [len(group)]
            lt["pf"] += _t() - t2
            pf_off = off
            samples.append((nxt_pf, pf_off))
            off += len(items)
            finished_pf = []
            for i, (slot, req) in enumerate(group):
                req.prefilled += L
                if req.prefilled >= len(req.tokens):
                    self._prefix_store(slot, req.tokens)
                    if req.prefill_only:
                        # PD prefill leg: announce the prefilled length, then
                        # the sampled first token with the PREFILLED flag, and
                        # PARK the slot for export_kv
                        del self.running[slot]
                        parked = [slot, req, None]
                        self._parked[req.rid] = parked
                        h.fills.append((parked, 2, pf_off + i))
                        h.events.append([req.rid, req.prefilled, self.PLEN_INFO])
                        row = [req.rid, None, self.PREFILLED]
                        h.fills.append((row, 1, pf_off + i))
                        h.events.append(row)
                        continue
                    # the final prefill chunk's logits sample the first
                    # generated token — decode then starts from the SAMPLED
                    # token at position n, instead of re-feeding the last
                    # prompt token (which wasted a KV slot and conditioned
                    # the first token on [.., t_{n-1}, t_{n-1}])
                    req.generated.append(-1)  # value patched in step_finish
                    h.fills.append((req.generated, len(req.generated) - 1, pf_off + i))
                    self._gen_cnt[slot] = len(req.generated)
                    h.produced += 1
                    self.total_generated += 1
                    if len(req.generated) >= req.max_new or self._seq_len_host[slot] >= c.max_seq - 2:
                        req.done = True
                        finished_pf.append(slot)
                    row = [req.rid, None, 1 if req.done else 0]
                    h.fills.append((row, 1, pf_off + i))
                    h.events.append(row)
            for s in finished_pf:
                del self.running[s]
                self._free_slots.append(s)
                self._seq_len_host[s] = 0
            if finished_pf:
                self.seq_len[self._h2d_i64(np.fromiter(finished_pf, dtype=np.int64))] = 0
            lt["pf_book"] += _t() - t2

        t3 = _t()
        for _ in range(max(1, decode_burst)):
            decode_slots = [s for s, r in self.running.items() if r.prefilled >= len(r.tokens)]
            if not decode_slots:
                break
            act_np = np.fromiter(decode_slots, dtype=np.int64, count=len(decode_slots))
            reqs = [self.running[s] for s in decode_slots]
            t4 = _t()
            nxt_full = self._decode_launch_np(act_np)  # device [max_slots]
            lt["decode"] += _t() - t4
            base = off
            samples.append((nxt_full, base))
            off += nxt_full.numel()
            # vectorized lifecycle: counts are authoritative in _gen_cnt, the
            # per-token event tuples are built in step_finish (overlapped
            # with the next tick's GPU work in the pipelined gateway)
            self._gen_cnt[act_np] += 1
            done_mask = ((self._gen_cnt[act_np] >= self._maxnew_arr[act_np])
                         | (self._seq_len_host[act_np] >= c.max_seq - 2))
            h.produced += len(decode_slots)
            self.total_generated += len(decode_slots)
            h.dec_batches.append((reqs, act_np, done_mask, base))
            if done_mask.any():
                finished = act_np[done_mask]
                for s in finished.tolist():
                    self.running[s].done = True
                    del self.running[s]
                    self._free_slots.append(s)
                    self._seq_len_host[s] = 0
                # reset the DEVICE positions too: the full-arena decode reads
                # seq_len[slot] as each slot's live window — a freed slot left
                # at its final position would keep streaming its whole stale
                # KV window every step (in steady state most of the arena is
                # free, so this is the difference between O(active) and
                # O(capacity) attention work)
                self.seq_len[self._h2d_i64(finished)] = 0

        lt["dec_book"] += _t() - t3
        if off:
            slab = self._get_slab(off)
            for t, base in samples:
                slab[base: base + t.numel()].copy_(t, non_blocking=True)
            h.slab, h.n = slab, off
            if self.device.type == "cuda":
                h.ev = torch.cuda.Event()
                h.ev.record()
        return h

    def step_finish(self, handle: "_StepHandle") -> int:
        """Resolve a step_launch handle: wait for its D2H copies, patch the
        token values into events/generated/parked, and publish the events.
        Waits only on the handle's own CUDA event — later launches keep
        running."""
        if handle is None:
            return 0
        if handle.ev is not None:
            handle.ev.synchronize()
        vals = None
        if handle.n:
            vals = handle.slab[: handle.n].numpy()  # pinned host view, no copy
        if handle.fills:
            for container, key, idx in handle.fills:
                container[key] = int(vals[idx])
        self._step_events.extend(tuple(r) for r in handle.events)
        ev = self._step_events
        for reqs, act_np, done_mask, base in handle.dec_batches:
            toks = vals[base + act_np].tolist()
            for r, t, d in zip(reqs, toks, done_mask.tolist()):
                r.generated.append(t)
                ev.append((r.rid, t, 1 if d else 0))
        return handle.produced

    def step(self, decode_burst: int = 1) -> int:
        """One engine iteration: admit + one prefill chunk + up to
        `decode_burst` decode tokens for every running slot.  Synchronous
        form of step_launch + step_finish; returns tokens produced."""
        return self.step_finish(self.step_launch(decode_burst))

    def _h2d_i64(self, arr) -> torch.Tensor:
        """Small int64 host->device transfer that does NOT block the host:
        pinned staging + non_blocking copy into a persistent device buffer
        (rotating 12-deep so in-flight DMAs are never overwritten within the
        pipeline's one-tick lag)."""
        arr = np.ascontiguousarray(arr, dtype=np.int64)
        n = arr.shape[0]
        if self.device.type != "cuda":
            return torch.from_numpy(arr)
        self._h2d_i = (self._h2d_i + 1) % len(self._h2d_bufs)
        entry = self._h2d_bufs[self._h2d_i]
        if entry is None or entry[0].numel() < n:
            cap = max(n, 3 * (self.cfg.max_slots + 8))
            entry = (torch.empty(cap, dtype=torch.long, pin_memory=True),
                     torch.empty(cap, dtype=torch.long, device=self.device))
            self._h2d_bufs[self._h2d_i] = entry
        pin, dev = entry
        pin[:n].numpy()[:] = arr
        dev[:n].copy_(pin[:n], non_blocking=True)
        return dev[:n]

    def _get_slab(self, n: int) -> torch.Tensor:
        """Rotating pinned host slabs for the deferred token reads (4 deep:
        the pipelined gateway holds at most one unresolved handle while the
        next launch writes a different slab)."""
        self._slab_i = (self._slab_i + 1) % len(self._slabs)
        s = self._slabs[self._slab_i]
        if s is None or s.numel() < n:
            s = torch.empty(max(n, 1024), dtype=torch.long,
                            pin_memory=self.device.type == "cuda")
            self._slabs[self._slab_i] = s
        return s

    # ---- forwards ----------------------------------------------------------
    def _kv_hist(self, li: int, which: int, slots):
        """Per-request KV history gather for the prefill attention; an fp8
        cache is dequantized to bf16 here (sdpa/flash consume bf16)."""
        sel = self.kv[li, which].index_select(0, slots)
        if sel.dtype == torch.float8_e4m3fn:
            sel = sel.to(self.dtype)
        return sel

    def _probe_flash_lse(self):
        """(q,k,v,is_causal) -> (out, logsumexp) via the flash kernel, or None
        when the aten op is unavailable.  The lse output lets the suffix-
        prefill path merge a no-mask history pass with a square causal chunk
        pass instead of taking the masked-sdpa math path.  Also probes
        native GQA support (mismatched q/kv head counts) — when the flash
        kernel handles it, prefill skips the repeat_interleave expansion
        (measured 11% of device time + 5% copyBuffer at group 4)."""

        def call(q, k, v, causal):
            r = torch.ops.aten._scaled_dot_product_flash_attention(q, k, v, 0.0, causal, False)
            return r[0], r[1]

        self._flash_gqa = False
        self._sdpa_gqa = False
        try:
            probe = torch.randn(1, 2, 8, 64, device=self.device, dtype=self.dtype)
            out, lse = call(probe, probe, probe, True)
            if out.shape != probe.shape or lse.shape[-1] != 8:
                return None
        except Exception:
            return None
        if self.cfg.gqa_group > 1:
            try:
                q4 = torch.randn(1, 4, 8, 64, device=self.device, dtype=self.dtype)
                kv = torch.randn(1, 2, 8, 64, device=self.device, dtype=self.dtype)
                out, lse = call(q4, kv, kv, False)
                ref = F.scaled_dot_product_attention(
                    q4.float(), kv.float().repeat_interleave(2, 1), kv.float().repeat_interleave(2, 1))
                self._flash_gqa = (out.shape == q4.shape
                                   and (out.float() - ref).abs().max().item() < 0.05)
            except Exception:
                self._flash_gqa = False
            try:
                out = F.scaled_dot_product_attention(q4, kv, kv, is_causal=True, enable_gqa=True)
                self._sdpa_gqa = out.shape == q4.shape
            except Exception:
                self._sdpa_gqa = False
        return call

    @staticmethod
    def _apply_rope(x, freqs):
        # x: [B, H, T, D]; freqs complex broadcastable to [B, H, T, D/2].
        # One complex multiply == the interleaved-pair rotation (3 kernels
        # instead of the 8 slice/mul/add kernels of the real-domain form).
        xc = torch.view_as_complex(x.float().reshape(*x.shape[:-1], -1, 2))
        return torch.view_as_real(xc * freqs).flatten(-2).to(x.dtype)

    # ---- prefix KV cache ---------------------------------------------------
    # Keys are the kvindex chain-hash schedule (kvindex/chain_keys.py — the
    # same content-addressed rolling hash the GPU tree uses): ONE O(n) pass
    # yields the key at every page boundary, vs the old
    # hash(tuple(tokens[:p]))-per-boundary which re-hashed the whole prefix
    # at each depth (O(n·pages), on the host, per admission).  Content
    # addressing also removes Python's per-process hash randomization, and a
    # token compare on hit makes a 64-bit collision restore impossible
    # (vLLM's content-addressed-hashing fix for cross-request KV leakage).
    def _prefix_lookup(self, tokens: List[int], keys: Optional[List[int]] = None):
        """Longest page-aligned cached prefix of `tokens` -> (arena_slot, plen).
        Capped at len(tokens)-1 so prefill always runs at least one token —
        the final-chunk logits sample the first generated token.  `keys`
        skips the O(n) hash when submit() precomputed it."""
        c = self.cfg
        page = c.prefix_cache_page
        if keys is None:
            pmax = min(((len(tokens) - 1) // page) * page, c.prefix_cache_max, c.max_seq - 1)
            if pmax < page:
                return None
            from ..kvindex.chain_keys import chain_keys

            keys = chain_keys(tokens[:pmax], page)
        if not keys:
            return None
        for i in range(len(keys) - 1, -1, -1):
            p = (i + 1) * page
            entry = self._pc_keys.get(keys[i])
            if entry is not None:
                slot, stored = entry
                st = self._pc_slot_tokens.get(slot)
                if (stored >= p and st is not None and len(st) >= p
                        and np.array_equal(st[:p], np.asarray(tokens[:p], dtype=np.int64))):
                    # LRU touch
                    try:
                        self._pc_lru.remove(slot)
                        self._pc_lru.append(slot)
                    except ValueError:
                        pass
                    return slot, p
        return None

    def _prefix_store(self, kv_slot: int, tokens: List[int]) -> None:
        if self._pc_arena is None:
            return
        c = self.cfg
        page = c.prefix_cache_page
        plen = min((len(tokens) // page) * page, c.prefix_cache_max, c.max_seq - 1)
        if plen < page:
            return
        from ..kvindex.chain_keys import chain_keys

        keys = chain_keys(tokens[:plen], page)
        if keys[-1] in self._pc_keys:
            return  # already cached
        # block-level dedup (vLLM-style): if the cache already covers all but
        # the last page or so of this prompt, a new entry would spend a
        # multi-MB arena copy + an eviction on a one-off suffix — skip it.
        # (Unchecked, the 500-stores/s churn evicted the hot shared prefixes
        # and the hit rate decayed over long runs.)
        import os as _os

        if not _os.environ.get("SMG_PC_NODEDUP"):
            existing = 0
            for i in range(len(keys) - 1, -1, -1):
                if keys[i] in self._pc_keys:
                    existing = (i + 1) * page
                    break
            if plen - existing < 2 * page:
                return
        victim = self._pc_lru.pop(0)
        for k in self._pc_slot_keys.pop(victim, []):
            # only drop keys the victim still owns: a later store of a longer
            # prompt sharing this prefix re-points the page key at its own
            # slot, and popping it here would orphan that live entry (the
            # decaying-hit-rate bug: shared-prefix keys vanished over time)
            if self._pc_keys.get(k, (None, 0))[0] == victim:
                del self._pc_keys[k]
        self._pc_arena[:, :, victim, :, :plen] = self.kv[:, :, kv_slot, :, :plen]
        for i, k in enumerate(keys):
            self._pc_keys[k] = (victim, (i + 1) * page)
        self._pc_slot_keys[victim] = list(keys)
        self._pc_slot_tokens[victim] = np.array(tokens[:plen], dtype=np.int64)
        self._pc_lru.append(victim)

    def _mlp(self, h, layer):
        x = _rms(h, layer.ln2)
        gu = x @ layer.w13
        if self._hip_silu_mul is not None:
            inner = gu.shape[-1] // 2
            rows = gu.numel() // gu.shape[-1]
            smul = torch.empty(*gu.shape[:-1], inner, device=gu.device, dtype=gu.dtype)
            self._hip_silu_mul(
                gu.data_ptr(), smul.data_ptr(), rows, inner,
                torch.cuda.current_stream().cuda_stream,
            )
            return torch.addmm(h.reshape(rows, -1), smul.view(rows, inner), layer.w2).view(h.shape)
        g, u = gu.chunk(2, dim=-1)
        return h + (F.silu(g) * u) @ layer.w2

    def _qkv(self, h, layer, freqs):
        c = self.cfg
        B, T, _ = h.shape
        qkv = _rms(h, layer.ln1) @ layer.wqkv
        q, k, v = qkv.split([c.d_model, c.kv_dim, c.kv_dim], dim=-1)
        q = q.view(B, T, c.n_heads, c.head_dim).transpose(1, 2)
        k = k.view(B, T, c.kv_heads, c.head_dim).transpose(1, 2)
        v = v.view(B, T, c.kv_heads, c.head_dim).transpose(1, 2)
        return self._apply_rope(q, freqs), self._apply_rope(k, freqs), v

    def _expand_kv(self, t):
        """[.., KVH, T, hd] -> [.., H, T, hd] for torch attention paths (the
        HIP decode kernel does the group mapping in-kernel instead)."""
        g = self.cfg.gqa_group
        return t if g == 1 else t.repeat_interleave(g, dim=-3)

    def _stage_tokens(self, rows: List, L: int) -> torch.Tensor:
        """[B, L] token tensor on device via rotating pinned staging buffers
        (3 deep: with the pipelined tick, tick N's H2D may still be in flight
        when tick N+1 builds its batch).  torch.tensor(list-of-lists) was the
        single most expensive host op in the launch phase."""
        stage = self._fill_stage(rows, L)
        if self.device.type != "cuda":
            return stage
        return stage.to(self.device, non_blocking=True)

    def _fill_stage(self, rows: List, L: int) -> torch.Tensor:
        """Fill a rotating pinned staging buffer with [B, L] token rows and
        return the (host) slice — callers H2D it into a fresh tensor (eager)
        or an existing graph input buffer (graphed)."""
        B = len(rows)
        if self.device.type != "cuda":
            return torch.from_numpy(
                np.stack([np.asarray(r, dtype=np.int64) for r in rows]))
        self._pf_stage_i = (self._pf_stage_i + 1) % len(self._pf_stages)
        stage = self._pf_stages[self._pf_stage_i]
        if stage is None or stage.shape[0] < B or stage.shape[1] < L:
            stage = torch.empty(max(B, self.cfg.prefill_group),
                                max(L, self.cfg.prefill_chunk),
                                dtype=torch.long, pin_memory=True)
            self._pf_stages[self._pf_stage_i] = stage
        sn = stage.numpy()
        for i, r in enumerate(rows):
            sn[i, :L] = r
        return stage[:B, :L]

    @torch.no_grad()
    def _prefill(self, slot: int, tokens: List[int], start: int) -> None:
        self._prefill_batch([(slot, start, np.asarray(tokens, dtype=np.int64))])

    @torch.no_grad()
    def _prefill_batch(self, items: List[tuple]) -> None:
        """Batched chunked prefill: `items` = [(slot, start, tokens)] with
        equal chunk lengths.  Per-request absolute positions drive rope and a
        per-request causal mask over each slot's own KV window.

        The device work lives in _prefill_forward; on GPU with graphs on,
        recurring (B, L, start0) shapes replay a captured hipGraph — the
        eager prefill was ~260 Python-dispatched launches per tick and the
        measured launch phase (18.2 ms/tick) was pure host time."""
        c = self.cfg
        B = len(items)
        L = len(items[0][2])
        starts_host = [int(st) for _, st, _ in items]
        max_start = max(starts_host)
        t_max = max_start + L
        fresh = max_start == 0  # cold prefill: flash causal path
        uniform_start = not fresh and all(st == starts_host[0] for st in starts_host)
        start0 = starts_host[0]
        nxt = None
        if self._pfg_enabled(B, fresh, uniform_start):
            nxt = self._prefill_graphed(items, B, L, t_max, fresh, uniform_start, start0)
        if nxt is None:
            self.pfg_eager += 1
        else:
            self.pfg_hits += 1
        if nxt is None:
            slots = self._h2d_i64(np.fromiter((s for s, _, _ in items), dtype=np.int64, count=B))
            starts = self._h2d_i64(np.asarray(starts_host, dtype=np.int64))
            t = self._stage_tokens([toks for _, _, toks in items], L)
            nxt = self._prefill_forward(slots, starts, t, B, L, t_max, fresh, uniform_start, start0)
        for slot, start, toks in items:
            self._seq_len_host[slot] = int(start) + L
        return nxt  # device tensor: the caller defers (or performs) the D2H read

    # ---- prefill hipGraph cache -------------------------------------------
    PFG_CAP = 16      # distinct (B, L, start0, fresh) shapes kept captured
    PFG_MIN_SEEN = 3  # capture a shape only once it repeats: a one-off group
    #                   shape captured mid-run costs ~60 ms of full-stream
    #                   syncs inside the timed region (measured: steps-12
    #                   bench 1189 with eager capture vs 1274 graphs-off);
    #                   hot shapes repeat during warmup and capture there

    def _pfg_enabled(self, B: int, fresh: bool, uniform_start: bool) -> bool:
        import os as _os

        if not (self.graphs and self.device.type == "cuda") or _os.environ.get("SMG_NO_PREFILL_GRAPH"):
            return False
        if B < 4:  # singles/tiny groups: capture overhead isn't worth it
            return False
        # graphable paths: cold flash-causal, or uniform-start suffix merge
        if fresh:
            return True
        use_fused = self._hip_fused is not None and getattr(self, "_hip_rope_prefill", None) is not None
        return bool(uniform_start and use_fused and self._flash_lse)

    def _prefill_graphed(self, items, B, L, t_max, fresh, uniform_start, start0):
        """Replay (or capture, first time per shape) the prefill graph.
        Returns the graph's output tensor, or None to fall back to eager.

        Capture protocol: warmup runs execute the REAL batch (prefill is
        idempotent — same inputs rewrite the same KV rows / seq_len /
        _last_tok values), then capture records without executing, then one
        replay performs this tick's work.  Input buffers (tokens / slots /
        starts) are content-dynamic; B, L and the history depth start0 are
        shape-defining and key the cache."""
        key = (B, L, 0 if fresh else start0, fresh)
        entry = self._pfg_cache.get(key)
        if entry is None:
            if key in self._pfg_bad or len(self._pfg_cache) >= self.PFG_CAP:
                return None
            if len(self._pfg_seen) > 512:  # arbitrary-L traffic: bound the counter map
                self._pfg_seen.clear()
            seen = self._pfg_seen.get(key, 0) + 1
            self._pfg_seen[key] = seen
            if seen < self.PFG_MIN_SEEN:
                return None
            try:
                entry = self._pfg_capture(key, items, B, L, t_max, fresh, uniform_start, start0)
            except Exception as exc:
                import os as _os
                import sys as _sys

                self._pfg_bad.add(key)
                if _os.environ.get("SMG_PFG_DEBUG"):
                    print(f"[pfg] capture failed for {key}: {exc!r}", file=_sys.stderr)
                # the failed attempt may have half-written state; the eager
                # fallback below rewrites it (idempotent)
                return None
            self._pfg_cache[key] = entry
        g, t_buf, slots_buf, starts_buf, nxt_out, slots_stage, starts_stage = entry
        stage = self._fill_stage([toks for _, _, toks in items], L)
        t_buf.copy_(stage, non_blocking=True)
        slots_stage.copy_(torch.from_numpy(np.fromiter((s for s, _, _ in items), dtype=np.int64, count=B)))
        starts_stage.copy_(torch.from_numpy(np.fromiter((st for _, st, _ in items), dtype=np.int64, count=B)))
        slots_buf.copy_(slots_stage, non_blocking=True)
        starts_buf.copy_(starts_stage, non_blocking=True)
        g.replay()
        return nxt_out

    def _pfg_capture(self, key, items, B, L, t_max, fresh, uniform_start, start0):
        dev = self.device
        t_buf = torch.zeros(B, L, dtype=torch.long, device=dev)
        slots_buf = torch.zeros(B, dtype=torch.long, device=dev)
        starts_buf = torch.zeros(B, dtype=torch.long, device=dev)
        slots_stage = torch.empty(B, dtype=torch.long, pin_memory=True)
        starts_stage = torch.empty(B, dtype=torch.long, pin_memory=True)
        # seed the buffers with the REAL batch so warmup executes this tick's
        # actual (idempotent) work
        stage = self._fill_stage([toks for _, _, toks in items], L)
        t_buf.copy_(stage)
        slots_buf.copy_(torch.tensor([s for s, _, _ in items], dtype=torch.long))
        starts_buf.copy_(torch.tensor([st for _, st, _ in items], dtype=torch.long))
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):  # warmup allocations outside capture
                self._prefill_forward(slots_buf, starts_buf, t_buf, B, L, t_max,
                                      fresh, uniform_start, start0)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            nxt_out = self._prefill_forward(slots_buf, starts_buf, t_buf, B, L, t_max,
                                            fresh, uniform_start, start0)
        return (g, t_buf, slots_buf, starts_buf, nxt_out, slots_stage, starts_stage)

    @torch.no_grad()
    def _prefill_embedded(self, slot: int, req) -> torch.Tensor:
        """EPD prefill: [E, D] vision embeddings at positions 0..E-1 followed
        by the prompt tokens; returns the sampled first token ([1], device)."""
        t = torch.from_numpy(np.ascontiguousarray(req.tokens, dtype=np.int64))
        if self.device.type == "cuda":
            t = t.to(self.device, non_blocking=True)
        h0 = torch.cat([req.mm_embed, self.embed[t]], 0).unsqueeze(0)  # [1, E+n, D]
        L = h0.shape[1]
        slots = self._h2d_i64(np.asarray([slot], dtype=np.int64))
        starts = self._h2d_i64(np.asarray([0], dtype=np.int64))
        nxt = self._prefill_forward(slots, starts, None, 1, L, L, True, False, 0, h0=h0)
        self._seq_len_host[slot] = L
        return nxt

    @torch.no_grad()
    def _prefill_forward(self, slots, starts, t, B: int, L: int, t_max: int,
                         fresh: bool, uniform_start: bool, start0: int, h0=None):
        """Device-only prefill forward (hipGraph-capturable: every input is a
        tensor whose CONTENT may change between replays; B/L/start0 are
        shape-defining and key the graph cache).  `h0` bypasses the token
        embedding lookup (EPD embedding-prefixed requests)."""
        c = self.cfg
        h = h0 if h0 is not None else self.embed[t]  # [B, L, D]
        pos = starts.unsqueeze(1) + torch.arange(L, device=self.device)  # [B, L]
        freqs = self.freqs_cis[pos].unsqueeze(1)  # [B, 1, L, D/2]
        if not fresh:
            kpos = torch.arange(t_max, device=self.device)
            mask = kpos.view(1, 1, 1, -1) <= pos.view(B, 1, L, 1)
        use_fused = self._hip_fused is not None and getattr(self, "_hip_rope_prefill", None) is not None
        if use_fused:
            # fused rope + KV scatter + [B,H,L,hd] emit (csrc/fused_decode.hip
            # smg_rope_prefill): replaces the rope complex-mul chains, the two
            # advanced-index scatters, and sdpa's transpose-contiguous copies
            slots_i32 = slots.to(torch.int32)
            starts_i32 = starts.to(torch.int32)
            qb = torch.empty(B, c.n_heads, L, c.head_dim, device=self.device, dtype=self.dtype)
            kb = torch.empty(B, c.kv_heads, L, c.head_dim, device=self.device, dtype=self.dtype)
            vb = torch.empty_like(kb)
            stream = torch.cuda.current_stream().cuda_stream
            freqs_ptr = self.freqs_cis.data_ptr()
        for li, layer in enumerate(self.layers):
            if use_fused:
                qkv = (_rms(h, layer.ln1) @ layer.wqkv).view(B, L, c.d_model + 2 * c.kv_dim)
                self._hip_rope_prefill(
                    qkv.data_ptr(), freqs_ptr, slots_i32.data_ptr(), starts_i32.data_ptr(),
                    self.kv[li, 0].data_ptr(), self.kv[li, 1].data_ptr(),
                    qb.data_ptr(), kb.data_ptr(), vb.data_ptr(),
                    B, L, c.n_heads, c.max_seq, c.head_dim, stream,
                    1 if self.kv.dtype == torch.float8_e4m3fn else 0,
                    c.kv_heads,
                )
                q, k, v = qb, kb, vb
            else:
                q, k, v = self._qkv(h, layer, freqs)
                # scatter this chunk's K/V into each request's slot window
                self.kv[li, 0][slots[:, None], :, pos] = k.permute(0, 2, 1, 3)
                self.kv[li, 1][slots[:, None], :, pos] = v.permute(0, 2, 1, 3)
            if fresh:
                # no history: attend within the chunk itself, flash kernel.
                # GQA stays native when the backend supports it (no
                # repeat_interleave expansion copies)
                if self.cfg.gqa_group > 1 and self._sdpa_gqa:
                    attn = F.scaled_dot_product_attention(q, k, v, is_causal=True, enable_gqa=True)
                else:
                    attn = F.scaled_dot_product_attention(
                        q, self._expand_kv(k), self._expand_kv(v), is_causal=True)
            elif use_fused and uniform_start and self._flash_lse:
                # uniform-start suffix chunk (the prefix-cache-hit fast path):
                # two FLASH passes — history cross-attention (all keys valid,
                # no mask) + square causal chunk — merged by logsumexp.
                # Replaces the masked-sdpa math path (bmm + 37 MB mask add +
                # softmax per layer).
                if self.cfg.gqa_group > 1 and not self._flash_gqa:
                    kk = self._expand_kv(self._kv_hist(li, 0, slots)[:, :, :start0])
                    vv = self._expand_kv(self._kv_hist(li, 1, slots)[:, :, :start0])
                    kc = self._expand_kv(k).contiguous()
                    vc = self._expand_kv(v).contiguous()
                else:
                    kk = self._kv_hist(li, 0, slots)[:, :, :start0]
                    vv = self._kv_hist(li, 1, slots)[:, :, :start0]
                    kc, vc = k, v
                o1, lse1 = self._flash_lse(q, kk, vv, False)
                o2, lse2 = self._flash_lse(q, kc, vc, True)
                merge = getattr(self, "_hip_lse_merge", None)
                if merge is not None and o1.is_contiguous() and o2.is_contiguous():
                    rows = B * c.n_heads * L
                    attn = torch.empty_like(o2)
                    merge(
                        o1.data_ptr(), o2.data_ptr(),
                        # aotriton may pad lse's last dim — slice to L first
                        lse1[..., :L].contiguous().data_ptr(),
                        lse2[..., :L].contiguous().data_ptr(),
                        attn.data_ptr(), rows, c.head_dim, stream,
                    )
                else:
                    lse_tot = torch.logaddexp(lse1, lse2)
                    attn = o1 * (lse1 - lse_tot).exp().unsqueeze(-1) + o2 * (lse2 - lse_tot).exp().unsqueeze(-1)
                    attn = attn.to(q.dtype)
            else:
                if self.cfg.gqa_group > 1 and self._sdpa_gqa:
                    kk = self._kv_hist(li, 0, slots)[:, :, :t_max]
                    vv = self._kv_hist(li, 1, slots)[:, :, :t_max]
                    attn = F.scaled_dot_product_attention(q, kk, vv, attn_mask=mask, enable_gqa=True)
                else:
                    kk = self._expand_kv(self._kv_hist(li, 0, slots)[:, :, :t_max])
                    vv = self._expand_kv(self._kv_hist(li, 1, slots)[:, :, :t_max])
                    attn = F.scaled_dot_product_attention(q, kk, vv, attn_mask=mask)
            attn2 = attn.transpose(1, 2).reshape(B * L, c.d_model)
            h = torch.addmm(h.view(B * L, c.d_model), attn2, layer.wo).view(B, L, c.d_model)
            h = self._mlp(h, layer)
        # final-position logits: one [B,D]@[D,V] GEMM samples the next token.
        # For requests whose prefill completes with this chunk this IS the
        # first generated token; for the rest _last_tok is overwritten by
        # their final chunk before decode ever reads it.
        nxt = (_rms(h[:, -1], self.ln_f) @ self.embed.t()).argmax(-1)  # [B]
        self._last_tok[slots] = nxt
        self.seq_len[slots] = starts + L
        return nxt  # device tensor: the caller defers (or performs) the D2H read

    def _decode_core(self, maxlen: int) -> torch.Tensor:
        """Full-arena decode forward: every slot participates with a static
        shape (no KV gather — the per-slot windows are views); inactive slots
        compute garbage that is masked out of the logical state.  A free
        slot's stray K/V write lands at its stale position and is always
        overwritten by the next prefill before it can be attended.  Static
        shapes per `maxlen` make this hipGraph-capturable."""
        c = self.cfg
        S = c.max_slots
        if self._hip_fused is not None and self._hip_attn is not None and self._hip_silu_mul is not None:
            return self._decode_core_fused()
        pos = self.seq_len  # [S] current length == write position
        freqs = self.freqs_cis[pos].view(S, 1, 1, -1)
        use_hip = self._hip_attn is not None
        if use_hip:
            self._pos_i32.copy_(pos.to(torch.int32))
            stream = torch.cuda.current_stream().cuda_stream
            scale = 1.0 / math.sqrt(c.head_dim)
        else:
            kpos = torch.arange(maxlen, device=self.device)
            mask = (kpos.unsqueeze(0) <= pos.unsqueeze(1)).unsqueeze(1).unsqueeze(1)
        h = self.embed[self._last_tok.unsqueeze(1)]  # [S, 1, D]
        for li, layer in enumerate(self.layers):
            q, k, v = self._qkv(h, layer, freqs)
            self.kv[li, 0, self._arange_slots, :, pos] = k[:, :, 0]
            self.kv[li, 1, self._arange_slots, :, pos] = v[:, :, 0]
            if use_hip:
                qc = q.reshape(S, c.n_heads, c.head_dim).contiguous()
                self._hip_attn(
                    qc.data_ptr(), self.kv[li, 0].data_ptr(), self.kv[li, 1].data_ptr(),
                    self._pos_i32.data_ptr(), self._attn_out.data_ptr(),
                    S, c.n_heads, c.max_seq, c.head_dim, scale, stream,
                    0, c.kv_heads,
                )
                attn_flat = self._attn_out.view(S, 1, c.d_model)
            else:
                kk = self._expand_kv(self.kv[li, 0][:, :, :maxlen])
                vv = self._expand_kv(self.kv[li, 1][:, :, :maxlen])
                attn = F.scaled_dot_product_attention(q, kk, vv, attn_mask=mask)
                attn_flat = attn.transpose(1, 2).reshape(S, 1, c.d_model)
            h = h + attn_flat @ layer.wo
            h = self._mlp(h, layer)
        h = _rms(h, self.ln_f)
        logits = h[:, 0] @ self.embed.t()
        return logits.argmax(-1)  # [S]

    @torch.no_grad()
    def _decode_core_fused(self) -> torch.Tensor:
        """GPU decode with the fused_decode.hip kernels: per layer the only
        torch-launched ops are rms_norm and the four hipBLASLt GEMMs (two of
        them addmm so the residual adds ride the GEMM epilogue); rope,
        KV-arena store, q pack and silu*mul are one HIP kernel each.
        hipGraph-capturable: every launch lands on the capturing stream."""
        c = self.cfg
        S = c.max_slots
        self._pos_i32.copy_(self.seq_len.to(torch.int32))
        stream = torch.cuda.current_stream().cuda_stream
        scale = 1.0 / math.sqrt(c.head_dim)
        h = self.embed[self._last_tok]  # [S, D]
        kv8 = 1 if self.kv.dtype == torch.float8_e4m3fn else 0
        freqs_ptr = self.freqs_cis.data_ptr()  # complex64 [T, hd/2] == float2
        use_mfma = self._use_mfma
        for li, layer in enumerate(self.layers):
            if use_mfma:
                # hand-written MFMA path: invrms(h) ⊙ (h @ (ln1 ⊙ wqkv)^T)
                self._hip_row_invrms(h.data_ptr(), self._invrms.data_ptr(),
                                     S, c.d_model, 1e-5, stream)
                self._hip_rms_gemm(h.data_ptr(), layer.wqkv_tg.data_ptr(),
                                   self._invrms.data_ptr(), self._qkv_buf.data_ptr(),
                                   S, c.d_model, c.d_model + 2 * c.kv_dim, stream)
                qkv = self._qkv_buf
            else:
                qkv = _rms(h, layer.ln1) @ layer.wqkv  # [S, D + 2*KD]
            self._hip_fused(
                qkv.data_ptr(), freqs_ptr, self._pos_i32.data_ptr(),
                self.kv[li, 0].data_ptr(), self.kv[li, 1].data_ptr(), self._q_buf.data_ptr(),
                S, c.n_heads, c.max_seq, c.head_dim, stream, kv8, c.kv_heads,
            )
            if self._attn_split > 1:
                self._hip_attn_split(
                    self._q_buf.data_ptr(), self.kv[li, 0].data_ptr(), self.kv[li, 1].data_ptr(),
                    self._pos_i32.data_ptr(), self._attn_out.data_ptr(),
                    self._attn_part.data_ptr(), self._attn_ml.data_ptr(),
                    S, c.n_heads, c.kv_heads, self._attn_split, c.max_seq, c.head_dim,
                    scale, stream, kv8,
                )
            else:
                self._hip_attn(
                    self._q_buf.data_ptr(), self.kv[li, 0].data_ptr(), self.kv[li, 1].data_ptr(),
                    self._pos_i32.data_ptr(), self._attn_out.data_ptr(),
                    S, c.n_heads, c.max_seq, c.head_dim, scale, stream, kv8, c.kv_heads,
                )
            h = torch.addmm(h, self._attn_out.view(S, c.d_model), layer.wo)
            if use_mfma:
                self._hip_row_invrms(h.data_ptr(), self._invrms.data_ptr(),
                                     S, c.d_model, 1e-5, stream)
                self._hip_rms_gemm(h.data_ptr(), layer.w13_tg.data_ptr(),
                                   self._invrms.data_ptr(), self._gu_buf.data_ptr(),
                                   S, c.d_model, 2 * c.d_ffn, stream)
                gu = self._gu_buf
            else:
                gu = _rms(h, layer.ln2) @ layer.w13  # [S, 2F]
            self._hip_silu_mul(gu.data_ptr(), self._smul_buf.data_ptr(), S, c.d_ffn, stream)
            h = torch.addmm(h, self._smul_buf, layer.w2)
        h = _rms(h, self.ln_f)
        return (h @ self.embed.t()).argmax(-1)  # [S]

    GRAPH_BUCKET = 256

    def _decode_graphed(self, maxlen: int) -> torch.Tensor:
        """hipGraph-captured decode keyed by the maxlen bucket: one replay
        instead of ~8 kernel launches per layer.  pos/_last_tok are read
        inside the graph from their persistent device tensors."""
        if self._hip_fused is not None and self._hip_attn is not None and self._hip_silu_mul is not None:
            bucket = self.cfg.max_seq  # fused path reads per-slot pos: one graph fits all
        else:
            bucket = min(((maxlen + self.GRAPH_BUCKET - 1) // self.GRAPH_BUCKET) * self.GRAPH_BUCKET,
                         self.cfg.max_seq)
        entry = self._graph_cache.get(bucket)
        if entry is None:
            torch.cuda.synchronize()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):  # warmup allocations outside capture
                    self._decode_core(bucket)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                out = self._decode_core(bucket)
            entry = (g, out)
            self._graph_cache[bucket] = entry
        g, out = entry
        g.replay()
        return out

    @torch.no_grad()
    def _decode_launch(self, slots: List[int]) -> torch.Tensor:
        """One decode pass for every slot in `slots`; returns the FULL
        [max_slots] sampled-token tensor on device without any host sync —
        the caller reads the values later (step_finish) so the host keeps
        launching while the GPU works."""
        return self._decode_launch_np(np.fromiter(slots, dtype=np.int64, count=len(slots)))

    @torch.no_grad()
    def _decode_launch_np(self, act_np: "np.ndarray") -> torch.Tensor:
        import os as _os

        maxlen = int(self._seq_len_host.max()) + 1
        if (self.graphs and self.device.type == "cuda"
                and not _os.environ.get("SMG_NO_DECODE_GRAPH")):
            nxt = self._decode_graphed(maxlen)
        else:
            nxt = self._decode_core(maxlen)
        act = self._h2d_i64(act_np)
        self._last_tok.index_copy_(0, act, nxt.index_select(0, act))
        self.seq_len.index_add_(0, act, torch.ones_like(act))
        self._seq_len_host[act_np] += 1
        return nxt

    @torch.no_grad()
    def _decode(self, slots: List[int]) -> List[int]:
        nxt = self._decode_launch(slots)
        act = torch.tensor(slots, device=self.device)
        return nxt.index_select(0, act).tolist()
