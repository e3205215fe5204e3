"""smg_amd — MI355X-native model-routing gateway (Shepherd Model Gateway capabilities).

A brand-new AMD-native LLM routing gateway: OpenAI/Anthropic-compatible HTTP surface,
ten load-balancing policies over a GPU-resident KV prefix index (gfx950 HIP kernels),
RCCL-over-xGMI fan-out to on-node workers, SWIM-gossip HA mesh, and full
observability.  Blueprint: SURVEY.md; reference behavior: lightseekorg/smg.
"""

__version__ = "0.1.0"
