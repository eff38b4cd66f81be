"""HyperSpot-AMD — MI355X-native multi-tenant LLM-serving platform.

A from-scratch rebuild of the capabilities of cyberfabric/cyberfabric-core
(reference: /root/reference — a Rust modkit SaaS host plane whose LLM-serving
data plane exists only as specification).  This package is the *serving plane*:

  - ``hyperspot.engine``   — paged-KV continuous-batching inference engine
  - ``hyperspot.models``   — model families (Llama-3, Mixtral MoE)
  - ``hyperspot.ops``      — op layer: hand-written CDNA4 HIP kernels (gfx950)
                             with PyTorch fp32 reference implementations for CPU
  - ``hyperspot.parallel`` — TP/EP process groups over RCCL (xGMI)
  - ``hyperspot.serving``  — the llm-gateway worker: OpenAI-style chat
                             completion semantics per the reference's GTS
                             schemas (reference modules/llm-gateway/docs/DESIGN.md)

The host plane (module runtime, api-gateway, auth/tenancy — reference
libs/modkit, modules/system/*) is the native C++ ``hyperspot-server`` under
``host/``.
"""

__version__ = "0.1.0"
