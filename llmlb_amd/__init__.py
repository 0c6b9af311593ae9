"""llmlb_amd — an MI355X-native LLM inference gateway + backend framework.

A from-scratch reimplementation of the capabilities of akiojin/llmlb
(reference: /root/reference, an OpenAI-compatible LLM gateway / load
balancer written in Rust) redesigned for a single 8x MI355X node:

  * ``gateway/``   — supervisor + client for the native C++ data-plane
                     proxy (zero-copy SSE, TPS-EMA routing; see /gateway).
  * ``backend/``   — the MI355X inference backend: continuous-batching
                     engine, paged KV cache, OpenAI-compatible server.
  * ``models/``    — model definitions (Llama-3 family), bf16, TP-aware.
  * ``ops/``       — hand-written HIP/CDNA4 kernels (gfx950) + CPU
                     reference fallbacks for tests.
  * ``parallel/``  — tensor parallelism over torch.distributed (RCCL on
                     ROCm; gloo for CPU tests) and xGMI topology-aware
                     replica placement.
  * ``telemetry/`` — amdsmi HBM/busy polling, hipEvent queue depth.

The reference gateway hosts no model kernels; the GPU
surface here (backends, kernels, telemetry, placement) is new work
specified by BASELINE.json.
"""

__version__ = "0.1.0"
