"""MI355X-native distributed diffusion orchestrator.

A from-scratch framework with the capabilities of ComfyUI-Distributed
(reference: /root/reference — see SURVEY.md), re-architected for AMD
Instinct MI355X (gfx950):

* seed-parallel multi-GPU generation with an RCCL-over-xGMI collector
  gather (reference: nodes/collector.py does this with HTTP + base64 PNG),
* tile-parallel Ultimate-SD-Upscale with a pull-queue tile scheduler,
  heartbeat/timeout/requeue fault tolerance and deterministic blend order
  (reference: upscale/modes/static.py, upscale/job_timeout.py),
* per-worker value overrides, batch dividers, delegate-only master and
  least-busy load balancing (reference: nodes/utilities.py,
  api/queue_orchestration.py),
* a wire-compatible ``POST /distributed/queue`` REST API for remote /
  cloud workers (reference: api/job_routes.py).

The compute substrate the reference borrows from ComfyUI (sampler, UNet,
VAE, tile blending) is implemented here natively: hand-written CDNA4 HIP
kernels (MFMA attention, fused GroupNorm+SiLU, tile extract/resize and
seam blend) driven by a PyTorch-ROCm host runtime, one process per GPU
with torch.distributed over RCCL.
"""

__version__ = "0.1.0"
