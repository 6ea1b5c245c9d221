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

import os as _os

# The handful of convs still on MIOpen (stride-2 downsamples, the K=4
# out_conv) trigger MIOpen's exhaustive kernel search on first use: ~220 s
# of naive-conv candidate runs in the flagship warmup canvas
# (profiles/r02_results.md). FAST find mode picks from heuristics instead;
# those convs are <1% of steady-state kernel time, so the tuned-vs-
# heuristic delta is noise while the warmup drops by minutes.
_os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")

# Node maps at package root (reference __init__.py:1-29 exposes the same
# names so a host app can discover the node classes; kept lazy so plain
# `import comfyui_distributed_amd` stays light).


def __getattr__(name):
    if name in ("NODE_CLASS_MAPPINGS", "NODE_DISPLAY_NAME_MAPPINGS"):
        from . import nodes as _nodes
        from .graph import builtin_nodes as _builtin

        if name == "NODE_CLASS_MAPPINGS":
            merged = dict(_builtin.BUILTIN_CLASS_MAPPINGS)
            merged.update(_nodes.NODE_CLASS_MAPPINGS)
            return merged
        return dict(_nodes.NODE_DISPLAY_NAME_MAPPINGS)
    raise AttributeError(name)
