"""Samplers + noise schedules (the common_ksampler equivalent).

Reference counterpart: ComfyUI's ``common_ksampler`` invoked per tile at
upscale/tile_ops.py:225-229. Implements the discrete
eps-prediction parameterization (linear beta schedule), karras/normal sigma
schedules, Euler / Euler-ancestral / DPM++ 2M samplers, classifier-free
guidance and partial denoise (the USDU ``denoise`` fraction).
"""

from __future__ import annotations

import math

import torch

SAMPLERS = ("euler", "euler_ancestral", "heun", "dpm_2", "dpmpp_2m",
            "dpmpp_2m_sde")
SCHEDULERS = ("normal", "karras", "exponential", "sgm_uniform", "simple",
              "beta")


def make_alphas_cumprod(n: int = 1000, beta_start: float = 0.00085,
                        beta_end: float = 0.012) -> torch.Tensor:
    betas = torch.linspace(beta_start**0.5, beta_end**0.5, n, dtype=torch.float64) ** 2
    return torch.cumprod(1.0 - betas, dim=0)


class NoiseSchedule:
    """sigma(t) = sqrt((1 - a_t) / a_t) over the discrete training schedule."""

    def __init__(self, n: int = 1000):
        ac = make_alphas_cumprod(n)
        self.sigmas_all = ((1 - ac) / ac).sqrt().float()  # ascending in t
        self.log_sigmas = self.sigmas_all.log()

    @property
    def sigma_max(self) -> float:
        return float(self.sigmas_all[-1])

    @property
    def sigma_min(self) -> float:
        return float(self.sigmas_all[0])

    def timestep(self, sigma: torch.Tensor) -> torch.Tensor:
        """Continuous timestep for a sigma (interpolated in log space)."""
        log_sigma = sigma.log()
        dists = log_sigma.reshape(-1, 1) - self.log_sigmas.to(sigma.device)[None]
        low_idx = dists.ge(0).cumsum(dim=1).argmax(dim=1).clamp(max=len(self.sigmas_all) - 2)
        high_idx = low_idx + 1
        low = self.log_sigmas.to(sigma.device)[low_idx]
        high = self.log_sigmas.to(sigma.device)[high_idx]
        w = ((low - log_sigma.reshape(-1)) / (low - high)).clamp(0, 1)
        return ((1 - w) * low_idx + w * high_idx).reshape(sigma.shape)

    def sigmas(self, steps: int, scheduler: str = "normal",
               denoise: float = 1.0) -> torch.Tensor:
        """Descending sigma sequence of length steps+1 ending at 0.

        ``denoise < 1`` keeps the tail: compute the full schedule at
        ``ceil(steps / denoise)`` and take the last steps+1 entries (USDU /
        ComfyUI partial-denoise behavior)."""
        if denoise <= 0:
            return torch.zeros(1)
        total = steps if denoise >= 1.0 else max(int(math.ceil(steps / denoise)), steps)
        n_train = len(self.sigmas_all)
        if scheduler == "karras":
            rho = 7.0
            ramp = torch.linspace(0, 1, total)
            mn, mx = self.sigma_min, self.sigma_max
            s = (mx ** (1 / rho) + ramp * (mn ** (1 / rho) - mx ** (1 / rho))) ** rho
        elif scheduler == "normal":
            t = torch.linspace(n_train - 1, 0, total)
            s = self.sigmas_all[t.long().clamp(0, n_train - 1)]
        elif scheduler == "exponential":
            s = torch.exp(torch.linspace(
                math.log(self.sigma_max), math.log(self.sigma_min), total))
        elif scheduler == "sgm_uniform":
            # uniform in timestep, excluding the final training step (the
            # SGM/EDM convention ComfyUI exposes under this name)
            t = torch.linspace(n_train - 1, 0, total + 1)[:-1]
            s = self.sigmas_all[t.long().clamp(0, n_train - 1)]
        elif scheduler == "simple":
            # fixed stride over the training schedule
            stride = n_train / total
            idx = [n_train - 1 - int(i * stride) for i in range(total)]
            s = self.sigmas_all[torch.tensor(idx).clamp(0, n_train - 1)]
        elif scheduler == "beta":
            # Beta(0.6, 0.6)-spaced timesteps (ComfyUI's "beta" scheduler):
            # clusters steps at both schedule ends
            try:
                from scipy.stats import beta as _beta

                ts = [round(_beta.ppf(1 - (i / total), 0.6, 0.6)
                            * (n_train - 1)) for i in range(total)]
            except Exception:  # scipy-less fallback: symmetric cosine
                u = torch.linspace(1, 0, total)
                ts = ((0.5 - 0.5 * torch.cos(u * math.pi))
                      * (n_train - 1)).round().tolist()
            s = self.sigmas_all[torch.tensor(
                [int(t) for t in ts]).clamp(0, n_train - 1)]
        else:
            raise ValueError(f"unknown scheduler {scheduler!r}")
        s = torch.cat([s, torch.zeros(1)])
        return s[-(steps + 1):]


#: cached graphs per wrapper before the least-recently-used is evicted —
#: each entry pins static I/O buffers plus a private memory pool, so an
#: unbounded dict would grow without limit on a long-running server that
#: sees many distinct shapes/configs
GRAPH_CACHE_CAP = 8


def _evict_lru(graphs: dict):
    if len(graphs) > GRAPH_CACHE_CAP:
        graphs.pop(next(iter(graphs)))


class GraphedModel:
    """hipGraph-captured model wrapper: one graph per input-shape set.

    The UNet denoise step launches hundreds of small kernels; on MI355X the
    step is launch-bound (see profiles/r01_usdu_kernel_stats.md: 66k
    dispatches, wall >> kernel time). Capturing the forward in a hipGraph
    replays the whole step as one submission. Inputs are copied into static
    buffers; the returned tensor is the static output (consumed immediately
    by the sampler math before the next replay).
    """

    def __init__(self, model):
        self.model = model
        self.graphs: dict = {}
        self.enabled = torch.cuda.is_available()
        self._failed = False

    def __call__(self, x, t, ctx, y=None):
        if not self.enabled or self._failed or not x.is_cuda:
            return self.model(x, t, ctx, y=y) if y is not None else self.model(x, t, ctx)
        key = (
            tuple(x.shape), tuple(ctx.shape),
            None if y is None else tuple(y.shape), x.dtype,
        )
        entry = self.graphs.pop(key, None)
        if entry is None:
            try:
                entry = self._capture(x, t, ctx, y)
            except Exception as exc:  # noqa: BLE001 - fall back to eager
                import warnings

                warnings.warn(f"hipGraph capture failed, running eager: {exc!r}")
                self._failed = True
                return self.model(x, t, ctx, y=y) if y is not None else self.model(x, t, ctx)
        self.graphs[key] = entry  # re-insert = most recently used
        _evict_lru(self.graphs)
        sx, st, sc, sy, sout, g = entry
        sx.copy_(x)
        st.copy_(t)
        sc.copy_(ctx)
        if y is not None:
            sy.copy_(y)
        g.replay()
        return sout

    def _capture(self, x, t, ctx, y):
        sx = x.detach().clone()
        st = t.detach().clone()
        sc = ctx.detach().clone()
        sy = y.detach().clone() if y is not None else None

        def fwd():
            return (self.model(sx, st, sc, y=sy) if sy is not None
                    else self.model(sx, st, sc))

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                out = fwd()
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            sout = fwd()
        return (sx, st, sc, sy, sout, g)


class GraphedSamplerLoop:
    """Captures an ENTIRE deterministic sampler loop — every UNet forward,
    the CFG combine and the sampler update math — as ONE hipGraph.

    Round-1 profiling (profiles/r01_prof_final_summary.csv header) showed a
    27% wall-over-kernel gap: per-step eager math (timestep interpolation,
    CFG cat/chunk, the x update) plus per-replay launch overhead dominate
    between the UNet graphs. One replay per CHUNK removes all of it. Sigma
    and timestep values are baked in as constants — the cache key includes
    the full sigma tuple — and conditioning/noise/latent are static
    copy-in buffers. Only deterministic samplers (no per-step randn) are
    captured; ancestral/SDE run the eager path.
    """

    SUPPORTED = ("euler", "heun", "dpm_2", "dpmpp_2m")

    def __init__(self, model, schedule: NoiseSchedule):
        self.model = model  # the RAW UNet module (not GraphedModel)
        self.schedule = schedule
        self.graphs: dict = {}
        self._failed = False

    def _denoise_const(self, x, i, entry_consts, s_ctx, s_y, cfg, need_cfg,
                       sig, t_tensor):
        """One CFG denoise with python-float sigma (mirrors CFGDenoiser)."""
        c_in = 1.0 / math.sqrt(1.0 + sig * sig)
        if need_cfg:
            x_in = torch.cat([x * c_in] * 2)
            eps = (self.model(x_in, t_tensor, s_ctx, y=s_y)
                   if s_y is not None else self.model(x_in, t_tensor, s_ctx))
            eps_c, eps_u = eps.chunk(2)
            eps = eps_u + cfg * (eps_c - eps_u)
        else:
            eps = (self.model(x * c_in, t_tensor, s_ctx, y=s_y)
                   if s_y is not None else self.model(x * c_in, t_tensor, s_ctx))
        return x - eps.float() * sig

    def _loop_body(self, sampler, x, sigs, denoise_at):
        """The sampler recurrence with float sigmas; ``denoise_at(x, i,
        sig)`` runs the model at schedule position i."""
        n = len(sigs) - 1
        if sampler == "euler":
            for i in range(n):
                sig, nxt = sigs[i], sigs[i + 1]
                denoised = denoise_at(x, i, sig)
                d = (x - denoised) / sig
                x = x + d * (nxt - sig)
            return x
        if sampler == "heun":
            for i in range(n):
                sig, nxt = sigs[i], sigs[i + 1]
                denoised = denoise_at(x, i, sig)
                d = (x - denoised) / sig
                dt = nxt - sig
                if nxt == 0:
                    x = x + d * dt
                    continue
                x2 = x + d * dt
                denoised2 = denoise_at(x2, i, nxt, aux=True)
                d2 = (x2 - denoised2) / nxt
                x = x + (d + d2) / 2 * dt
            return x
        if sampler == "dpm_2":
            for i in range(n):
                sig, nxt = sigs[i], sigs[i + 1]
                denoised = denoise_at(x, i, sig)
                d = (x - denoised) / sig
                if nxt == 0:
                    x = x + d * (nxt - sig)
                    continue
                sig_mid = math.exp(
                    math.log(sig) + (math.log(nxt) - math.log(sig)) * 0.5)
                x2 = x + d * (sig_mid - sig)
                denoised2 = denoise_at(x2, i, sig_mid, aux=True)
                d2 = (x2 - denoised2) / sig_mid
                x = x + d2 * (nxt - sig)
            return x
        if sampler == "dpmpp_2m":
            old_denoised = None
            for i in range(n):
                sig, nxt = sigs[i], sigs[i + 1]
                denoised = denoise_at(x, i, sig)
                if nxt == 0:
                    x = denoised
                elif old_denoised is None:
                    h = -math.log(nxt) + math.log(sig)
                    x = (nxt / sig) * x - (math.expm1(-h)) * denoised
                else:
                    h = -math.log(nxt) + math.log(sig)
                    h_last = -math.log(sig) + math.log(sigs[i - 1])
                    r = h_last / h
                    denoised_d = (1 + 1 / (2 * r)) * denoised - \
                        (1 / (2 * r)) * old_denoised
                    x = (nxt / sig) * x - (math.expm1(-h)) * denoised_d
                old_denoised = denoised
            return x
        raise ValueError(sampler)

    def run(self, cond, uncond, cfg_scale, noise, sigmas, sampler,
            start_from_latent):
        """Returns the sampled latent, or None when this config can't be
        captured (caller falls back to eager)."""
        if self._failed or sampler not in self.SUPPORTED or not noise.is_cuda:
            return None
        b = noise.shape[0]
        need_cfg = cfg_scale != 1.0 and uncond is not None
        sigs = [float(s) for s in sigmas]
        ctx_c = cond["context"]
        y_c = cond.get("y")
        key = (
            sampler, tuple(noise.shape), tuple(sigs), bool(need_cfg),
            float(cfg_scale), tuple(ctx_c.shape),
            None if y_c is None else tuple(y_c.shape),
            start_from_latent is not None,
        )
        entry = self.graphs.pop(key, None)
        if entry is None:
            try:
                entry = self._capture(key, cond, uncond, cfg_scale, noise,
                                      sigs, sampler, start_from_latent)
            except Exception as exc:  # noqa: BLE001
                import warnings

                warnings.warn(
                    f"sampler-loop hipGraph capture failed, eager: {exc!r}")
                self._failed = True
                return None
        self.graphs[key] = entry  # re-insert = most recently used
        _evict_lru(self.graphs)
        (s_noise, s_lat, s_ctx, s_y, s_out, g) = entry
        s_noise.copy_(noise)
        if s_lat is not None:
            s_lat.copy_(start_from_latent)
        s_ctx.copy_(self._cat_ctx(cond, uncond, b,
                                  cfg_scale != 1.0 and uncond is not None))
        if s_y is not None:
            s_y.copy_(self._cat_y(cond, uncond, b,
                                  cfg_scale != 1.0 and uncond is not None))
        g.replay()
        # clone: the static output buffer is overwritten by the next replay
        return s_out.clone()

    def _cat_ctx(self, cond, uncond, b, need_cfg):
        ctx = cond["context"].expand(b, -1, -1)
        if need_cfg:
            ctx = torch.cat([ctx, uncond["context"].expand(b, -1, -1)])
        return ctx

    def _cat_y(self, cond, uncond, b, need_cfg):
        y = cond.get("y")
        if y is None:
            return None
        y = y.expand(b, -1)
        if need_cfg:
            y = torch.cat([y, uncond["y"].expand(b, -1)])
        return y

    def _capture(self, key, cond, uncond, cfg_scale, noise, sigs, sampler,
                 start_from_latent):
        b = noise.shape[0]
        need_cfg = cfg_scale != 1.0 and uncond is not None
        s_noise = noise.detach().float().clone()
        s_lat = (None if start_from_latent is None
                 else start_from_latent.detach().float().clone())
        s_ctx = self._cat_ctx(cond, uncond, b, need_cfg).detach().clone()
        s_y = self._cat_y(cond, uncond, b, need_cfg)
        s_y = None if s_y is None else s_y.detach().clone()
        # timestep tensors per schedule position: constants of the capture
        dev = noise.device
        tb = 2 * b if need_cfg else b
        sig_t = torch.tensor(sigs[:-1], dtype=torch.float32)
        t_vals = self.schedule.timestep(sig_t)
        t_ts = [torch.full((tb,), float(t), device=dev) for t in t_vals]
        # heun/dpm_2 evaluate at a second sigma inside the step; map those
        # to their own timestep tensors lazily
        aux_cache: dict[float, torch.Tensor] = {}

        def t_for(sig_val: float):
            t = aux_cache.get(sig_val)
            if t is None:
                tv = self.schedule.timestep(
                    torch.tensor([sig_val], dtype=torch.float32))
                t = torch.full((tb,), float(tv[0]), device=dev)
                aux_cache[sig_val] = t
            return t

        def denoise_at(x, i, sig, aux=False):
            t_tensor = t_for(sig) if aux else t_ts[i]
            return self._denoise_const(x, i, None, s_ctx, s_y, cfg_scale,
                                       need_cfg, sig, t_tensor)

        def fwd():
            x = (s_lat + s_noise * sigs[0] if s_lat is not None
                 else s_noise * sigs[0])
            return self._loop_body(sampler, x, sigs, denoise_at)

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                fwd()
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            s_out = fwd()
        return (s_noise, s_lat, s_ctx, s_y, s_out, g)


class CFGDenoiser:
    """eps-model + classifier-free guidance -> denoised prediction x0."""

    def __init__(self, unet, schedule: NoiseSchedule, cond, uncond, cfg_scale: float,
                 use_graph: bool = True):
        if use_graph:
            # one GraphedModel per underlying module, cached across jobs so
            # capture cost is paid once per shape
            wrapper = getattr(unet, "_graphed_wrapper", None)
            if wrapper is None or wrapper.model is not unet:
                wrapper = GraphedModel(unet)
                object.__setattr__(unet, "_graphed_wrapper", wrapper)
            self.unet = wrapper
        else:
            self.unet = unet
        self.schedule = schedule
        self.cond = cond
        self.uncond = uncond
        self.cfg_scale = cfg_scale

    def __call__(self, x: torch.Tensor, sigma: torch.Tensor) -> torch.Tensor:
        b = x.shape[0]
        sig = sigma.reshape(-1).to(x.device)
        if sig.numel() == 1:
            sig = sig.expand(b)
        t = self.schedule.timestep(sig)
        # eps parameterization: x_t = x0 + sigma * eps, model input is
        # x_t / sqrt(1 + sigma^2) (the "v-scaling" of discrete eps models).
        c_in = (1.0 / (1.0 + sig**2).sqrt()).reshape(-1, 1, 1, 1).to(x.dtype)
        # the UNet casts inputs to its own parameter dtype (bf16 on GPU)
        need_cfg = self.cfg_scale != 1.0 and self.uncond is not None
        if need_cfg:
            x_in = torch.cat([x * c_in] * 2)
            t_in = torch.cat([t] * 2)
            ctx = torch.cat([self.cond["context"].expand(b, -1, -1),
                             self.uncond["context"].expand(b, -1, -1)])
            y = None
            if self.cond.get("y") is not None:
                y = torch.cat([self.cond["y"].expand(b, -1),
                               self.uncond["y"].expand(b, -1)])
            eps = self.unet(x_in, t_in, ctx, y=y)
            eps_c, eps_u = eps.chunk(2)
            eps = eps_u + self.cfg_scale * (eps_c - eps_u)
        else:
            ctx = self.cond["context"].expand(b, -1, -1)
            y = self.cond.get("y")
            if y is not None:
                y = y.expand(b, -1)
            eps = self.unet(x * c_in, t, ctx, y=y)
        return x - eps.float() * sig.reshape(-1, 1, 1, 1)


class _MaskedDenoiser:
    """Inpainting wrapper: outside the mask the latent is pinned to the
    original re-noised at the current sigma before every model call
    (ComfyUI's latent-preservation behavior for LATENT noise_mask)."""

    def __init__(self, inner, orig, noise, mask):
        self.inner = inner
        self.orig = orig
        self.noise = noise
        self.mask = mask

    def __call__(self, x, sigma):
        keep = self.orig + self.noise * sigma
        x = x * self.mask + keep * (1 - self.mask)
        return self.inner(x, sigma)


class _InterruptibleDenoiser:
    """Checks the user-interrupt flag before every model call, so a long
    sampler loop aborts promptly in every mode (ComfyUI checks
    model_management interrupts per step the same way)."""

    def __init__(self, inner, runtime):
        self.inner = inner
        self.runtime = runtime

    def __call__(self, x, sigma):
        self.runtime.throw_if_interrupted()
        return self.inner(x, sigma)


def sample(denoiser, noise_or_latent: torch.Tensor, sigmas: torch.Tensor,
           sampler: str = "euler", seed: int | None = None,
           start_from_latent: torch.Tensor | None = None,
           denoise_mask: torch.Tensor | None = None) -> torch.Tensor:
    """Run the sampler loop. ``noise_or_latent`` is pure noise for txt2img;
    for img2img pass ``start_from_latent`` and the noised start is formed
    here as latent + noise * sigmas[0]. ``denoise_mask`` ([B or 1, 1, h, w],
    1 = denoise, 0 = keep) enables inpainting: the unmasked region of
    ``start_from_latent`` is preserved exactly."""
    x = noise_or_latent.float() * sigmas[0]
    if start_from_latent is not None:
        x = start_from_latent.float() + noise_or_latent.float() * sigmas[0]
    if denoise_mask is not None:
        if start_from_latent is None:
            raise ValueError("denoise_mask needs start_from_latent")
        m = denoise_mask.to(x.device, torch.float32)
        denoiser = _MaskedDenoiser(denoiser, start_from_latent.float(),
                                   noise_or_latent.float(), m)
    gen = None
    if seed is not None:
        gen = torch.Generator(device="cpu").manual_seed(seed)
    from ..nodes.runtime import get_runtime

    # whole-loop hipGraph fast path: deterministic samplers on GPU with a
    # plain CFGDenoiser replay the full loop as one graph (interrupt is
    # checked once per call — per-chunk abort granularity)
    import os as _os

    if (
        isinstance(denoiser, CFGDenoiser)
        and denoise_mask is None
        and noise_or_latent.is_cuda
        and sampler in GraphedSamplerLoop.SUPPORTED
        and _os.environ.get("DISTGPU_GRAPH_LOOP", "1") == "1"
    ):
        get_runtime().throw_if_interrupted()
        raw = getattr(denoiser.unet, "model", denoiser.unet)
        loop = getattr(raw, "_loop_graph", None)
        if loop is None or loop.model is not raw:
            loop = GraphedSamplerLoop(raw, denoiser.schedule)
            object.__setattr__(raw, "_loop_graph", loop)
        out = loop.run(denoiser.cond, denoiser.uncond, denoiser.cfg_scale,
                       noise_or_latent.float(), sigmas, sampler,
                       None if start_from_latent is None
                       else start_from_latent.float())
        if out is not None:
            return out

    denoiser = _InterruptibleDenoiser(denoiser, get_runtime())
    if sampler == "euler":
        out = _sample_euler(denoiser, x, sigmas)
    elif sampler == "euler_ancestral":
        out = _sample_euler_ancestral(denoiser, x, sigmas, gen)
    elif sampler == "heun":
        out = _sample_heun(denoiser, x, sigmas)
    elif sampler == "dpm_2":
        out = _sample_dpm_2(denoiser, x, sigmas)
    elif sampler == "dpmpp_2m":
        out = _sample_dpmpp_2m(denoiser, x, sigmas)
    elif sampler == "dpmpp_2m_sde":
        out = _sample_dpmpp_2m_sde(denoiser, x, sigmas, gen)
    else:
        raise ValueError(f"unknown sampler {sampler!r}")
    if denoise_mask is not None:
        m = denoise_mask.to(out.device, torch.float32)
        out = out * m + start_from_latent.float() * (1 - m)
    return out


def _sample_euler(denoiser, x, sigmas):
    for i in range(len(sigmas) - 1):
        sigma = sigmas[i]
        denoised = denoiser(x, sigma)
        d = (x - denoised) / sigma
        x = x + d * (sigmas[i + 1] - sigma)
    return x


def _sample_euler_ancestral(denoiser, x, sigmas, gen):
    for i in range(len(sigmas) - 1):
        sigma, sigma_next = sigmas[i], sigmas[i + 1]
        denoised = denoiser(x, sigma)
        if sigma_next == 0:
            x = denoised
            continue
        sigma_up = (sigma_next**2 * (sigma**2 - sigma_next**2) / sigma**2).sqrt()
        sigma_down = (sigma_next**2 - sigma_up**2).sqrt()
        d = (x - denoised) / sigma
        x = x + d * (sigma_down - sigma)
        noise = torch.randn(x.shape, generator=gen, dtype=torch.float32).to(x.device)
        x = x + noise * sigma_up
    return x


def _sample_heun(denoiser, x, sigmas):
    """Heun's 2nd-order method: Euler predictor + trapezoidal corrector
    (2 model calls per step except the final sigma-0 step)."""
    for i in range(len(sigmas) - 1):
        sigma, sigma_next = sigmas[i], sigmas[i + 1]
        denoised = denoiser(x, sigma)
        d = (x - denoised) / sigma
        dt = sigma_next - sigma
        if sigma_next == 0:
            x = x + d * dt
            continue
        x2 = x + d * dt
        denoised2 = denoiser(x2, sigma_next)
        d2 = (x2 - denoised2) / sigma_next
        x = x + (d + d2) / 2 * dt
    return x


def _sample_dpm_2(denoiser, x, sigmas):
    """DPM-Solver-2: midpoint (log-sigma geometric mean) 2nd-order step."""
    for i in range(len(sigmas) - 1):
        sigma, sigma_next = sigmas[i], sigmas[i + 1]
        denoised = denoiser(x, sigma)
        d = (x - denoised) / sigma
        if sigma_next == 0:
            x = x + d * (sigma_next - sigma)
            continue
        sigma_mid = sigma.log().lerp(sigma_next.log(), 0.5).exp()
        x2 = x + d * (sigma_mid - sigma)
        denoised2 = denoiser(x2, sigma_mid)
        d2 = (x2 - denoised2) / sigma_mid
        x = x + d2 * (sigma_next - sigma)
    return x


def _sample_dpmpp_2m_sde(denoiser, x, sigmas, gen, eta: float = 1.0):
    """DPM++ 2M SDE (midpoint noise addition), seeded like the ancestral
    sampler so distributed results stay deterministic."""
    old_denoised = None
    h_last = None
    for i in range(len(sigmas) - 1):
        sigma, sigma_next = sigmas[i], sigmas[i + 1]
        denoised = denoiser(x, sigma)
        if sigma_next == 0:
            x = denoised
        else:
            t, s = -sigma.log(), -sigma_next.log()
            h = s - t
            eta_h = eta * h
            x = (sigma_next / sigma) * (-eta_h).exp() * x + \
                (-h - eta_h).expm1().neg() * denoised
            if old_denoised is not None:
                r = h_last / h
                x = x + ((-h - eta_h).expm1().neg() / (-h - eta_h) + 1) * \
                    (1 / r) * (denoised - old_denoised)
            if eta > 0:
                noise = torch.randn(x.shape, generator=gen,
                                    dtype=torch.float32).to(x.device)
                x = x + noise * sigma_next * (-2 * eta_h).expm1().neg().sqrt()
            h_last = h
        old_denoised = denoised
    return x


def _sample_dpmpp_2m(denoiser, x, sigmas):
    old_denoised = None
    for i in range(len(sigmas) - 1):
        sigma, sigma_next = sigmas[i], sigmas[i + 1]
        denoised = denoiser(x, sigma)
        t, t_next = -sigma.log(), -sigma_next.log() if sigma_next > 0 else None
        if sigma_next == 0:
            x = denoised
        elif old_denoised is None:
            h = t_next - t
            x = (sigma_next / sigma) * x - (-h).expm1() * denoised
        else:
            h = t_next - t
            h_last = t - (-sigmas[i - 1].log())
            r = h_last / h
            denoised_d = (1 + 1 / (2 * r)) * denoised - (1 / (2 * r)) * old_denoised
            x = (sigma_next / sigma) * x - (-h).expm1() * denoised_d
        old_denoised = denoised
    return x
