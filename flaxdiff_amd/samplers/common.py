"""Sampler base: CFG-batched model evaluation + timestep spacing + sample loop.

Math contract: reference /root/reference/flaxdiff/samplers/common.py:17-433.
MI355X execution notes:
  * the per-step body (one CFG-doubled model forward + a few AXPYs) is
    captured in a hipGraph by `enable_graph_capture()` and replayed per step
    with the timestep fed through a device buffer — sampling becomes
    launch-overhead-free (SURVEY.md §2.5 plan);
  * everything runs under torch.no_grad().
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..predictors import DiffusionPredictionTransform
from ..schedulers import NoiseScheduler
from ..utils import MarkovState, RandomMarkovState, clip_images


class DiffusionSampler:
    """Base class for diffusion samplers (reference samplers/common.py:17)."""

    def __init__(self,
                 model,
                 noise_schedule: NoiseScheduler,
                 model_output_transform: DiffusionPredictionTransform,
                 input_config=None,
                 guidance_scale: float = 0.0,
                 autoencoder=None,
                 timestep_spacing: str = "linear"):
        self.model = model
        self.noise_schedule = noise_schedule
        self.model_output_transform = model_output_transform
        self.guidance_scale = guidance_scale
        self.autoencoder = autoencoder
        self.timestep_spacing = timestep_spacing
        self.input_config = input_config

        self.unconditionals = (input_config.get_unconditionals()
                               if input_config is not None else [])

        if hasattr(noise_schedule, "min_inv_rho"):
            self.min_inv_rho = noise_schedule.min_inv_rho
            self.max_inv_rho = noise_schedule.max_inv_rho

        self._graph_capture = False
        self._graphed: Optional["_GraphedSampleModel"] = None

    # _uncond_cache holds device-resident copies of the null embeddings (an
    # H2D copy inside hipGraph capture would be illegal). Reassigning
    # unconditionals (DiffusionTrainer.validation_sample does this after
    # construction) must invalidate the cache, so it is a managed property.
    @property
    def unconditionals(self):
        return self._unconditionals

    @unconditionals.setter
    def unconditionals(self, value):
        self._unconditionals = value
        self._uncond_cache = {}

    def enable_graph_capture(self, enabled: bool = True):
        """Capture the per-step CFG-doubled model forward in a hipGraph and
        replay it each step (timesteps fed through a device buffer). The
        per-sampler take_next_step AXPYs stay eager — they are a few
        elementwise kernels on the sample tensor. No-op without a GPU."""
        self._graph_capture = enabled and torch.cuda.is_available()
        if not enabled:
            self._graphed = None
        return self

    # ------------------------------------------------------------------
    # model evaluation (CFG batch doubling: reference common.py:60-103)
    # ------------------------------------------------------------------
    @torch.no_grad()
    def sample_model(self, x_t: torch.Tensor, t: torch.Tensor, *conditioning_inputs):
        ns = self.noise_schedule
        tr = self.model_output_transform
        if self.guidance_scale > 0:
            x_cat = torch.cat([x_t, x_t], dim=0)
            t_cat = torch.cat([t, t], dim=0)
            rates_cat = ns.get_rates(t_cat)
            c_in = tr.get_input_scale(tuple(r.to(x_cat.device) for r in rates_cat))
            if torch.is_tensor(c_in):
                c_in = c_in.to(x_cat.dtype)
            finals = []
            for i, (cond, uncond) in enumerate(zip(conditioning_inputs,
                                                   self.unconditionals)):
                # cache the null embedding device-resident: an H2D copy here
                # would be illegal inside hipGraph capture
                key = (i, cond.device, cond.dtype)
                uncond_dev = self._uncond_cache.get(key)
                if uncond_dev is None:
                    uncond_dev = torch.as_tensor(uncond, device=cond.device,
                                                 dtype=cond.dtype)
                    self._uncond_cache[key] = uncond_dev
                finals.append(torch.cat(
                    [cond, uncond_dev.broadcast_to(cond.shape)], dim=0))
            xin, tin = ns.transform_inputs(x_cat * c_in, t_cat)
            out = self.model(xin, tin.to(x_cat.device) if torch.is_tensor(tin) else tin, *finals)
            out_cond, out_uncond = out.chunk(2, dim=0)
            out = out_uncond + self.guidance_scale * (out_cond - out_uncond)
        else:
            rates = ns.get_rates(t)
            c_in = tr.get_input_scale(tuple(r.to(x_t.device) for r in rates))
            if torch.is_tensor(c_in):
                c_in = c_in.to(x_t.dtype)
            xin, tin = ns.transform_inputs(x_t * c_in, t)
            out = self.model(xin, tin.to(x_t.device) if torch.is_tensor(tin) else tin,
                             *conditioning_inputs)
        x_0, eps = tr(x_t.float(), out.float(), t, ns)
        return x_0, eps, out

    def post_process(self, samples: torch.Tensor) -> torch.Tensor:
        if self.autoencoder is not None:
            samples = self.autoencoder.decode(samples)
        return clip_images(samples)

    # ------------------------------------------------------------------
    # single step (reference common.py:117-157)
    # ------------------------------------------------------------------
    def sample_step(self, sample_model_fn, current_samples, current_step,
                    model_conditioning_inputs, next_step=None,
                    state: RandomMarkovState = None):
        B = current_samples.shape[0]
        dev = current_samples.device
        step_ones = torch.ones(B, device=dev, dtype=torch.float32)
        cur = step_ones * float(current_step)
        nxt = step_ones * float(next_step)
        pred_images, pred_noise, _ = sample_model_fn(current_samples, cur,
                                                     *model_conditioning_inputs)
        return self.take_next_step(
            current_samples=current_samples, reconstructed_samples=pred_images,
            pred_noise=pred_noise, current_step=cur, next_step=nxt, state=state,
            model_conditioning_inputs=model_conditioning_inputs,
            sample_model_fn=sample_model_fn)

    def take_next_step(self, current_samples, reconstructed_samples,
                       model_conditioning_inputs, pred_noise, current_step,
                       state: RandomMarkovState, sample_model_fn, next_step=1):
        raise NotImplementedError

    # ------------------------------------------------------------------
    # timestep spacing (reference common.py:178-245)
    # ------------------------------------------------------------------
    def scale_steps(self, steps):
        scale_factor = self.noise_schedule.max_timesteps / 1000
        return steps * scale_factor

    def get_steps(self, start_step, end_step, diffusion_steps):
        step_range = start_step - end_step
        if not diffusion_steps:
            diffusion_steps = step_range
        diffusion_steps = min(diffusion_steps, step_range)

        spacing = getattr(self, "timestep_spacing", "linear")
        if spacing == "quadratic":
            s = torch.linspace(0, 1, diffusion_steps) ** 2
            steps = ((start_step - end_step) * s + end_step).flip(0)
        elif spacing == "karras":
            # NOTE: for KarrasVE/EDM schedulers the scheduler's own get_sigmas
            # already applies the rho ramp, so LINEAR spacing is the Karras
            # schedule; this option re-warps the step domain (reference
            # samplers/common.py:214-230). end_step=0 would make sigma_min=0
            # and the log-space degenerate (every interior point -inf), so
            # floor it at one step.
            sigma_min = max(float(end_step), 1.0) / start_step
            rho = 7.0
            sigmas = torch.exp(torch.linspace(torch.log(torch.tensor(1.0)),
                                              torch.log(torch.tensor(float(sigma_min))),
                                              diffusion_steps))
            steps = torch.clamp(
                (sigmas ** (1 / rho) - self.min_inv_rho) / (self.max_inv_rho - self.min_inv_rho),
                0, 1) * start_step
        elif spacing == "exponential":
            s = torch.linspace(0, 1, diffusion_steps)
            steps = torch.exp(s * torch.log(torch.tensor((start_step + 1) / (end_step + 1)))) \
                * (end_step + 1) - 1
            steps = torch.clamp(steps, end_step, start_step).flip(0)
        else:  # linear
            steps = torch.linspace(end_step, start_step, diffusion_steps).flip(0)
        steps = steps.round().long()
        # Repeated consecutive steps would give dtau == 0 in the ODE samplers
        # (division by zero -> NaN); only degenerate spacings produce them.
        return torch.unique_consecutive(steps)

    # ------------------------------------------------------------------
    # sample loop (reference common.py:248-389)
    # ------------------------------------------------------------------
    @torch.no_grad()
    def generate_samples(self,
                         num_samples: int,
                         resolution: int,
                         sequence_length: Optional[int] = None,
                         diffusion_steps: int = 1000,
                         start_step: Optional[int] = None,
                         end_step: int = 0,
                         steps_override=None,
                         priors=None,
                         rngstate: Optional[RandomMarkovState] = None,
                         conditioning: Optional[List] = None,
                         model_conditioning_inputs: Optional[Tuple] = None,
                         device=None,
                         dtype=torch.float32,
                         progress: bool = False) -> torch.Tensor:
        if rngstate is None:
            rngstate = RandomMarkovState(42)
        if device is None:
            device = (next(self.model.parameters()).device
                      if hasattr(self.model, "parameters") else "cpu")
        if start_step is None:
            # Callers operate in the 1000-step convention; scale_steps maps to
            # the scheduler range (reference samplers/common.py:178-181 — the
            # reference README's inference passes start_step=1000 explicitly).
            # Continuous schedulers report max_timesteps == 1, so using it as
            # the step range would collapse sampling to a single model eval.
            mt = self.noise_schedule.max_timesteps
            start_step = 1000 if (isinstance(mt, (int, float)) and mt <= 1) \
                else int(mt)

        if priors is None:
            rngstate, key = rngstate.get_random_key()
            samples = self._get_initial_samples(resolution, num_samples, sequence_length,
                                                key, start_step, device, dtype)
        else:
            if self.autoencoder is not None:
                priors = self.autoencoder.encode(priors)
            samples = priors.to(device=device, dtype=dtype)

        if conditioning is not None:
            if model_conditioning_inputs is not None:
                raise ValueError("Cannot provide both conditioning and model_conditioning_inputs")
            model_conditioning_inputs = self.input_config.encode_conditions(
                conditioning, device=device, dtype=dtype)
        if model_conditioning_inputs is None:
            if self.unconditionals:
                # unconditional generation of a conditional model: feed the
                # null embeddings (reference samplers/common.py:315-349)
                model_conditioning_inputs = tuple(
                    torch.as_tensor(u, device=device).unsqueeze(0)
                    .expand(num_samples, *u.shape).to(dtype)
                    for u in self.unconditionals)
            else:
                model_conditioning_inputs = ()

        if self._graph_capture and samples.is_cuda:
            if (self._graphed is None
                    or not self._graphed.matches(samples, model_conditioning_inputs)):
                self._graphed = _GraphedSampleModel(self, samples,
                                                   model_conditioning_inputs)
            else:
                # Same shapes as the captured graph, possibly different
                # conditioning contents (new prompts on a cached sampler):
                # refresh the static conditioning buffers before replaying.
                self._graphed.update_conds(model_conditioning_inputs)

            def sample_model_fn(x_t, t, *cond):
                return self._graphed(x_t, t)
        else:
            def sample_model_fn(x_t, t, *cond):
                return self.sample_model(x_t, t, *cond)

        steps = steps_override if steps_override is not None else \
            self.get_steps(start_step, end_step, diffusion_steps)

        it = range(len(steps))
        if progress:
            import tqdm
            it = tqdm.tqdm(it)
        for i in it:
            current_step = self.scale_steps(float(steps[i]))
            next_step = self.scale_steps(float(steps[i + 1]) if i + 1 < len(steps) else 0.0)
            if i != len(steps) - 1:
                samples, rngstate = self.sample_step(
                    sample_model_fn=sample_model_fn, current_samples=samples,
                    current_step=current_step, next_step=next_step,
                    model_conditioning_inputs=model_conditioning_inputs,
                    state=rngstate)
            else:
                step_ones = torch.ones(samples.shape[0], device=samples.device)
                samples, _, _ = sample_model_fn(samples, step_ones * current_step,
                                                *model_conditioning_inputs)
        return self.post_process(samples)

    generate_images = generate_samples

    # ------------------------------------------------------------------
    def _noise_parameters(self, resolution, start_step):
        start_step = self.scale_steps(start_step)
        steps = torch.tensor([float(start_step)])
        alpha_n, sigma_n = self.noise_schedule.get_rates(steps)
        variance = torch.sqrt(alpha_n ** 2 + sigma_n ** 2).item()
        image_size = resolution
        image_channels = 3
        if self.autoencoder is not None:
            image_size = image_size // self.autoencoder.downscale_factor
            image_channels = self.autoencoder.latent_channels
        return variance, image_size, image_channels

    def _get_initial_samples(self, resolution, batch_size, sequence_length, key,
                             start_step, device, dtype):
        variance, image_size, image_channels = self._noise_parameters(resolution, start_step)
        if sequence_length is not None:
            shape = (batch_size, sequence_length, image_size, image_size, image_channels)
        else:
            shape = (batch_size, image_size, image_size, image_channels)
        return key.normal(shape, device=device).to(dtype) * variance


class _GraphedSampleModel:
    """hipGraph-captured sample_model: static input/cond/output buffers, one
    graph replay per model evaluation (SURVEY.md §2.5 MI355X plan).

    Conditioning tensors live in static buffers refreshed via update_conds()
    at the start of every generate_samples call (constant across the sampling
    loop, so one copy per call is capture-safe); x_t and t stream through
    static buffers each call.
    """

    def __init__(self, sampler: DiffusionSampler, x: torch.Tensor,
                 conds: Tuple[torch.Tensor, ...]):
        self.sampler = sampler
        self.shape = tuple(x.shape)
        self.dtype = x.dtype
        dev = x.device
        self.static_x = torch.zeros_like(x)
        self.static_t = torch.zeros(x.shape[0], device=dev, dtype=torch.float32)
        self.static_conds = tuple(c.clone() for c in conds)

        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):  # warmup allocations outside the graph
                sampler.sample_model(self.static_x, self.static_t,
                                     *self.static_conds)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out_x0, self.out_eps, self.out_raw = sampler.sample_model(
                self.static_x, self.static_t, *self.static_conds)

    def matches(self, x: torch.Tensor, conds) -> bool:
        return (tuple(x.shape) == self.shape and x.dtype == self.dtype
                and len(conds) == len(self.static_conds)
                and all(tuple(c.shape) == tuple(s.shape) and c.dtype == s.dtype
                        for c, s in zip(conds, self.static_conds)))

    def update_conds(self, conds):
        for s, c in zip(self.static_conds, conds):
            s.copy_(c)

    def __call__(self, x_t: torch.Tensor, t: torch.Tensor):
        self.static_x.copy_(x_t)
        self.static_t.copy_(t)
        self.graph.replay()
        # out_raw is cloned too: the next replay overwrites the static output
        # in place, so an unprotected reference would be silently corrupted.
        return self.out_x0.clone(), self.out_eps.clone(), self.out_raw.clone()
