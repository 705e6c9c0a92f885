"""Noise schedule: SD scaled-linear betas -> k-diffusion sigma space.

All samplers run in sigma space (x = x0 + sigma*eps). The UNet is an
epsilon-predictor taking the *scaled* input x/sqrt(sigma^2+1) and the
(possibly fractional, for Karras sigmas) timestep matching sigma.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch

TRAIN_STEPS = 1000
BETA_START = 0.00085
BETA_END = 0.012


def make_sigmas_full() -> torch.Tensor:
    """Per-train-step sigma table, fp64 [1000]."""
    betas = (
        torch.linspace(
            BETA_START**0.5, BETA_END**0.5, TRAIN_STEPS, dtype=torch.float64
        )
        ** 2
    )
    alphas_cum = torch.cumprod(1.0 - betas, dim=0)
    return ((1 - alphas_cum) / alphas_cum).sqrt()


_table_cache: torch.Tensor | None = None


def sigma_for_t(t: float) -> float:
    """Inverse of _timesteps_for: fractional train-timestep -> sigma
    (log-sigma linear interpolation on the train table)."""
    global _table_cache
    if _table_cache is None:
        _table_cache = make_sigmas_full().log()
    logt = _table_cache
    lo = int(t)
    lo = max(0, min(lo, len(logt) - 1))
    hi = min(lo + 1, len(logt) - 1)
    frac = float(t) - lo
    return float((logt[lo] * (1 - frac) + logt[hi] * frac).exp())


@dataclass
class Schedule:
    sigmas: torch.Tensor  # [steps+1] descending, last = 0
    timesteps: torch.Tensor  # [steps] fractional train-step indices

    @property
    def steps(self) -> int:
        return len(self.timesteps)


def _timesteps_for(sigmas: torch.Tensor, table: torch.Tensor) -> torch.Tensor:
    """Map sigmas -> fractional train-timesteps by log-sigma interpolation."""
    log_t = table.log()
    log_s = sigmas.clamp_min(table[0]).log()
    ts = torch.zeros(len(sigmas), dtype=torch.float64)
    for i, ls in enumerate(log_s):
        idx = torch.searchsorted(log_t, ls).clamp(1, len(table) - 1)
        lo, hi = log_t[idx - 1], log_t[idx]
        w = ((ls - lo) / (hi - lo)).clamp(0, 1)
        ts[i] = (idx - 1).double() + w
    return ts


def discrete_schedule(steps: int) -> Schedule:
    """Evenly spaced (leading) train timesteps, as sdwui's default."""
    table = make_sigmas_full()
    idx = torch.linspace(0, TRAIN_STEPS - 1, steps, dtype=torch.float64).round().long()
    idx = idx.flip(0)  # descending t = descending sigma
    sig = table[idx]
    sigmas = torch.cat([sig, torch.zeros(1, dtype=torch.float64)])
    return Schedule(sigmas=sigmas.float(), timesteps=idx.double().float())


def karras_schedule(steps: int, rho: float = 7.0) -> Schedule:
    table = make_sigmas_full()
    smin, smax = float(table[0]), float(table[-1])
    ramp = torch.linspace(0, 1, steps, dtype=torch.float64)
    inv_rho = 1.0 / rho
    sig = (smax**inv_rho + ramp * (smin**inv_rho - smax**inv_rho)) ** rho
    ts = _timesteps_for(sig, table)
    sigmas = torch.cat([sig, torch.zeros(1, dtype=torch.float64)])
    return Schedule(sigmas=sigmas.float(), timesteps=ts.float())


def exponential_schedule(steps: int) -> Schedule:
    """Log-linear sigmas (k-diffusion get_sigmas_exponential)."""
    table = make_sigmas_full()
    smin, smax = float(table[0]), float(table[-1])
    sig = torch.linspace(
        math.log(smax), math.log(smin), steps, dtype=torch.float64
    ).exp()
    ts = _timesteps_for(sig, table)
    sigmas = torch.cat([sig, torch.zeros(1, dtype=torch.float64)])
    return Schedule(sigmas=sigmas.float(), timesteps=ts.float())


def sgm_uniform_schedule(steps: int) -> Schedule:
    """Uniform in train-timestep WITHOUT the final zero-sigma training step
    (sgm convention: endpoints chosen so the last denoise lands on the
    lowest-noise trained step)."""
    table = make_sigmas_full()
    idx = torch.linspace(
        TRAIN_STEPS - 1, 0, steps + 1, dtype=torch.float64
    )[:-1].round().long()
    sig = table[idx]
    sigmas = torch.cat([sig, torch.zeros(1, dtype=torch.float64)])
    return Schedule(sigmas=sigmas.float(), timesteps=idx.double().float())


def polyexponential_schedule(steps: int, rho: float = 1.0) -> Schedule:
    """Log-linear ramp raised to rho (k-diffusion
    get_sigmas_polyexponential; rho=1 equals Exponential)."""
    table = make_sigmas_full()
    smin, smax = float(table[0]), float(table[-1])
    ramp = torch.linspace(1, 0, steps, dtype=torch.float64) ** rho
    sig = (ramp * (math.log(smax) - math.log(smin)) + math.log(smin)).exp()
    ts = _timesteps_for(sig, table)
    sigmas = torch.cat([sig, torch.zeros(1, dtype=torch.float64)])
    return Schedule(sigmas=sigmas.float(), timesteps=ts.float())


def kl_optimal_schedule(steps: int) -> Schedule:
    """KL-optimal spacing (arctan-uniform; sdwui "KL Optimal",
    from "Align Your Steps"' analysis)."""
    table = make_sigmas_full()
    smin, smax = float(table[0]), float(table[-1])
    t = torch.linspace(
        math.atan(smax), math.atan(smin), steps, dtype=torch.float64
    )
    sig = torch.tan(t)
    ts = _timesteps_for(sig, table)
    sigmas = torch.cat([sig, torch.zeros(1, dtype=torch.float64)])
    return Schedule(sigmas=sigmas.float(), timesteps=ts.float())


SCHEDULERS = {
    "automatic": None,  # resolved from the sampler name
    "uniform": discrete_schedule,
    "karras": karras_schedule,
    "exponential": exponential_schedule,
    "sgm uniform": sgm_uniform_schedule,
    "sgm_uniform": sgm_uniform_schedule,
    "polyexponential": polyexponential_schedule,
    "kl optimal": kl_optimal_schedule,
    "kl_optimal": kl_optimal_schedule,
}


def scheduler_names() -> list:
    return ["Automatic", "Uniform", "Karras", "Exponential", "SGM Uniform",
            "Polyexponential", "KL Optimal"]


def schedule_for(
    sampler_name: str, steps: int, scheduler: str = "Automatic"
) -> Schedule:
    """sdwui scheduler dropdown semantics: an explicit scheduler wins;
    "Automatic" keeps the sampler-name convention ("... Karras")."""
    key = (scheduler or "Automatic").strip().lower()
    fn = SCHEDULERS.get(key)
    if fn is not None:
        return fn(steps)
    if key not in SCHEDULERS:
        raise KeyError(f"unknown scheduler '{scheduler}'")
    if "Karras" in sampler_name:
        return karras_schedule(steps)
    if "Exponential" in sampler_name:
        return exponential_schedule(steps)
    return discrete_schedule(steps)
