"""ETA prediction with rolling mean-percent-error correction.

Capability parity with the reference's worker.py:176-286 (eta/eta_hr/eta_mpe)
and :475-492 (error sampling): predicted seconds scale the rank's measured
canonical speed (images/minute at 512x512, 20 steps) by step count, pixel
count and sampler cost, then regress by the rolling mean percent error of the
last few predictions (window 5, outliers beyond 500% discarded).

Unlike the reference's empirically-fitted sampler-speed table
(worker.py:75-94, measured on unstated hardware), sampler cost here is
derived from the sampler's model-evaluations-per-step — exact for the
UNet-bound regime this engine runs in — and the MPE loop absorbs the rest.
"""
from __future__ import annotations

from typing import Deque, Optional
import collections

MPE_WINDOW = 5
MPE_OUTLIER_PCT = 500.0
CANONICAL_PIXELS = 512 * 512
CANONICAL_STEPS = 20

# model evaluations per sampler step (2nd-order samplers call the UNet twice)
SAMPLER_EVALS_PER_STEP = {
    "Euler": 1.0,
    "Euler a": 1.0,
    "DDIM": 1.0,
    "DDPM": 1.0,
    "LMS": 1.0,
    "DPM++ 2M": 1.0,
    "DPM++ 2M Karras": 1.0,
    "DPM++ 2M SDE": 1.0,
    "DPM++ 2M SDE Karras": 1.0,
    "DPM++ 2M SDE Heun": 1.0,
    "DPM++ 2M SDE Heun Karras": 1.0,
    "DPM++ 2M SDE Heun Exponential": 1.0,
    "DPM++ 3M SDE": 1.0,
    "DPM++ 3M SDE Karras": 1.0,
    "LCM": 1.0,
    "DPM++ SDE": 2.0,
    "DPM++ SDE Karras": 2.0,
    "Heun": 2.0,
    "DPM2": 2.0,
    "DPM2 a": 2.0,
    "DPM++ 2S a": 2.0,
    "DPM++ 2S a Karras": 2.0,
    "UniPC": 1.0,  # corrector eval is reused as the next predictor eval
    "Restart": 2.6,  # Heun cost + ~30% restart overhead at 20 steps
    "DPM fast": 1.0,      # fixed eval budget == steps (ref: +15.5% vs Euler a)
    "DPM adaptive": 2.5,  # error-controlled evals (ref table: -61% vs Euler a)
    "PLMS": 1.0,
    "DPM2 Karras": 2.0,
    "DPM2 a Karras": 2.0,
}


def sampler_cost(sampler_name: str) -> float:
    return SAMPLER_EVALS_PER_STEP.get(sampler_name, 1.0)


class EtaPredictor:
    """Per-rank ETA model fed by the benchmark engine and request outcomes."""

    def __init__(self, avg_ipm: float = 0.0,
                 history: Optional[list] = None) -> None:
        self.avg_ipm = avg_ipm  # canonical images/minute; 0 = unbenchmarked
        self.percent_errors: Deque[float] = collections.deque(
            history or [], maxlen=MPE_WINDOW
        )

    # -- prediction ---------------------------------------------------------
    def eta(
        self,
        images: int,
        steps: int = CANONICAL_STEPS,
        width: int = 512,
        height: int = 512,
        sampler_name: str = "Euler a",
        hr_scale: float = 0.0,
        hr_steps: int = 0,
        correct: bool = True,
    ) -> float:
        """Predicted seconds for this rank to produce ``images`` images.

        hr_scale/hr_steps model a hires-fix second pass at scaled resolution
        (ref worker.py:205-228): hr pass cost = hr_steps at (scale^2) pixels.
        """
        if self.avg_ipm <= 0 or images <= 0:
            return 0.0
        spi_canonical = 60.0 / self.avg_ipm  # seconds per canonical image
        pixel_f = (width * height) / CANONICAL_PIXELS
        step_f = steps / CANONICAL_STEPS
        samp_f = sampler_cost(sampler_name)
        seconds = images * spi_canonical * pixel_f * step_f * samp_f
        if hr_scale and hr_steps:
            hr_pixel_f = pixel_f * hr_scale * hr_scale
            seconds += images * spi_canonical * hr_pixel_f * (
                hr_steps / CANONICAL_STEPS
            ) * samp_f
        if correct:
            seconds *= self.correction_factor()
        return seconds

    # -- error feedback -----------------------------------------------------
    def record_outcome(self, predicted: float, actual: float) -> None:
        """Push one (predicted, actual) pair into the MPE window.

        percent error > 0 means the prediction was too LOW (ran longer).
        Outliers beyond ±500% are discarded (ref worker.py:475-492).
        """
        if predicted <= 0 or actual <= 0:
            return
        pct = (actual - predicted) / predicted * 100.0
        if abs(pct) > MPE_OUTLIER_PCT:
            return
        self.percent_errors.append(pct)

    def reset_errors(self) -> None:
        """Drop the error-correction history (ref 2.0.0 'debug option for
        resetting error correction at runtime')."""
        self.percent_errors.clear()

    def mpe(self) -> float:
        if not self.percent_errors:
            return 0.0
        return sum(self.percent_errors) / len(self.percent_errors)

    def correction_factor(self) -> float:
        return max(0.1, 1.0 + self.mpe() / 100.0)
