from .pipeline import PipelineRequest, PipelineResult, StableDiffusionPipeline
from .samplers import build_sampler, sampler_names
from .schedule import Schedule, discrete_schedule, karras_schedule, schedule_for

__all__ = [
    "PipelineRequest",
    "PipelineResult",
    "StableDiffusionPipeline",
    "build_sampler",
    "sampler_names",
    "Schedule",
    "discrete_schedule",
    "karras_schedule",
    "schedule_for",
]
