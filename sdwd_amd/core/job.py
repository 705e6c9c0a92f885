"""Job model (ref world.py:37-72).

One Job per participating rank and generation: how many gallery images the
rank owes, whether it is complementary (bonus images beyond the requested
batch), an optional per-job step override (step scaling), and the gallery
slots its output lands in.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class Job:
    worker_label: str
    batch_size: int = 0  # images this rank owes
    complementary: bool = False
    step_override: Optional[int] = None
    gallery_offset: int = 0  # first gallery slot (non-complementary jobs)
    seeds: List[int] = field(default_factory=list)
    subseeds: List[int] = field(default_factory=list)
    # measured on completion:
    predicted_eta: float = 0.0
    elapsed: float = 0.0
    # live progress (denoise steps done / total), updated by the executor
    steps_done: int = 0
    steps_total: int = 0

    def add_work(self, images: int, width: int, height: int,
                 pixel_cap: int = 0) -> int:
        """Add up to ``images`` images, honoring the rank's pixel cap
        (ref world.py:62-72). Returns how many were actually accepted."""
        if images <= 0:
            return 0
        if pixel_cap and pixel_cap > 0:
            budget = pixel_cap - self.batch_size * width * height
            affordable = max(0, budget // (width * height))
            images = min(images, int(affordable))
        self.batch_size += images
        return images

    @property
    def empty(self) -> bool:
        return self.batch_size <= 0
