"""Deterministic seed plan for sharded batches (ref distributed.py:251-254,
297-305).

The contract: an N-GPU gallery must be image-for-image identical to a 1-GPU
run of the same batch. The master fixes the base seed once, then shard k
covering gallery positions [off, off+n) gets per-image seeds
``base_seed + off + i`` — or, when subseed variation is active
(subseed_strength > 0), a FIXED seed with subseed offsets instead, matching
sdwui semantics where variation walks the subseed.
"""
from __future__ import annotations

import random
from dataclasses import dataclass
from typing import List, Optional


@dataclass
class SeedPlan:
    seeds: List[int]
    subseeds: List[int]
    subseed_strength: float = 0.0


def fix_seed(seed: int) -> int:
    """-1 means 'random': draw once so every shard agrees on the base."""
    if seed is None or int(seed) == -1:
        return random.randrange(0, 2**32 - 1)
    return int(seed)


def shard_seeds(
    base_seed: int,
    offset: int,
    count: int,
    subseed: Optional[int] = None,
    subseed_strength: float = 0.0,
) -> SeedPlan:
    """Seeds for the shard covering gallery slots [offset, offset+count)."""
    if subseed_strength and subseed_strength > 0:
        sub = fix_seed(subseed if subseed is not None else -1)
        return SeedPlan(
            seeds=[base_seed] * count,
            subseeds=[sub + offset + i for i in range(count)],
            subseed_strength=subseed_strength,
        )
    return SeedPlan(
        seeds=[base_seed + offset + i for i in range(count)],
        subseeds=[-1] * count,
        subseed_strength=0.0,
    )
