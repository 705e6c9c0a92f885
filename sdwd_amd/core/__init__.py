from .eta import EtaPredictor, sampler_cost
from .job import Job
from .seeds import SeedPlan, fix_seed, shard_seeds
from .state import IllegalTransition, State, StateMachine
from .worker import Worker
from .world import GenRequest, World

__all__ = [
    "EtaPredictor",
    "sampler_cost",
    "Job",
    "SeedPlan",
    "fix_seed",
    "shard_seeds",
    "IllegalTransition",
    "State",
    "StateMachine",
    "Worker",
    "GenRequest",
    "World",
]
