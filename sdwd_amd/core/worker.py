"""Per-GPU rank descriptor ("worker") — ref worker.py:51-758 re-designed.

Where the reference's Worker wrapped a remote sdwui instance behind an
authenticated HTTPS session (request(), benchmark(), interrupt(), reachable(),
load_options()), this Worker is a GPU rank of the local node: identity is a
device ordinal, the transport is in-process (or RCCL when running one process
per GPU), interrupt is an Event checked between denoise steps, liveness is a
device memory query instead of GET /memory, and model sync is a broadcast
instead of POST /options.
"""
from __future__ import annotations

import threading
from typing import Callable, Optional, Tuple

from ..config import (
    BENCHMARK_TIMED_SAMPLES,
    BENCHMARK_WARMUP_SAMPLES,
    BenchmarkPayload,
    WorkerModel,
)
from ..utils import get_logger
from .eta import EtaPredictor
from .state import State, StateMachine

log = get_logger("worker")

# A benchmark runner executes the canonical payload on a device and returns
# elapsed seconds; injected by the engine so core stays model-agnostic.
BenchmarkRunner = Callable[["Worker", BenchmarkPayload], float]


class Worker:
    def __init__(
        self,
        label: str,
        device: int = 0,
        avg_ipm: float = 0.0,
        pixel_cap: int = 0,
        model_override: Optional[str] = None,
        disabled: bool = False,
        is_master: bool = False,
    ) -> None:
        self.label = label
        self.device = device
        self.pixel_cap = pixel_cap
        self.model_override = model_override
        self.is_master = is_master
        self.sm = StateMachine(State.DISABLED if disabled else State.IDLE)
        self.eta = EtaPredictor(avg_ipm=avg_ipm)
        self.interrupt_event = threading.Event()
        self.loaded_model: Optional[str] = None  # ref worker.py loaded_model cache
        self.loaded_vae: Optional[str] = None

    # -- state --------------------------------------------------------------
    @property
    def state(self) -> State:
        return self.sm.state

    def set_state(self, new: State, strict: bool = False) -> bool:
        ok = self.sm.set(new, strict=strict)
        if ok and new is State.UNAVAILABLE:
            # force model resync on recovery (ref worker.py:747-755)
            self.loaded_model = None
            self.loaded_vae = None
        return ok

    @property
    def available(self) -> bool:
        return self.state in (State.IDLE, State.WORKING, State.INTERRUPTED)

    # -- interrupt (ref worker.py:595-603) -----------------------------------
    def interrupt(self) -> None:
        self.interrupt_event.set()
        self.set_state(State.INTERRUPTED)

    def clear_interrupt(self) -> None:
        self.interrupt_event.clear()
        if self.state is State.INTERRUPTED:
            self.set_state(State.IDLE)

    # -- liveness (ref worker.py:605-621 GET /memory) ------------------------
    def reachable(self) -> bool:
        try:
            free, total = self.memory()
            return total > 0
        except Exception as exc:  # device lost / HIP error
            log.warning("rank %s unreachable: %s", self.label, exc)
            return False

    def memory(self) -> Tuple[int, int]:
        """(free, total) bytes on this rank's device; (0, 0) without a GPU."""
        try:
            import torch

            if torch.cuda.is_available():
                with torch.cuda.device(self.device):
                    return torch.cuda.mem_get_info()
        except Exception:
            pass
        return (0, 0)

    # -- benchmark (ref worker.py:506-575) -----------------------------------
    def benchmark(self, runner: BenchmarkRunner, payload: BenchmarkPayload) -> float:
        """2 warmup + 3 timed samples of the canonical payload -> avg_ipm."""
        if not self.available:
            return self.eta.avg_ipm
        self.set_state(State.WORKING)
        try:
            for _ in range(BENCHMARK_WARMUP_SAMPLES):
                runner(self, payload)
            samples = []
            for _ in range(BENCHMARK_TIMED_SAMPLES):
                elapsed = runner(self, payload)
                if elapsed > 0:
                    samples.append(payload.batch_size * 60.0 / elapsed)
            if samples:
                self.eta.avg_ipm = sum(samples) / len(samples)
                log.info("rank %s benchmark: %.2f ipm", self.label, self.eta.avg_ipm)
        finally:
            self.set_state(State.IDLE)
        return self.eta.avg_ipm

    # -- persistence ---------------------------------------------------------
    def to_model(self) -> WorkerModel:
        return WorkerModel(
            label=self.label,
            device=self.device,
            avg_ipm=self.eta.avg_ipm,
            eta_percent_error=list(self.eta.percent_errors),
            last_mpe=self.eta.mpe() if self.eta.percent_errors else None,
            state=self.state.value,
            disabled=self.state is State.DISABLED,
            pixel_cap=self.pixel_cap,
            model_override=self.model_override,
        )

    @classmethod
    def from_model(cls, m: WorkerModel, is_master: bool = False) -> "Worker":
        w = cls(
            label=m.label,
            device=m.device,
            avg_ipm=m.avg_ipm,
            pixel_cap=m.pixel_cap,
            model_override=m.model_override,
            disabled=m.disabled,
            is_master=is_master,
        )
        w.eta.percent_errors.extend(m.eta_percent_error[-5:])
        return w

    def __repr__(self) -> str:  # pragma: no cover
        return f"Worker({self.label}, dev={self.device}, ipm={self.eta.avg_ipm:.1f}, {self.state.value})"
