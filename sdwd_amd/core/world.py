"""World — the benchmark-weighted in-process batch sharder.

Capability parity with the reference's world.py:75-811 (worker registry,
make_jobs, optimize_jobs with stall-based deferral / redistribution /
remainder round-robin / complementary production / step scaling, benchmark
orchestration, config persistence, liveness sweeps, interrupts), re-designed
for one node of identical-image GPUs: a "worker" is a GPU rank, dispatch is
in-process (or RCCL collectives under torchrun), and failed shards are
REQUEUED onto surviving ranks instead of silently dropped (the reference
lost them — worker.py:498-500).

On homogeneous ranks the optimizer reduces to an equal split plus
remainder round-robin; the heterogeneous logic (deferral, complementary
jobs, step scaling, pixel caps) is kept and unit-tested with synthetic
speed skews.
"""
from __future__ import annotations

import concurrent.futures
import threading
from dataclasses import dataclass
from typing import Dict, List, Optional

from ..config import (
    BenchmarkPayload,
    ConfigModel,
    SettingsModel,
    load_config,
    save_config,
)
from ..utils import get_logger

from .job import Job
from .seeds import fix_seed, shard_seeds
from .state import State
from .worker import BenchmarkRunner, Worker

log = get_logger("world")


@dataclass
class GenRequest:
    """The slice of a generation request the scheduler cares about."""

    task: str = "txt2img"  # or "img2img" (per-task distribution toggles)
    batch_size: int = 1
    width: int = 512
    height: int = 512
    steps: int = 20
    sampler_name: str = "Euler a"
    seed: int = -1
    subseed: int = -1
    subseed_strength: float = 0.0
    hr_scale: float = 0.0
    hr_steps: int = 0


class World:
    def __init__(
        self,
        config_path: Optional[str] = None,
        settings: Optional[SettingsModel] = None,
    ) -> None:
        self._workers: List[Worker] = []
        self.config_path = config_path
        self.settings = settings or SettingsModel()
        self.benchmark_payload = BenchmarkPayload()
        self._lock = threading.Lock()
        self.interrupted = threading.Event()

    # -- registry (ref world.py:91, 146-171, 405-416) ------------------------
    def add_worker(self, worker: Worker) -> Worker:
        with self._lock:
            self._workers = [w for w in self._workers if w.label != worker.label]
            self._workers.append(worker)
        return worker

    def remove_worker(self, label: str) -> None:
        with self._lock:
            self._workers = [w for w in self._workers if w.label != label]

    def get_worker(self, label: str) -> Optional[Worker]:
        for w in self._workers:
            if w.label == label:
                return w
        return None

    @property
    def workers(self) -> List[Worker]:
        return list(self._workers)

    def active_workers(self) -> List[Worker]:
        """Ranks eligible for new jobs (ref world.py:405-416): available and,
        in thin-client mode, not the master."""
        out = []
        for w in self._workers:
            if not w.available:
                continue
            if self.settings.thin_client and w.is_master:
                continue
            out.append(w)
        return out

    def default_batch_size(self) -> int:
        """ref world.py:111-115 — one image per active rank by default."""
        return max(1, len(self.active_workers()))

    # -- scheduling ----------------------------------------------------------
    def make_jobs(self, request: GenRequest) -> List[Job]:
        """Build one Job per active rank and run the optimizer
        (ref world.py:378-392 + 418-601)."""
        jobs = self.optimize_jobs(request)
        self._assign_seeds(jobs, request)
        return jobs

    def optimize_jobs(
        self, request: GenRequest, allow_complementary: bool = True
    ) -> List[Job]:
        workers = self.active_workers()
        if not workers:
            raise RuntimeError("no available ranks")
        total = int(request.batch_size)

        distribute = (
            self.settings.distribute_img2img
            if request.task == "img2img"
            else self.settings.distribute_txt2img
        )
        if not distribute or len(workers) == 1:
            # single-rank mode: the whole batch on the first available rank
            job = Job(worker_label=workers[0].label, batch_size=total)
            job.predicted_eta = self._predict(workers[0], total, request)
            self._log_distribution([job], request)
            return [job]

        benchmarked = all(w.eta.avg_ipm > 0 for w in workers)
        if not benchmarked or len(workers) == 1:
            plan = self._equal_split(workers, total, request)
        else:
            plan = self._weighted_split(workers, total, request)

        # pixel caps: clamp each job and requeue the overflow round-robin
        self._apply_pixel_caps(plan, workers, request)

        # complementary production for deferred ranks
        if self.settings.complement_production and allow_complementary:
            self._add_complementary(plan, workers, request)

        jobs = [j for j in plan.values() if not j.empty]
        self._log_distribution(jobs, request)
        return jobs

    # the split helpers return {label: Job}
    def _equal_split(
        self, workers: List[Worker], total: int, request: GenRequest
    ) -> Dict[str, Job]:
        plan = {w.label: Job(worker_label=w.label) for w in workers}
        n = len(workers)
        base, rem = divmod(total, n)
        order = sorted(workers, key=lambda w: -w.eta.avg_ipm)  # fastest first
        for i, w in enumerate(order):
            plan[w.label].batch_size = base + (1 if i < rem else 0)
        return plan

    def _weighted_split(
        self, workers: List[Worker], total: int, request: GenRequest
    ) -> Dict[str, Job]:
        """Benchmark-weighted shard sizing with stall-based deferral
        (ref world.py:418-557)."""
        realtime = list(workers)
        # Defer ranks whose single-image ETA would stall the rest beyond
        # job_timeout (ref job_stall, world.py:363-376). The candidate is
        # the rank with the largest SINGLE-IMAGE time: the proportional
        # split hands slow ranks near-zero shares, so ranking candidates
        # by their share's eta (as round 1 did) never surfaces them and
        # complementary production silently never fired.
        while len(realtime) > 1:
            worst = max(
                realtime, key=lambda w: self._predict(w, 1, request)
            )
            one_img = self._predict(worst, 1, request)
            others = [w for w in realtime if w is not worst]
            t_without = max(
                self._predict(w, s, request)
                for w, s in zip(
                    others, self._proportional(others, total).values()
                )
            )
            if one_img > t_without + self.settings.job_timeout:
                realtime.remove(worst)  # deferred -> complementary candidate
                continue
            break

        plan = {w.label: Job(worker_label=w.label) for w in workers}
        shares = self._proportional(realtime, total)
        for w in realtime:
            plan[w.label].batch_size = shares[w.label]
            plan[w.label].predicted_eta = self._predict(
                w, shares[w.label], request
            )
        for w in workers:
            if w not in realtime:
                plan[w.label].complementary = True
        return plan

    def _proportional(self, workers: List[Worker], total: int) -> Dict[str, int]:
        """Integer speed-proportional split; remainder round-robin to the
        fastest ranks (ref world.py:482-510)."""
        weights = [max(w.eta.avg_ipm, 1e-6) for w in workers]
        wsum = sum(weights)
        shares = {
            w.label: int(total * wt / wsum) for w, wt in zip(workers, weights)
        }
        assigned = sum(shares.values())
        order = sorted(workers, key=lambda w: -w.eta.avg_ipm)
        i = 0
        while assigned < total:
            shares[order[i % len(order)].label] += 1
            assigned += 1
            i += 1
        return shares

    def _apply_pixel_caps(
        self, plan: Dict[str, Job], workers: List[Worker], request: GenRequest
    ) -> None:
        by_label = {w.label: w for w in workers}
        overflow = 0
        for job in plan.values():
            w = by_label[job.worker_label]
            if w.pixel_cap and job.batch_size * request.width * request.height > w.pixel_cap:
                affordable = w.pixel_cap // (request.width * request.height)
                overflow += job.batch_size - affordable
                job.batch_size = int(affordable)
        if overflow <= 0:
            return
        order = sorted(workers, key=lambda w: -w.eta.avg_ipm)
        i, spins = 0, 0
        while overflow > 0 and spins < 10_000:
            w = order[i % len(order)]
            job = plan[w.label]
            cap_ok = (
                not w.pixel_cap
                or (job.batch_size + 1) * request.width * request.height
                <= w.pixel_cap
            )
            if cap_ok and not job.complementary:
                job.batch_size += 1
                overflow -= 1
            i += 1
            spins += 1
        # every realtime rank capped out: spill onto deferred ranks (their
        # jobs were marked complementary) and PROMOTE them — these images
        # are part of the requested batch and must own real gallery slots,
        # arriving late beats not arriving (batch conservation is what the
        # seed plan and the gather math are built on)
        if overflow > 0:
            for w in sorted(workers, key=lambda w: -w.eta.avg_ipm):
                job = plan[w.label]
                if not job.complementary:
                    continue
                while overflow > 0:
                    if (
                        w.pixel_cap
                        and (job.batch_size + 1)
                        * request.width * request.height > w.pixel_cap
                    ):
                        break
                    job.batch_size += 1
                    overflow -= 1
                if job.batch_size > 0:
                    job.complementary = False
                    job.step_override = None
                    job.predicted_eta = self._predict(
                        w, job.batch_size, request
                    )
                if overflow == 0:
                    break
        if overflow > 0:
            log.warning("pixel caps too tight: %d images dropped", overflow)

    def _add_complementary(
        self, plan: Dict[str, Job], workers: List[Worker], request: GenRequest
    ) -> None:
        """Size bonus images to the realtime slack (ref world.py:519-557),
        optionally scaling steps down to fit (ref world.py:547-557)."""
        realtime_etas = [
            j.predicted_eta
            for j in plan.values()
            if not j.complementary and j.batch_size > 0
        ]
        if not realtime_etas:
            return
        t_max = max(realtime_etas)
        by_label = {w.label: w for w in workers}
        for job in plan.values():
            if not job.complementary:
                continue
            w = by_label[job.worker_label]
            one = self._predict(w, 1, request)
            if one <= 0:
                continue
            bonus = int(t_max // one)
            if bonus >= 1:
                job.batch_size = bonus
                job.predicted_eta = self._predict(w, bonus, request)
            elif self.settings.step_scaling:
                scaled = max(1, int(request.steps * t_max / one))
                if scaled < request.steps:
                    job.batch_size = 1
                    job.step_override = scaled
                    job.predicted_eta = t_max
            # else: job stays empty and is filtered out

    def _predict(self, worker: Worker, images: int, request: GenRequest) -> float:
        return worker.eta.eta(
            images=images,
            steps=request.steps,
            width=request.width,
            height=request.height,
            sampler_name=request.sampler_name,
            hr_scale=request.hr_scale,
            hr_steps=request.hr_steps,
        )

    def _assign_seeds(self, jobs: List[Job], request: GenRequest) -> None:
        """Contiguous gallery offsets + the C22 seed plan. Non-complementary
        jobs (rank order) own slots [0, batch); complementary jobs append."""
        base_seed = fix_seed(request.seed)
        request.seed = base_seed
        offset = 0
        ordered = [j for j in jobs if not j.complementary] + [
            j for j in jobs if j.complementary
        ]
        for job in ordered:
            job.gallery_offset = offset
            sp = shard_seeds(
                base_seed,
                offset,
                job.batch_size,
                subseed=request.subseed,
                subseed_strength=request.subseed_strength,
            )
            job.seeds = sp.seeds
            job.subseeds = sp.subseeds
            offset += job.batch_size

    def _log_distribution(self, jobs: List[Job], request: GenRequest) -> None:
        parts = []
        for j in jobs:
            tag = "+" if j.complementary else ""
            st = f"@{j.step_override}s" if j.step_override else ""
            parts.append(f"{j.worker_label}:{j.batch_size}{tag}{st}")
        log.debug(
            "distribution for batch=%d %dx%d: %s",
            request.batch_size,
            request.width,
            request.height,
            " ".join(parts),
        )

    # -- failure recovery (improves on ref worker.py:498-500) ----------------
    def requeue_failed(
        self, failed_job: Job, request: GenRequest
    ) -> List[Job]:
        """Re-shard a failed rank's images onto the surviving ranks, keeping
        the original seeds so the gallery stays deterministic."""
        failed = self.get_worker(failed_job.worker_label)
        if failed is not None:
            failed.set_state(State.UNAVAILABLE)
        survivors = self.active_workers()
        if not survivors:
            raise RuntimeError("no surviving ranks to requeue onto")
        sub = GenRequest(**{**request.__dict__, "batch_size": failed_job.batch_size})
        # no complementary/bonus jobs on a requeue: the retry plan must total
        # EXACTLY the failed shard's batch so the seed slices below align
        # (a bonus job would read past failed_job.seeds and fall back to 0)
        jobs = self.optimize_jobs(sub, allow_complementary=False)
        # keep the failed shard's slots and seeds
        pos = 0
        for job in jobs:
            job.gallery_offset = failed_job.gallery_offset + pos
            job.seeds = failed_job.seeds[pos : pos + job.batch_size]
            job.subseeds = failed_job.subseeds[pos : pos + job.batch_size]
            job.step_override = failed_job.step_override
            pos += job.batch_size
        return jobs

    # -- benchmark orchestration (ref world.py:199-278) ----------------------
    def benchmark(
        self,
        runner: BenchmarkRunner,
        rebenchmark: bool = False,
        parallel: bool = True,
    ) -> Dict[str, float]:
        targets = [
            w
            for w in self.workers
            if w.available and (rebenchmark or w.eta.avg_ipm <= 0)
        ]
        if not targets:
            return {w.label: w.eta.avg_ipm for w in self.workers}
        if parallel and len(targets) > 1:
            with concurrent.futures.ThreadPoolExecutor(len(targets)) as pool:
                futs = {
                    pool.submit(w.benchmark, runner, self.benchmark_payload): w
                    for w in targets
                }
                for f in concurrent.futures.as_completed(futs):
                    f.result()
        else:
            for w in targets:
                w.benchmark(runner, self.benchmark_payload)
        return {w.label: w.eta.avg_ipm for w in self.workers}

    def speed_summary(self) -> str:
        """ref world.py:297-315."""
        lines = []
        total = 0.0
        for w in sorted(self.workers, key=lambda w: -w.eta.avg_ipm):
            total += w.eta.avg_ipm
            lines.append(
                f"{w.label}: {w.eta.avg_ipm:.2f} ipm, mpe {w.eta.mpe():+.1f}%, "
                f"{w.state.value}"
            )
        lines.append(f"total: {total:.2f} ipm")
        return "\n".join(lines)

    # -- liveness (ref world.py:724-778) -------------------------------------
    def ping(self, indiscriminate: bool = False) -> Dict[str, bool]:
        results = {}
        for w in self.workers:
            if w.state is State.DISABLED and not indiscriminate:
                results[w.label] = False
                continue
            ok = w.reachable()
            results[w.label] = ok
            if ok and w.state is State.UNAVAILABLE:
                w.set_state(State.IDLE)
            elif not ok and w.state is not State.DISABLED:
                w.set_state(State.UNAVAILABLE)
        return results

    # -- interrupt (ref world.py:173-179) ------------------------------------
    def interrupt_all(self) -> None:
        self.interrupted.set()
        for w in self.workers:
            if w.state is State.WORKING:
                w.interrupt()

    def clear_interrupt(self) -> None:
        self.interrupted.clear()
        for w in self.workers:
            w.clear_interrupt()

    # -- config persistence (ref world.py:616-722) ---------------------------
    def load(self, path: Optional[str] = None) -> ConfigModel:
        cfg = load_config(path or self.config_path)
        self.settings = cfg.settings
        self.benchmark_payload = cfg.benchmark_payload
        with self._lock:
            self._workers = [
                Worker.from_model(m, is_master=(i == 0))
                for i, m in enumerate(cfg.workers)
            ]
        return cfg

    def save(self, path: Optional[str] = None) -> str:
        cfg = ConfigModel(
            workers=[w.to_model() for w in self.workers],
            benchmark_payload=self.benchmark_payload,
            settings=self.settings,
        )
        return save_config(cfg, path or self.config_path)

    @classmethod
    def from_devices(
        cls, n_devices: int, config_path: Optional[str] = None
    ) -> "World":
        world = cls(config_path=config_path)
        for i in range(n_devices):
            world.add_worker(
                Worker(label=f"gpu{i}", device=i, is_master=(i == 0))
            )
        return world
