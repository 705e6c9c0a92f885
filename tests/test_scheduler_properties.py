"""Property-based invariants for the job optimizer (hypothesis).

The example-based tests in test_core_scheduler.py pin known scenarios;
these sweep RANDOM fleets (speeds spanning 4 orders of magnitude, caps,
deferral, complementary production, step scaling) and assert the
invariants the rest of the system depends on:

  I1  non-complementary shards conserve the requested batch exactly;
  I2  gallery offsets are contiguous and every slot has a seed, with
      non-complementary seeds == [seed, seed+1, ..., seed+batch-1] (the
      C22 determinism contract: an N-GPU gallery equals a 1-GPU batch);
  I3  every job targets an active worker, at most one job per rank;
  I4  planning is deterministic (same fleet + request -> same plan);
  I5  pixel caps are never exceeded (when satisfiable);
  I6  a requeue plan reproduces the failed shard's seeds exactly and
      never invents complementary work.
"""
import pytest

try:
    from hypothesis import HealthCheck, given, settings
    from hypothesis import strategies as st
except ImportError:  # pragma: no cover
    pytest.skip("hypothesis not installed", allow_module_level=True)

from sdwd_amd.config.models import SettingsModel
from sdwd_amd.core import World, Worker
from sdwd_amd.core.world import GenRequest

ipm_strategy = st.floats(
    min_value=0.01, max_value=100.0, allow_nan=False, allow_infinity=False
)
fleet_strategy = st.lists(ipm_strategy, min_size=1, max_size=9)


def make_world(ipms, **settings_kw):
    w = World(settings=SettingsModel(**settings_kw))
    for i, ipm in enumerate(ipms):
        w.add_worker(
            Worker(label=f"gpu{i}", device=i, avg_ipm=ipm, is_master=(i == 0))
        )
    return w


def plan_signature(jobs):
    return sorted(
        (j.worker_label, j.batch_size, j.complementary, j.step_override,
         j.gallery_offset, tuple(j.seeds))
        for j in jobs
    )


common = settings(
    derandomize=True,
    max_examples=60,
    deadline=None,
    suppress_health_check=[HealthCheck.function_scoped_fixture],
)


class TestPlanInvariants:
    @common
    @given(
        ipms=fleet_strategy,
        batch=st.integers(min_value=1, max_value=64),
        steps=st.integers(min_value=1, max_value=50),
        step_scaling=st.booleans(),
        complement=st.booleans(),
    )
    def test_batch_conserved_and_seeds_contiguous(
        self, ipms, batch, steps, step_scaling, complement
    ):
        world = make_world(
            ipms,
            step_scaling=step_scaling,
            complement_production=complement,
            job_timeout=2.0,
        )
        req = GenRequest(batch_size=batch, steps=steps, seed=1234)
        jobs = world.make_jobs(req)
        real = [j for j in jobs if not j.complementary]
        comp = [j for j in jobs if j.complementary]
        # I1: the requested batch is conserved exactly by realtime shards
        assert sum(j.batch_size for j in real) == batch
        # I3: one job per rank, all of them registered and active
        labels = [j.worker_label for j in jobs]
        assert len(labels) == len(set(labels))
        active = {w.label for w in world.active_workers()}
        assert set(labels) <= active
        # complementary work only exists when the feature is on
        if not complement:
            assert comp == []
        for j in comp:
            assert j.batch_size >= 1
            if j.step_override is not None:
                assert 1 <= j.step_override < steps
        # I2: contiguous offsets, aligned seeds, no gaps or overlaps
        ordered = sorted(jobs, key=lambda j: j.gallery_offset)
        offset = 0
        for j in ordered:
            assert j.gallery_offset == offset
            assert len(j.seeds) == j.batch_size
            assert j.seeds == [1234 + offset + i for i in range(j.batch_size)]
            offset += j.batch_size
        real_seeds = [
            s
            for j in sorted(real, key=lambda j: j.gallery_offset)
            for s in j.seeds
        ]
        assert real_seeds == [1234 + i for i in range(batch)]

    @common
    @given(
        ipms=fleet_strategy,
        batch=st.integers(min_value=1, max_value=48),
    )
    def test_plan_deterministic(self, ipms, batch):
        # I4: two worlds built from the same fleet produce identical plans
        a = make_world(ipms).make_jobs(GenRequest(batch_size=batch, seed=7))
        b = make_world(ipms).make_jobs(GenRequest(batch_size=batch, seed=7))
        assert plan_signature(a) == plan_signature(b)

    @common
    @given(
        ipms=st.lists(ipm_strategy, min_size=2, max_size=6),
        batch=st.integers(min_value=1, max_value=32),
        cap_images=st.integers(min_value=1, max_value=8),
    )
    def test_pixel_caps_respected(self, ipms, batch, cap_images):
        world = make_world(ipms, complement_production=False)
        capped = world.get_worker("gpu1")
        capped.pixel_cap = cap_images * 512 * 512
        jobs = world.make_jobs(GenRequest(batch_size=batch, seed=3))
        by = {j.worker_label: j for j in jobs}
        # I5: the cap holds whenever the rest of the fleet can absorb the
        # overflow (it always can here: gpu0 is uncapped)
        if "gpu1" in by:
            assert by["gpu1"].batch_size <= cap_images
        assert sum(j.batch_size for j in jobs) == batch

    @common
    @given(
        ipms=st.lists(ipm_strategy, min_size=2, max_size=6),
        batch=st.integers(min_value=2, max_value=32),
        victim=st.integers(min_value=0, max_value=5),
        complement=st.booleans(),
    )
    def test_requeue_preserves_failed_seeds(
        self, ipms, batch, victim, complement
    ):
        world = make_world(ipms, complement_production=complement)
        req = GenRequest(batch_size=batch, seed=999)
        jobs = world.make_jobs(req)
        real = [j for j in jobs if not j.complementary and j.batch_size > 0]
        victim_job = real[victim % len(real)]
        replacement = world.requeue_failed(victim_job, req)
        # I6: exact seed reproduction, no bonus work on the retry path
        assert sum(j.batch_size for j in replacement) == victim_job.batch_size
        assert not any(j.complementary for j in replacement)
        seeds = [
            s
            for j in sorted(replacement, key=lambda j: j.gallery_offset)
            for s in j.seeds
        ]
        assert seeds == victim_job.seeds
        for j in replacement:
            assert j.worker_label != victim_job.worker_label
