"""Property-based scheduler invariants (hypothesis).

The reference's optimizer (world.py:418-601) was only example-tested; these
pin the invariants every plan must satisfy for ANY speed distribution,
batch size, settings combination:

- coverage: non-complementary jobs tile [0, batch) exactly (no image lost,
  none duplicated) — the silent-drop class of bug;
- seeds: image k always gets seed base+k regardless of sharding (C22);
- caps: pixel caps are respected whenever any uncapped rank exists;
- determinism: the same inputs always produce the same plan.
"""
import pytest
from hypothesis import HealthCheck, given, settings as hsettings
from hypothesis import strategies as st

from sdwd_amd.config.models import SettingsModel
from sdwd_amd.core import GenRequest, World


def build_world(ipms, pixel_caps=None, settings=None):
    w = World.from_devices(len(ipms), config_path=None)
    if settings is not None:
        w.settings = settings
    for i, worker in enumerate(w.workers):
        worker.eta.avg_ipm = ipms[i]
        if pixel_caps:
            worker.pixel_cap = pixel_caps[i]
    return w


ipms_strategy = st.lists(
    st.one_of(st.just(0.0), st.floats(1.0, 300.0)), min_size=1, max_size=8
)


@hsettings(max_examples=120, deadline=None,
           suppress_health_check=[HealthCheck.too_slow])
@given(
    ipms=ipms_strategy,
    batch=st.integers(1, 128),
    steps=st.integers(1, 50),
    size=st.sampled_from([(512, 512), (640, 448), (1024, 1024), (64, 64)]),
    complement=st.booleans(),
    step_scaling=st.booleans(),
    thin=st.booleans(),
    seed=st.integers(0, 2**31 - 1),
)
def test_plan_invariants(ipms, batch, steps, size, complement, step_scaling,
                         thin, seed):
    cfg = SettingsModel(
        complement_production=complement,
        step_scaling=step_scaling,
        thin_client=thin and len(ipms) > 1,
    )
    world = build_world(ipms, settings=cfg)
    req = GenRequest(batch_size=batch, width=size[0], height=size[1],
                     steps=steps, seed=seed)
    jobs = world.make_jobs(req)

    primary = [j for j in jobs if not j.complementary]
    # coverage: offsets tile [0, batch) exactly
    primary.sort(key=lambda j: j.gallery_offset)
    pos = 0
    for j in primary:
        assert j.batch_size >= 0
        if j.batch_size == 0:
            continue
        assert j.gallery_offset == pos, (
            [(x.gallery_offset, x.batch_size) for x in primary]
        )
        pos += j.batch_size
    assert pos == batch

    # seed plan: image k gets seed base+k no matter which rank runs it
    for j in primary:
        for i, s in enumerate(j.seeds):
            assert s == seed + j.gallery_offset + i

    # complementary jobs never claim gallery space beyond the batch
    for j in jobs:
        if j.complementary:
            assert j.gallery_offset >= batch or j.batch_size >= 0

    # thin client: the master takes no primary work when others exist
    if cfg.thin_client and len(ipms) > 1:
        master = [j for j in primary
                  if j.worker_label == "gpu0" and j.batch_size > 0]
        assert not master


@hsettings(max_examples=60, deadline=None)
@given(
    ipms=st.lists(st.floats(1.0, 300.0), min_size=2, max_size=8),
    batch=st.integers(1, 64),
    seed=st.integers(0, 2**30),
)
def test_plan_deterministic(ipms, batch, seed):
    req = GenRequest(batch_size=batch, seed=seed)
    a = build_world(ipms).make_jobs(req)
    b = build_world(ipms).make_jobs(req)
    assert [(j.worker_label, j.gallery_offset, j.batch_size, j.seeds)
            for j in a] == [
        (j.worker_label, j.gallery_offset, j.batch_size, j.seeds) for j in b
    ]


@hsettings(max_examples=60, deadline=None)
@given(
    ipms=st.lists(st.floats(10.0, 100.0), min_size=2, max_size=6),
    batch=st.integers(2, 64),
)
def test_faster_ranks_get_no_less(ipms, batch):
    """Monotonicity: sorting workers by speed, primary shard sizes are
    non-increasing as speed decreases (the weighted split's whole point)."""
    world = build_world(ipms)
    jobs = world.make_jobs(GenRequest(batch_size=batch, seed=1))
    by_label = {
        j.worker_label: j.batch_size for j in jobs if not j.complementary
    }
    order = sorted(range(len(ipms)), key=lambda i: -ipms[i])
    sizes = [by_label.get(f"gpu{i}", 0) for i in order]
    assert all(a >= b for a, b in zip(sizes, sizes[1:])), (ipms, by_label)


@hsettings(max_examples=40, deadline=None)
@given(
    batch=st.integers(1, 64),
    cap_images=st.integers(1, 8),
)
def test_pixel_caps_respected(batch, cap_images):
    """A capped rank never gets more than its cap while an uncapped rank
    exists to absorb the overflow."""
    caps = [0, cap_images * 512 * 512]  # gpu0 uncapped, gpu1 capped
    world = build_world([60.0, 60.0], pixel_caps=caps)
    jobs = world.make_jobs(GenRequest(batch_size=batch, seed=0))
    for j in jobs:
        if j.worker_label == "gpu1" and not j.complementary:
            assert j.batch_size <= cap_images, (batch, cap_images, j)
