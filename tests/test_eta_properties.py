"""Property-based invariants for the ETA predictor and the seed plan.

The shard planner leans on the predictor's SHAPE (monotone in images,
steps, pixels and speed) more than on its absolute accuracy; the seed
plan must tile gallery slots without gaps for any (offset, count) split.
"""
import pytest

try:
    from hypothesis import given, settings
    from hypothesis import strategies as st
except ImportError:  # pragma: no cover
    pytest.skip("hypothesis not installed", allow_module_level=True)

from sdwd_amd.core.eta import EtaPredictor
from sdwd_amd.core.seeds import shard_seeds

common = settings(max_examples=60, deadline=None, derandomize=True)
ipm = st.floats(min_value=0.01, max_value=500.0,
                allow_nan=False, allow_infinity=False)


class TestEtaProperties:
    @common
    @given(
        speed=ipm,
        images=st.integers(min_value=1, max_value=64),
        steps=st.integers(min_value=1, max_value=100),
        side=st.sampled_from([256, 512, 768, 1024]),
    )
    def test_monotone_in_work_and_speed(self, speed, images, steps, side):
        e = EtaPredictor(avg_ipm=speed)
        base = e.eta(images=images, steps=steps, width=side, height=side)
        assert base > 0
        # more images / more steps / more pixels never predict faster
        assert e.eta(images=images + 1, steps=steps, width=side,
                     height=side) > base
        assert e.eta(images=images, steps=steps + 1, width=side,
                     height=side) > base
        assert e.eta(images=images, steps=steps, width=side * 2,
                     height=side) > base
        # a faster rank never predicts slower
        faster = EtaPredictor(avg_ipm=speed * 2)
        assert faster.eta(images=images, steps=steps, width=side,
                          height=side) < base
        # linear in images (the proportional splitter assumes this)
        two = e.eta(images=2 * images, steps=steps, width=side, height=side)
        assert two == pytest.approx(2 * base, rel=1e-6)

    @common
    @given(
        speed=ipm,
        images=st.integers(min_value=1, max_value=16),
        hr_scale=st.floats(min_value=1.0, max_value=4.0, allow_nan=False),
        hr_steps=st.integers(min_value=1, max_value=50),
    )
    def test_hires_pass_only_adds(self, speed, images, hr_scale, hr_steps):
        e = EtaPredictor(avg_ipm=speed)
        plain = e.eta(images=images)
        with_hr = e.eta(images=images, hr_scale=hr_scale, hr_steps=hr_steps)
        assert with_hr > plain

    @common
    @given(
        speed=ipm,
        predicted=st.floats(min_value=0.1, max_value=1e4, allow_nan=False),
        ratio=st.floats(min_value=0.2, max_value=4.0, allow_nan=False),
    )
    def test_mpe_feedback_bounded(self, speed, predicted, ratio):
        e = EtaPredictor(avg_ipm=speed)
        for _ in range(8):
            e.record_outcome(predicted, predicted * ratio)
        f = e.correction_factor()
        # the rolling-MPE correction stays positive and finite, and pulls
        # the prediction TOWARD the observed ratio
        assert 0.0 < f < 100.0
        if ratio > 1.05:
            assert f > 1.0
        elif ratio < 0.95:
            assert f < 1.0


class TestSeedPlanProperties:
    @common
    @given(
        base=st.integers(min_value=0, max_value=2**31 - 1),
        splits=st.lists(st.integers(min_value=1, max_value=12),
                        min_size=1, max_size=6),
    )
    def test_shards_tile_the_gallery(self, base, splits):
        offset = 0
        all_seeds = []
        for count in splits:
            plan = shard_seeds(base, offset, count)
            assert len(plan.seeds) == count
            all_seeds.extend(plan.seeds)
            offset += count
        total = sum(splits)
        # any split of the batch yields the same flat seed sequence
        assert all_seeds == [base + i for i in range(total)]

    @common
    @given(
        base=st.integers(min_value=0, max_value=2**31 - 1),
        sub=st.integers(min_value=0, max_value=2**31 - 1),
        splits=st.lists(st.integers(min_value=1, max_value=12),
                        min_size=1, max_size=6),
        strength=st.floats(min_value=0.01, max_value=1.0, allow_nan=False),
    )
    def test_subseed_variation_tiles_subseeds(
        self, base, sub, splits, strength
    ):
        offset = 0
        seeds, subs = [], []
        for count in splits:
            plan = shard_seeds(
                base, offset, count, subseed=sub, subseed_strength=strength
            )
            seeds.extend(plan.seeds)
            subs.extend(plan.subseeds)
            offset += count
        total = sum(splits)
        # variation mode: one shared base seed, per-slot subseeds
        assert seeds == [base] * total
        assert subs == [sub + i for i in range(total)]
