"""Property-based tests for the pure-python codecs and prompt parsers."""
import pytest
import torch
from hypothesis import given, settings as hsettings
from hypothesis import strategies as st

from sdwd_amd.utils.images import decode_png, encode_png, png_parameters


@hsettings(max_examples=40, deadline=None)
@given(
    h=st.integers(1, 40),
    w=st.integers(1, 40),
    seed=st.integers(0, 2**31 - 1),
)
def test_png_round_trip_exact(h, w, seed):
    g = torch.Generator().manual_seed(seed)
    img = torch.randint(0, 256, (h, w, 3), generator=g, dtype=torch.int64).to(
        torch.uint8
    )
    assert torch.equal(decode_png(encode_png(img)), img)


@hsettings(max_examples=40, deadline=None)
@given(text=st.text(min_size=0, max_size=200))
def test_png_parameters_round_trip(text):
    img = torch.zeros(2, 2, 3, dtype=torch.uint8)
    data = encode_png(img, parameters=text)
    got = png_parameters(data)
    # latin-1 replacement is the only allowed lossiness
    expected = text.encode("latin-1", "replace").decode("latin-1")
    if text:
        assert got == expected
    else:
        assert got is None


_plain_words = st.text(
    alphabet=st.sampled_from("abcdefghij "), min_size=1, max_size=30
)


@hsettings(max_examples=60, deadline=None)
@given(text=_plain_words)
def test_parse_weighted_preserves_plain_text(text):
    from sdwd_amd.models.tokenizer import parse_weighted

    frags = parse_weighted(text)
    assert "".join(f for f, _ in frags) == text or not text.strip()
    assert all(w == 1.0 for _, w in frags)


@hsettings(max_examples=60, deadline=None)
@given(
    text=_plain_words,
    steps=st.integers(1, 30),
)
def test_prompt_schedule_plain_is_single_segment(text, steps):
    from sdwd_amd.pipeline.prompt_schedule import prompt_schedule

    assert prompt_schedule(text, steps) == [(0, text)]


@hsettings(max_examples=60, deadline=None)
@given(
    frm=st.text(alphabet=st.sampled_from("abc"), min_size=1, max_size=8),
    to=st.text(alphabet=st.sampled_from("xyz"), min_size=1, max_size=8),
    when=st.floats(0.05, 0.95),
    steps=st.integers(2, 30),
)
def test_prompt_schedule_switch_consistency(frm, to, when, steps):
    """[from:to:when] yields exactly the 'from' text before the threshold
    and the 'to' text at/after it, for every step."""
    from sdwd_amd.pipeline.prompt_schedule import prompt_at_step

    thr = when * steps
    for i in range(steps):
        got = prompt_at_step(f"[{frm}:{to}:{when}]", i, steps)
        assert got == (frm if i < thr else to), (i, thr, got)


@hsettings(max_examples=40, deadline=None)
@given(
    seed=st.integers(0, 2**31 - 1),
    h=st.integers(4, 32),
    w=st.integers(4, 32),
)
def test_color_correct_is_stat_projection(seed, h, w):
    from sdwd_amd.utils.images import color_correct

    g = torch.Generator().manual_seed(seed)
    out = torch.randint(30, 220, (h, w, 3), generator=g,
                        dtype=torch.int64).to(torch.uint8)
    ref = torch.randint(30, 220, (h, w, 3), generator=g,
                        dtype=torch.int64).to(torch.uint8)
    c = color_correct(out, ref)
    assert c.shape == out.shape and c.dtype == torch.uint8
    # idempotent up to rounding: correcting twice changes little
    c2 = color_correct(c, ref)
    assert (c2.float() - c.float()).abs().mean() < 3.0


@hsettings(max_examples=40, deadline=None)
@given(
    steps=st.integers(1, 150),
    seed=st.integers(0, 2**31 - 1),
    cfg=st.floats(1.0, 30.0),
    w=st.sampled_from([64, 512, 768, 1024]),
    h=st.sampled_from([64, 512, 768, 1024]),
    sampler=st.sampled_from(["Euler a", "DPM++ 2M Karras", "UniPC"]),
    prompt=st.text(alphabet=st.sampled_from("abc xyz,"), min_size=1,
                   max_size=40),
)
def test_infotext_emit_parse_round_trip(steps, seed, cfg, w, h, sampler,
                                        prompt):
    """The infotext our pipeline emits must be recoverable by our own
    parser (the sdwui 'send to txt2img' loop)."""
    from sdwd_amd.utils.images import parse_infotext

    prompt = prompt.replace("\n", " ").strip() or "x"
    text = (f"{prompt}\nNegative prompt: neg\n"
            f"Steps: {steps}, Sampler: {sampler}, CFG scale: {cfg}, "
            f"Seed: {seed}, Size: {w}x{h}, Model: sd15")
    p = parse_infotext(text)
    assert p["prompt"] == prompt
    assert p["steps"] == steps
    assert p["seed"] == seed
    assert abs(p["cfg_scale"] - cfg) < 1e-6 or str(cfg) in str(p["cfg_scale"])
    assert p["width"] == w and p["height"] == h
