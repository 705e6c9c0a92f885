"""Unit tests for the utility layers: logging ring buffer, CLI flags,
process-group helpers in single-process mode, tokenizer determinism."""
import argparse

import torch

from sdwd_amd.config import add_flags, export_env
from sdwd_amd.models import tokenizer
from sdwd_amd.parallel import broadcast_object, gather_images
from sdwd_amd.parallel.group import allgather_floats
from sdwd_amd.utils import get_logger, ring_buffer


class TestLogging:
    def test_ring_buffer_captures(self):
        log = get_logger("ringtest")
        log.info("ring-entry-%d", 42)
        lines = ring_buffer()
        assert any("ring-entry-42" in ln for ln in lines)

    def test_ring_buffer_bounded(self):
        log = get_logger("ringtest")
        for i in range(100):
            log.info("flood %d", i)
        assert len(ring_buffer()) <= 64


class TestFlags:
    def test_defaults_and_export(self, monkeypatch, tmp_path):
        monkeypatch.delenv("SDWD_CONFIG", raising=False)
        ap = argparse.ArgumentParser()
        add_flags(ap)
        args = ap.parse_args(
            ["--sdwd-config", str(tmp_path / "c.json"), "--sdwd-debug"]
        )
        assert args.sdwd_debug
        export_env(args)
        import os

        assert os.environ["SDWD_CONFIG"].endswith("c.json")
        assert os.environ["SDWD_DEBUG"] == "1"


class TestGroupSingleProcess:
    def test_broadcast_object_identity(self):
        obj = {"a": [1, 2, 3]}
        assert broadcast_object(obj) is obj

    def test_gather_images_passthrough(self):
        shard = torch.randint(0, 255, (3, 8, 8, 3), dtype=torch.uint8)
        out = gather_images(shard, [3], "cpu")
        assert torch.equal(out, shard)

    def test_allgather_floats_single(self):
        assert allgather_floats([1.5, 2.5], "cpu") == [[1.5, 2.5]]


class TestTokenizer:
    def test_deterministic_and_shaped(self):
        a = tokenizer.encode("A herd of cows")
        b = tokenizer.encode("A herd of cows")
        assert a == b
        assert len(a) == 77
        assert a[0] == tokenizer.BOS
        assert tokenizer.EOS in a

    def test_batch(self):
        t = tokenizer.encode_batch(["one", "two words here"])
        assert t.shape == (2, 77)
        assert t.dtype == torch.long

    def test_ids_in_vocab(self):
        ids = tokenizer.encode("punctuation, too! and-hyphens 123")
        assert all(0 <= i < tokenizer.VOCAB_SIZE for i in ids)


class TestBpeTokenizer:
    """Real CLIP byte-level BPE path (synthetic tiny vocab on disk)."""

    @staticmethod
    def _write_fixture(tmp_path):
        import json

        from sdwd_amd.models.tokenizer import _bytes_to_unicode

        # base alphabet: every byte symbol and its </w> form, then the
        # merged tokens the merges below can produce.
        syms = list(_bytes_to_unicode().values())
        vocab = {}
        for s in syms:
            vocab[s] = len(vocab)
        for s in syms:
            vocab[s + "</w>"] = len(vocab)
        merges = [("l", "o"), ("lo", "w</w>"), ("c", "o"), ("co", "w</w>")]
        for a, b in merges:
            if a + b not in vocab:
                vocab[a + b] = len(vocab)
        vpath = tmp_path / "vocab.json"
        mpath = tmp_path / "merges.txt"
        vpath.write_text(json.dumps(vocab))
        mpath.write_text(
            "#version: test\n" + "\n".join(f"{a} {b}" for a, b in merges)
        )
        return str(vpath), str(mpath), vocab

    def test_merges_apply_in_rank_order(self, tmp_path):
        vpath, mpath, vocab = self._write_fixture(tmp_path)
        try:
            bpe = tokenizer.use_bpe(vpath, mpath)
            # "low" -> l+o merge first (rank 0), then lo+w</w> (rank 1)
            assert bpe.encode_text("low") == [vocab["low</w>"]]
            assert bpe.encode_text("cow") == [vocab["cow</w>"]]
            # unmergeable word falls apart into symbols ending in </w>
            ids = bpe.encode_text("ab")
            assert ids == [vocab["a"], vocab["b</w>"]]
        finally:
            tokenizer.use_hash()

    def test_module_encode_uses_bpe_when_active(self, tmp_path):
        vpath, mpath, vocab = self._write_fixture(tmp_path)
        try:
            tokenizer.use_bpe(vpath, mpath)
            ids = tokenizer.encode("low cow")
            assert ids[0] == tokenizer.BOS
            assert ids[1] == vocab["low</w>"]
            assert ids[2] == vocab["cow</w>"]
            assert ids[3] == tokenizer.EOS
            assert len(ids) == 77
            # weighted path shares the BPE fragments
            wids, wts = tokenizer.encode_weighted("low (cow:1.5)")
            assert wids[1] == vocab["low</w>"]
            assert wids[2] == vocab["cow</w>"]
            assert wts[1] == 1.0 and wts[2] == 1.5
        finally:
            tokenizer.use_hash()
        # reverted: hash path again, still deterministic
        assert tokenizer.encode("low cow") == tokenizer.encode("low cow")

    def test_whitespace_and_case_normalised(self, tmp_path):
        vpath, mpath, _ = self._write_fixture(tmp_path)
        try:
            bpe = tokenizer.use_bpe(vpath, mpath)
            assert bpe.encode_text("  LOW\n\tcow ") == bpe.encode_text("low cow")
        finally:
            tokenizer.use_hash()


class TestEtaDefaults:
    def test_unknown_sampler_cost(self):
        from sdwd_amd.core import sampler_cost

        assert sampler_cost("Some Future Sampler") == 1.0
        assert sampler_cost("Heun") == 2.0


class TestColorCorrect:
    def test_matches_reference_stats(self):
        from sdwd_amd.utils.images import color_correct

        g = torch.Generator().manual_seed(1)
        out = (torch.randn(32, 32, 3, generator=g) * 20 + 60).clamp(
            0, 255
        ).to(torch.uint8)
        ref = (torch.randn(32, 32, 3, generator=g) * 40 + 180).clamp(
            0, 255
        ).to(torch.uint8)
        corrected = color_correct(out, ref)
        cm = corrected.float().mean(dim=(0, 1))
        rm = ref.float().mean(dim=(0, 1))
        assert (cm - rm).abs().max() < 8.0  # uint8 rounding/clamping slack


class TestParseInfotext:
    def test_round_trip_fields(self):
        from sdwd_amd.utils.images import parse_infotext

        text = ("a cow, detailed\nNegative prompt: blurry, bad\n"
                "Steps: 20, Sampler: Euler a, CFG scale: 7.0, Seed: 42, "
                "Size: 512x768, Model: sd15, Clip skip: 2")
        p = parse_infotext(text)
        assert p["prompt"] == "a cow, detailed"
        assert p["negative_prompt"] == "blurry, bad"
        assert p["steps"] == 20 and p["seed"] == 42
        assert p["cfg_scale"] == 7.0
        assert p["width"] == 512 and p["height"] == 768
        assert p["sampler"] == "Euler a"
        assert p["clip_skip"] == 2

    def test_no_negative(self):
        from sdwd_amd.utils.images import parse_infotext

        p = parse_infotext("just a prompt\nSteps: 4, Seed: 1")
        assert p["prompt"] == "just a prompt"
        assert p["negative_prompt"] == ""


class TestPngMetadata:
    def test_parameters_round_trip(self):
        from sdwd_amd.utils.images import encode_png, png_parameters

        img = torch.randint(0, 255, (8, 8, 3), dtype=torch.uint8)
        info = "a cow\nSteps: 20, Sampler: Euler a, Seed: 7"
        data = encode_png(img, parameters=info)
        assert png_parameters(data) == info
        # image still decodes
        from sdwd_amd.utils.images import decode_png

        assert torch.equal(decode_png(data), img)

    def test_no_metadata_is_none(self):
        from sdwd_amd.utils.images import encode_png, png_parameters

        img = torch.zeros(4, 4, 3, dtype=torch.uint8)
        assert png_parameters(encode_png(img)) is None


class TestDynamicPrompts:
    """Native execution of the sd-dynamic-prompts syntax (C18)."""

    def test_variants_deterministic(self):
        from sdwd_amd.pipeline.wildcards import expand

        a = expand("a {red|green|blue} hat", 42)
        b = expand("a {red|green|blue} hat", 42)
        assert a == b
        assert a in ("a red hat", "a green hat", "a blue hat")
        seen = {expand("a {red|green|blue} hat", s) for s in range(30)}
        assert len(seen) > 1  # different seeds draw different variants

    def test_multi_select(self):
        from sdwd_amd.pipeline.wildcards import expand

        out = expand("{2$$a|b|c}", 7)
        parts = out.split(", ")
        assert len(parts) == 2 and len(set(parts)) == 2
        assert set(parts) <= {"a", "b", "c"}

    def test_weights(self):
        from sdwd_amd.pipeline.wildcards import expand

        hits = sum(
            expand("{99::heavy|1::light}", s) == "heavy" for s in range(100)
        )
        assert hits > 90

    def test_nesting(self):
        from sdwd_amd.pipeline.wildcards import expand

        out = expand("{a {big|small} cat|a dog}", 3)
        assert out in ("a big cat", "a small cat", "a dog")

    def test_wildcard_files(self, tmp_path):
        from sdwd_amd.pipeline.wildcards import expand

        (tmp_path / "animals.txt").write_text("# comment\ncow\nhorse\n")
        sub = tmp_path / "style"
        sub.mkdir()
        (sub / "mood.txt").write_text("calm\n")
        out = expand("a __animals__, __style/mood__", 5, root=str(tmp_path))
        assert out in ("a cow, calm", "a horse, calm")
        # missing wildcard degrades to empty, no raise
        assert expand("x __nope__ y", 1, root=str(tmp_path)) == "x  y"

    def test_plain_text_untouched(self):
        from sdwd_amd.pipeline.wildcards import expand, has_dynamic_syntax

        assert expand("plain prompt", 1) == "plain prompt"
        assert not has_dynamic_syntax("plain prompt")
        assert has_dynamic_syntax("{a|b}")
        assert has_dynamic_syntax("__cards__")


class TestGnPartialHandshake:
    """The conv->GroupNorm partials side-channel plumbing (pure logic;
    kernel numerics are covered on GPU)."""

    def _fake(self, t, tiles, c):
        import torch

        gnp = torch.zeros(tiles, 2, c)
        t._sdwd_gnp = (gnp, 1, t._version)
        return gnp

    def test_cat_merges_valid_partials(self):
        import torch

        from sdwd_amd.ops import cat_channels_gn

        a = torch.zeros(1, 8, 16, 16)
        b = torch.zeros(1, 4, 16, 16)
        ga = self._fake(a, 2, 8)
        gb = self._fake(b, 2, 4)
        out = cat_channels_gn(a, b)
        assert out.shape[1] == 12
        meta = getattr(out, "_sdwd_gnp", None)
        assert meta is not None
        assert meta[0].shape == (2, 2, 12)
        assert torch.equal(meta[0][:, :, :8], ga)
        assert torch.equal(meta[0][:, :, 8:], gb)

    def test_cat_drops_on_missing_or_stale(self):
        import torch

        from sdwd_amd.ops import cat_channels_gn

        a = torch.zeros(1, 8, 16, 16)
        b = torch.zeros(1, 4, 16, 16)
        self._fake(a, 2, 8)  # b has no partials
        out = cat_channels_gn(a, b)
        assert getattr(out, "_sdwd_gnp", None) is None
        # stale version: in-place mutation invalidates
        a2 = torch.zeros(1, 8, 16, 16)
        b2 = torch.zeros(1, 4, 16, 16)
        self._fake(a2, 2, 8)
        self._fake(b2, 2, 4)
        a2.add_(1.0)  # bumps _version
        out2 = cat_channels_gn(a2, b2)
        assert getattr(out2, "_sdwd_gnp", None) is None

    def test_cat_drops_on_tile_mismatch(self):
        import torch

        from sdwd_amd.ops import cat_channels_gn

        a = torch.zeros(1, 8, 16, 16)
        b = torch.zeros(1, 4, 16, 16)
        self._fake(a, 2, 8)
        gb = torch.zeros(3, 2, 4)
        b._sdwd_gnp = (gb, 1, b._version)
        out = cat_channels_gn(a, b)
        assert getattr(out, "_sdwd_gnp", None) is None


class TestInfotextParse:
    """parse_infotext inverts the pipeline's emitted 'parameters' text —
    the grammar sdwui's own parser (and every 'send to txt2img' tool)
    expects."""

    def test_basic_grammar(self):
        from sdwd_amd.utils.infotext import parse_infotext

        out = parse_infotext(
            "a cow\nover two lines\nNegative prompt: blurry\n"
            "Steps: 20, Sampler: Euler a, CFG scale: 7.0, Seed: 5, "
            'Size: 512x512, RP Ratios: "1,2", Worker Label: gpu1'
        )
        assert out["prompt"] == "a cow\nover two lines"
        assert out["negative_prompt"] == "blurry"
        assert out["Steps"] == "20"
        assert out["Sampler"] == "Euler a"
        assert out["RP Ratios"] == "1,2"  # quoted comma survives
        assert out["Worker Label"] == "gpu1"

    def test_no_negative(self):
        from sdwd_amd.utils.infotext import parse_infotext

        out = parse_infotext("just a cow\nSteps: 4, Seed: 1, Size: 64x64")
        assert out["prompt"] == "just a cow"
        assert out["negative_prompt"] == ""
        assert out["Seed"] == "1"

    def test_engine_round_trip(self):
        """A fully-optioned engine run emits infotext this parser (and so
        sdwui's) reads back field-for-field."""
        import torch

        from sdwd_amd.parallel import GenerationRequest, LocalEngine
        from sdwd_amd.utils.infotext import parse_infotext

        eng = LocalEngine(model="tiny", devices=["cpu", "cpu"])
        for w in eng.world.workers:
            w.eta.avg_ipm = 60.0
        res = eng.generate(GenerationRequest(
            prompt="a cow", negative_prompt="blurry", batch_size=2,
            width=64, height=64, steps=2, seed=9, cfg_scale=4.5,
            sampler_name="Heun", clip_skip=2, eta=0.5, tiling=True,
            enable_hr=True, hr_scale=0.0, hr_resize_x=96, hr_resize_y=96,
            hr_steps=1, denoising_strength=0.6,
        ))
        out = parse_infotext(res.infotexts[1])
        assert out["prompt"] == "a cow"
        assert out["negative_prompt"] == "blurry"
        assert out["Steps"] == "2"
        assert out["Sampler"] == "Heun"
        assert out["CFG scale"] == "4.5"
        assert out["Seed"] == "10"
        assert out["Clip skip"] == "2"
        assert out["Eta"] == "0.5"
        assert out["Tiling"] == "True"
        assert out["Hires resize"] == "96x96"
        assert out["Worker Label"].startswith("gpu")
