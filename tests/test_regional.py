"""Regional Prompter (matrix mode) executed natively: mask construction,
prompt parsing, the masked cross-attention blend, and the pipeline path."""
import pytest
import torch

from sdwd_amd.models.unet import CrossAttention, RegionalContext
from sdwd_amd.pipeline.pipeline import _parse_regional, _region_masks


class _Req:
    def __init__(self, prompt, ratios="1,1", base=0.2):
        self.prompt = prompt
        self.regional_ratios = ratios
        self.regional_base_ratio = base


class TestRegionMasks:
    def test_columns_partition(self):
        m = _region_masks("columns", [1.0, 1.0], 8, 8)
        assert m.shape == (2, 8, 8)
        assert torch.equal(m.sum(0), torch.ones(8, 8))  # exact partition
        assert m[0, :, :4].all() and m[1, :, 4:].all()

    def test_rows_ratios(self):
        m = _region_masks("rows", [1.0, 3.0], 8, 8)
        assert m[0, :2].all() and m[1, 2:].all()
        assert torch.equal(m.sum(0), torch.ones(8, 8))

    def test_thin_region_never_empty(self):
        m = _region_masks("columns", [100.0, 0.001], 8, 8)
        assert m[1].sum() > 0  # at least one column survives rounding


class TestParseRegional:
    def test_base_plus_regions(self):
        got = _parse_regional(_Req("base BREAK left BREAK right"))
        assert got == ("base", ["left", "right"], [1.0, 1.0], 0.2)

    def test_no_base(self):
        got = _parse_regional(_Req("left BREAK right", base=0.0))
        assert got == ("", ["left", "right"], [1.0, 1.0], 0.0)

    def test_mismatch_disables(self):
        assert _parse_regional(_Req("a BREAK b", ratios="1,1,1")) is None

    def test_bad_ratios_disable(self):
        assert _parse_regional(_Req("a BREAK b BREAK c", ratios="1,x")) is None
        assert _parse_regional(_Req("a BREAK b BREAK c", ratios="1,-1")) is None


def _ref_attention(q, k, v, heads):
    b, s, d = q.shape
    dh = d // heads
    q = q.view(b, s, heads, dh).permute(0, 2, 1, 3)
    k = k.view(b, -1, heads, dh).permute(0, 2, 1, 3)
    v = v.view(b, -1, heads, dh).permute(0, 2, 1, 3)
    a = torch.softmax(q @ k.transpose(-1, -2) / dh**0.5, dim=-1)
    return (a @ v).permute(0, 2, 1, 3).reshape(b, s, d)


class TestRegionalCrossAttention:
    def test_blend_matches_manual_reference(self):
        torch.manual_seed(0)
        d, heads, s, b = 32, 4, 16, 3  # s = 4x4 grid
        attn = CrossAttention(d, d, heads).eval()
        x = torch.randn(b, s, d)
        plain = torch.randn(b, 77, d)
        regions = torch.randn(2, 77, d)
        masks = _region_masks("columns", [1.0, 1.0], 4, 4)
        rc = RegionalContext(plain, regions, masks, rows=2,
                             base_ratio=0.25, lat_hw=(4, 4))
        with torch.no_grad():
            out = attn(x, rc)

            def att(xx, cc):
                q = attn.to_q(xx)
                k = attn.to_k(cc)
                v = attn.to_v(cc)
                return _ref_attention(q, k, v, heads)

            base = att(x, plain)
            m = rc.masks_for(s, x.device, x.dtype)
            cover = m.sum(0).clamp(0, 1)
            keep = (0.25 + 0.75 * (1.0 - cover))[None, :, None]
            reg = sum(
                m[r][None, :, None]
                * att(x[:2], regions[r : r + 1].expand(2, -1, -1))
                for r in range(2)
            )
            want = torch.cat([keep * base[:2] + 0.75 * reg, base[2:]])
            want = attn.to_out(want)
        assert torch.allclose(out, want, atol=1e-5), (
            (out - want).abs().max()
        )

    def test_masks_for_resolution_inference(self):
        masks = _region_masks("rows", [1.0, 1.0], 8, 6)
        rc = RegionalContext(torch.zeros(1, 77, 8), torch.zeros(2, 77, 8),
                             masks, rows=1, base_ratio=0.0, lat_hw=(8, 6))
        assert rc.masks_for(48, torch.device("cpu"), torch.float32).shape == (2, 48)
        assert rc.masks_for(12, torch.device("cpu"), torch.float32).shape == (2, 12)
        # pooled masks still partition the canvas
        m = rc.masks_for(12, torch.device("cpu"), torch.float32)
        assert torch.allclose(m.sum(0), torch.ones(12))

    def test_getitem_slices_rows(self):
        masks = _region_masks("columns", [1.0], 4, 4)
        rc = RegionalContext(torch.zeros(6, 77, 8), torch.zeros(1, 77, 8),
                             masks, rows=3, base_ratio=0.2, lat_hw=(4, 4))
        sub = rc[:2]
        assert sub.plain.shape[0] == 2 and sub.rows == 2
        assert sub.shape == (2, 77, 8)


class TestRegionalPipeline:
    @pytest.fixture(scope="class")
    def pipe(self):
        from sdwd_amd.pipeline import StableDiffusionPipeline

        return StableDiffusionPipeline("tiny", device="cpu")

    def _req(self, prompt, **kw):
        from sdwd_amd.pipeline import PipelineRequest

        args = dict(
            prompt=prompt, steps=2, width=64, height=64, seeds=[11],
            regional_mode="columns", regional_ratios="1,1",
            regional_base_ratio=0.2,
        )
        args.update(kw)
        return PipelineRequest(**args)

    def test_runs_and_deterministic(self, pipe):
        req = self._req("sky BREAK red tree BREAK blue lake")
        a = pipe.generate(req)
        b = pipe.generate(req)
        assert torch.equal(a.images, b.images)
        assert "RP Active: True" in a.infotexts[0]

    def test_regions_change_output_and_order_matters(self, pipe):
        base = pipe.generate(self._req("sky", regional_mode="")).images
        ab = pipe.generate(self._req("sky BREAK red tree BREAK blue lake"))
        ba = pipe.generate(self._req("sky BREAK blue lake BREAK red tree"))
        assert not torch.equal(base, ab.images)
        assert not torch.equal(ab.images, ba.images)

    def test_invalid_spec_falls_back_to_plain(self, pipe):
        from sdwd_amd.pipeline import PipelineRequest

        plain = pipe.generate(PipelineRequest(
            prompt="sky BREAK x", steps=2, width=64, height=64, seeds=[11],
        ))
        broken = pipe.generate(self._req("sky BREAK x", regional_ratios="1,1,1"))
        assert torch.equal(plain.images, broken.images)
        assert "RP Active" not in broken.infotexts[0]

    def test_cfg1_fast_path(self, pipe):
        req = self._req("a BREAK b BREAK c", cfg_scale=1.0)
        out = pipe.generate(req)
        assert out.images.shape == (1, 64, 64, 3)

    def test_regional_img2img_with_mask(self, pipe):
        # regional conditioning composes with the img2img/inpaint path
        init = torch.full((1, 64, 64, 3), 190, dtype=torch.uint8)
        lat = pipe.encode_image(init, seeds=[11])
        mask = torch.zeros(64, 64, dtype=torch.uint8)
        mask[:, 32:] = 255
        req = self._req(
            "base BREAK left BREAK right",
            init_latents=lat, mask_image=mask, denoising_strength=0.9,
        )
        a = pipe.generate(req)
        b = pipe.generate(req)
        assert torch.equal(a.images, b.images)
        assert "RP Active: True" in a.infotexts[0]

    def test_regional_with_controlnet_unit(self, pipe):
        # the ControlNet copy receives the BASE rows (it is a 4ch
        # base-UNet clone and knows nothing of region masks)
        hint = torch.randint(0, 255, (1, 64, 64, 3), dtype=torch.uint8)
        req = self._req(
            "base BREAK a BREAK b",
            control_image=hint, control_model="controlnet-tiny",
            control_scale=0.7,
        )
        out = pipe.generate(req)
        no_cn = pipe.generate(self._req("base BREAK a BREAK b"))
        assert not torch.equal(out.images, no_cn.images)

    def test_regional_with_hires(self, pipe):
        req = self._req(
            "base BREAK a BREAK b", enable_hr=True, hr_scale=2.0,
            hr_steps=2, denoising_strength=0.6,
        )
        out = pipe.generate(req)
        assert out.images.shape == (1, 128, 128, 3)
        assert torch.isfinite(out.images.float()).all()


class TestRegionalAPIParse:
    def test_dict_payload(self):
        from sdwd_amd.api.server import _parse_regional_prompter

        got = _parse_regional_prompter({
            "Regional Prompter": {"args": [{
                "active": True, "mode": "Columns", "ratios": "1,2",
                "base_ratio": 0.3,
            }]}
        })
        assert got == {"regional_mode": "columns",
                       "regional_ratios": "1,2",
                       "regional_base_ratio": 0.3}

    def test_inactive_and_unsupported_mode(self):
        from sdwd_amd.api.server import _parse_regional_prompter

        assert _parse_regional_prompter({
            "regional prompter": {"args": [{"active": False}]}
        }) == {}
        assert _parse_regional_prompter({
            "regional prompter": {"args": [{"mode": "Mask"}]}
        }) == {}
        assert _parse_regional_prompter({}) == {}

    def test_use_base_false(self):
        from sdwd_amd.api.server import _parse_regional_prompter

        got = _parse_regional_prompter({
            "regional prompter": {"args": [{"mode": "rows",
                                            "use_base": False}]}
        })
        assert got["regional_base_ratio"] == 0.0
