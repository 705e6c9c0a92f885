"""ldm/sdwui checkpoint-format conversion tests.

No real checkpoint is available offline, so correctness is pinned two ways:
round-tripping (export -> load must be bit-exact, including the quant-conv
folds and CLIP qkv split/fuse), and canary checks that the exported key set
for the sd15 config is exactly the canonical SD1.5 layout (686 UNet / 248
VAE / 196 CLIP tensors, spot-checked names and shapes).
"""
import pytest
import torch

from sdwd_amd.models.convert import (
    CLIP_PREFIX,
    UNET_PREFIX,
    VAE_PREFIX,
    load_ldm_state_dict,
    to_ldm_state_dict,
)
from sdwd_amd.models.registry import load_model


def _perturb(bundle):
    with torch.no_grad():
        for p in bundle.unet.parameters():
            p.add_(torch.randn_like(p) * 0.1)
        for p in bundle.vae.parameters():
            p.add_(torch.randn_like(p) * 0.1)
        for p in bundle.text_encoder.parameters():
            p.add_(torch.randn_like(p) * 0.1)


class TestRoundTrip:
    def test_tiny_bit_exact(self):
        a = load_model("tiny", device="cpu", cache=False)
        exported = to_ldm_state_dict(a)
        b = load_model("tiny", device="cpu", cache=False)
        _perturb(b)  # make sure the load actually has to do something
        report = load_ldm_state_dict(b, exported)
        assert not report["missing"], report["missing"][:5]
        assert not report["unexpected"], report["unexpected"][:5]
        for part in ("unet", "vae", "text_encoder"):
            sa = getattr(a, part).state_dict()
            sb = getattr(b, part).state_dict()
            for k in sa:
                assert torch.equal(sa[k], sb[k]), f"{part}.{k}"

    def test_quant_conv_fold_is_exact(self):
        """A non-identity quant/post_quant conv folds into conv_out/conv_in
        so that encode/decode outputs match applying the two convs."""
        import torch.nn.functional as F

        a = load_model("tiny", device="cpu", cache=False)
        exported = to_ldm_state_dict(a)
        lat = 2 * a.vae.cfg.latent_channels
        g = torch.Generator().manual_seed(5)
        qw = torch.randn(lat, lat, 1, 1, generator=g) * 0.2 + torch.eye(
            lat
        ).reshape(lat, lat, 1, 1)
        qb = torch.randn(lat, generator=g) * 0.1
        exported[VAE_PREFIX + "quant_conv.weight"] = qw
        exported[VAE_PREFIX + "quant_conv.bias"] = qb

        b = load_model("tiny", device="cpu", cache=False)
        _perturb(b)
        load_ldm_state_dict(b, exported)

        x = torch.randn(1, 3, 32, 32, generator=g)
        with torch.no_grad():
            ref = F.conv2d(a.vae.encoder(x), qw, qb)
            got = b.vae.encoder(x)
        assert torch.allclose(ref, got, atol=1e-4), (ref - got).abs().max()

    def test_forward_equivalent(self):
        a = load_model("tiny", device="cpu", cache=False)
        b = load_model("tiny", device="cpu", cache=False)
        _perturb(b)
        load_ldm_state_dict(b, to_ldm_state_dict(a))
        x = torch.randn(1, 4, 16, 16)
        t = torch.tensor([3.0])
        ctx = torch.randn(1, 77, a.unet.cfg.context_dim)
        with torch.no_grad():
            assert torch.equal(a.unet(x, t, ctx), b.unet(x, t, ctx))


class TestSdxlRoundTrip:
    def test_tiny_xl_bit_exact(self):
        """SDXL-layout export/load: dual encoders (CLIP-L split-qkv +
        open_clip fused-qkv), label_emb, text_projection."""
        a = load_model("tiny-xl", device="cpu", cache=False)
        exported = to_ldm_state_dict(a)
        assert any(k.startswith("conditioner.embedders.0.") for k in exported)
        assert any(
            ".attn.in_proj_weight" in k
            for k in exported if k.startswith("conditioner.embedders.1.")
        )
        b = load_model("tiny-xl", device="cpu", cache=False)
        _perturb(b)
        with torch.no_grad():
            for p in b.text_encoder_2.parameters():
                p.add_(torch.randn_like(p) * 0.1)
        report = load_ldm_state_dict(b, exported)
        assert not report["missing"], report["missing"][:5]
        assert not report["unexpected"], report["unexpected"][:5]
        for part in ("unet", "vae", "text_encoder"):
            sa = getattr(a, part).state_dict()
            sb = getattr(b, part).state_dict()
            for k in sa:
                assert torch.equal(sa[k], sb[k]), f"{part}.{k}"
        sa = a.text_encoder_2.state_dict()
        sb = b.text_encoder_2.state_dict()
        for k in sa:  # b additionally carries the identity text_projection
            assert torch.equal(sa[k], sb[k]), f"text_encoder_2.{k}"
        # pooled output unchanged by the identity projection
        tokens = torch.randint(0, 100, (2, 77))
        tokens[:, -1] = 49407
        with torch.no_grad():
            ha = a.text_encoder_2(tokens)
            hb = b.text_encoder_2(tokens)
            assert torch.equal(ha, hb)
            assert torch.allclose(
                a.text_encoder_2.pooled(tokens, ha),
                b.text_encoder_2.pooled(tokens, hb), atol=1e-5,
            )


class TestSd2RoundTrip:
    @staticmethod
    def _small_sd2_bundle():
        """Hand-built SD2-shaped bundle: 1024-wide open_clip-style text
        tower (the export heuristic keys off d_model == 1024), tiny UNet/VAE."""
        import zlib

        from sdwd_amd.models.clip import CLIPTextEncoder
        from sdwd_amd.models.registry import ModelBundle, _seeded_init
        from sdwd_amd.models.unet import UNetConfig, UNetModel
        from sdwd_amd.models.vae import AutoencoderKL, VAEConfig

        te = CLIPTextEncoder(d_model=1024, layers=2, heads=4)
        unet = UNetModel(
            UNetConfig(model_channels=32, channel_mult=[1, 2],
                       num_res_blocks=1, transformer_depth=[1, 1],
                       context_dim=1024, num_heads=2, groups=8)
        )
        vae = AutoencoderKL(VAEConfig.tiny())
        for off, m in enumerate((te, unet, vae)):
            _seeded_init(m, zlib.crc32(b"sd2test") % (2 ** 31) + off)
        return ModelBundle("sd2test", te, None, unet, vae, context_dim=1024)

    def test_sd2_layout_round_trip(self):
        a = self._small_sd2_bundle()
        exported = to_ldm_state_dict(a)
        # SD2 naming: open_clip tower under cond_stage_model.model.
        assert any(
            k.startswith("cond_stage_model.model.transformer.resblocks.")
            for k in exported
        )
        assert "cond_stage_model.model.text_projection" in exported
        assert not any(
            k.startswith("cond_stage_model.transformer.") for k in exported
        )
        b = self._small_sd2_bundle()
        _perturb(b)
        report = load_ldm_state_dict(b, exported)
        assert not report["missing"], report["missing"][:5]
        assert not report["unexpected"], report["unexpected"][:5]
        for part in ("unet", "vae", "text_encoder"):
            sa = getattr(a, part).state_dict()
            sb = getattr(b, part).state_dict()
            for k in sa:
                assert torch.equal(sa[k], sb[k]), f"{part}.{k}"

    def test_sd21_builder_registered(self):
        from sdwd_amd.models.registry import available_models

        assert "sd21" in available_models()


class TestCheckpointFile:
    def test_ldm_safetensors_auto_detected(self, tmp_path):
        """An sdwui-format .safetensors file loads through the normal
        load_checkpoint entry point."""
        from safetensors.torch import save_file

        from sdwd_amd.models.registry import load_checkpoint

        a = load_model("tiny", device="cpu", cache=False)
        exported = {
            k: v.contiguous().clone() for k, v in to_ldm_state_dict(a).items()
        }
        path = str(tmp_path / "sdwui_style.safetensors")
        save_file(exported, path)
        b = load_checkpoint(path)
        x = torch.randn(1, 4, 16, 16)
        t = torch.tensor([3.0])
        ctx = torch.randn(1, 77, a.unet.cfg.context_dim)
        with torch.no_grad():
            assert torch.equal(a.unet(x, t, ctx), b.unet(x, t, ctx))


class TestInpaintArchInference:
    def test_9ch_file_loads_as_inpaint_arch(self, tmp_path):
        from safetensors.torch import save_file

        from sdwd_amd.models.registry import load_checkpoint

        a = load_model("tiny-inpaint", device="cpu", cache=False)
        exported = {
            k: v.contiguous().clone() for k, v in to_ldm_state_dict(a).items()
        }
        path = str(tmp_path / "inpaint.safetensors")
        save_file(exported, path)
        b = load_checkpoint(path)
        assert b.unet.cfg.in_channels == 9
        assert b.latent_channels == 4


class TestCheckpointDir:
    def test_file_backed_models(self, tmp_path, monkeypatch):
        """*.safetensors in SDWD_CHECKPOINT_DIR become loadable model names
        (sdwui checkpoint-folder semantics)."""
        from sdwd_amd.models.registry import (
            available_models,
            load_model as lm,
            refresh_checkpoint_files,
            save_checkpoint,
        )

        a = lm("tiny", device="cpu", cache=False)
        d = tmp_path / "ckpts"
        d.mkdir()
        save_checkpoint(a, str(d / "my-custom-model.safetensors"))
        monkeypatch.setenv("SDWD_CHECKPOINT_DIR", str(d))
        try:
            names = refresh_checkpoint_files()
            assert names == ["my-custom-model"]
            assert "my-custom-model" in available_models()
            b = lm("my-custom-model", cache=False)
            x = torch.randn(1, 4, 16, 16)
            t = torch.tensor([3.0])
            ctx = torch.randn(1, 77, a.unet.cfg.context_dim)
            with torch.no_grad():
                assert torch.equal(a.unet(x, t, ctx), b.unet(x, t, ctx))
        finally:
            refresh_checkpoint_files(dirpath=str(tmp_path / "none"))
        assert "my-custom-model" not in available_models()


@pytest.fixture(scope="module")
def sd15_keys():
    bundle = load_model("sd15", device="cpu", cache=False)
    exported = to_ldm_state_dict(bundle)
    return {k: tuple(v.shape) for k, v in exported.items()}


class TestSd15Layout:
    """Canonical SD1.5 checkpoint layout canaries."""

    def test_tensor_counts(self, sd15_keys):
        unet = [k for k in sd15_keys if k.startswith(UNET_PREFIX)]
        vae = [k for k in sd15_keys if k.startswith(VAE_PREFIX)]
        clip = [k for k in sd15_keys if k.startswith(CLIP_PREFIX)]
        assert len(unet) == 686
        assert len(vae) == 248
        assert len(clip) == 196

    @pytest.mark.parametrize(
        "key,shape",
        [
            ("model.diffusion_model.input_blocks.0.0.weight", (320, 4, 3, 3)),
            ("model.diffusion_model.input_blocks.1.0.in_layers.2.weight",
             (320, 320, 3, 3)),
            ("model.diffusion_model.input_blocks.1.1.transformer_blocks.0."
             "attn2.to_k.weight", (320, 768)),
            ("model.diffusion_model.input_blocks.3.0.op.weight",
             (320, 320, 3, 3)),
            ("model.diffusion_model.input_blocks.4.0.skip_connection.weight",
             (640, 320, 1, 1)),
            ("model.diffusion_model.middle_block.1.proj_in.weight",
             (1280, 1280, 1, 1)),
            ("model.diffusion_model.output_blocks.2.1.conv.weight",
             (1280, 1280, 3, 3)),
            ("model.diffusion_model.output_blocks.5.2.conv.weight",
             (1280, 1280, 3, 3)),
            ("model.diffusion_model.output_blocks.11.0.in_layers.2.weight",
             (320, 640, 3, 3)),
            ("model.diffusion_model.out.2.weight", (4, 320, 3, 3)),
            ("first_stage_model.encoder.down.0.block.0.norm1.weight", (128,)),
            ("first_stage_model.encoder.down.1.block.0.nin_shortcut.weight",
             (256, 128, 1, 1)),
            ("first_stage_model.encoder.mid.attn_1.q.weight",
             (512, 512, 1, 1)),
            ("first_stage_model.decoder.up.1.upsample.conv.weight",
             (256, 256, 3, 3)),
            ("first_stage_model.quant_conv.weight", (8, 8, 1, 1)),
            ("first_stage_model.post_quant_conv.weight", (4, 4, 1, 1)),
            ("cond_stage_model.transformer.text_model.embeddings."
             "token_embedding.weight", (49408, 768)),
            ("cond_stage_model.transformer.text_model.encoder.layers.11."
             "self_attn.q_proj.weight", (768, 768)),
            ("cond_stage_model.transformer.text_model.encoder.layers.0."
             "mlp.fc1.weight", (3072, 768)),
            ("cond_stage_model.transformer.text_model.final_layer_norm.weight",
             (768,)),
        ],
    )
    def test_canonical_key_shapes(self, sd15_keys, key, shape):
        assert key in sd15_keys, key
        assert sd15_keys[key] == shape

    def test_no_upsample_at_level0(self, sd15_keys):
        assert ("first_stage_model.decoder.up.0.upsample.conv.weight"
                not in sd15_keys)
        assert ("model.diffusion_model.output_blocks.11.1.conv.weight"
                not in sd15_keys)
        # but level-0 output blocks do carry attention in SD1.5
        assert ("model.diffusion_model.output_blocks.11.1.transformer_blocks"
                ".0.attn1.to_q.weight" in sd15_keys)


class TestModelCacheIsolation:
    def test_cache_keys_by_device_and_dtype(self):
        """Two holders with different dtype targets must not alias — a
        shared instance would be .to()-moved under the first holder."""
        import torch as _t

        a = load_model("tiny", device="cpu", dtype=_t.float32)
        b = load_model("tiny", device="cpu", dtype=_t.float64)
        assert a.unet is not b.unet
        assert next(a.unet.parameters()).dtype == _t.float32
        assert next(b.unet.parameters()).dtype == _t.float64
        # same key -> same instance (the cache still caches)
        c = load_model("tiny", device="cpu", dtype=_t.float32)
        assert c.unet is a.unet


class TestRefinerConvert:
    def test_refiner_round_trip_and_prefix(self):
        """The refiner's CLIP-G exports under conditioner.embedders.0
        (it has no CLIP-L) and reloads with full key consumption."""
        a = load_model("tiny-xl-refiner", device="cpu", cache=False)
        state = to_ldm_state_dict(a)
        assert any(
            k.startswith("conditioner.embedders.0.model.") for k in state
        )
        assert not any(
            k.startswith("conditioner.embedders.1.") for k in state
        )
        from sdwd_amd.models.convert import load_ldm_state_dict

        b = load_model("tiny-xl-refiner", device="cpu", cache=False)
        rep = load_ldm_state_dict(b, state)
        assert not rep["missing"] and not rep["unexpected"]


class TestStandaloneVae:
    """sdwui "SD VAE" dropdown files (ref C13 synced VAE by name through
    load_options, worker.py:646-688) — standalone-file load + engine swap."""

    def _vae_file(self, tmp_path, bare=False):
        from safetensors.torch import save_file

        a = load_model("tiny", device="cpu", cache=False)
        with torch.no_grad():
            for p in a.vae.parameters():
                p.add_(torch.randn_like(p) * 0.05)
        full = to_ldm_state_dict(a)
        vae_sd = {
            (k[len("first_stage_model."):] if bare else k): v.contiguous().clone()
            for k, v in full.items()
            if k.startswith("first_stage_model.")
        }
        path = str(tmp_path / ("bare.safetensors" if bare else "pfx.safetensors"))
        save_file(vae_sd, path)
        return a, path

    @pytest.mark.parametrize("bare", [False, True])
    def test_load_vae_state_dict(self, tmp_path, bare):
        """Both key spellings (first_stage_model.* and bare ldm) load and
        reproduce the source VAE's decode exactly."""
        from safetensors.torch import load_file

        from sdwd_amd.models.convert import load_vae_state_dict

        a, path = self._vae_file(tmp_path, bare=bare)
        b = load_model("tiny", device="cpu", cache=False)
        z = torch.randn(1, 4, 8, 8, generator=torch.Generator().manual_seed(5))
        with torch.no_grad():
            before = b.vae.decode(z)
            report = load_vae_state_dict(b.vae, load_file(path))
            after = b.vae.decode(z)
            want = a.vae.decode(z)
        assert not report["missing"], report["missing"][:5]
        assert not report["unexpected"], report["unexpected"][:5]
        assert not torch.equal(before, after)
        assert torch.equal(after, want)

    def test_engine_set_vae_and_restore(self, tmp_path, monkeypatch):
        from sdwd_amd.models.registry import refresh_vae_files
        from sdwd_amd.parallel import GenerationRequest, LocalEngine

        _, path = self._vae_file(tmp_path)
        monkeypatch.setenv("SDWD_VAE_DIR", str(tmp_path))
        refresh_vae_files()
        eng = LocalEngine(model="tiny", devices=["cpu"])
        eng.world.workers[0].eta.avg_ipm = 60.0
        req = GenerationRequest(
            prompt="v", batch_size=1, width=64, height=64, steps=1, seed=8
        )
        auto = eng.generate(req).images
        eng.set_vae("pfx")
        swapped = eng.generate(req).images
        assert not torch.equal(auto, swapped)
        eng.set_vae("auto")
        restored = eng.generate(req).images
        assert torch.equal(auto, restored)

    def test_unknown_vae_raises(self):
        from sdwd_amd.parallel import LocalEngine

        eng = LocalEngine(model="tiny", devices=["cpu"])
        with pytest.raises(KeyError):
            eng.set_vae("no-such-vae")
