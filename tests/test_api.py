"""API surface tests (sdapi/v1-compatible routes) at tiny scale on CPU."""
import base64
import json

import pytest
import torch
from fastapi.testclient import TestClient

from sdwd_amd.api import create_app
from sdwd_amd.parallel import LocalEngine
from sdwd_amd.utils.images import decode_png, encode_png


@pytest.fixture(scope="module")
def monkeypatch_module():
    from _pytest.monkeypatch import MonkeyPatch

    mp = MonkeyPatch()
    yield mp
    mp.undo()


@pytest.fixture(scope="module")
def client(tmp_path_factory, monkeypatch_module):
    monkeypatch_module.setenv(
        "SDWD_CONFIG", str(tmp_path_factory.mktemp("cfg") / "c.json")
    )
    engine = LocalEngine(model="tiny", devices=["cpu", "cpu"])
    for w in engine.world.workers:
        w.eta.avg_ipm = 60.0
    app = create_app(engine=engine)
    return TestClient(app)


class TestTxt2Img:
    def test_basic(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "a cow",
                "steps": 2,
                "width": 64,
                "height": 64,
                "batch_size": 2,
                "seed": 77,
            },
        )
        assert r.status_code == 200
        body = r.json()
        # grid + 2 images
        assert len(body["images"]) == 3
        img = decode_png(base64.b64decode(body["images"][1]))
        assert img.shape == (64, 64, 3)
        info = json.loads(body["info"])
        assert info["all_seeds"] == [77, 78]
        assert len(info["infotexts"]) == 2


class TestBatchCount:
    def test_n_iter_folds_into_gallery(self, client):
        """sdwui batch_count (n_iter): N sequential batches; here the sharder
        runs them as one contiguous gallery with the same seed plan."""
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "a cow",
                "steps": 2,
                "width": 64,
                "height": 64,
                "batch_size": 2,
                "n_iter": 2,
                "seed": 300,
            },
        )
        assert r.status_code == 200
        body = r.json()
        assert len(body["images"]) == 5  # grid + 4
        info = json.loads(body["info"])
        assert info["all_seeds"] == [300, 301, 302, 303]
        assert len(info["infotexts"]) == 4


class TestImg2Img:
    def test_round_trip(self, client):
        init = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
        b64 = base64.b64encode(encode_png(init)).decode()
        r = client.post(
            "/sdapi/v1/img2img",
            json={
                "prompt": "re",
                "steps": 2,
                "width": 64,
                "height": 64,
                "batch_size": 1,
                "seed": 5,
                "init_images": [b64],
                "denoising_strength": 0.6,
            },
        )
        assert r.status_code == 200
        assert len(r.json()["images"]) >= 1

    def test_missing_init(self, client):
        r = client.post(
            "/sdapi/v1/img2img", json={"prompt": "x", "init_images": []}
        )
        assert r.status_code == 422

    def test_soft_inpainting_payload(self, client):
        init = torch.full((64, 64, 3), 170, dtype=torch.uint8)
        mask = torch.zeros(64, 64, 3, dtype=torch.uint8)
        mask[:, 32:] = 255
        body = {
            "prompt": "re", "steps": 2, "width": 64, "height": 64,
            "seed": 5, "denoising_strength": 1.0,
            "init_images": [base64.b64encode(encode_png(init)).decode()],
            "mask": base64.b64encode(encode_png(mask)).decode(),
            "alwayson_scripts": {
                "soft inpainting": {
                    "args": [{
                        "Soft inpainting": True,
                        "Schedule bias": 1.5,
                        "Preservation strength": 0.5,
                    }]
                }
            },
        }
        r = client.post("/sdapi/v1/img2img", json=body)
        assert r.status_code == 200, r.text
        info = json.loads(r.json()["info"])
        assert "Soft inpainting: True" in info["infotexts"][0]
        assert "Schedule bias: 1.5" in info["infotexts"][0]
        # same request without the script: plain hard-mask inpaint
        body2 = dict(body)
        body2.pop("alwayson_scripts")
        r2 = client.post("/sdapi/v1/img2img", json=body2)
        assert r2.status_code == 200
        assert r.json()["images"][-1] != r2.json()["images"][-1]

    def test_soft_inpainting_positional_and_disabled(self):
        from sdwd_amd.api.server import _parse_soft_inpainting

        got = _parse_soft_inpainting(
            {"soft inpainting": {"args": [True, 2.0, 0.25]}}
        )
        assert got["soft_inpainting"] is True
        assert got["si_schedule_bias"] == 2.0
        assert got["si_preservation_strength"] == 0.25
        assert _parse_soft_inpainting(
            {"soft inpainting": {"args": [{"Soft inpainting": False}]}}
        ) == {}
        assert _parse_soft_inpainting({}) == {}


class TestControl:
    def test_models_and_samplers(self, client):
        models = client.get("/sdapi/v1/sd-models").json()
        assert any(m["model_name"] == "sd15" for m in models)
        samplers = client.get("/sdapi/v1/samplers").json()
        assert any(s["name"] == "Euler a" for s in samplers)

    def test_memory(self, client):
        r = client.get("/sdapi/v1/memory")
        assert r.status_code == 200

    def test_interrupt(self, client):
        assert client.post("/sdapi/v1/interrupt").status_code == 200

    def test_progress(self, client):
        r = client.get("/sdapi/v1/progress").json()
        assert "progress" in r and "state" in r

    def test_options_unknown_model(self, client):
        r = client.post(
            "/sdapi/v1/options", json={"sd_model_checkpoint": "nope"}
        )
        assert r.status_code == 404

    def test_settings_round_trip(self, client):
        import json as _json
        import os

        before = client.get("/sdwd/settings").json()
        assert before["complement_production"] is True
        r = client.post(
            "/sdwd/settings",
            json={"job_timeout": 7.5, "distribute_img2img": False},
        )
        assert r.status_code == 200
        after = client.get("/sdwd/settings").json()
        assert after["job_timeout"] == 7.5
        assert after["distribute_img2img"] is False
        # persisted to the config file
        with open(os.environ["SDWD_CONFIG"]) as fh:
            on_disk = _json.load(fh)
        assert on_disk["settings"]["job_timeout"] == 7.5
        # restore for other tests
        client.post(
            "/sdwd/settings",
            json={"job_timeout": 3.0, "distribute_img2img": True},
        )

    def test_override_settings_model_and_clip_skip(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "o", "steps": 1, "width": 64, "height": 64,
                "seed": 3,
                "override_settings": {
                    "sd_model_checkpoint": "tiny-xl",
                    "CLIP_stop_at_last_layers": 2,
                },
            },
        )
        assert r.status_code == 200
        # the swap persisted (sdwui override_settings default behavior)
        opts = client.get("/sdapi/v1/options").json()
        assert opts["sd_model_checkpoint"] == "tiny-xl"
        # switch back for the other tests
        assert client.post(
            "/sdapi/v1/options", json={"sd_model_checkpoint": "tiny"}
        ).status_code == 200

    def test_override_settings_unknown_model_404(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "o", "steps": 1, "width": 64, "height": 64,
                "override_settings": {"sd_model_checkpoint": "nope"},
            },
        )
        assert r.status_code == 404

    def test_img2img_inpaint_full_res_end_to_end(self, client):
        from sdwd_amd.utils.images import decode_png, encode_png

        init = torch.full((64, 64, 3), 90, dtype=torch.uint8)
        mask = torch.zeros(64, 64, 3, dtype=torch.uint8)
        mask[24:40, 24:40] = 255
        r = client.post(
            "/sdapi/v1/img2img",
            json={
                "prompt": "patch", "steps": 2, "width": 64, "height": 64,
                "seed": 5, "denoising_strength": 1.0,
                "init_images": [base64.b64encode(encode_png(init)).decode()],
                "mask": base64.b64encode(encode_png(mask)).decode(),
                "inpaint_full_res": True, "inpaint_full_res_padding": 8,
                "mask_blur": 2, "inpainting_fill": 2,
                "resize_mode": 0,
            },
        )
        assert r.status_code == 200
        imgs = r.json()["images"]
        out = decode_png(base64.b64decode(imgs[-1]))
        assert out.shape == (64, 64, 3)
        # far corner is pasted from the original init exactly
        assert torch.equal(out[:8, :8], init[:8, :8])

    def test_save_and_send_images_flags(self, client, tmp_path,
                                        monkeypatch_module):
        import os

        monkeypatch_module.setenv("SDWD_OUTPUT_DIR", str(tmp_path / "out"))
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "s", "steps": 1, "width": 64, "height": 64,
                  "seed": 8, "send_images": False, "save_images": True},
        )
        assert r.status_code == 200
        assert r.json()["images"] == []
        saved = os.listdir(tmp_path / "out")
        assert len(saved) == 1 and saved[0].endswith(".png")

    def test_unknown_fields_ignored(self, client):
        """sdwui clients send fields we don't implement (restore_faces,
        script_args, ...); they must not 422."""
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "x", "steps": 1, "width": 64, "height": 64,
                  "seed": 1, "restore_faces": False, "script_args": [],
                  "script_name": None, "comments": {}},
        )
        assert r.status_code == 200

    def test_all_knobs_accepted(self, client):
        """Every sdwui-compat request field plumbs through end to end."""
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "a [cat:dog:0.5] AND a cow:0.7 <lora:style-z:0.5>",
                "negative_prompt": "blurry",
                "steps": 2, "width": 64, "height": 64, "seed": 11,
                "sampler_name": "Heun", "scheduler": "Exponential",
                "cfg_scale": 4.0, "clip_skip": 2, "tiling": True,
                "s_churn": 0.5, "s_noise": 1.1, "s_min_uncond": 0.1,
                "enable_hr": True, "hr_scale": 2.0,
                "hr_upscaler": "Latent (bilinear)",
                "hr_second_pass_steps": 1,
                "refiner_checkpoint": "tiny-xl", "refiner_switch_at": 0.5,
            },
        )
        assert r.status_code == 200
        body = r.json()
        assert len(body["images"]) >= 1
        # unmerge the <lora:...> tag so the shared engine weights are
        # pristine for whatever test runs next
        client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "plain", "steps": 1, "width": 64, "height": 64,
                  "seed": 1},
        )

    def test_basic_auth(self, monkeypatch):
        import base64 as b64

        from fastapi.testclient import TestClient

        from sdwd_amd.api import create_app
        from sdwd_amd.parallel import LocalEngine

        monkeypatch.setenv("SDWD_API_AUTH", "user:secret")
        eng = LocalEngine(model="tiny", devices=["cpu"])
        c = TestClient(create_app(engine=eng))
        assert c.get("/sdapi/v1/sd-models").status_code == 401
        hdr = {"Authorization": "Basic " + b64.b64encode(
            b"user:secret").decode()}
        assert c.get("/sdapi/v1/sd-models", headers=hdr).status_code == 200
        bad = {"Authorization": "Basic " + b64.b64encode(
            b"user:wrong").decode()}
        assert c.get("/sdapi/v1/sd-models", headers=bad).status_code == 401

    def test_styles(self, client, tmp_path, monkeypatch_module):
        from sdwd_amd.pipeline.styles import apply_styles, refresh_styles

        p = tmp_path / "styles.csv"
        p.write_text(
            'name,prompt,negative_prompt\n'
            'cinematic,"{prompt}, film grain, 35mm",blurry\n'
            'plainsuffix,"golden hour",\n'
        )
        monkeypatch_module.setenv("SDWD_STYLES_FILE", str(p))
        try:
            names = refresh_styles()
            assert names == ["cinematic", "plainsuffix"]
            pr, ng = apply_styles("a cow", "bad", ["cinematic"])
            assert pr == "a cow, film grain, 35mm"
            assert ng == "bad, blurry"
            pr2, _ = apply_styles("a cow", "", ["plainsuffix"])
            assert pr2 == "a cow, golden hour"
            r = client.get("/sdapi/v1/prompt-styles")
            assert any(e["name"] == "cinematic" for e in r.json())
            # end-to-end through txt2img (styled prompt lands in infotext)
            import json as _json

            resp = client.post(
                "/sdapi/v1/txt2img",
                json={"prompt": "a cow", "steps": 1, "width": 64,
                      "height": 64, "seed": 5, "styles": ["cinematic"]},
            )
            info = _json.loads(resp.json()["info"])
            assert "film grain" in info["infotexts"][0]
        finally:
            refresh_styles(path=str(tmp_path / "none.csv"))

    def test_extras_upscale(self, client):
        from sdwd_amd.utils.images import decode_png, encode_png

        img = torch.randint(0, 255, (16, 16, 3), dtype=torch.uint8)
        data = base64.b64encode(encode_png(img)).decode()
        r = client.post(
            "/sdapi/v1/extra-single-image",
            json={"image": data, "upscaling_resize": 2.0,
                  "upscaler_1": "Nearest"},
        )
        assert r.status_code == 200
        out = decode_png(base64.b64decode(r.json()["image"]))
        assert out.shape == (32, 32, 3)
        # nearest 2x: every 2x2 block equals the source pixel
        assert torch.equal(out[::2, ::2], img)
        # explicit target size
        r2 = client.post(
            "/sdapi/v1/extra-single-image",
            json={"image": data, "upscaling_resize_w": 24,
                  "upscaling_resize_h": 40, "upscaler_1": "Bicubic"},
        )
        out2 = decode_png(base64.b64decode(r2.json()["image"]))
        assert out2.shape == (40, 24, 3)

    def test_startup_enumeration_routes(self, client):
        """The static lists sdwui GUIs fetch at startup all answer."""
        for route in ("upscalers", "latent-upscale-modes", "face-restorers",
                      "hypernetworks", "scripts", "embeddings", "sd-vae",
                      "script-info", "cmd-flags"):
            r = client.get(f"/sdapi/v1/{route}")
            assert r.status_code == 200, route
        ups = [u["name"] for u in client.get("/sdapi/v1/upscalers").json()]
        assert "Lanczos" in ups

    def test_schedulers_route(self, client):
        names = [s["label"] for s in client.get("/sdapi/v1/schedulers").json()]
        assert "Karras" in names and "Exponential" in names

    def test_png_info_round_trip(self, client):
        from sdwd_amd.utils.images import encode_png

        img = torch.zeros(8, 8, 3, dtype=torch.uint8)
        data = base64.b64encode(
            encode_png(img, parameters="a cow\nSteps: 4, Seed: 9")
        ).decode()
        r = client.post("/sdapi/v1/png-info", json={"image": data})
        assert r.status_code == 200
        assert "Steps: 4" in r.json()["info"]

    def test_loras_routes(self, client, tmp_path, monkeypatch_module):
        import torch as _t

        from sdwd_amd.models.lora import (
            make_random_lora, refresh_lora_files, save_lora_file,
        )

        d = tmp_path / "loras"
        d.mkdir()
        pipe = next(iter(client.app.state.engine.pipes.values()))
        save_lora_file(
            make_random_lora("filelora", pipe.model.unet), str(d / "filelora.safetensors")
        )
        monkeypatch_module.setenv("SDWD_LORA_DIR", str(d))
        r = client.post("/sdapi/v1/refresh-loras")
        assert r.json()["found"] == ["filelora"]
        names = [e["name"] for e in client.get("/sdapi/v1/loras").json()]
        assert "filelora" in names
        refresh_lora_files(dirpath=str(tmp_path / "none"))

    def test_reset_mpe(self, client):
        eng = client.app.state.engine
        eng.world.get_worker("gpu0").eta.record_outcome(1.0, 2.0)
        assert eng.world.get_worker("gpu0").eta.mpe() != 0.0
        assert client.post("/sdwd/reset-mpe").status_code == 200
        assert eng.world.get_worker("gpu0").eta.mpe() == 0.0

    def test_release_lock(self, client):
        assert client.post("/sdwd/release-lock").status_code == 200
        # engine still serves afterwards
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "l", "steps": 1, "width": 64, "height": 64,
                  "seed": 2},
        )
        assert r.status_code == 200

    def test_restart_workers(self, client):
        from sdwd_amd.core.state import State

        eng = client.app.state.engine
        eng.world.get_worker("gpu0").set_state(State.UNAVAILABLE)
        eng.world.interrupted.set()
        r = client.post("/sdwd/restart-workers")
        assert r.status_code == 200
        assert "gpu0" in r.json()["restarted"]
        assert eng.world.get_worker("gpu0").state is State.IDLE
        assert not eng.world.interrupted.is_set()
        # idempotent: nothing left to restart
        assert client.post("/sdwd/restart-workers").json()["restarted"] == []

    def test_benchmark_payload_round_trip(self, client):
        before = client.get("/sdwd/benchmark-payload").json()
        assert before["steps"] == 20
        r = client.post(
            "/sdwd/benchmark-payload", json={"steps": 8, "batch_size": 2}
        )
        assert r.status_code == 200
        after = client.get("/sdwd/benchmark-payload").json()
        assert after["steps"] == 8 and after["batch_size"] == 2
        assert client.post(
            "/sdwd/benchmark-payload", json={"warp": 1}
        ).status_code == 422
        client.post("/sdwd/benchmark-payload",
                    json={"steps": 20, "batch_size": 1})

    def test_settings_rejects_unknown(self, client):
        r = client.post("/sdwd/settings", json={"warp_factor": 9})
        assert r.status_code == 422

    def test_status(self, client):
        r = client.get("/sdwd/status").json()
        assert len(r["workers"]) == 2
        assert r["workers"][0]["state"] in (
            "IDLE", "WORKING", "INTERRUPTED", "UNAVAILABLE", "DISABLED"
        )
        assert "speed_summary" in r

    def test_worker_disable_enable(self, client):
        assert client.post("/sdwd/worker/gpu1/disable").status_code == 200
        st = client.get("/sdwd/status").json()
        w1 = [w for w in st["workers"] if w["label"] == "gpu1"][0]
        assert w1["state"] == "DISABLED"
        assert client.post("/sdwd/worker/gpu1/enable").status_code == 200

    def test_worker_config_pixel_cap_and_override(self, client):
        """Ref ui.py:161-171/313-319 parity: pixel cap + checkpoint
        override are editable per worker through the API/UI."""
        r = client.post(
            "/sdwd/worker/gpu1/config",
            json={"pixel_cap": 512 * 512 * 2, "model_override": "tiny-xl"},
        )
        assert r.status_code == 200, r.text
        st = client.get("/sdwd/status").json()
        w1 = [w for w in st["workers"] if w["label"] == "gpu1"][0]
        assert w1["pixel_cap"] == 512 * 512 * 2
        assert w1["model_override"] == "tiny-xl"
        # clearing works; unknown model rejected; unknown field rejected
        r = client.post(
            "/sdwd/worker/gpu1/config",
            json={"pixel_cap": 0, "model_override": ""},
        )
        assert r.status_code == 200
        w1 = [
            w for w in client.get("/sdwd/status").json()["workers"]
            if w["label"] == "gpu1"
        ][0]
        assert w1["pixel_cap"] == 0 and w1["model_override"] == ""
        assert client.post(
            "/sdwd/worker/gpu1/config", json={"model_override": "nope"}
        ).status_code == 404
        assert client.post(
            "/sdwd/worker/gpu1/config", json={"bogus": 1}
        ).status_code == 422
        assert client.post(
            "/sdwd/worker/nope/config", json={"pixel_cap": 1}
        ).status_code == 404


class TestDynamicPromptsAPI:
    def test_per_image_expansion(self, client):
        body = {
            "prompt": "a {red|green|blue|black|white} cow",
            "steps": 1, "width": 64, "height": 64, "seed": 77,
            "batch_size": 3,
            "alwayson_scripts": {"dynamic prompts": {"args": [True]}},
        }
        r = client.post("/sdapi/v1/txt2img", json=body)
        assert r.status_code == 200, r.text
        info = r.json()["info"]
        # per-image prompts recorded; deterministic across repeat calls
        import json as _json
        prompts = _json.loads(info)["all_prompts"]
        assert len(prompts) == 3
        assert all(" cow" in p and "{" not in p for p in prompts)
        r2 = client.post("/sdapi/v1/txt2img", json=body)
        assert _json.loads(r2.json()["info"])["all_prompts"] == prompts

    def test_disabled_passthrough(self, client):
        body = {
            "prompt": "a {red|blue} cow", "steps": 1, "width": 64,
            "height": 64, "seed": 7,
            "alwayson_scripts": {"dynamic prompts": {"args": [False]}},
        }
        r = client.post("/sdapi/v1/txt2img", json=body)
        assert r.status_code == 200
        import json as _json
        ps = _json.loads(r.json()["info"])["all_prompts"]
        assert ps[0] == "a {red|blue} cow"  # literal, not expanded


class TestAlwaysonControlNet:
    def test_controlnet_unit_applied(self, client):
        hint = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
        b64 = base64.b64encode(encode_png(hint)).decode()
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "cn",
                "steps": 2,
                "width": 64,
                "height": 64,
                "seed": 3,
                "alwayson_scripts": {
                    "controlnet": {
                        "args": [
                            {"input_image": b64, "model": "controlnet-tiny",
                             "weight": 0.8}
                        ]
                    }
                },
            },
        )
        assert r.status_code == 200

    def test_hires_fields(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "hr", "steps": 2, "width": 64, "height": 64,
                "seed": 3, "enable_hr": True, "hr_scale": 2.0,
                "hr_second_pass_steps": 2,
            },
        )
        assert r.status_code == 200
        img = decode_png(base64.b64decode(r.json()["images"][0]))
        assert img.shape[0] == 128  # grid of one 128x128 image


class TestWebUI:
    def test_index_page(self, client):
        r = client.get("/")
        assert r.status_code == 200
        assert "sdwd_amd" in r.text
        assert "/sdwd/status" in r.text
        assert "/sdwd/settings" in r.text  # settings tab (ref Settings tab)


class TestRestart:
    def test_soft_restart(self, client):
        r = client.post("/sdapi/v1/server-restart")
        assert r.status_code == 200
        # still serves after restart
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "post-restart", "steps": 1, "width": 64,
                  "height": 64, "seed": 1},
        )
        assert r.status_code == 200


class TestBadInput:
    def test_malformed_init_image_422(self, client):
        r = client.post(
            "/sdapi/v1/img2img",
            json={"prompt": "x", "init_images": ["bm90YXBuZw=="],
                  "steps": 1, "width": 64, "height": 64},
        )
        assert r.status_code == 422


class TestGlobalOptions:
    def test_clip_skip_and_ensd_via_options(self, client):
        r = client.post(
            "/sdapi/v1/options",
            json={"CLIP_stop_at_last_layers": 2, "eta_noise_seed_delta": 31337},
        )
        assert r.status_code == 200
        opts = client.get("/sdapi/v1/options").json()
        assert opts["CLIP_stop_at_last_layers"] == 2
        assert opts["eta_noise_seed_delta"] == 31337
        # per-request override still wins
        r2 = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "g", "steps": 1, "width": 64, "height": 64,
                  "seed": 4,
                  "override_settings": {"CLIP_stop_at_last_layers": 1}},
        )
        assert r2.status_code == 200
        # reset
        client.post(
            "/sdapi/v1/options",
            json={"CLIP_stop_at_last_layers": 1, "eta_noise_seed_delta": 0},
        )

    def test_sd_vae_selection(self, client, tmp_path, monkeypatch):
        """sdwui SD-VAE dropdown flow: list, select, restore (ref C13:
        load_options synced vae by name, worker.py:646-688)."""
        from safetensors.torch import save_file

        from sdwd_amd.models.convert import to_ldm_state_dict
        from sdwd_amd.models.registry import load_model

        a = load_model("tiny", device="cpu", cache=False)
        with torch.no_grad():
            for p in a.vae.parameters():
                p.add_(torch.randn_like(p) * 0.05)
        vae_sd = {
            k: v.contiguous().clone()
            for k, v in to_ldm_state_dict(a).items()
            if k.startswith("first_stage_model.")
        }
        save_file(vae_sd, str(tmp_path / "anime.safetensors"))
        monkeypatch.setenv("SDWD_VAE_DIR", str(tmp_path))
        names = [e["model_name"] for e in client.get("/sdapi/v1/sd-vae").json()]
        assert names == ["auto", "anime"]
        assert client.post(
            "/sdapi/v1/options", json={"sd_vae": "anime"}
        ).status_code == 200
        assert client.get("/sdapi/v1/options").json()["sd_vae"] == "anime"
        assert client.post(
            "/sdapi/v1/options", json={"sd_vae": "missing-vae"}
        ).status_code == 404
        # "Automatic" restores the checkpoint's own VAE
        assert client.post(
            "/sdapi/v1/options", json={"sd_vae": "Automatic"}
        ).status_code == 200
        assert client.get("/sdapi/v1/options").json()["sd_vae"] == "auto"
        # per-request override: applies for one generation, restores after
        body = {"prompt": "v", "steps": 1, "width": 64, "height": 64,
                "seed": 6}
        plain = client.post("/sdapi/v1/txt2img", json=body).json()["images"]
        over = client.post(
            "/sdapi/v1/txt2img",
            json={**body, "override_settings": {"sd_vae": "anime"}},
        ).json()["images"]
        assert over[-1] != plain[-1]  # different VAE decodes differently
        assert client.get("/sdapi/v1/options").json()["sd_vae"] == "auto"
        again = client.post("/sdapi/v1/txt2img", json=body).json()["images"]
        assert again[-1] == plain[-1]  # global VAE untouched
        r = client.post(
            "/sdapi/v1/txt2img",
            json={**body, "override_settings": {"sd_vae": "nope"}},
        )
        assert r.status_code == 404


class TestResponseHygiene:
    def test_parameters_never_echo_tensors(self, client):
        init = torch.full((64, 64, 3), 90, dtype=torch.uint8)
        mask = torch.zeros(64, 64, 3, dtype=torch.uint8)
        mask[8:16, 8:16] = 255
        r = client.post(
            "/sdapi/v1/img2img",
            json={
                "prompt": "t", "steps": 1, "width": 64, "height": 64,
                "seed": 2, "init_images": [
                    base64.b64encode(encode_png(init)).decode()
                ],
                "mask": base64.b64encode(encode_png(mask)).decode(),
            },
        )
        assert r.status_code == 200
        params = r.json()["parameters"]
        assert params["init_images"] is None
        assert params["mask_image"] is None
        # the whole response stays small (no tensor-as-list blowup)
        assert len(r.content) < 200_000


class TestUnloadReload:
    def test_unload_then_reload_then_generate(self, client):
        assert client.post("/sdapi/v1/unload-checkpoint").status_code == 200
        assert client.post("/sdapi/v1/reload-checkpoint").status_code == 200
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "u", "steps": 1, "width": 64, "height": 64,
                  "seed": 3},
        )
        assert r.status_code == 200

    def test_interrogate_graceful(self, client):
        assert client.post("/sdapi/v1/interrogate").status_code == 501


class TestEmbeddingsApi:
    def test_loaded_listing_and_refresh(self, client, tmp_path,
                                        monkeypatch_module):
        from safetensors.torch import save_file

        from sdwd_amd.models import embeddings

        d = tmp_path / "embs"
        d.mkdir()
        save_file({"emb_params": torch.randn(2, 64)},
                  str(d / "apitrigger.safetensors"))
        monkeypatch_module.setenv("SDWD_EMBEDDINGS_DIR", str(d))
        try:
            r = client.post("/sdapi/v1/refresh-embeddings")
            assert r.json()["found"] == ["apitrigger"]
            listed = client.get("/sdapi/v1/embeddings").json()["loaded"]
            assert listed["apitrigger"]["vectors"] == 2
            # generation with the trigger goes through
            g = client.post(
                "/sdapi/v1/txt2img",
                json={"prompt": "an apitrigger cow", "steps": 1,
                      "width": 64, "height": 64, "seed": 9},
            )
            assert g.status_code == 200
        finally:
            embeddings.clear()


class TestExtrasBatch:
    def test_batch_upscale(self, client):
        from sdwd_amd.utils.images import decode_png, encode_png

        imgs = [torch.randint(0, 255, (8, 8, 3), dtype=torch.uint8)
                for _ in range(3)]
        payload = {
            "upscaling_resize": 2.0, "upscaler_1": "Nearest",
            "imageList": [
                {"data": base64.b64encode(encode_png(i)).decode(),
                 "name": f"i{n}.png"}
                for n, i in enumerate(imgs)
            ],
        }
        r = client.post("/sdapi/v1/extra-batch-images", json=payload)
        assert r.status_code == 200
        outs = r.json()["images"]
        assert len(outs) == 3
        for src, b in zip(imgs, outs):
            dec = decode_png(base64.b64decode(b))
            assert dec.shape == (16, 16, 3)
            assert torch.equal(dec[::2, ::2], src)


class TestGoldenWorkflow:
    """One realistic session end to end through the HTTP surface: the
    flows a reference user strings together (benchmark -> generate with
    hires -> iterate with soft inpaint -> inspect parameters -> utils)."""

    def test_session(self, client):
        import json as _json

        # 1. status + a re-benchmark on an edited (tiny) payload
        st = client.get("/sdwd/status").json()
        assert len(st["workers"]) == 2
        assert client.post("/sdwd/benchmark-payload", json={
            "width": 64, "height": 64, "steps": 1,
        }).status_code == 200
        r0 = client.post("/sdwd/benchmark")
        assert r0.status_code == 200
        assert all(v > 0 for v in r0.json()["speeds"].values())

        # 2. txt2img with hires fix + per-request model override fields
        r = client.post("/sdapi/v1/txt2img", json={
            "prompt": "a golden cow", "steps": 2, "width": 64,
            "height": 64, "seed": 41, "batch_size": 2,
            "enable_hr": True, "hr_scale": 2.0, "hr_second_pass_steps": 2,
            "denoising_strength": 0.6,
        })
        assert r.status_code == 200, r.text
        body = r.json()
        info = _json.loads(body["info"])
        assert info["all_seeds"] == [41, 42]
        gen_png = body["images"][-1]  # last = an image (first may be grid)

        # 3. png-info round-trips the parameters line
        pi = client.post("/sdapi/v1/png-info", json={"image": gen_png})
        assert pi.status_code == 200
        assert "a golden cow" in pi.json()["info"]
        assert "Hires upscale: 2.0" in pi.json()["info"]

        # 4. img2img soft-inpaint iteration on the generated image
        mask = torch.zeros(64, 64, 3, dtype=torch.uint8)
        mask[:, :32] = 255
        # the hires output is 128x128; send it back at 64x64
        from sdwd_amd.utils.images import decode_png, encode_png

        img = decode_png(base64.b64decode(gen_png))[::2, ::2].contiguous()
        r2 = client.post("/sdapi/v1/img2img", json={
            "prompt": "a silver cow", "steps": 2, "width": 64,
            "height": 64, "seed": 43, "denoising_strength": 0.9,
            "init_images": [base64.b64encode(encode_png(img)).decode()],
            "mask": base64.b64encode(encode_png(mask)).decode(),
            "alwayson_scripts": {
                "soft inpainting": {"args": [{"Soft inpainting": True}]}
            },
        })
        assert r2.status_code == 200, r2.text
        info2 = _json.loads(r2.json()["info"])
        assert "Soft inpainting: True" in info2["infotexts"][0]

        # 5. utils: interrupt (idle no-op) + restart-workers + progress
        assert client.post("/sdapi/v1/interrupt").status_code == 200
        assert client.post("/sdwd/restart-workers").status_code == 200
        pr = client.get("/sdapi/v1/progress").json()
        assert pr["progress"] in (0, 0.0) or pr["progress"] <= 1.0
        # the session leaves every rank schedulable
        st2 = client.get("/sdwd/status").json()
        assert all(w["state"] in ("IDLE",) for w in st2["workers"])


class TestWebUIWiring:
    def test_every_ui_fetch_target_exists(self, client):
        """The single-page UI's JS calls must map to real routes — catches
        a renamed endpoint breaking the page silently (no browser in CI)."""
        import re

        html = client.get("/").text
        static = set(re.findall(r"fetch\('([^']+)'", html))
        dynamic = set(re.findall(r"fetch\(`([^`]+)`", html))
        assert static, "UI lost its JS wiring?"
        known_paths = {r.path for r in client.app.routes}
        for ep in static:
            path = ep.split("?")[0]
            assert path in known_paths, f"UI references missing route {path}"
        for ep in dynamic:
            # template routes: substitute the parameters the JS uses
            path = (ep.split("?")[0]
                    .replace("${label}", "gpu0")
                    .replace("${act}", "enable"))
            r = client.post(path, json={})
            assert r.status_code < 500, (ep, r.status_code)
