"""X/Y/Z plot (native selectable script) — value parsing, sweep
execution through the engine, API dispatch (script_name/script_args)."""
import base64
import json

import pytest
import torch

from sdwd_amd.parallel.xyz import (
    AXIS_OPTIONS,
    parse_axis_values,
    resolve_axis,
    run_xyz,
)


class TestValueParsing:
    def test_int_list(self):
        assert parse_axis_values("int", "1, 2, 3") == [1, 2, 3]

    def test_int_range(self):
        assert parse_axis_values("int", "1-5") == [1, 2, 3, 4, 5]

    def test_int_range_step(self):
        assert parse_axis_values("int", "1-10 (+3)") == [1, 4, 7, 10]

    def test_int_range_count(self):
        assert parse_axis_values("int", "1-9 [3]") == [1, 5, 9]

    def test_int_mixed(self):
        assert parse_axis_values("int", "7, 1-3") == [7, 1, 2, 3]

    def test_float_range_step(self):
        vals = parse_axis_values("float", "1.0-2.0 (+0.5)")
        assert vals == [1.0, 1.5, 2.0]

    def test_float_count(self):
        vals = parse_axis_values("float", "0-1 [5]")
        assert vals == [0.0, 0.25, 0.5, 0.75, 1.0]

    def test_str_csv(self):
        assert parse_axis_values("str", "Euler a, DDIM") == ["Euler a", "DDIM"]

    def test_str_quoted_comma(self):
        assert parse_axis_values("str", '"a, b", c') == ["a, b", "c"]

    def test_list_passthrough(self):
        assert parse_axis_values("str", ["x", "y"]) == ["x", "y"]

    def test_negative_range(self):
        assert parse_axis_values("int", "3-1 (-1)") == [3, 2, 1]


class TestResolveAxis:
    def test_by_index(self):
        assert resolve_axis(4).name == "Steps"
        assert resolve_axis(0).name == "Nothing"

    def test_by_name(self):
        assert resolve_axis("CFG Scale").name == "CFG Scale"
        assert resolve_axis("steps").name == "Steps"

    def test_alias(self):
        assert resolve_axis("cfg").name == "CFG Scale"
        assert resolve_axis("model").name == "Checkpoint name"

    def test_unknown(self):
        with pytest.raises(ValueError):
            resolve_axis("No Such Axis")
        with pytest.raises(ValueError):
            resolve_axis(len(AXIS_OPTIONS))

    def test_published_axes_have_kinds(self):
        for o in AXIS_OPTIONS:
            assert o.kind in ("int", "float", "str")


@pytest.fixture(scope="module")
def engine():
    from sdwd_amd.parallel import LocalEngine

    eng = LocalEngine(model="tiny", devices=["cpu", "cpu"])
    for w in eng.world.workers:
        w.eta.avg_ipm = 60.0
    return eng


def _gen(**kw):
    from sdwd_amd.parallel import GenerationRequest

    base = dict(
        prompt="a cow", batch_size=1, width=64, height=64, steps=2, seed=11
    )
    base.update(kw)
    return GenerationRequest(**base)


class TestRunXYZ:
    def test_2x2_sweep(self, engine):
        out = run_xyz(
            engine, _gen(),
            "Steps", "1,2", "CFG Scale", "5.0, 7.0",
        )
        # grid: y rows x x cols of 64x64 cells
        assert out["grid"].shape == (128, 128, 3)
        assert len(out["images"]) == 4
        assert out["labels"]["x_axis"] == "Steps"
        assert out["labels"]["x_values"] == ["1", "2"]
        assert out["labels"]["y_values"] == ["5.0", "7.0"]
        # all cells share the request's fixed seed
        assert out["seeds"] == [11, 11, 11, 11]
        assert all("XYZ: Steps:" in i for i in out["infotexts"])

    def test_cells_actually_vary(self, engine):
        out = run_xyz(engine, _gen(), "Seed", "11, 12", 0, "")
        a = out["grid"][:, :64]
        b = out["grid"][:, 64:]
        assert not torch.equal(a, b)
        assert out["seeds"] == [11, 12]

    def test_z_axis_stacks(self, engine):
        out = run_xyz(
            engine, _gen(),
            "Steps", "1,2", 0, "", "CFG Scale", "5,7",
            include_sub_grids=True,
        )
        assert out["grid"].shape == (128, 128, 3)  # two 64x128 z grids
        assert len(out["sub_grids"]) == 2

    def test_prompt_sr(self, engine):
        out = run_xyz(
            engine, _gen(prompt="a red cow"),
            "Prompt S/R", "red, blue", 0, "",
        )
        assert len(out["images"]) == 2
        assert "a blue cow" in out["infotexts"][1]
        assert not torch.equal(out["images"][0], out["images"][1])

    def test_seed_fixed_once_when_random(self, engine):
        out = run_xyz(engine, _gen(seed=-1), "Steps", "1,2", 0, "")
        assert out["seeds"][0] == out["seeds"][1] != -1

    def test_nothing_axis(self, engine):
        out = run_xyz(engine, _gen(), "Steps", "1,2", "Nothing", "ignored")
        assert len(out["images"]) == 2
        assert out["labels"]["y_values"] == []

    def test_cell_cap(self, engine):
        with pytest.raises(ValueError):
            run_xyz(engine, _gen(), "Seed", "1-64", "Steps", "1-64", 0, "")

    def test_batch_lone_images(self, engine):
        out = run_xyz(
            engine, _gen(batch_size=2), "Steps", "1,2", 0, "",
            include_lone_images=True,
        )
        assert len(out["images"]) == 4  # 2 cells x batch 2
        assert out["grid"].shape == (64, 128, 3)  # first image per cell


class TestXYZApi:
    @pytest.fixture(scope="class")
    def client(self, tmp_path_factory):
        from fastapi.testclient import TestClient

        from sdwd_amd.api import create_app
        from sdwd_amd.parallel import LocalEngine

        import os

        os.environ["SDWD_CONFIG"] = str(
            tmp_path_factory.mktemp("cfg") / "c.json"
        )
        eng = LocalEngine(model="tiny", devices=["cpu", "cpu"])
        for w in eng.world.workers:
            w.eta.avg_ipm = 60.0
        return TestClient(create_app(engine=eng))

    def test_old_layout(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "a cow", "steps": 2, "width": 64, "height": 64,
                "seed": 3,
                "script_name": "X/Y/Z plot",
                "script_args": ["Steps", "1,2", "CFG Scale", "5,7",
                                 0, "", True, True, False, False],
            },
        )
        assert r.status_code == 200, r.text
        body = r.json()
        info = json.loads(body["info"])
        assert info["xyz_plot"]["x_axis"] == "Steps"
        # grid + 4 lone images
        assert len(body["images"]) == 5
        grid = body["images"][0]
        assert len(base64.b64decode(grid)) > 100

    def test_dropdown_layout(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "a cow", "steps": 2, "width": 64, "height": 64,
                "seed": 3,
                "script_name": "xyz plot",
                "script_args": [
                    4, "1,2", None,           # Steps (index), values
                    8, "", ["Euler a", "DDIM"],  # Sampler via dropdown
                    0, "", None,
                    True, False, False, False, 0, False,
                ],
            },
        )
        assert r.status_code == 200, r.text
        info = json.loads(r.json()["info"])
        assert info["xyz_plot"]["y_axis"] == "Sampler"
        assert info["xyz_plot"]["y_values"] == ["Euler a", "DDIM"]
        assert len(info["all_seeds"]) == 4

    def test_img2img_script(self, client):
        from sdwd_amd.utils.images import encode_png

        init = torch.randint(0, 255, (64, 64, 3), dtype=torch.uint8)
        b64 = base64.b64encode(encode_png(init)).decode()
        r = client.post(
            "/sdapi/v1/img2img",
            json={
                "prompt": "re", "steps": 2, "width": 64, "height": 64,
                "seed": 5, "init_images": [b64],
                "script_name": "x/y/z plot",
                "script_args": ["Denoising", "0.4, 0.8", 0, "", 0, ""],
            },
        )
        assert r.status_code == 200, r.text
        assert len(json.loads(r.json()["info"])["all_seeds"]) == 2

    def test_unknown_script_rejected(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "x", "script_name": "Ultimate SD upscale",
                  "script_args": []},
        )
        assert r.status_code == 422
        assert "not available" in r.json()["detail"]

    def test_missing_args_rejected(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "x", "script_name": "x/y/z plot"},
        )
        assert r.status_code == 422

    def test_bad_axis_rejected(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "x", "script_name": "x/y/z plot",
                  "script_args": ["Bogus Axis", "1", 0, "", 0, ""]},
        )
        assert r.status_code == 422

    def test_scripts_route_reports_xyz(self, client):
        r = client.get("/sdapi/v1/scripts")
        assert "x/y/z plot" in r.json()["txt2img"]
        info = client.get("/sdapi/v1/script-info").json()
        xyz = [e for e in info if e["name"] == "x/y/z plot"]
        assert xyz and xyz[0]["args"]

    def test_skip_route(self, client):
        assert client.post("/sdapi/v1/skip").status_code == 200

    def test_unknown_sampler_axis_value_rejected(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "x", "script_name": "x/y/z plot",
                  "script_args": ["Sampler", "Euler a, NotASampler",
                                   0, "", 0, ""]},
        )
        assert r.status_code == 422
        assert "NotASampler" in r.json()["detail"]

    def test_unknown_checkpoint_axis_value_rejected(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "x", "script_name": "x/y/z plot",
                  "script_args": ["Checkpoint name", "no-such-model",
                                   0, "", 0, ""]},
        )
        assert r.status_code == 422
        assert "no-such-model" in r.json()["detail"]


class TestValueParsingProperties:
    """Hypothesis laws for the sdwui range grammar."""

    from hypothesis import given, settings as hsettings
    from hypothesis import strategies as st

    @given(a=st.integers(-50, 50), b=st.integers(-50, 50))
    @hsettings(max_examples=60, deadline=None)
    def test_int_range_is_inclusive_and_monotone(self, a, b):
        if a > b:
            a, b = b, a
        vals = parse_axis_values("int", f"{a}-{b}")
        assert vals[0] == a and vals[-1] == b
        assert vals == sorted(vals)
        assert len(vals) == b - a + 1

    @given(a=st.integers(-20, 20), b=st.integers(-20, 20),
           n=st.integers(2, 9))
    @hsettings(max_examples=60, deadline=None)
    def test_count_form_hits_endpoints(self, a, b, n):
        vals = parse_axis_values("int", f"{a}-{b} [{n}]")
        assert len(vals) == n
        assert vals[0] == a and vals[-1] == b

    @given(a=st.integers(-20, 20), b=st.integers(0, 40),
           s=st.integers(1, 7))
    @hsettings(max_examples=60, deadline=None)
    def test_step_form_never_overshoots(self, a, b, s):
        hi = a + b
        vals = parse_axis_values("int", f"{a}-{hi} (+{s})")
        assert vals[0] == a
        assert all(v <= hi for v in vals)
        assert all(y - x == s for x, y in zip(vals, vals[1:]))

    @given(xs=st.lists(st.integers(-99, 99), min_size=1, max_size=6))
    @hsettings(max_examples=60, deadline=None)
    def test_comma_list_round_trips(self, xs):
        vals = parse_axis_values("int", ", ".join(str(x) for x in xs))
        assert vals == xs
