"""Prompt matrix + Prompts-from-file (native sdwui built-in selectable
scripts) — combination math, line parsing, engine runs, API dispatch."""
import base64
import json

import pytest
import torch

from sdwd_amd.parallel.builtin_scripts import (
    parse_prompt_line,
    prompt_matrix_prompts,
    run_prompt_matrix,
    run_prompts_from_file,
)


class TestMatrixPrompts:
    def test_combinations(self):
        assert prompt_matrix_prompts("a cow|red|hat") == [
            "a cow", "a cow, red", "a cow, hat", "a cow, red, hat",
        ]

    def test_put_at_start(self):
        assert prompt_matrix_prompts("a cow|red", put_at_start=True) == [
            "a cow", "red, a cow",
        ]

    def test_space_delimiter(self):
        assert prompt_matrix_prompts("a|b", delimiter=" ") == ["a", "a b"]

    def test_no_options(self):
        assert prompt_matrix_prompts("just a cow") == ["just a cow"]


class TestLineParsing:
    def test_bare_prompt(self):
        assert parse_prompt_line("a cow in a field") == {
            "prompt": "a cow in a field"
        }

    def test_options(self):
        out = parse_prompt_line('--prompt "a dog" --steps 3 --cfg_scale 4.5')
        assert out == {"prompt": "a dog", "steps": 3, "cfg_scale": 4.5}

    def test_unquoted_multiword(self):
        out = parse_prompt_line("--prompt a big dog --seed 7")
        assert out == {"prompt": "a big dog", "seed": 7}

    def test_dashes_alias(self):
        assert parse_prompt_line("--cfg-scale 3 --prompt x") == {
            "cfg_scale": 3.0, "prompt": "x"
        }

    def test_unknown_option(self):
        with pytest.raises(ValueError):
            parse_prompt_line("--bogus 1")

    def test_missing_value(self):
        with pytest.raises(ValueError):
            parse_prompt_line("--steps")


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


class TestRunMatrix:
    def test_two_by_two(self, engine):
        out = run_prompt_matrix(engine, _gen(prompt="a cow|red"))
        assert len(out["images"]) == 2
        assert out["prompts"] == ["a cow", "a cow, red"]
        # same fixed seed on every combo (sdwui default)
        assert out["seeds"] == [11, 11]
        assert out["grid"].shape[2] == 3
        assert not torch.equal(out["images"][0], out["images"][1])

    def test_different_seeds(self, engine):
        out = run_prompt_matrix(
            engine, _gen(prompt="a cow|red"), different_seeds=True
        )
        assert out["seeds"] == [11, 12]

    def test_negative_matrix(self, engine):
        out = run_prompt_matrix(
            engine, _gen(prompt="a cow", negative_prompt="blurry|dark"),
            prompt_type="negative",
        )
        assert len(out["images"]) == 2
        assert out["prompts"] == ["blurry", "blurry, dark"]

    def test_cap(self, engine):
        with pytest.raises(ValueError):
            run_prompt_matrix(engine, _gen(prompt="|".join("x" * 10)))


class TestRunFromFile:
    def test_two_lines(self, engine):
        out = run_prompts_from_file(
            engine, _gen(), 'a cow\n--prompt "a dog" --steps 3'
        )
        assert len(out["images"]) == 2
        assert not torch.equal(out["images"][0], out["images"][1])

    def test_iterate_seeds(self, engine):
        out = run_prompts_from_file(
            engine, _gen(batch_size=2), "a\nb\nc", checkbox_iterate=True
        )
        # 3 lines x batch 2; line seeds advance by the batch size
        assert out["seeds"] == [11, 12, 13, 14, 15, 16]

    def test_fixed_seeds_without_iterate(self, engine):
        out = run_prompts_from_file(engine, _gen(), "a\nb")
        assert out["seeds"] == [11, 11]

    def test_per_line_size_mixed(self, engine):
        out = run_prompts_from_file(
            engine, _gen(),
            "--prompt a --width 64 --height 64\n"
            "--prompt b --width 64 --height 32",
        )
        assert out["grid"] is None  # mixed sizes: no grid
        assert out["images"][0].shape == (64, 64, 3)
        assert out["images"][1].shape == (32, 64, 3)

    def test_empty(self, engine):
        with pytest.raises(ValueError):
            run_prompts_from_file(engine, _gen(), "   \n  ")


class TestScriptsApi:
    @pytest.fixture(scope="class")
    def client(self, tmp_path_factory):
        import os

        from fastapi.testclient import TestClient

        from sdwd_amd.api import create_app
        from sdwd_amd.parallel import LocalEngine

        os.environ["SDWD_CONFIG"] = str(
            tmp_path_factory.mktemp("cfg") / "c.json"
        )
        eng = LocalEngine(model="tiny", devices=["cpu", "cpu"])
        for w in eng.world.workers:
            w.eta.avg_ipm = 60.0
        return TestClient(create_app(engine=eng))

    def test_prompt_matrix(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "a cow|red", "steps": 2, "width": 64,
                "height": 64, "seed": 3,
                "script_name": "Prompt matrix",
                "script_args": [False, False, "positive", "comma", 0],
            },
        )
        assert r.status_code == 200, r.text
        body = r.json()
        info = json.loads(body["info"])
        assert info["all_prompts"] == ["a cow", "a cow, red"]
        assert len(body["images"]) == 3  # grid + 2
        assert len(base64.b64decode(body["images"][0])) > 100

    def test_prompts_from_file(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={
                "prompt": "ignored", "steps": 2, "width": 64, "height": 64,
                "seed": 5,
                "script_name": "Prompts from file or textbox",
                "script_args": [False, False, "a cow\n--prompt x --seed 9"],
            },
        )
        assert r.status_code == 200, r.text
        info = json.loads(r.json()["info"])
        assert info["all_seeds"] == [5, 9]
        assert info["all_prompts"] == ["a cow", "x"]  # per-line prompts

    def test_bad_line_rejected(self, client):
        r = client.post(
            "/sdapi/v1/txt2img",
            json={"prompt": "x",
                  "script_name": "prompts from file",
                  "script_args": [False, False, "--bogus 1"]},
        )
        assert r.status_code == 422

    def test_scripts_listed(self, client):
        names = client.get("/sdapi/v1/scripts").json()["txt2img"]
        assert "prompt matrix" in names
        assert "prompts from file or textbox" in names
