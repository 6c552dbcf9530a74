"""CLI command mirror (reference `cli.py`) via click's CliRunner."""


import pandas as pd
import pytest
from click.testing import CliRunner

from sutro_amd.cli import cli


@pytest.fixture()
def runner(sutro_home, monkeypatch, tmp_path):
    # isolate ~/.sutro config too
    monkeypatch.setenv("HOME", str(tmp_path))
    import sutro_amd.validation as v

    monkeypatch.setattr(v, "CONFIG_DIR", str(tmp_path / ".sutro"))
    monkeypatch.setattr(v, "CONFIG_PATH", str(tmp_path / ".sutro" / "config.json"))
    return CliRunner()


def test_models_lists_registry(runner):
    res = runner.invoke(cli, ["models"])
    assert res.exit_code == 0
    assert "qwen-3-32b" in res.output
    assert "mixtral-8x7b" in res.output


def test_quotas(runner):
    res = runner.invoke(cli, ["quotas"])
    assert res.exit_code == 0
    assert "p0:" in res.output and "p1:" in res.output


def test_login_and_set_base_url(runner):
    res = runner.invoke(cli, ["login"], input="my-key\n")
    assert res.exit_code == 0
    assert "Authenticated" in res.output
    res = runner.invoke(cli, ["set-base-url", "local"])
    assert res.exit_code == 0
    from sutro_amd.validation import load_config

    cfg = load_config()
    assert cfg["api_key"] == "my-key"
    assert cfg["base_url"] == "local"


def test_jobs_list_empty(runner):
    res = runner.invoke(cli, ["jobs", "list"])
    assert res.exit_code == 0


def test_datasets_roundtrip_cli(runner, tmp_path):
    res = runner.invoke(cli, ["datasets", "create"])
    assert res.exit_code == 0
    ds = res.output.strip().splitlines()[-1]
    assert ds.startswith("dataset-")
    p = tmp_path / "in.csv"
    pd.DataFrame({"t": ["a", "b"]}).to_csv(p, index=False)
    res = runner.invoke(cli, ["datasets", "upload", ds, str(p)])
    assert res.exit_code == 0
    res = runner.invoke(cli, ["datasets", "files", ds])
    assert "in.csv" in res.output
    res = runner.invoke(cli, ["datasets", "list"])
    assert ds in res.output


def test_cache_commands(runner):
    res = runner.invoke(cli, ["cache", "show"])
    assert res.exit_code == 0
    res = runner.invoke(cli, ["cache", "clear"])
    assert res.exit_code == 0
    assert "removed" in res.output


def test_docs(runner):
    assert runner.invoke(cli, ["docs"]).exit_code == 0
