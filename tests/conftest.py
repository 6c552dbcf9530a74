
import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that require an MI355X GPU")


@pytest.fixture()
def sutro_home(tmp_path, monkeypatch):
    """Isolated service home so tests never touch ~/.sutro-amd."""
    home = str(tmp_path / "sutro-home")
    monkeypatch.setenv("SUTRO_AMD_HOME", home)
    return home


@pytest.fixture()
def local_client(sutro_home):
    """A Sutro client bound to an isolated home with tiny CPU engines."""
    from sutro_amd.sdk import Sutro

    client = Sutro(home=sutro_home, device="cpu",
                   engine_kwargs={"num_kv_blocks": 256, "max_model_len": 2048})
    yield client
    client.shutdown()


@pytest.fixture()
def tiny_engine():
    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.models.registry import tiny_spec_for_tests

    cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                       max_model_len=512, num_kv_blocks=128,
                       max_tokens_per_step=256)
    return LLMEngine(cfg)
