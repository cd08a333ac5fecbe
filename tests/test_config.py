"""Config precedence tests (parity with the reference's
tests/test_key_precedence.py, tests/test_env_config*.py intent)."""

import os

import pytest

from fei_amd.utils.config import Config


@pytest.fixture
def clean_env(monkeypatch):
    for var in list(os.environ):
        if var.startswith("FEI_") or var.endswith("_API_KEY") or var == "LLM_API_KEY":
            monkeypatch.delenv(var, raising=False)
    return monkeypatch


def test_schema_default(clean_env, tmp_path):
    cfg = Config(ini_path=str(tmp_path / "fei.ini"), load_dotenv=False)
    assert cfg.get("llm.provider") == "local"
    assert cfg.get_typed("engine.tp") == 1
    assert cfg.get_typed("engine.use_hip_graph") is True


def test_env_beats_ini(clean_env, tmp_path):
    ini = tmp_path / "fei.ini"
    ini.write_text("[llm]\nprovider = stub\n")
    cfg = Config(ini_path=str(ini), load_dotenv=False)
    assert cfg.get("llm.provider") == "stub"
    clean_env.setenv("FEI_LLM_PROVIDER", "scripted")
    assert cfg.get("llm.provider") == "scripted"


def test_provider_api_key_env(clean_env, tmp_path):
    clean_env.setenv("LLM_API_KEY", "fallback-key")
    cfg = Config(ini_path=str(tmp_path / "fei.ini"), load_dotenv=False)
    assert cfg.get("llm.api_key") == "fallback-key"
    clean_env.setenv("LLM_API_KEY", "fallback-key")
    clean_env.setenv("ANTHROPIC_API_KEY", "provider-key")
    assert cfg.get("anthropic.api_key") == "provider-key"
    clean_env.setenv("FEI_ANTHROPIC_API_KEY", "fei-key")
    assert cfg.get("anthropic.api_key") == "fei-key"


def test_set_and_persist(clean_env, tmp_path):
    ini = tmp_path / "sub" / "fei.ini"
    cfg = Config(ini_path=str(ini), load_dotenv=False)
    cfg.set("llm.model", "llama3-70b", persist=True)
    cfg2 = Config(ini_path=str(ini), load_dotenv=False)
    assert cfg2.get("llm.model") == "llama3-70b"
    mode = os.stat(ini).st_mode & 0o777
    assert mode == 0o600


def test_typed_coercion(clean_env, tmp_path):
    clean_env.setenv("FEI_ENGINE_TP", "8")
    clean_env.setenv("FEI_ENGINE_USE_HIP_GRAPH", "false")
    cfg = Config(ini_path=str(tmp_path / "fei.ini"), load_dotenv=False)
    assert cfg.get_typed("engine.tp") == 8
    assert cfg.get_typed("engine.use_hip_graph") is False


def test_bad_value_falls_back(clean_env, tmp_path):
    clean_env.setenv("FEI_ENGINE_TP", "not-a-number")
    cfg = Config(ini_path=str(tmp_path / "fei.ini"), load_dotenv=False)
    assert cfg.get_typed("engine.tp") == 1


def test_dotenv_loaded_but_env_wins(clean_env, tmp_path, monkeypatch):
    """.env values load but never override pre-set environment variables
    (reference contract: config.py:320-365)."""
    monkeypatch.chdir(tmp_path)
    (tmp_path / ".env").write_text("FEI_LLM_MODEL=model-from-dotenv\n"
                                   "LLM_API_KEY=dotenv-key\n")
    cfg = Config(ini_path=str(tmp_path / "fei.ini"), load_dotenv=True)
    assert cfg.get("llm.model") == "model-from-dotenv"
    assert cfg.get("llm.api_key") == "dotenv-key"
    clean_env.setenv("FEI_LLM_MODEL", "model-from-env")
    assert cfg.get("llm.model") == "model-from-env"


def test_get_section_and_delete(clean_env, tmp_path):
    cfg = Config(ini_path=str(tmp_path / "fei.ini"), load_dotenv=False)
    cfg.set("engine.tp", 4)
    section = cfg.get_section("engine")
    assert section["tp"] == 4
    assert cfg.delete("engine.tp") is True
    assert cfg.get_typed("engine.tp") == 1
