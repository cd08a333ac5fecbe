"""Layered configuration for fei_amd.

Precedence (reference parity — fei/utils/config.py:406-501):

    1. environment variables:
         ``FEI_<SECTION>_<OPTION>`` (e.g. ``FEI_ENGINE_MODEL``)
         ``<PROVIDER>_API_KEY`` for ``<provider>.api_key`` lookups
         ``LLM_API_KEY`` as the final api-key fallback
    2. ``.env`` files (multi-location, *never* overriding pre-set env vars)
    3. the ini file (``~/.fei.ini`` by default)
    4. schema defaults

The reference hardens file permissions on the ini path (config.py:293-318)
— we keep that (chmod 600 on write).
"""

from __future__ import annotations

import configparser
import os
import threading
from typing import Any, Dict, List, Optional

from fei_amd.utils.logging import get_logger

logger = get_logger("utils.config")

# Schema: section.option -> (default, type). Types: str, int, float, bool.
CONFIG_SCHEMA: Dict[str, Dict[str, Any]] = {
    "llm": {
        "provider": ("local", str),       # "local" (engine) | "stub" (echo) | "scripted"
        "model": ("llama3-8b", str),
        "max_tokens": (4000, int),
        "temperature": (0.0, float),
        "api_key": ("", str),
    },
    "engine": {
        "tp": (1, int),
        "dtype": ("bf16", str),
        "max_seq_len": (4096, int),
        "kv_block_size": (16, int),
        "gpu_mem_fraction": (0.9, float),
        "use_hip_graph": (True, bool),
    },
    "memdir": {
        "base": ("", str),                 # empty -> ./Memdir (memdir.utils default)
        "server_port": (5000, int),
        "api_key": ("", str),
    },
    "memorychain": {
        "node": ("localhost:6789", str),
        "port": (6789, int),
        "difficulty": (2, int),
    },
    "log": {
        "level": ("WARNING", str),
        "file": ("", str),
    },
}

_DOTENV_LOCATIONS = [".env", os.path.join(os.path.expanduser("~"), ".fei", ".env")]


def _parse_dotenv(path: str) -> Dict[str, str]:
    out: Dict[str, str] = {}
    try:
        with open(path, "r", encoding="utf-8") as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#") or "=" not in line:
                    continue
                key, _, value = line.partition("=")
                key = key.strip()
                value = value.strip().strip("'\"")
                if key:
                    out[key] = value
    except OSError:
        pass
    return out


def _coerce(value: Any, typ: type) -> Any:
    if value is None:
        return None
    if typ is bool:
        if isinstance(value, bool):
            return value
        return str(value).strip().lower() in ("1", "true", "yes", "on")
    if typ is int:
        return int(str(value).strip())
    if typ is float:
        return float(str(value).strip())
    return str(value)


class Config:
    """Layered config. Thread-safe for get/set."""

    def __init__(self, ini_path: Optional[str] = None, load_dotenv: bool = True):
        self._lock = threading.RLock()
        self.ini_path = ini_path or os.path.join(os.path.expanduser("~"), ".fei.ini")
        self._ini = configparser.ConfigParser()
        self._dotenv: Dict[str, str] = {}
        if load_dotenv:
            self._load_dotenv_files()
        self._load_ini()

    # -- loading -------------------------------------------------------------

    def _load_dotenv_files(self) -> None:
        """Load .env values; pre-set environment variables always win
        (reference parity: config.py:320-365 preserves pre-set env vars)."""
        for loc in _DOTENV_LOCATIONS:
            for key, value in _parse_dotenv(loc).items():
                if key not in self._dotenv:
                    self._dotenv[key] = value

    def _load_ini(self) -> None:
        try:
            if os.path.exists(self.ini_path):
                self._ini.read(self.ini_path)
        except (OSError, configparser.Error) as e:
            logger.warning("failed reading ini %s: %s", self.ini_path, e)

    def _secure_path(self, path: str) -> None:
        try:
            os.chmod(path, 0o600)
        except OSError:
            pass

    # -- env resolution ------------------------------------------------------

    def _get_from_env(self, section: str, option: str) -> Optional[str]:
        """Env var resolution incl. provider api-key conventions
        (reference: config.py:470-501)."""
        candidates: List[str] = [f"FEI_{section.upper()}_{option.upper()}"]
        if option == "api_key":
            candidates.append(f"{section.upper()}_API_KEY")
            candidates.append("LLM_API_KEY")
        for name in candidates:
            if name in os.environ:
                return os.environ[name]
            if name in self._dotenv:
                return self._dotenv[name]
        return None

    # -- public API ----------------------------------------------------------

    def get(self, key: str, default: Any = None) -> Any:
        """Look up ``section.option`` through the precedence chain."""
        if "." not in key:
            raise ValueError(f"config key must be 'section.option', got {key!r}")
        section, option = key.split(".", 1)
        with self._lock:
            env_val = self._get_from_env(section, option)
            if env_val is not None:
                return env_val
            if self._ini.has_option(section, option):
                return self._ini.get(section, option)
            schema = CONFIG_SCHEMA.get(section, {}).get(option)
            if schema is not None:
                return schema[0]
            return default

    def get_typed(self, key: str, default: Any = None) -> Any:
        """Like :meth:`get` but coerced to the schema type."""
        section, option = key.split(".", 1)
        schema = CONFIG_SCHEMA.get(section, {}).get(option)
        value = self.get(key, default)
        if schema is None or value is None:
            return value
        try:
            return _coerce(value, schema[1])
        except (TypeError, ValueError):
            logger.warning("bad value for %s: %r, using default", key, value)
            return schema[0]

    def get_int(self, key: str, default: int = 0) -> int:
        try:
            return int(self.get(key, default))
        except (TypeError, ValueError):
            return default

    def get_bool(self, key: str, default: bool = False) -> bool:
        return bool(_coerce(self.get(key, default), bool))

    def get_section(self, section: str) -> Dict[str, Any]:
        out: Dict[str, Any] = {}
        for option in CONFIG_SCHEMA.get(section, {}):
            out[option] = self.get_typed(f"{section}.{option}")
        if self._ini.has_section(section):
            for option in self._ini.options(section):
                out.setdefault(option, self._ini.get(section, option))
        return out

    def set(self, key: str, value: Any, persist: bool = False) -> None:
        section, option = key.split(".", 1)
        with self._lock:
            if not self._ini.has_section(section):
                self._ini.add_section(section)
            self._ini.set(section, option, str(value))
            if persist:
                os.makedirs(os.path.dirname(self.ini_path) or ".", exist_ok=True)
                with open(self.ini_path, "w", encoding="utf-8") as f:
                    self._ini.write(f)
                self._secure_path(self.ini_path)

    def delete(self, key: str, persist: bool = False) -> bool:
        section, option = key.split(".", 1)
        with self._lock:
            removed = self._ini.remove_option(section, option) if self._ini.has_section(section) else False
            if removed and persist:
                with open(self.ini_path, "w", encoding="utf-8") as f:
                    self._ini.write(f)
                self._secure_path(self.ini_path)
            return removed


_GLOBAL: Optional[Config] = None
_GLOBAL_LOCK = threading.Lock()


def get_config(reload: bool = False) -> Config:
    """Global config singleton (reference: config.py:240-258)."""
    global _GLOBAL
    with _GLOBAL_LOCK:
        if _GLOBAL is None or reload:
            _GLOBAL = Config()
        return _GLOBAL
