"""traceml.yaml config resolution with four-level precedence:
explicit CLI arg > TRACEML_* env > traceml.yaml (walk-up <= 10 dirs) >
built-in defaults (reference: config/yaml_loader.py:1-68).
"""

from __future__ import annotations

import os
from dataclasses import fields
from typing import Optional

from traceml_amd.runtime.settings import ENV_PREFIX, TraceMLSettings

YAML_FILENAME = "traceml.yaml"
MAX_WALK_UP = 10

#: yaml key -> settings field (yaml uses the same names as settings fields)
YAML_KEYS = {f.name for f in fields(TraceMLSettings)}


def find_yaml(start_dir: Optional[str] = None) -> Optional[str]:
    current = os.path.abspath(start_dir or os.getcwd())
    for _ in range(MAX_WALK_UP):
        candidate = os.path.join(current, YAML_FILENAME)
        if os.path.isfile(candidate):
            return candidate
        parent = os.path.dirname(current)
        if parent == current:
            break
        current = parent
    return None


def load_yaml_settings(path: Optional[str] = None) -> dict:
    path = path or find_yaml()
    if path is None:
        return {}
    try:
        import yaml

        with open(path, "r", encoding="utf-8") as f:
            data = yaml.safe_load(f) or {}
    except Exception:
        return {}
    if not isinstance(data, dict):
        return {}
    return {k: v for k, v in data.items() if k in YAML_KEYS and v is not None}


def resolve_config(
    cli_overrides: Optional[dict] = None,
    environ=None,
    start_dir: Optional[str] = None,
) -> TraceMLSettings:
    environ = os.environ if environ is None else environ
    settings = TraceMLSettings()  # defaults
    for key, value in load_yaml_settings(find_yaml(start_dir)).items():
        setattr(settings, key, value)
    env_settings = TraceMLSettings.from_env(environ)
    for f in fields(TraceMLSettings):
        env_name = ENV_PREFIX + f.name.upper()
        if env_name in environ:
            setattr(settings, f.name, getattr(env_settings, f.name))
    for key, value in (cli_overrides or {}).items():
        if value is not None and key in YAML_KEYS:
            setattr(settings, key, value)
    return settings
