"""Runtime settings + the TRACEML_* environment-variable contract.

Precedence (applied in config/yaml_loader.resolve_config): explicit arg/CLI >
``TRACEML_*`` env > ``traceml.yaml`` > defaults
(reference: runtime/settings.py:22-89, config/yaml_loader.py:1-66).
The env-var set is the inter-process contract between the launcher, the
aggregator process and the training ranks (reference commands.py:371-415).
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field, fields
from typing import Optional

DEFAULT_AGGREGATOR_PORT = 29765
DEFAULT_DASHBOARD_PORT = 8765
DEFAULT_INTERVAL_SEC = 2.0
DEFAULT_MODE = "summary"
DEFAULT_LOGS_DIR = "./logs"
DEFAULT_FINALIZE_TIMEOUT_SEC = 300.0

ENV_PREFIX = "TRACEML_"


def _env_name(field_name: str) -> str:
    return ENV_PREFIX + field_name.upper()


@dataclass
class TraceMLSettings:
    session_id: Optional[str] = None
    run_name: Optional[str] = None
    mode: str = DEFAULT_MODE  # cli | dashboard | summary
    interval: float = DEFAULT_INTERVAL_SEC
    logs_dir: str = DEFAULT_LOGS_DIR
    aggregator_host: str = "127.0.0.1"
    aggregator_bind: str = "127.0.0.1"
    aggregator_port: int = DEFAULT_AGGREGATOR_PORT
    dashboard_port: int = DEFAULT_DASHBOARD_PORT
    finalize_timeout: float = DEFAULT_FINALIZE_TIMEOUT_SEC
    trace_max_steps: Optional[int] = None
    enable_logging: bool = True
    disabled: bool = False
    html_report: bool = False
    expected_ranks: Optional[int] = None
    rank_stats_rccl: bool = True  # MI355X: RCCL-over-xGMI rank-stats exchange
    ddp_comm_timing: bool = True  # explicit ddp_comm phase via comm-stream stamps

    def to_env(self) -> dict:
        """Serialize to the TRACEML_* env contract (skip Nones)."""
        out = {}
        for f in fields(self):
            value = getattr(self, f.name)
            if value is None:
                continue
            if isinstance(value, bool):
                value = "1" if value else "0"
            out[_env_name(f.name)] = str(value)
        return out

    @classmethod
    def from_env(cls, environ=None) -> "TraceMLSettings":
        environ = os.environ if environ is None else environ
        kwargs = {}
        for f in fields(cls):
            raw = environ.get(_env_name(f.name))
            if raw is None:
                continue
            kwargs[f.name] = _coerce(f, raw)
        return cls(**kwargs)


def _coerce(f, raw: str):
    type_name = str(f.type)
    if "bool" in type_name:
        return raw.strip().lower() in ("1", "true", "yes", "on")
    if "int" in type_name:
        try:
            return int(raw)
        except ValueError:
            return None
    if "float" in type_name:
        try:
            return float(raw)
        except ValueError:
            return None
    return raw


def apply_settings_to_env(settings: TraceMLSettings, environ=None) -> None:
    environ = os.environ if environ is None else environ
    environ.update(settings.to_env())
