"""Launch-context capture: cwd, sys.path, argv, env subset — recorded into
the run manifest so a session is reproducible/debuggable after the fact
(reference: runtime/launch_context.py)."""

from __future__ import annotations

import os
import sys
from dataclasses import dataclass, field
from typing import Dict, List


_CAPTURED_ENV_PREFIXES = ("TRACEML_", "TORCHELASTIC_", "MASTER_", "HIP_",
                          "ROCR_", "HSA_", "PYTORCH_", "MIOPEN_", "RCCL_",
                          "NCCL_")
_CAPTURED_ENV_KEYS = ("RANK", "LOCAL_RANK", "WORLD_SIZE", "LOCAL_WORLD_SIZE",
                      "GROUP_RANK", "NODE_RANK", "CUDA_VISIBLE_DEVICES",
                      "ROCR_VISIBLE_DEVICES")


@dataclass
class LaunchContext:
    cwd: str
    argv: List[str]
    sys_path_head: List[str]
    python: str
    env: Dict[str, str] = field(default_factory=dict)

    @classmethod
    def capture(cls) -> "LaunchContext":
        env = {}
        for key, value in os.environ.items():
            if key in _CAPTURED_ENV_KEYS or key.startswith(_CAPTURED_ENV_PREFIXES):
                env[key] = value
        return cls(
            cwd=os.getcwd(),
            argv=list(sys.argv),
            sys_path_head=sys.path[:5],
            python=sys.version.split()[0],
            env=env,
        )

    def to_payload(self) -> dict:
        return {
            "cwd": self.cwd,
            "argv": self.argv,
            "sys_path_head": self.sys_path_head,
            "python": self.python,
            "env": self.env,
        }
