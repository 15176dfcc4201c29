"""Typed launch configuration → commands (reference: launcher/launch_config.py:91-216)."""

from __future__ import annotations

import sys
from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class TorchrunLaunchConfig:
    nproc_per_node: int = 1
    nnodes: int = 1
    node_rank: int = 0
    master_addr: str = "127.0.0.1"
    master_port: int = 29500
    script: str = ""
    script_args: List[str] = field(default_factory=list)

    def to_command(self, executor_path: str) -> List[str]:
        cmd = [
            sys.executable,
            "-m",
            "torch.distributed.run",
            f"--nnodes={self.nnodes}",
            f"--nproc-per-node={self.nproc_per_node}",
            f"--node-rank={self.node_rank}",
            f"--master-addr={self.master_addr}",
            f"--master-port={self.master_port}",
            executor_path,
            self.script,
        ]
        cmd.extend(self.script_args)
        return cmd

    @property
    def world_size(self) -> int:
        return self.nnodes * self.nproc_per_node


@dataclass
class AggregatorLaunchConfig:
    node_rank: int = 0

    @property
    def is_owner(self) -> bool:
        """Only node 0 runs the aggregator (reference: launch_config.py:143)."""
        return self.node_rank == 0


@dataclass
class RunIdentity:
    run_name: Optional[str] = None
    session_id: Optional[str] = None

    def validate(self, nnodes: int) -> None:
        if nnodes > 1 and not self.run_name:
            raise ValueError(
                "--run-name is required for multi-node runs (it keys the "
                "shared session directory)"
            )
