"""Rank identity resolution: env > torch.distributed > single-process defaults.

Resolves global/local rank, world/local-world size, node rank, hostname and
pid for this process. Dependency-injected (env mapping + torch loader) so
DDP identities are testable with fake environments and without process
groups (reference: runtime/identity.py:90-232).
"""

from __future__ import annotations

import os
import socket
from dataclasses import dataclass, field
from typing import Callable, Mapping, Optional


@dataclass(frozen=True)
class RuntimeIdentity:
    global_rank: int = 0
    local_rank: int = 0
    world_size: int = 1
    local_world_size: int = 1
    node_rank: int = 0
    hostname: str = field(default_factory=socket.gethostname)
    pid: int = field(default_factory=os.getpid)

    @property
    def rank(self) -> int:
        """Compat alias: the per-node (local) rank (reference identity.py:102-110)."""
        return self.local_rank

    def to_meta(self) -> dict:
        return {
            "rank": self.rank,
            "global_rank": self.global_rank,
            "local_rank": self.local_rank,
            "world_size": self.world_size,
            "local_world_size": self.local_world_size,
            "node_rank": self.node_rank,
            "hostname": self.hostname,
            "pid": self.pid,
        }


def _default_torch_loader():
    try:
        import torch.distributed as dist

        return dist
    except Exception:
        return None


def _int_env(env: Mapping[str, str], *names: str) -> Optional[int]:
    for name in names:
        raw = env.get(name)
        if raw is None or raw == "":
            continue
        try:
            return int(raw)
        except ValueError:
            continue
    return None


def resolve_runtime_identity(
    env: Optional[Mapping[str, str]] = None,
    torch_dist_loader: Callable = _default_torch_loader,
    hostname: Optional[str] = None,
) -> RuntimeIdentity:
    env = os.environ if env is None else env

    global_rank = _int_env(env, "RANK")
    local_rank = _int_env(env, "LOCAL_RANK")
    world_size = _int_env(env, "WORLD_SIZE")
    local_world_size = _int_env(env, "LOCAL_WORLD_SIZE")
    node_rank = _int_env(env, "GROUP_RANK", "NODE_RANK")

    if global_rank is None or world_size is None:
        dist = torch_dist_loader()
        if dist is not None:
            try:
                if dist.is_available() and dist.is_initialized():
                    if global_rank is None:
                        global_rank = dist.get_rank()
                    if world_size is None:
                        world_size = dist.get_world_size()
            except Exception:
                pass

    global_rank = 0 if global_rank is None else global_rank
    world_size = 1 if world_size is None else world_size
    if local_world_size is None:
        local_world_size = world_size if world_size <= 8 else 8
    if local_rank is None:
        local_rank = global_rank % max(1, local_world_size)
    if node_rank is None:
        node_rank = global_rank // max(1, local_world_size)

    return RuntimeIdentity(
        global_rank=global_rank,
        local_rank=local_rank,
        world_size=world_size,
        local_world_size=local_world_size,
        node_rank=node_rank,
        hostname=hostname if hostname is not None else socket.gethostname(),
        pid=os.getpid(),
    )
