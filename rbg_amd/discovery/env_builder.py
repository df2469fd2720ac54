"""Env injection for engine processes — the RBG_* contract.

Mirrors reference pkg/discovery/env_builder.go:33-141 (identity env via
downward API + RBG_LWP_* for leader-worker rank bootstrap) and
injector.go:183-246 (ordered merge preserving user-set values).  MI355X
additions: RBG_GPU_IDS / RBG_MASTER_* so an engine can hipSetDevice and join
its RCCL communicator without any service discovery round-trip.
"""
from __future__ import annotations

from typing import Dict, List

from ..api import constants as C
from ..api.types import EngineSpec, EnvVar


def identity_env(group_name: str, role_name: str, role_index: int,
                 instance_name: str, component_name: str = "",
                 component_index: int = 0, group_size: int = 0,
                 config_path: str = "") -> Dict[str, str]:
    env = {
        C.ENV_GROUP_NAME: group_name,
        C.ENV_ROLE_NAME: role_name,
        C.ENV_ROLE_INDEX: str(role_index),
        C.ENV_ROLE_INSTANCE_NAME: instance_name,
    }
    if component_name:
        env[C.ENV_COMPONENT_NAME] = component_name
        env[C.ENV_COMPONENT_INDEX] = str(component_index)
    if group_size:
        env[C.ENV_GROUP_SIZE] = str(group_size)
    if config_path:
        env[C.ENV_CONFIG_PATH] = config_path
    return env


def leader_worker_env(leader_address: str, worker_index: int,
                      group_size: int) -> Dict[str, str]:
    """Rank topology for TP/PP groups (reference env_builder.go:50-74). The
    engine derives: rank = RBG_LWP_WORKER_INDEX, world = RBG_LWP_GROUP_SIZE,
    rendezvous at RBG_LWP_LEADER_ADDRESS -> RCCL communicator over xGMI."""
    return {
        C.ENV_LWP_LEADER_ADDRESS: leader_address,
        C.ENV_LWP_WORKER_INDEX: str(worker_index),
        C.ENV_LWP_GROUP_SIZE: str(group_size),
    }


def device_env(gpu_ids: List[int], master_addr: str = "127.0.0.1",
               master_port: int = 0) -> Dict[str, str]:
    env = {C.ENV_GPU_IDS: ",".join(str(g) for g in gpu_ids)}
    if master_port:
        env[C.ENV_MASTER_ADDR] = master_addr
        env[C.ENV_MASTER_PORT] = str(master_port)
    return env


def merge_env(engine: EngineSpec, injected: Dict[str, str]) -> List[EnvVar]:
    """Ordered merge: injected identity env first, then the user's env so
    `$(VAR)` references to injected values resolve and user values win on
    collision (reference injector.go:183-246)."""
    out: List[EnvVar] = [EnvVar(name=k, value=v) for k, v in injected.items()]
    user_names = {e.name for e in engine.env}
    out = [e for e in out if e.name not in user_names]
    out.extend(EnvVar(name=e.name, value=e.value) for e in engine.env)
    return out


def env_as_dict(env: List[EnvVar]) -> Dict[str, str]:
    return {e.name: e.value for e in env}
