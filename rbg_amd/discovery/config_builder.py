"""Discovery config builder — the group-topology `config.yaml`.

Keeps the schema of the reference's discovery ConfigMap (reference:
pkg/discovery/config_builder.go:54-145, mounted at /etc/rbg): group
name/size/roles plus per-role instance addresses & ports — except that on one
MI355X node an "address" is a local endpoint (127.0.0.1:port) plus the GPU
ids the instance owns, so engines can open xGMI peer copies and join RCCL
groups directly from the topology file.
"""
from __future__ import annotations

import os
from typing import Any, Dict, List

import yaml



def instance_name(rbg_name: str, role: str, index: int) -> str:
    """Stable per-replica name `{rbg}-{role}-{i}` (reference helper.go:106-116)."""
    return f"{rbg_name}-{role}-{index}"


def service_name(rbg_name: str, role: str) -> str:
    """The headless-service-name analog `s-{rbg}-{role}`."""
    return f"s-{rbg_name}-{role}"


def build_config(rbg, instances: Dict[str, List[Dict[str, Any]]]) -> Dict[str, Any]:
    """Build the discovery document.

    ``instances`` maps role name -> list of dicts with keys
    {name, address, ports, gpu_ids, ready}; caller (the RBG controller)
    collects these from RoleInstance statuses.
    """
    roles = []
    group_size = 0
    for role in rbg.spec.roles:
        insts = instances.get(role.name, [])
        group_size += len(insts)
        roles.append({
            "name": role.name,
            "replicas": role.replicas,
            "service": service_name(rbg.metadata.name, role.name),
            "instances": [
                {
                    "name": i["name"],
                    "address": i.get("address", ""),
                    "ports": i.get("ports", []),
                    "gpuIds": i.get("gpu_ids", []),
                    "ready": bool(i.get("ready", False)),
                }
                for i in insts
            ],
        })
    return {
        "group": {
            "name": rbg.metadata.name,
            "size": group_size,
            "roles": roles,
        }
    }


def dump_config(doc: Dict[str, Any]) -> str:
    return yaml.safe_dump(doc, sort_keys=False)


def write_config(doc: Dict[str, Any], path: str) -> None:
    os.makedirs(os.path.dirname(path), exist_ok=True)
    tmp = path + ".tmp"
    with open(tmp, "w") as f:
        f.write(dump_config(doc))
    os.replace(tmp, path)      # atomic swap so engines never read a torn file


def load_config(path: str) -> Dict[str, Any]:
    with open(path) as f:
        return yaml.safe_load(f)


class TopologyRegistry:
    """In-proc topology store serialized to the config.yaml schema
    (SURVEY §2.3 "Topology registry"). The controller updates it each
    reconcile; engines read the file (path from RBG_CONFIG_PATH) or query a
    live registry in-process."""

    def __init__(self, root_dir: str):
        self.root_dir = root_dir

    def path_for(self, namespace: str, rbg_name: str) -> str:
        return os.path.join(self.root_dir, namespace, rbg_name, "config.yaml")

    def role_path_for(self, namespace: str, rbg_name: str,
                      role: str) -> str:
        """Legacy discovery mode: one config per role (the reference's
        pre-KEP-133 per-role ConfigMaps `{rbg}-{role}`)."""
        return os.path.join(self.root_dir, namespace, rbg_name,
                            f"config-{role}.yaml")

    def publish(self, rbg, instances: Dict[str, List[Dict[str, Any]]],
                mode: str = "refined") -> str:
        doc = build_config(rbg, instances)
        path = self.path_for(rbg.metadata.namespace, rbg.metadata.name)
        write_config(doc, path)
        if mode == "legacy":
            # legacy groups ALSO get per-role files so engines written
            # against the old per-role schema keep working (reference
            # ensureDiscoveryConfigMode keeps pre-existing groups on the
            # legacy ConfigMaps; KEP-133)
            for role_doc in doc["group"]["roles"]:
                write_config(
                    {"group": {"name": rbg.metadata.name,
                               "size": len(role_doc["instances"]),
                               "roles": [role_doc]}},
                    self.role_path_for(rbg.metadata.namespace,
                                       rbg.metadata.name,
                                       role_doc["name"]))
        return path

    def remove(self, namespace: str, rbg_name: str) -> None:
        import glob
        path = self.path_for(namespace, rbg_name)
        for p in [path] + glob.glob(os.path.join(
                os.path.dirname(path), "config-*.yaml")):
            try:
                os.remove(p)
            except FileNotFoundError:
                pass
