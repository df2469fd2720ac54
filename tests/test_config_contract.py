"""Discovery config.yaml wire contract (reference pkg/discovery/
config_builder.go:54-145 and the service naming of api helper.go:106-116).
Engines parse this file from RBG_CONFIG_PATH; key names and the
`{rbg}-{role}-{i}` / `s-{rbg}-{role}` naming are load-bearing."""
import os

from rbg_amd.discovery.config_builder import (TopologyRegistry, build_config,
                                              dump_config, instance_name,
                                              load_config, service_name)
from tests.test_api_types import make_rbg


def test_naming_contract():
    assert instance_name("grp", "decode", 2) == "grp-decode-2"
    assert service_name("grp", "decode") == "s-grp-decode"


def test_build_config_schema():
    rbg = make_rbg("grp")
    role = rbg.spec.roles[0].name
    doc = build_config(rbg, {role: [
        {"name": f"grp-{role}-0", "address": "127.0.0.1:9001",
         "ports": [9001], "gpu_ids": [0], "ready": True},
        {"name": f"grp-{role}-1", "ready": False},
    ]})
    grp = doc["group"]
    assert grp["name"] == "grp" and grp["size"] == 2
    r = next(x for x in grp["roles"] if x["name"] == role)
    assert r["service"] == f"s-grp-{role}"
    first, second = r["instances"]
    assert set(first) == {"name", "address", "ports", "gpuIds", "ready"}
    assert first["gpuIds"] == [0] and first["ready"] is True
    # absent fields default rather than KeyError (partial status rows)
    assert second["address"] == "" and second["ports"] == [] \
        and second["ready"] is False


def test_registry_publish_roundtrip_and_remove(tmp_path):
    rbg = make_rbg("grp")
    role = rbg.spec.roles[0].name
    reg = TopologyRegistry(str(tmp_path))
    insts = {role: [{"name": f"grp-{role}-0", "ready": True}]}
    path = reg.publish(rbg, insts)
    assert path == reg.path_for("default", "grp")
    doc = load_config(path)
    assert doc == build_config(rbg, insts)
    # yaml round-trips bit-exactly through dump_config
    assert load_config(path) == doc and dump_config(doc)
    reg.remove("default", "grp")
    assert not os.path.exists(path)


def test_registry_legacy_mode_emits_per_role_files(tmp_path):
    """KEP-133 compatibility: legacy groups also get per-role config files
    so engines written against the old schema keep working."""
    rbg = make_rbg("grp")
    role = rbg.spec.roles[0].name
    reg = TopologyRegistry(str(tmp_path))
    reg.publish(rbg, {role: [{"name": f"grp-{role}-0"}]}, mode="legacy")
    role_doc = load_config(reg.role_path_for("default", "grp", role))
    assert role_doc["group"]["roles"][0]["name"] == role
    assert role_doc["group"]["size"] == 1
    reg.remove("default", "grp")
    assert not os.path.exists(reg.role_path_for("default", "grp", role))
