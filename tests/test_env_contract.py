"""RBG_* env wire contract — the engine-facing names an external engine
scripts against (reference pkg/discovery/env_builder.go:33-141 and the
ordered merge of injector.go:183-246).  These are literal-string pins:
renaming a key is a breaking change for every engine launch script."""
from rbg_amd.api.types import EngineSpec, EnvVar
from rbg_amd.discovery.env_builder import (device_env, env_as_dict,
                                           identity_env, leader_worker_env,
                                           merge_env)


def test_identity_env_wire_names():
    env = identity_env("grp", "decode", 3, "grp-decode-3",
                       component_name="leader", component_index=0,
                       group_size=5, config_path="/etc/rbg/config.yaml")
    assert env == {
        "RBG_GROUP_NAME": "grp",
        "RBG_ROLE_NAME": "decode",
        "RBG_ROLE_INDEX": "3",
        "RBG_ROLE_INSTANCE_NAME": "grp-decode-3",
        "RBG_COMPONENT_NAME": "leader",
        "RBG_COMPONENT_INDEX": "0",
        "RBG_GROUP_SIZE": "5",
        "RBG_CONFIG_PATH": "/etc/rbg/config.yaml",
    }
    # optional fields are OMITTED, not emitted empty (downward-API parity)
    bare = identity_env("g", "r", 0, "g-r-0")
    assert set(bare) == {"RBG_GROUP_NAME", "RBG_ROLE_NAME",
                         "RBG_ROLE_INDEX", "RBG_ROLE_INSTANCE_NAME"}


def test_leader_worker_rank_env():
    env = leader_worker_env("10.0.0.1:29500", worker_index=2, group_size=4)
    assert env == {
        "RBG_LWP_LEADER_ADDRESS": "10.0.0.1:29500",
        "RBG_LWP_WORKER_INDEX": "2",
        "RBG_LWP_GROUP_SIZE": "4",
    }


def test_device_env_csv_and_optional_master():
    assert device_env([0, 2, 5]) == {"RBG_GPU_IDS": "0,2,5"}
    env = device_env([1], master_port=29501)
    assert env == {"RBG_GPU_IDS": "1",
                   "RBG_MASTER_ADDR": "127.0.0.1",
                   "RBG_MASTER_PORT": "29501"}


def test_merge_env_injected_first_user_wins():
    engine = EngineSpec(name="e", env=[
        EnvVar(name="RBG_ROLE_NAME", value="user-override"),
        EnvVar(name="EXTRA", value="$(RBG_GROUP_NAME)-suffix"),
    ])
    merged = merge_env(engine, identity_env("grp", "prefill", 0,
                                            "grp-prefill-0"))
    names = [e.name for e in merged]
    # injected identity env comes first so later $(VAR) references resolve
    assert names.index("RBG_GROUP_NAME") < names.index("EXTRA")
    # user value wins on collision and the key is not duplicated
    assert names.count("RBG_ROLE_NAME") == 1
    assert env_as_dict(merged)["RBG_ROLE_NAME"] == "user-override"
    assert env_as_dict(merged)["RBG_GROUP_NAME"] == "grp"
