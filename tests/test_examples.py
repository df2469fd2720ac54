"""Every shipped example YAML must parse, validate and carry meaningful
knobs — the rot guard for `examples/` (the reference ships its examples
under examples/inference/ and CI-applies them)."""
import glob
import os

import pytest
import yaml

from rbg_amd.api import constants as C
from rbg_amd.api.types import RoleBasedGroup, load_object
from rbg_amd.api.validation import validate_rbg

EXAMPLES = sorted(glob.glob(
    os.path.join(os.path.dirname(__file__), "..", "examples", "*.yaml")))


def _docs(path):
    with open(path) as f:
        return [d for d in yaml.safe_load_all(f) if d]


@pytest.mark.parametrize("path", EXAMPLES, ids=os.path.basename)
def test_example_parses_and_validates(path):
    assert EXAMPLES, "examples/ directory is empty"
    for doc in _docs(path):
        obj = load_object(doc)
        assert obj.kind in C.ALL_KINDS
        assert obj.metadata.name
        if isinstance(obj, RoleBasedGroup):
            validate_rbg(obj)      # raises on any schema violation
            assert obj.spec.roles


def test_showcase_example_round_trips_annotations():
    """pd-sticky-exclusive.yaml exercises the full annotation vocabulary;
    each key must survive the load and be one the controllers consume."""
    path = os.path.join(os.path.dirname(__file__), "..",
                        "examples", "pd-sticky-exclusive.yaml")
    (doc,) = _docs(path)
    rbg = load_object(doc)
    annos = rbg.metadata.annotations
    assert annos[C.ANNO_GANG_SCHEDULING] == "true"
    assert annos[C.ANNO_GANG_TIMEOUT] == "120"
    assert annos[C.ANNO_EXCLUSIVE_TOPOLOGY] == "xgmi-hive"
    assert annos[C.ANNO_KV_TRANSFER] == "peer"
    assert annos[C.ANNO_INPLACE_SCHEDULING] == "required"
    assert annos[C.ANNO_INPLACE_GRANULARITY] == "instance"
    assert annos[C.ANNO_DISCOVERY_MODE] == "refined"
    decode = rbg.spec.role("decode")
    assert decode.rollout_strategy.rolling_update.max_surge == 1
    assert decode.rollout_strategy.rolling_update.max_unavailable == 0
