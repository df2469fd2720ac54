"""Property tests: serde round-trips for arbitrary specs.

The wire format (camelCase dicts, the YAML/RPC interchange) must be
lossless for any object the dataclasses can represent — the analog of the
reference's generated deepcopy/conversion being exercised by fuzzed
apply-configurations.
"""
from hypothesis import given, settings, strategies as st

from rbg_amd.api import constants as C
from rbg_amd.api.serde import asdict, fromdict
from rbg_amd.api.types import (ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec,
                               RoleSpec, RollingUpdate, RolloutStrategy,
                               load_object)

names = st.from_regex(r"[a-z][a-z0-9-]{0,14}[a-z0-9]", fullmatch=True)
labels = st.dictionaries(
    st.from_regex(r"[a-z][a-z0-9./-]{0,30}", fullmatch=True),
    st.from_regex(r"[A-Za-z0-9._-]{0,20}", fullmatch=True), max_size=4)


@st.composite
def role_specs(draw):
    return RoleSpec(
        name=draw(names),
        replicas=draw(st.integers(min_value=0, max_value=64)),
        pod_management_policy=draw(st.sampled_from(
            [C.POD_MANAGEMENT_PARALLEL, C.POD_MANAGEMENT_ORDERED_READY])),
        update_strategy_type=draw(st.sampled_from(
            [C.UPDATE_IN_PLACE_IF_POSSIBLE, C.UPDATE_IN_PLACE_ONLY,
             C.UPDATE_RECREATE])),
        rollout_strategy=RolloutStrategy(rolling_update=RollingUpdate(
            max_unavailable=draw(st.integers(min_value=0, max_value=8)),
            max_surge=draw(st.integers(min_value=0, max_value=8)),
            partition=draw(st.integers(min_value=0, max_value=8)))),
    )


@st.composite
def rbgs(draw):
    roles = draw(st.lists(role_specs(), min_size=1, max_size=4,
                          unique_by=lambda r: r.name))
    return RoleBasedGroup(
        metadata=ObjectMeta(name=draw(names), labels=draw(labels),
                            annotations=draw(labels)),
        spec=RoleBasedGroupSpec(roles=roles))


@given(rbgs())
@settings(max_examples=60, deadline=None)
def test_rbg_wire_roundtrip_lossless(rbg):
    wire = asdict(rbg)
    back = fromdict(RoleBasedGroup, wire)
    assert asdict(back) == wire
    # and the kind-dispatching loader agrees
    again = load_object(wire)
    assert asdict(again) == wire


@given(rbgs())
@settings(max_examples=30, deadline=None)
def test_rbg_wire_is_camel_case_only(rbg):
    def walk(d):
        if isinstance(d, dict):
            for k, v in d.items():
                assert "_" not in k or "/" in k or "." in k, k
                walk(v)
        elif isinstance(d, list):
            for v in d:
                walk(v)
    walk(asdict(rbg))
