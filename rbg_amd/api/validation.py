"""Admission validation for v1alpha2 objects.

Mirrors the checks of the reference's validating webhook (reference:
api/workloads/v1alpha2/rolebasedgroup_validation.go:33-215,
roletemplate_validation.go:27-166, rolebasedgroup_admission.go:46-98):
name legality, duplicate/unknown roles, dependency references, rolling-update
parameter sanity, pattern exclusivity, scaling-adapter replica immutability —
plus MI355X-native checks (GPU budget fits the node).
"""
from __future__ import annotations

import re
from typing import List, Optional

from . import constants as C
from .types import RoleBasedGroup, RoleSpec, EngineTemplate

_NAME_RE = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")
MAX_NAME_LEN = 63


class ValidationError(ValueError):
    def __init__(self, errors: List[str]):
        self.errors = errors
        super().__init__("; ".join(errors))


def validate_name(name: str, what: str, errs: List[str]) -> None:
    if not name:
        errs.append(f"{what}: name must not be empty")
    elif len(name) > MAX_NAME_LEN:
        errs.append(f"{what}: name {name!r} exceeds {MAX_NAME_LEN} characters")
    elif not _NAME_RE.match(name):
        errs.append(f"{what}: name {name!r} must be lowercase DNS-1123")


def _validate_template(tmpl: Optional[EngineTemplate], where: str, errs: List[str]) -> None:
    if tmpl is None:
        return
    seen = set()
    for e in tmpl.engines:
        if e.name in seen:
            errs.append(f"{where}: duplicate engine name {e.name!r}")
        seen.add(e.name)
        if not e.runner and not e.command:
            errs.append(f"{where}: engine {e.name!r} needs a runner or a command")
        if e.resources.gpus < 0:
            errs.append(f"{where}: engine {e.name!r} gpus must be >= 0")
        if e.resources.gpus > C.MI355X_GPUS_PER_NODE:
            errs.append(
                f"{where}: engine {e.name!r} requests {e.resources.gpus} GPUs "
                f"but the node has {C.MI355X_GPUS_PER_NODE}")
        if e.resources.hbm_bytes > C.MI355X_HBM_BYTES_PER_GPU * max(1, e.resources.gpus):
            errs.append(f"{where}: engine {e.name!r} HBM request exceeds device capacity")


def _validate_role(role: RoleSpec, role_names: set, errs: List[str]) -> None:
    where = f"role {role.name!r}"
    validate_name(role.name, where, errs)
    if role.replicas < 0:
        errs.append(f"{where}: replicas must be >= 0")
    for dep in role.dependencies:
        if dep not in role_names:
            errs.append(f"{where}: dependency {dep!r} is not a role in this group")
        if dep == role.name:
            errs.append(f"{where}: cannot depend on itself")
    ru = role.rollout_strategy.rolling_update
    if ru.max_unavailable < 0 or ru.max_surge < 0 or ru.partition < 0:
        errs.append(f"{where}: rollingUpdate parameters must be >= 0")
    if ru.max_unavailable == 0 and ru.max_surge == 0:
        errs.append(f"{where}: maxUnavailable and maxSurge cannot both be 0")
    if role.pattern not in (C.PATTERN_STANDALONE, C.PATTERN_LEADER_WORKER,
                            C.PATTERN_CUSTOM_COMPONENTS):
        errs.append(f"{where}: unknown pattern {role.pattern!r}")
    if role.pattern == C.PATTERN_LEADER_WORKER:
        lwp = role.leader_worker_pattern
        if lwp is None:
            errs.append(f"{where}: leaderWorkerPattern required for pattern=leaderWorker")
        elif lwp.size < 1:
            errs.append(f"{where}: leaderWorkerPattern.size must be >= 1")
    if role.pattern == C.PATTERN_CUSTOM_COMPONENTS:
        ccp = role.custom_components_pattern
        if ccp is None or not ccp.components:
            errs.append(f"{where}: customComponentsPattern.components required")
        else:
            seen = set()
            for comp in ccp.components:
                if comp.name in seen:
                    errs.append(f"{where}: duplicate component {comp.name!r}")
                seen.add(comp.name)
                if comp.size < 1:
                    errs.append(f"{where}: component {comp.name!r} size must be >= 1")
                _validate_template(comp.template, f"{where}/component {comp.name!r}", errs)
    if role.template is None and role.template_ref is None and \
            role.pattern != C.PATTERN_CUSTOM_COMPONENTS:
        errs.append(f"{where}: template or templateRef required")
    _validate_template(role.template, where, errs)
    if role.update_strategy_type not in (C.UPDATE_IN_PLACE_IF_POSSIBLE,
                                         C.UPDATE_IN_PLACE_ONLY, C.UPDATE_RECREATE):
        errs.append(f"{where}: unknown updateStrategy {role.update_strategy_type!r}")
    if role.restart_policy not in (C.RESTART_POLICY_NONE,
                                   C.RESTART_POLICY_RECREATE_INSTANCE):
        errs.append(f"{where}: unknown restartPolicy {role.restart_policy!r}")


def validate_rbg(rbg: RoleBasedGroup) -> None:
    """Raise ValidationError with every problem found (create/update admission)."""
    errs: List[str] = []
    validate_name(rbg.metadata.name, "metadata", errs)
    if not rbg.spec.roles:
        errs.append("spec.roles must not be empty")
    names = [r.name for r in rbg.spec.roles]
    role_names = set(names)
    if len(role_names) != len(names):
        errs.append("spec.roles contains duplicate role names")
    for role in rbg.spec.roles:
        if role.template_ref is not None and role.template_ref.name not in rbg.spec.role_templates:
            errs.append(f"role {role.name!r}: templateRef {role.template_ref.name!r} "
                        "not found in roleTemplates")
        _validate_role(role, role_names, errs)
    if errs:
        raise ValidationError(errs)


def validate_rbg_update(old: RoleBasedGroup, new: RoleBasedGroup) -> None:
    """Update admission: metadata.name immutable; when a role is driven by a
    bound ScalingAdapter its replicas may only be changed via the adapter
    (reference rolebasedgroup_validation.go scaling-adapter immutability)."""
    errs: List[str] = []
    if old.metadata.name != new.metadata.name:
        errs.append("metadata.name is immutable")
    for new_role in new.spec.roles:
        old_role = old.spec.role(new_role.name)
        if old_role is None:
            continue
        if (old_role.scaling_adapter and old_role.scaling_adapter.enable
                and new_role.scaling_adapter and new_role.scaling_adapter.enable
                and new_role.replicas != old_role.replicas):
            errs.append(f"role {new_role.name!r}: replicas is managed by its "
                        "scalingAdapter and cannot be set directly")
    validate_rbg(new)
    if errs:
        raise ValidationError(errs)
