"""v1alpha1 — the deprecated legacy API + bidirectional conversion.

The reference keeps a full v1alpha1 package (api/workloads/v1alpha1/
rolebasedgroup_types.go, rolebasedgroup_conversion.go:42-180) whose RBG
schema predates the v1alpha2 Pattern model:

  * role-level ``workload: {apiVersion, kind}`` (StatefulSet | Deployment |
    LeaderWorkerSet | InstanceSet) instead of ``pattern``
  * role-level ``leaderWorkerSet: {size, patchLeaderTemplate,
    patchWorkerTemplate}`` + top-level ``restartPolicy`` enum
  * role-level inline ``template`` / ``templateRef`` + ``templatePatch``
    (v1alpha2 moves template sourcing into the Pattern)
  * spec-level ``podGroupPolicy`` (kubeScheduling | volcanoScheduling)
    → in v1alpha2 gang scheduling is annotation-driven
  * spec-level ``coordination[]`` rules → in v1alpha2 a separate
    CoordinatedPolicy object (synthesized by the migration controller,
    reference coordinatedpolicy_migration_controller.go:59-156)

Conversion here follows the reference's webhook-conversion semantics but
operates at the serialized-dict level (the natural seam in this framework:
objects enter through YAML/JSON):

  ``to_v2(dict) -> dict``    v1alpha1 RBG doc → v1alpha2 RBG doc.  Lossy
                             fields (podGroupPolicy, coordination, per-role
                             workload kind) are preserved in annotations so
                             ``from_v2`` round-trips, mirroring
                             preserveV1alpha1Fields / restoreV1alpha1Fields.
  ``from_v2(dict) -> dict``  v1alpha2 RBG doc → v1alpha1 doc for legacy
                             clients (ConvertFrom analog).

`load_object` dispatches v1alpha1 RoleBasedGroup docs through ``to_v2``
automatically, so `rbgctl apply -f legacy.yaml` just works.
"""
from __future__ import annotations

import json
from typing import Any, Dict

from . import constants as C

API_VERSION_V1ALPHA1 = "workloads.x-k8s.io/v1alpha1"

# conversion-only annotations (reference RoleWorkloadTypeAnnotationKey et al.)
ANNO_WORKLOAD_TYPE = f"{C.PREFIX}/role-workload-type"
ANNO_POD_GROUP_POLICY = f"{C.PREFIX}/v1alpha1-pod-group-policy"
ANNO_COORDINATION = f"{C.PREFIX}/v1alpha1-coordination"
# original per-role rolloutStrategy + LWS patch templates, preserved so the
# documented v1 -> v2 -> v1 round-trip is lossless for them (the reference
# keeps these via annotations too; round-1 advisor finding)
ANNO_ROLLOUT = f"{C.PREFIX}/v1alpha1-rollout-strategy"
ANNO_LWS = f"{C.PREFIX}/v1alpha1-lws"

_LEGACY_RESTART = {
    "RecreateRoleInstanceOnPodRestart": C.RESTART_POLICY_RECREATE_INSTANCE,
    "None": C.RESTART_POLICY_NONE,
}
_LEGACY_RESTART_BACK = {v: k for k, v in _LEGACY_RESTART.items()}


def is_v1alpha1(doc: Dict[str, Any]) -> bool:
    return doc.get("apiVersion", "") == API_VERSION_V1ALPHA1


def set_to_v2(doc: Dict[str, Any]) -> Dict[str, Any]:
    """v1alpha1 RoleBasedGroupSet doc → v1alpha2 (the template is an RBG
    spec; reference rolebasedgroupset_conversion.go)."""
    spec = doc.get("spec") or {}
    inner = {"apiVersion": API_VERSION_V1ALPHA1, "kind": C.KIND_RBG,
             "metadata": dict(doc.get("metadata") or {}),
             "spec": spec.get("template") or {}}
    conv = to_v2(inner)
    out_meta = conv["metadata"]
    return {"apiVersion": C.API_VERSION, "kind": C.KIND_RBG_SET,
            "metadata": out_meta,
            "spec": {"replicas": spec.get("replicas", 1),
                     "template": conv["spec"]}}


def set_from_v2(doc: Dict[str, Any]) -> Dict[str, Any]:
    """v1alpha2 RoleBasedGroupSet doc → v1alpha1 for legacy clients."""
    spec = doc.get("spec") or {}
    inner = {"apiVersion": C.API_VERSION, "kind": C.KIND_RBG,
             "metadata": dict(doc.get("metadata") or {}),
             "spec": spec.get("template") or {}}
    conv = from_v2(inner)
    return {"apiVersion": API_VERSION_V1ALPHA1, "kind": C.KIND_RBG_SET,
            "metadata": conv["metadata"],
            "spec": {"replicas": spec.get("replicas", 1),
                     "template": conv["spec"]}}


def _merge_patch(base: Dict[str, Any], patch: Dict[str, Any]) -> Dict[str, Any]:
    """JSON-merge-patch-lite used for patchLeaderTemplate/patchWorkerTemplate
    over the role template (good enough for EngineTemplate docs: dicts merge
    recursively, engine lists merge by name)."""
    if not patch:
        return base
    out = dict(base)
    for k, v in patch.items():
        if k == "engines" and isinstance(v, list):
            merged = [dict(e) for e in out.get("engines", [])]
            by_name = {e.get("name"): e for e in merged}
            for p in v:
                tgt = by_name.get(p.get("name"))
                if tgt is None:
                    merged.append(p)
                else:
                    for pk, pv in p.items():
                        if pk == "args" and isinstance(pv, dict):
                            tgt.setdefault("args", {}).update(pv)
                        else:
                            tgt[pk] = pv
            out["engines"] = merged
        elif isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _merge_patch(out[k], v)
        else:
            out[k] = v
    return out


def _convert_role_to_v2(role: Dict[str, Any],
                        annos: Dict[str, str]) -> Dict[str, Any]:
    workload = role.get("workload") or {}
    kind = workload.get("kind", "StatefulSet")
    out: Dict[str, Any] = {
        "name": role.get("name", ""),
        "replicas": role.get("replicas", 1),
    }
    for k in ("dependencies", "servicePorts", "minReadySeconds",
              "template", "dataclass_ignore"):
        if k in role and k != "dataclass_ignore":
            out[k] = role[k]
    if role.get("templateRef"):
        out["templateRef"] = {"name": role["templateRef"].get("name", ""),
                              "patch": role.get("templatePatch") or {}}
    # engineRuntimes: [{profileName, ...}] -> [profileName]
    if role.get("engineRuntimes"):
        out["engineRuntimes"] = [er.get("profileName", "")
                                 for er in role["engineRuntimes"]]
    if role.get("scalingAdapter"):
        out["scalingAdapter"] = role["scalingAdapter"]
    if role.get("rolloutStrategy"):
        ru = (role["rolloutStrategy"] or {}).get("rollingUpdate") or {}
        out["rolloutStrategy"] = {"rollingUpdate": {
            k: ru[k] for k in ("maxUnavailable", "maxSurge", "partition")
            if k in ru}}
        if ru.get("type"):
            out["updateStrategyType"] = ru["type"]
        # keep the original doc for a lossless round-trip
        saved = json.loads(annos.get(ANNO_ROLLOUT, "{}"))
        saved[role.get("name", "")] = role["rolloutStrategy"]
        annos[ANNO_ROLLOUT] = json.dumps(saved, sort_keys=True)
    restart = _LEGACY_RESTART.get(role.get("restartPolicy", ""),
                                  C.RESTART_POLICY_RECREATE_INSTANCE)
    out["restartPolicy"] = restart

    lws = role.get("leaderWorkerSet")
    if lws is not None or kind == "LeaderWorkerSet":
        lws = lws or {}
        base = role.get("template") or {}
        lwp: Dict[str, Any] = {
            "size": lws.get("size", 1),
            "restartPolicy": restart,
        }
        if lws.get("patchLeaderTemplate"):
            lwp["leaderTemplate"] = _merge_patch(base,
                                                 lws["patchLeaderTemplate"])
        if lws.get("patchWorkerTemplate"):
            lwp["workerTemplate"] = _merge_patch(base,
                                                 lws["patchWorkerTemplate"])
        out["pattern"] = C.PATTERN_LEADER_WORKER
        out["leaderWorkerPattern"] = lwp
        if lws:
            # preserve the original patches (merge output is not invertible)
            saved = json.loads(annos.get(ANNO_LWS, "{}"))
            saved[out["name"]] = lws
            annos[ANNO_LWS] = json.dumps(saved, sort_keys=True)
    elif role.get("components"):
        out["pattern"] = C.PATTERN_CUSTOM_COMPONENTS
        out["customComponentsPattern"] = {"components": role["components"]}
    else:
        out["pattern"] = C.PATTERN_STANDALONE
    # preserve the declared workload kind for round-trip (reference stores it
    # in a conversion-only role annotation; we key it by role name group-wide)
    if workload:
        types = json.loads(annos.get(ANNO_WORKLOAD_TYPE, "{}"))
        types[out["name"]] = f"{workload.get('apiVersion', 'apps/v1')}/{kind}"
        annos[ANNO_WORKLOAD_TYPE] = json.dumps(types, sort_keys=True)
    return out


def to_v2(doc: Dict[str, Any]) -> Dict[str, Any]:
    """v1alpha1 RoleBasedGroup doc → v1alpha2 doc (ConvertTo analog)."""
    spec = doc.get("spec") or {}
    meta = dict(doc.get("metadata") or {})
    annos = dict(meta.get("annotations") or {})
    roles = [_convert_role_to_v2(r, annos) for r in spec.get("roles", [])]
    out_spec: Dict[str, Any] = {"roles": roles}
    if spec.get("roleTemplates"):
        out_spec["roleTemplates"] = {
            rt["name"]: rt.get("template", {})
            for rt in spec["roleTemplates"]}
    pgp = spec.get("podGroupPolicy")
    if pgp:
        # semantic mapping: podGroupPolicy → gang-scheduling annotations
        annos[C.ANNO_GANG_SCHEDULING] = "true"
        timeout = (pgp.get("kubeScheduling") or {}).get(
            "scheduleTimeoutSeconds")
        if timeout:
            annos[C.ANNO_GANG_TIMEOUT] = str(timeout)
        annos[ANNO_POD_GROUP_POLICY] = json.dumps(pgp, sort_keys=True)
    if spec.get("coordination"):
        annos[ANNO_COORDINATION] = json.dumps(spec["coordination"],
                                              sort_keys=True)
    if annos:
        meta["annotations"] = annos
    return {"apiVersion": C.API_VERSION, "kind": C.KIND_RBG,
            "metadata": meta, "spec": out_spec,
            **({"status": doc["status"]} if "status" in doc else {})}


def _convert_role_from_v2(role: Dict[str, Any],
                          workload_types: Dict[str, str],
                          rollouts: Dict[str, Any],
                          lws_saved: Dict[str, Any]) -> Dict[str, Any]:
    out: Dict[str, Any] = {"name": role.get("name", ""),
                           "replicas": role.get("replicas", 1)}
    for k in ("dependencies", "servicePorts", "minReadySeconds", "template"):
        if k in role:
            out[k] = role[k]
    # restore rolloutStrategy: the preserved original if present, else
    # synthesize from the v2 fields
    if out["name"] in rollouts:
        out["rolloutStrategy"] = rollouts[out["name"]]
    elif role.get("rolloutStrategy"):
        ru = dict((role["rolloutStrategy"] or {}).get("rollingUpdate") or {})
        if role.get("updateStrategyType"):
            ru["type"] = role["updateStrategyType"]
        out["rolloutStrategy"] = {"rollingUpdate": ru}
    if role.get("templateRef"):
        out["templateRef"] = {"name": role["templateRef"].get("name", "")}
        if role["templateRef"].get("patch"):
            out["templatePatch"] = role["templateRef"]["patch"]
    if role.get("engineRuntimes"):
        out["engineRuntimes"] = [{"profileName": n}
                                 for n in role["engineRuntimes"]]
    if role.get("scalingAdapter"):
        out["scalingAdapter"] = role["scalingAdapter"]
    if role.get("restartPolicy"):
        out["restartPolicy"] = _LEGACY_RESTART_BACK.get(
            role["restartPolicy"], "None")
    wt = workload_types.get(out["name"], "")
    pattern = role.get("pattern", C.PATTERN_STANDALONE)
    if pattern == C.PATTERN_LEADER_WORKER:
        lwp = role.get("leaderWorkerPattern") or {}
        out["workload"] = {"apiVersion": "leaderworkerset.x-k8s.io/v1",
                           "kind": "LeaderWorkerSet"}
        if out["name"] in lws_saved:
            # restore the original patch templates (preserved: the merged
            # leader/worker templates are not invertible)
            out["leaderWorkerSet"] = lws_saved[out["name"]]
            out["leaderWorkerSet"].setdefault("size", lwp.get("size", 1))
        else:
            out["leaderWorkerSet"] = {"size": lwp.get("size", 1)}
    elif pattern == C.PATTERN_CUSTOM_COMPONENTS:
        out["workload"] = {"apiVersion": C.API_VERSION, "kind": "InstanceSet"}
        out["components"] = (role.get("customComponentsPattern") or {}).get(
            "components", [])
    elif wt:
        ver, _, kind = wt.rpartition("/")
        out["workload"] = {"apiVersion": ver, "kind": kind}
    else:
        out["workload"] = {"apiVersion": "apps/v1", "kind": "StatefulSet"}
    return out


def from_v2(doc: Dict[str, Any]) -> Dict[str, Any]:
    """v1alpha2 RoleBasedGroup doc → v1alpha1 doc (ConvertFrom analog)."""
    spec = doc.get("spec") or {}
    meta = dict(doc.get("metadata") or {})
    annos = dict(meta.get("annotations") or {})
    workload_types = json.loads(annos.pop(ANNO_WORKLOAD_TYPE, "{}"))
    rollouts = json.loads(annos.pop(ANNO_ROLLOUT, "{}"))
    lws_saved = json.loads(annos.pop(ANNO_LWS, "{}"))
    out_spec: Dict[str, Any] = {
        "roles": [_convert_role_from_v2(r, workload_types, rollouts,
                                        lws_saved)
                  for r in spec.get("roles", [])]}
    if spec.get("roleTemplates"):
        out_spec["roleTemplates"] = [
            {"name": n, "template": t}
            for n, t in sorted(spec["roleTemplates"].items())]
    if ANNO_POD_GROUP_POLICY in annos:
        out_spec["podGroupPolicy"] = json.loads(
            annos.pop(ANNO_POD_GROUP_POLICY))
        annos.pop(C.ANNO_GANG_SCHEDULING, None)
        annos.pop(C.ANNO_GANG_TIMEOUT, None)
    if ANNO_COORDINATION in annos:
        out_spec["coordination"] = json.loads(annos.pop(ANNO_COORDINATION))
    meta["annotations"] = annos
    if not annos:
        meta.pop("annotations", None)
    return {"apiVersion": API_VERSION_V1ALPHA1, "kind": C.KIND_RBG,
            "metadata": meta, "spec": out_spec,
            **({"status": doc["status"]} if "status" in doc else {})}


# ---------------------------------------------------------------------------
# Legacy InstanceSet / Instance kinds (reference api/workloads/v1alpha1/
# instanceset_types.go, instance_types.go).  The reference serves these as
# standalone v1alpha1 CRDs — the predecessors of RoleInstanceSet /
# RoleInstance — gated by the deprecated-workload toggle.  Here a legacy doc
# converts structurally on the way in (load_object) and back out for legacy
# clients.
# ---------------------------------------------------------------------------

ANNO_IS_LIFECYCLE = f"{C.PREFIX}/v1alpha1-lifecycle"
ANNO_IS_SCALE_MAXUNAVAILABLE = f"{C.PREFIX}/v1alpha1-scale-max-unavailable"
ANNO_READY_POLICY = f"{C.PREFIX}/v1alpha1-ready-policy"

_LEGACY_UPDATE_TYPE = {
    "InPlaceIfPossible": C.UPDATE_IN_PLACE_IF_POSSIBLE,
    "InPlaceOnly": C.UPDATE_IN_PLACE_ONLY,
    "RecreatePod": C.UPDATE_RECREATE,
    "RecreateInstance": C.UPDATE_RECREATE,
}
_LEGACY_UPDATE_TYPE_BACK = {
    C.UPDATE_IN_PLACE_IF_POSSIBLE: "InPlaceIfPossible",
    C.UPDATE_IN_PLACE_ONLY: "InPlaceOnly",
    C.UPDATE_RECREATE: "RecreatePod",
}


def _int_or_pct(v: Any, replicas: int, default: int) -> int:
    """IntOrString: plain int passes through; "N%" rounds UP against
    replicas (apimachinery intstr semantics for maxUnavailable)."""
    if v is None:
        return default
    if isinstance(v, int):
        return v
    s = str(v).strip()
    if s.endswith("%"):
        pct = float(s[:-1])
        return max(0, int(-(-replicas * pct // 100)))
    return int(s)


def instanceset_to_v2(doc: Dict[str, Any]) -> Dict[str, Any]:
    """v1alpha1 InstanceSet doc -> v1alpha2 RoleInstanceSet doc."""
    spec = doc.get("spec") or {}
    meta = dict(doc.get("metadata") or {})
    annos = dict(meta.get("annotations") or {})
    replicas = int(spec.get("replicas", 1))
    out_spec: Dict[str, Any] = {"replicas": replicas}
    sel = (spec.get("selector") or {}).get("matchLabels")
    if sel:
        out_spec["selector"] = dict(sel)
    tmpl = spec.get("instanceTemplate") or {}
    out_spec["template"] = {
        "metadata": tmpl.get("metadata") or {},
        "components": tmpl.get("components") or
                      (tmpl.get("spec") or {}).get("components") or [],
    }
    rp = (tmpl.get("spec") or {}).get("restartPolicy") or \
        tmpl.get("restartPolicy")
    if rp:
        out_spec["template"]["restartPolicy"] = _LEGACY_RESTART.get(
            rp, C.RESTART_POLICY_RECREATE_INSTANCE)
    scale = spec.get("scaleStrategy") or {}
    if scale.get("instanceToDelete"):
        annos[C.ANNO_ROLE_INSTANCE_TO_DELETE] = ",".join(
            scale["instanceToDelete"])
    if scale.get("maxUnavailable") is not None:
        annos[ANNO_IS_SCALE_MAXUNAVAILABLE] = str(scale["maxUnavailable"])
    upd = spec.get("updateStrategy") or {}
    out_spec["updateStrategy"] = {
        "type": _LEGACY_UPDATE_TYPE.get(upd.get("type", ""),
                                        C.UPDATE_IN_PLACE_IF_POSSIBLE),
        "partition": _int_or_pct(upd.get("partition"), replicas, 0),
        "maxUnavailable": _int_or_pct(upd.get("maxUnavailable"),
                                      replicas, 1),
        "maxSurge": _int_or_pct(upd.get("maxSurge"), replicas, 0),
        "paused": bool(upd.get("paused", False)),
        "gracePeriodSeconds": int(
            (upd.get("inPlaceUpdateStrategy") or {}).get(
                "gracePeriodSeconds", 0)),
    }
    if spec.get("revisionHistoryLimit") is not None:
        out_spec["revisionHistoryLimit"] = spec["revisionHistoryLimit"]
    if spec.get("minReadySeconds") is not None:
        out_spec["minReadySeconds"] = spec["minReadySeconds"]
    if spec.get("lifecycle"):
        annos[ANNO_IS_LIFECYCLE] = json.dumps(spec["lifecycle"],
                                              sort_keys=True)
    if annos:
        meta["annotations"] = annos
    return {"apiVersion": C.API_VERSION, "kind": C.KIND_ROLE_INSTANCE_SET,
            "metadata": meta, "spec": out_spec,
            **({"status": doc["status"]} if "status" in doc else {})}


def instanceset_from_v2(doc: Dict[str, Any]) -> Dict[str, Any]:
    """v1alpha2 RoleInstanceSet doc -> v1alpha1 InstanceSet doc."""
    spec = doc.get("spec") or {}
    meta = dict(doc.get("metadata") or {})
    annos = dict(meta.get("annotations") or {})
    out_spec: Dict[str, Any] = {"replicas": spec.get("replicas", 1)}
    if spec.get("selector"):
        out_spec["selector"] = {"matchLabels": dict(spec["selector"])}
    tmpl = spec.get("template") or {}
    out_tmpl: Dict[str, Any] = {
        "metadata": tmpl.get("metadata") or {},
        "components": tmpl.get("components") or [],
    }
    if tmpl.get("restartPolicy"):
        out_tmpl["restartPolicy"] = _LEGACY_RESTART_BACK.get(
            tmpl["restartPolicy"], "None")
    out_spec["instanceTemplate"] = out_tmpl
    scale: Dict[str, Any] = {}
    if C.ANNO_ROLE_INSTANCE_TO_DELETE in annos:
        scale["instanceToDelete"] = [
            n for n in annos.pop(C.ANNO_ROLE_INSTANCE_TO_DELETE).split(",")
            if n]
    if ANNO_IS_SCALE_MAXUNAVAILABLE in annos:
        raw = annos.pop(ANNO_IS_SCALE_MAXUNAVAILABLE)
        scale["maxUnavailable"] = int(raw) if raw.lstrip("-").isdigit() \
            else raw
    if scale:
        out_spec["scaleStrategy"] = scale
    upd = spec.get("updateStrategy") or {}
    out_upd: Dict[str, Any] = {
        "type": _LEGACY_UPDATE_TYPE_BACK.get(
            upd.get("type", ""), "InPlaceIfPossible"),
        "partition": upd.get("partition", 0),
        "maxUnavailable": upd.get("maxUnavailable", 1),
        "maxSurge": upd.get("maxSurge", 0),
        "paused": upd.get("paused", False),
    }
    if upd.get("gracePeriodSeconds"):
        out_upd["inPlaceUpdateStrategy"] = {
            "gracePeriodSeconds": upd["gracePeriodSeconds"]}
    out_spec["updateStrategy"] = out_upd
    for k in ("revisionHistoryLimit", "minReadySeconds"):
        if k in spec:
            out_spec[k] = spec[k]
    if ANNO_IS_LIFECYCLE in annos:
        out_spec["lifecycle"] = json.loads(annos.pop(ANNO_IS_LIFECYCLE))
    meta["annotations"] = annos
    if not annos:
        meta.pop("annotations", None)
    return {"apiVersion": API_VERSION_V1ALPHA1, "kind": "InstanceSet",
            "metadata": meta, "spec": out_spec,
            **({"status": doc["status"]} if "status" in doc else {})}


def instance_to_v2(doc: Dict[str, Any]) -> Dict[str, Any]:
    """v1alpha1 Instance doc -> v1alpha2 RoleInstance doc."""
    spec = doc.get("spec") or {}
    meta = dict(doc.get("metadata") or {})
    annos = dict(meta.get("annotations") or {})
    out_spec: Dict[str, Any] = {
        "components": spec.get("components") or [],
    }
    if spec.get("restartPolicy"):
        out_spec["restartPolicy"] = _LEGACY_RESTART.get(
            spec["restartPolicy"], C.RESTART_POLICY_RECREATE_INSTANCE)
    if spec.get("readinessGates"):
        out_spec["readinessGates"] = spec["readinessGates"]
    if spec.get("readyPolicy"):
        annos[ANNO_READY_POLICY] = spec["readyPolicy"]
    pgp = spec.get("podGroupPolicy")
    if pgp:
        annos[C.ANNO_GANG_SCHEDULING] = "true"
        annos[ANNO_POD_GROUP_POLICY] = json.dumps(pgp, sort_keys=True)
    if annos:
        meta["annotations"] = annos
    return {"apiVersion": C.API_VERSION, "kind": C.KIND_ROLE_INSTANCE,
            "metadata": meta, "spec": out_spec,
            **({"status": doc["status"]} if "status" in doc else {})}


def instance_from_v2(doc: Dict[str, Any]) -> Dict[str, Any]:
    """v1alpha2 RoleInstance doc -> v1alpha1 Instance doc."""
    spec = doc.get("spec") or {}
    meta = dict(doc.get("metadata") or {})
    annos = dict(meta.get("annotations") or {})
    out_spec: Dict[str, Any] = {"components": spec.get("components") or []}
    if spec.get("restartPolicy"):
        out_spec["restartPolicy"] = _LEGACY_RESTART_BACK.get(
            spec["restartPolicy"], "None")
    if spec.get("readinessGates"):
        out_spec["readinessGates"] = spec["readinessGates"]
    if ANNO_READY_POLICY in annos:
        out_spec["readyPolicy"] = annos.pop(ANNO_READY_POLICY)
    if ANNO_POD_GROUP_POLICY in annos:
        out_spec["podGroupPolicy"] = json.loads(
            annos.pop(ANNO_POD_GROUP_POLICY))
        annos.pop(C.ANNO_GANG_SCHEDULING, None)
    meta["annotations"] = annos
    if not annos:
        meta.pop("annotations", None)
    return {"apiVersion": API_VERSION_V1ALPHA1, "kind": "Instance",
            "metadata": meta, "spec": out_spec,
            **({"status": doc["status"]} if "status" in doc else {})}
