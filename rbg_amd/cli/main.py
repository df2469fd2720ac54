"""rbgctl — the kubectl-rbg plugin analog.

Commands (reference cmd/cli/cmd/{root,status,rollout}):
  apply -f FILE              create or update objects from YAML
  get KIND [NAME]            list / show objects
  status RBG                 role/worker table with readiness
  scale ADAPTER --replicas N drive a scaling adapter
  delete KIND NAME
  rollout history RBG        ControllerRevision list
  rollout diff RBG [--to N]  diff current spec vs a revision
  rollout undo RBG [--to N]  restore a previous revision's spec

Talks to a running rbgd daemon (--port) via the RemoteClient.
"""
from __future__ import annotations

import argparse
import difflib
import sys
from typing import List, Optional

import yaml

from ..api import constants as C
from ..api.serde import asdict, fromdict
from ..api.types import RoleBasedGroupSpec, load_object
from ..client.client import (BaseClient, RemoteClient,
                             update_with_retry)


def _print_table(rows: List[List[str]], header: List[str]) -> None:
    widths = [max(len(str(r[i])) for r in [header] + rows)
              for i in range(len(header))]
    fmt = "  ".join("{:<%d}" % w for w in widths)
    print(fmt.format(*header))
    for r in rows:
        print(fmt.format(*[str(c) for c in r]))


def cmd_apply(client: BaseClient, args) -> int:
    with open(args.file) as f:
        docs = [d for d in yaml.safe_load_all(f) if d]
    for doc in docs:
        obj = load_object(doc)
        existing = client.get(obj.kind, obj.metadata.name,
                              obj.metadata.namespace)
        if existing is None:
            client.create(obj)
            print(f"{obj.kind}/{obj.metadata.name} created")
        else:
            def configure(cur, obj=obj):
                cur.spec = obj.spec
                cur.metadata.labels = obj.metadata.labels
                cur.metadata.annotations = obj.metadata.annotations
            update_with_retry(client, obj.kind, obj.metadata.name,
                              obj.metadata.namespace, configure)
            print(f"{obj.kind}/{obj.metadata.name} configured")
    return 0


def cmd_get(client: BaseClient, args) -> int:
    kind = _resolve_kind(args.kind)
    if args.name:
        obj = client.get(kind, args.name, args.namespace)
        if obj is None:
            print(f"{kind}/{args.name} not found", file=sys.stderr)
            return 1
        doc = asdict(obj)
        if getattr(args, "output", "") == "v1alpha1":
            # serve the deprecated API version on the way out (the
            # conversion-webhook read path; conversion annotations written
            # by apply make the round-trip lossless)
            from ..api import v1alpha1 as _legacy
            if kind == C.KIND_RBG:
                doc = _legacy.from_v2(doc)
            elif kind == C.KIND_RBG_SET:
                doc = _legacy.set_from_v2(doc)
            else:
                print(f"-o v1alpha1 supports {C.KIND_RBG} and "
                      f"{C.KIND_RBG_SET} only", file=sys.stderr)
                return 1
        print(yaml.safe_dump(doc, sort_keys=False))
        return 0
    if kind == C.KIND_EVENT:
        import time as _t
        evs = sorted(client.list(kind, args.namespace),
                     key=lambda e: e.last_timestamp)
        rows = [[f"{_t.time() - e.last_timestamp:.0f}s", e.type, e.reason,
                 f"{e.involved_object.kind}/{e.involved_object.name}",
                 e.count, e.message[:60]] for e in evs]
        _print_table(rows, ["AGE", "TYPE", "REASON", "OBJECT", "COUNT",
                            "MESSAGE"])
        return 0
    rows = []
    for obj in client.list(kind, args.namespace):
        ready = ""
        for cond in getattr(obj.status, "conditions", []):
            if cond.type == C.COND_READY:
                ready = cond.status
        rows.append([obj.metadata.name, ready,
                     obj.metadata.resource_version])
    _print_table(rows, ["NAME", "READY", "RV"])
    return 0


def cmd_status(client: BaseClient, args) -> int:
    rbg = client.get(C.KIND_RBG, args.name, args.namespace)
    if rbg is None:
        print(f"rolebasedgroup {args.name} not found", file=sys.stderr)
        return 1
    print(f"RoleBasedGroup: {rbg.metadata.name}")
    for cond in rbg.status.conditions:
        print(f"  {cond.type}: {cond.status} ({cond.reason}) {cond.message}")
    rows = []
    for rs in rbg.status.role_statuses:
        role = rbg.spec.role(rs.name)
        rows.append([rs.name, f"{rs.ready_replicas}/{role.replicas if role else '?'}",
                     rs.updated_replicas])
    print()
    _print_table(rows, ["ROLE", "READY", "UPDATED"])
    # surface stalled role workloads (e.g. Progressing=False when an
    # in-place-only update is infeasible) right in `rbgctl status`
    for rs in rbg.status.role_statuses:
        ris = client.get(C.KIND_ROLE_INSTANCE_SET,
                         f"{args.name}-{rs.name}", args.namespace)
        for cond in getattr(getattr(ris, "status", None), "conditions", []):
            if cond.type != C.COND_READY and cond.status != "True":
                print(f"  role {rs.name}: {cond.type}={cond.status} "
                      f"({cond.reason}) {cond.message}")
    insts = client.list(C.KIND_ROLE_INSTANCE, args.namespace,
                        selector={C.LABEL_GROUP_NAME: args.name})
    rows = []
    for inst in sorted(insts, key=lambda i: i.metadata.name):
        for w in inst.status.workers:
            rows.append([inst.metadata.name, w.name, w.phase, w.pid,
                         ",".join(map(str, w.gpu_ids)) or "-",
                         inst.status.restart_count])
    print()
    _print_table(rows, ["INSTANCE", "WORKER", "PHASE", "PID", "GPUS",
                        "RESTARTS"])
    return 0


def cmd_describe(client: BaseClient, args) -> int:
    """kubectl-describe analog: spec summary + conditions + recent events
    about the object."""
    kind = _resolve_kind(args.kind)
    obj = client.get(kind, args.name, args.namespace)
    if obj is None:
        print(f"{args.kind}/{args.name} not found", file=sys.stderr)
        return 1
    print(f"Name:      {obj.metadata.name}")
    print(f"Namespace: {obj.metadata.namespace}")
    print(f"Kind:      {obj.kind}")
    if obj.metadata.labels:
        print(f"Labels:    {obj.metadata.labels}")
    for cond in getattr(obj.status, "conditions", []):
        print(f"Condition: {cond.type}={cond.status} ({cond.reason}) "
              f"{cond.message}")
    import time as _t
    evs = [e for e in client.list(C.KIND_EVENT, args.namespace)
           if e.involved_object.kind == kind and
           e.involved_object.name == args.name]
    evs.sort(key=lambda e: e.last_timestamp)
    if evs:
        print("Events:")
        rows = [[f"{_t.time() - e.last_timestamp:.0f}s", e.type, e.reason,
                 e.count, e.message[:60]] for e in evs[-15:]]
        _print_table(rows, ["AGE", "TYPE", "REASON", "COUNT", "MESSAGE"])
    return 0


def cmd_scale(client: BaseClient, args) -> int:
    client.scale(args.name, args.replicas, args.namespace)
    print(f"scalingadapter/{args.name} scaled to {args.replicas}")
    return 0


def cmd_delete(client: BaseClient, args) -> int:
    ok = client.delete(_resolve_kind(args.kind), args.name, args.namespace)
    print(f"{args.kind}/{args.name} " + ("deleted" if ok else "not found"))
    return 0 if ok else 1


def _spec_yaml(data: dict) -> str:
    return yaml.safe_dump(data, sort_keys=True)


def cmd_rollout(client: BaseClient, args) -> int:
    revs = client.revisions(args.name, args.namespace)
    if args.action == "history":
        _print_table([[r.revision, r.metadata.name,
                       r.metadata.labels.get(C.LABEL_REVISION_HASH, "")]
                      for r in revs], ["REVISION", "NAME", "HASH"])
        return 0
    rbg = client.get(C.KIND_RBG, args.name, args.namespace)
    if rbg is None:
        print(f"rolebasedgroup {args.name} not found", file=sys.stderr)
        return 1
    if not revs:
        print("no revisions recorded", file=sys.stderr)
        return 1
    target = None
    if args.to:
        target = next((r for r in revs if r.revision == args.to), None)
    else:
        # default: previous revision (reference rollout_undo.go semantics)
        target = revs[-2] if len(revs) >= 2 else revs[-1]
    if target is None:
        print(f"revision {args.to} not found", file=sys.stderr)
        return 1
    if args.action == "diff":
        cur = _spec_yaml(asdict(rbg.spec)).splitlines(keepends=True)
        old = _spec_yaml(target.data).splitlines(keepends=True)
        sys.stdout.writelines(difflib.unified_diff(
            old, cur, fromfile=f"revision-{target.revision}",
            tofile="current"))
        return 0
    if args.action == "undo":
        def roll_back(cur, data=target.data):
            cur.spec = fromdict(RoleBasedGroupSpec, data)
        update_with_retry(client, C.KIND_RBG, args.name, args.namespace,
                           roll_back)
        print(f"rolebasedgroup/{args.name} rolled back to revision "
              f"{target.revision}")
        return 0
    print(f"unknown rollout action {args.action}", file=sys.stderr)
    return 1


_KIND_ALIASES = {
    "rbg": C.KIND_RBG, "rolebasedgroup": C.KIND_RBG,
    "rbgset": C.KIND_RBG_SET, "ris": C.KIND_ROLE_INSTANCE_SET,
    "roleinstanceset": C.KIND_ROLE_INSTANCE_SET,
    "roleinstance": C.KIND_ROLE_INSTANCE, "ri": C.KIND_ROLE_INSTANCE,
    "scalingadapter": C.KIND_SCALING_ADAPTER,
    "rbgsa": C.KIND_SCALING_ADAPTER,
    "coordinatedpolicy": C.KIND_COORDINATED_POLICY,
    "warmup": C.KIND_WARMUP,
    "revision": C.KIND_CONTROLLER_REVISION,
    "event": C.KIND_EVENT, "events": C.KIND_EVENT, "ev": C.KIND_EVENT,
}


def _resolve_kind(kind: str) -> str:
    k = _KIND_ALIASES.get(kind.lower())
    if k is None and kind in C.ALL_KINDS:
        return kind
    if k is None:
        raise SystemExit(f"unknown kind {kind!r}")
    return k


def build_parser() -> argparse.ArgumentParser:
    ap = argparse.ArgumentParser(prog="rbgctl")
    ap.add_argument("--port", type=int, default=7471)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("-n", "--namespace", default="default")
    sub = ap.add_subparsers(dest="cmd", required=True)
    p = sub.add_parser("apply")
    p.add_argument("-f", "--file", required=True)
    p = sub.add_parser("get")
    p.add_argument("kind")
    p.add_argument("name", nargs="?")
    p.add_argument("-o", "--output", default="",
                   help="'v1alpha1' emits the deprecated API version")
    p = sub.add_parser("status")
    p.add_argument("name")
    p = sub.add_parser("describe")
    p.add_argument("kind")
    p.add_argument("name")
    p = sub.add_parser("scale")
    p.add_argument("name")
    p.add_argument("--replicas", type=int, required=True)
    p = sub.add_parser("delete")
    p.add_argument("kind")
    p.add_argument("name")
    p = sub.add_parser("rollout")
    p.add_argument("action", choices=["history", "diff", "undo"])
    p.add_argument("name")
    p.add_argument("--to", type=int, default=0)
    return ap


def main(argv: Optional[List[str]] = None,
         client: Optional[BaseClient] = None) -> int:
    args = build_parser().parse_args(argv)
    if client is None:
        client = RemoteClient(args.host, args.port)
    return {
        "apply": cmd_apply, "get": cmd_get, "status": cmd_status,
        "describe": cmd_describe, "scale": cmd_scale, "delete": cmd_delete,
        "rollout": cmd_rollout,
    }[args.cmd](client, args)


if __name__ == "__main__":
    sys.exit(main())
