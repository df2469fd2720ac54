"""ControllerRevision machinery — spec checkpointing and rollback.

Mirrors reference pkg/utils/revision_utils.go:50-393: every RBG spec change
is snapshotted as a content-hashed revision object; identical snapshots
dedupe onto the existing revision; history is truncated to a limit; the CLI's
``rollout history/diff/undo`` reads these (reference cmd/cli/cmd/rollout).
"""
from __future__ import annotations

import hashlib
import json
from typing import Any, Dict, List, Optional

from ..api import constants as C
from ..api.serde import asdict, fromdict
from ..api.types import ControllerRevision, ObjectMeta, RoleBasedGroupSpec
from .store import Store, set_owner


def hash_spec(data: Dict[str, Any]) -> str:
    blob = json.dumps(data, sort_keys=True, separators=(",", ":"))
    return hashlib.sha256(blob.encode()).hexdigest()[:10]


def snapshot_rbg_spec(rbg) -> Dict[str, Any]:
    return asdict(rbg.spec)


def revision_name(owner_name: str, h: str) -> str:
    return f"{owner_name}-{h}"


class RevisionManager:
    def __init__(self, store: Store, history_limit: int = 10):
        self.store = store
        self.history_limit = history_limit

    def list_for(self, owner) -> List[ControllerRevision]:
        revs = self.store.list_owned(C.KIND_CONTROLLER_REVISION,
                                     owner.metadata.uid, owner.metadata.namespace)
        revs.sort(key=lambda r: r.revision)
        return revs

    def ensure_current(self, owner, spec_data: Dict[str, Any]) -> ControllerRevision:
        """Create (or dedupe onto) the revision matching spec_data; returns it.
        The returned revision's hash is the per-role revision label value
        (reference roleinstanceset_reconciler.go:116-125)."""
        h = hash_spec(spec_data)
        name = revision_name(owner.metadata.name, h)
        existing = self.store.try_get(C.KIND_CONTROLLER_REVISION, name,
                                      owner.metadata.namespace)
        revs = self.list_for(owner)
        if existing is not None:
            # dedupe: bump to latest revision number if it is being re-adopted
            if revs and revs[-1].metadata.name != name:
                next_num = revs[-1].revision + 1
                self.store.apply(C.KIND_CONTROLLER_REVISION, name,
                                 lambda r: setattr(r, "revision", next_num) or r,
                                 owner.metadata.namespace)
                existing.revision = next_num
            return existing
        next_num = (revs[-1].revision + 1) if revs else 1
        rev = ControllerRevision(
            metadata=ObjectMeta(name=name, namespace=owner.metadata.namespace,
                                labels={C.LABEL_GROUP_NAME: owner.metadata.name,
                                        C.LABEL_REVISION_HASH: h}),
            data=spec_data, revision=next_num)
        set_owner(rev, owner)
        created = self.store.create(rev)
        self.truncate(owner)
        return created

    def truncate(self, owner) -> None:
        revs = self.list_for(owner)
        excess = len(revs) - self.history_limit
        for rev in revs[:max(0, excess)]:
            self.store.try_delete(C.KIND_CONTROLLER_REVISION, rev.metadata.name,
                                  owner.metadata.namespace)

    def get_by_number(self, owner, number: int) -> Optional[ControllerRevision]:
        for rev in self.list_for(owner):
            if rev.revision == number:
                return rev
        return None

    def restore_spec(self, rev: ControllerRevision) -> RoleBasedGroupSpec:
        return fromdict(RoleBasedGroupSpec, rev.data)
