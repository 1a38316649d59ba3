"""ControllerRevision history for LeaderWorkerSet templates.

Behavioral port of reference pkg/utils/revision/revision_utils.go: template
snapshots are stored as replace-patches of {leaderWorkerTemplate,
networkConfig}, named ``<lws>-<hash>-<revnum>``, labeled with the revision
key, owned by the LWS, applied back for old-revision group rebuilds and
truncated when a rollout completes.
"""
from __future__ import annotations

from typing import Optional

from ..api import leaderworkerset as lwsapi
from ..api import serde
from ..api.core import ControllerRevision
from ..api.leaderworkerset import LeaderWorkerSet, NetworkConfig
from ..api.meta import OwnerReference
from ..cluster.store import AlreadyExistsError, NotFoundError, Store
from .hashutil import canonical_json, fnv32a, safe_encode_uint32


def get_revision_key(obj) -> str:
    """revision_utils.go:106-110 — read the revision label."""
    return (obj.metadata.labels or {}).get(lwsapi.REVISION_KEY, "")


def get_patch(lws: LeaderWorkerSet) -> dict:
    """revision_utils.go:265-297 — replace-patch of the revisioned state
    ({leaderWorkerTemplate, networkConfig})."""
    clone = serde.deep_copy(lws)
    if clone.spec.network_config is None:
        clone.spec.network_config = NetworkConfig(
            subdomain_policy=lwsapi.SubdomainPolicy.Shared)
    raw = serde.to_dict(clone)
    spec = raw.get("spec", {})
    network_config = dict(spec.get("networkConfig", {}))
    template = dict(spec.get("leaderWorkerTemplate", {}))
    network_config["$patch"] = "replace"
    template["$patch"] = "replace"
    return {"spec": {"networkConfig": network_config,
                     "leaderWorkerTemplate": template}}


def hash_revision_data(data: dict) -> str:
    """revision_utils.go:333-342 — FNV-32a over the patch + safe-encode."""
    return safe_encode_uint32(fnv32a(canonical_json(data).encode()))


def list_revisions(store: Store, lws: LeaderWorkerSet) -> list[ControllerRevision]:
    """Owned revisions for the LWS (revision_utils.go ListRevisions)."""
    out = []
    for cr in store.list("ControllerRevision", lws.metadata.namespace,
                         label_selector={lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name}):
        ctrl_ref = next((r for r in cr.metadata.owner_references if r.controller), None)
        if ctrl_ref is None or ctrl_ref.uid == lws.metadata.uid:
            out.append(cr)
    return out


def new_revision(store: Store, lws: LeaderWorkerSet,
                 revision_key: str = "") -> ControllerRevision:
    """revision_utils.go:52-94."""
    revisions = list_revisions(store, lws)
    revision_num = 1 + max((r.revision for r in revisions), default=0)
    patch = get_patch(lws)
    h = hash_revision_data(patch)
    if not revision_key:
        revision_key = h
    cr = ControllerRevision(data=patch, revision=revision_num)
    cr.metadata.name = f"{lws.metadata.name}-{h}-{revision_num}"
    cr.metadata.namespace = lws.metadata.namespace
    cr.metadata.labels = {lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
                          lwsapi.REVISION_KEY: revision_key}
    cr.metadata.owner_references = [OwnerReference(
        api_version=lws.api_version, kind=lws.kind, name=lws.metadata.name,
        uid=lws.metadata.uid, controller=True, block_owner_deletion=True)]
    return cr


def create_revision(store: Store, cr: ControllerRevision) -> ControllerRevision:
    try:
        return store.create(cr)
    except AlreadyExistsError:
        return store.get("ControllerRevision", cr.metadata.namespace,
                         cr.metadata.name)


def get_revision(store: Store, lws: LeaderWorkerSet,
                 revision_key: str) -> Optional[ControllerRevision]:
    """revision_utils.go:112-135 — latest revision matching the key."""
    if not revision_key:
        return None
    matches = [r for r in list_revisions(store, lws)
               if get_revision_key(r) == revision_key]
    if not matches:
        return None
    return max(matches, key=lambda r: r.revision)


def apply_revision(lws: LeaderWorkerSet,
                   revision: ControllerRevision) -> LeaderWorkerSet:
    """revision_utils.go:168-184 — restore the revisioned template onto a
    copy of the live object.  The stored patch uses $patch: replace on
    leaderWorkerTemplate and networkConfig, i.e. wholesale substitution."""
    restored = serde.deep_copy(lws)
    spec_patch = revision.data.get("spec", {})
    template = {k: v for k, v in spec_patch.get("leaderWorkerTemplate", {}).items()
                if k != "$patch"}
    network = {k: v for k, v in spec_patch.get("networkConfig", {}).items()
               if k != "$patch"}
    from ..api.leaderworkerset import LeaderWorkerTemplate
    restored.spec.leader_worker_template = serde.from_dict(
        LeaderWorkerTemplate, template)
    restored.spec.network_config = serde.from_dict(NetworkConfig, network)
    return restored


def equal_revision(lhs: Optional[ControllerRevision],
                   rhs: Optional[ControllerRevision]) -> bool:
    """revision_utils.go:188-194."""
    if lhs is None or rhs is None:
        return lhs is rhs
    return canonical_json(lhs.data) == canonical_json(rhs.data)


def set_matches_revision(lws: LeaderWorkerSet, proposed: ControllerRevision,
                         existing: ControllerRevision,
                         cache: Optional[dict] = None) -> bool:
    """revision_utils.go:199-235 — semantic equality robust to serializer
    drift: apply the existing revision back and re-generate the patch in
    the current format before comparing."""
    cache_key = (lws.metadata.uid, lws.metadata.generation,
                 existing.metadata.resource_version)
    if cache is not None and cache_key in cache:
        return True
    latest = apply_revision(lws, existing)
    reconstructed = get_patch(latest)
    if canonical_json(proposed.data) == canonical_json(reconstructed):
        if cache is not None:
            cache[cache_key] = True
            while len(cache) > 10000:  # LRU bound (reference cache size 10k)
                cache.pop(next(iter(cache)))
        return True
    return False


def truncate_revisions(store: Store, lws: LeaderWorkerSet,
                       revision_key: str) -> None:
    """revision_utils.go:239-259."""
    for r in list_revisions(store, lws):
        if get_revision_key(r) != revision_key:
            try:
                store.delete("ControllerRevision", r.metadata.namespace,
                             r.metadata.name)
            except NotFoundError:
                pass
