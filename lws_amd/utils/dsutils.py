"""DisaggregatedSet domain utilities.

Behavioral port of reference pkg/utils/disaggregatedset/utils.go (naming,
labels, revision hashing, revision-role grouping, initial-replicas
annotation) and affinity.go (placement-policy affinity builder).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from ..api import disaggregatedset as dsapi
from ..api import leaderworkerset as lwsapi
from ..api import serde
from ..api.core import (Affinity, PodAffinity, PodAffinityTerm,
                        PodAntiAffinity, PodSpec)
from ..api.leaderworkerset import LeaderWorkerSet
from ..api.meta import LabelSelector, LabelSelectorRequirement
from .hashutil import canonical_json, sha256_short

NUM_REQUIRED_ROLES = 2
REVISION_LENGTH = 8


# -- initial-replicas annotation (utils.go:33-81) --------------------------

def get_initial_replicas(lws: LeaderWorkerSet) -> Optional[int]:
    value = (lws.metadata.annotations or {}).get(
        dsapi.INITIAL_REPLICAS_ANNOTATION_KEY, "")
    if not value:
        return None
    try:
        return int(value)
    except ValueError:
        return None


def set_initial_replicas(lws: LeaderWorkerSet, replicas: int) -> None:
    lws.metadata.annotations[dsapi.INITIAL_REPLICAS_ANNOTATION_KEY] = \
        str(replicas)


# -- naming / labels (utils.go:93-111) -------------------------------------

def generate_name(base: str, slice_: int, revision: str, role: str) -> str:
    return f"{base}-{slice_}-{revision}-{role}"


def generate_legacy_name(base: str, revision: str, role: str) -> str:
    return f"{base}-{revision}-{role}"


def generate_labels(base: str, slice_: int, revision: str,
                    role: str) -> dict[str, str]:
    return {
        "app": f"{base}-{slice_}-{role}",
        dsapi.ROLE_LABEL_KEY: role,
        dsapi.SLICE_LABEL_KEY: str(slice_),
        dsapi.SET_NAME_LABEL_KEY: base,
        dsapi.REVISION_LABEL_KEY: revision,
    }


def get_slices(ds) -> int:
    return ds.spec.slices if ds.spec.slices is not None else 1


def slice_label_matches(labels: dict[str, str], slice_: int) -> bool:
    """utils.go:124-133 — slice<0 matches all; slice 0 adopts label-less
    legacy objects."""
    if slice_ < 0:
        return True
    value = (labels or {}).get(dsapi.SLICE_LABEL_KEY, "")
    if not value:
        return slice_ == 0
    return value == str(slice_)


def has_slice_label(labels: dict[str, str]) -> bool:
    return bool((labels or {}).get(dsapi.SLICE_LABEL_KEY, ""))


# -- revision hash (utils.go:142-169) ---------------------------------------

def compute_revision(roles) -> str:
    templates = [{"name": r.name,
                  "template": serde.to_dict(r.spec.leader_worker_template)}
                 for r in roles]
    return sha256_short(canonical_json(templates), REVISION_LENGTH)


def get_role_names(ds) -> list[str]:
    return [r.name for r in ds.spec.roles]


def get_role_configs(ds) -> dict:
    return {r.name: r for r in ds.spec.roles}


# -- revision-role grouping (utils.go:190-248) ------------------------------

def get_lws_replicas(lws: LeaderWorkerSet) -> int:
    return lws.spec.replicas if lws.spec.replicas is not None else 1


@dataclass
class RevisionRoles:
    revision: str = ""
    roles: dict[str, LeaderWorkerSet] = field(default_factory=dict)


class RevisionRolesList(list):
    def total_replicas(self, role: str) -> int:
        return sum(get_lws_replicas(rev.roles[role]) for rev in self
                   if role in rev.roles)

    def total_initial_replicas(self, role: str) -> int:
        total = 0
        for rev in self:
            lws = rev.roles.get(role)
            if lws is None:
                continue
            init = get_initial_replicas(lws)
            total += init if init is not None else get_lws_replicas(lws)
        return total


def group_by_revision(lws_list) -> RevisionRolesList:
    by_rev: dict[str, RevisionRoles] = {}
    for lws in lws_list:
        revision = (lws.metadata.labels or {}).get(dsapi.REVISION_LABEL_KEY, "")
        role = (lws.metadata.labels or {}).get(dsapi.ROLE_LABEL_KEY, "")
        rr = by_rev.setdefault(revision, RevisionRoles(revision=revision))
        rr.roles[role] = lws
    return RevisionRolesList(by_rev.values())


def compute_initial_replica_state(lws_list) -> dict[str, int]:
    """utils.go ComputeInitialReplicaState."""
    state: dict[str, int] = {}
    for lws in lws_list:
        role = (lws.metadata.labels or {}).get(dsapi.ROLE_LABEL_KEY, "")
        if not role:
            continue
        init = get_initial_replicas(lws)
        replicas = init if init is not None else get_lws_replicas(lws)
        state[role] = state.get(role, 0) + replicas
    return state


# -- placement affinity (affinity.go:37-115) --------------------------------

DS_SLICE_AFFINITY_KEY = dsapi.SLICE_LABEL_KEY
DS_NAME_AFFINITY_KEY = dsapi.SET_NAME_LABEL_KEY


def set_placement_affinities(pod_spec: PodSpec, ds_name: str, slice_: int,
                             policy) -> None:
    """ExclusiveSlice: co-locate the slice's roles on one topology domain +
    anti-affinity vs other slices of the same DS.  ExclusiveTopology adds
    anti-affinity vs ALL other DS slices (1:1 domain<->slice).  Slice-0
    legacy objects carry no slice label, handled via DoesNotExist terms."""
    if policy is None or policy.type in ("", dsapi.PlacementType.NoneType):
        return
    topo = policy.topology
    if pod_spec.affinity is None:
        pod_spec.affinity = Affinity()
    if pod_spec.affinity.pod_affinity is None:
        pod_spec.affinity.pod_affinity = PodAffinity()
    if pod_spec.affinity.pod_anti_affinity is None:
        pod_spec.affinity.pod_anti_affinity = PodAntiAffinity()

    slice_str = str(slice_)
    # co-locate with this DS+slice's pods
    slice_match = [LabelSelectorRequirement(key=DS_NAME_AFFINITY_KEY,
                                            operator="In", values=[ds_name]),
                   LabelSelectorRequirement(key=DS_SLICE_AFFINITY_KEY,
                                            operator="In",
                                            values=[slice_str])]
    pod_spec.affinity.pod_affinity.required_during_scheduling_ignored_during_execution.append(
        PodAffinityTerm(label_selector=LabelSelector(
            match_expressions=slice_match), topology_key=topo))

    if policy.type == dsapi.PlacementType.ExclusiveSlice:
        # repel other slices of the SAME DS
        pod_spec.affinity.pod_anti_affinity.required_during_scheduling_ignored_during_execution.append(
            PodAffinityTerm(label_selector=LabelSelector(match_expressions=[
                LabelSelectorRequirement(key=DS_NAME_AFFINITY_KEY,
                                         operator="In", values=[ds_name]),
                LabelSelectorRequirement(key=DS_SLICE_AFFINITY_KEY,
                                         operator="NotIn",
                                         values=[slice_str]),
            ]), topology_key=topo))
    else:  # ExclusiveTopology: repel every other DS slice (any DS)
        pod_spec.affinity.pod_anti_affinity.required_during_scheduling_ignored_during_execution.append(
            PodAffinityTerm(label_selector=LabelSelector(match_expressions=[
                LabelSelectorRequirement(key=DS_NAME_AFFINITY_KEY,
                                         operator="Exists"),
                LabelSelectorRequirement(key=DS_NAME_AFFINITY_KEY,
                                         operator="NotIn", values=[ds_name]),
            ]), topology_key=topo))
        pod_spec.affinity.pod_anti_affinity.required_during_scheduling_ignored_during_execution.append(
            PodAffinityTerm(label_selector=LabelSelector(match_expressions=[
                LabelSelectorRequirement(key=DS_NAME_AFFINITY_KEY,
                                         operator="In", values=[ds_name]),
                LabelSelectorRequirement(key=DS_SLICE_AFFINITY_KEY,
                                         operator="NotIn",
                                         values=[slice_str]),
            ]), topology_key=topo))
