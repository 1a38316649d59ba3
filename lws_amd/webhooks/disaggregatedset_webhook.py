"""DisaggregatedSet validating admission.

Behavioral port of reference
pkg/webhooks/disaggregatedset/disaggregatedset_webhook.go: role rollout
restrictions, generated-name DNS length math, scaler name limits, the
alpha External+slices>1 restriction, the all-or-nothing replicas CEL rule
(disaggregatedset_types.go:102), and placement-policy validation.
"""
from __future__ import annotations

import re
from typing import Optional

from ..api import disaggregatedset as dsapi
from ..api import leaderworkerset as lwsapi
from ..api.disaggregatedset import DisaggregatedSet
from ..api.meta import get_int_or_percent
from ..cluster.store import InvalidError

DNS1035_MAX = 63
SCALER_NAME_MAX = 253
REVISION_LEN = 8
SERVICE_SUFFIX_LEN = 4  # len("-prv")
SEPARATORS = 3
ROLE_NAME_RE = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")


def _validate_role_rollout_strategy(role, errs: list[str]) -> None:
    rs = role.spec.rollout_strategy
    if rs.type and rs.type != lwsapi.RolloutStrategyType.RollingUpdate:
        errs.append(f"roles[{role.name}].spec.rolloutStrategy.type: unsupported "
                    f"value {rs.type!r} (must be RollingUpdate or empty)")
    ruc = rs.rolling_update_configuration
    if ruc is not None:
        if ruc.partition:
            errs.append(f"roles[{role.name}]: partition is not supported by "
                        "DisaggregatedSet; rolling updates are managed across "
                        "roles by the DisaggregatedSet controller")
        replicas = role.spec.replicas
        if replicas:
            mu = ruc.max_unavailable if ruc.max_unavailable is not None else 1
            ms = ruc.max_surge if ruc.max_surge is not None else 0
            mu_v = get_int_or_percent(mu, replicas, False)
            ms_v = get_int_or_percent(ms, replicas, True)
            if mu_v == 0 and ms_v == 0:
                errs.append(f"roles[{role.name}]: maxUnavailable must not be 0 "
                            "when `maxSurge` is 0")


def _role_exclusive_topology_annotation(role) -> Optional[str]:
    keys = (lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY,
            lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY)
    t = role.spec.leader_worker_template
    for key in keys:
        if key in (role.metadata.annotations or {}):
            return key
        if t.leader_template is not None and \
                key in (t.leader_template.metadata.annotations or {}):
            return key
        if key in (t.worker_template.metadata.annotations or {}):
            return key
    return None


def validate_ds(ds: DisaggregatedSet, old: Optional[DisaggregatedSet]) -> None:
    errs: list[str] = []
    roles = ds.spec.roles or []
    if len(roles) < dsapi.MIN_ROLES:
        errs.append(f"spec.roles: at least {dsapi.MIN_ROLES} roles required")
    if len(roles) > dsapi.MAX_ROLES:
        errs.append(f"spec.roles: at most {dsapi.MAX_ROLES} roles allowed")
    names = [r.name for r in roles]
    if len(set(names)) != len(names):
        errs.append("spec.roles: role names must be unique")
    slices = ds.spec.slices if ds.spec.slices is not None else 1
    if slices < dsapi.MIN_SLICES or slices > dsapi.MAX_SLICES:
        errs.append(f"spec.slices: must be within [{dsapi.MIN_SLICES}, "
                    f"{dsapi.MAX_SLICES}]")

    has_external = False
    for role in roles:
        if not role.name or len(role.name) > 63 or not ROLE_NAME_RE.match(role.name):
            errs.append(f"spec.roles[{role.name!r}].name: must be a DNS-1035 label")
        _validate_role_rollout_strategy(role, errs)
        if dsapi.role_scaling_mode(role) == dsapi.RoleScalingMode.External:
            has_external = True
            scaler_name = f"{ds.metadata.name}-{role.name}"
            if len(scaler_name) > SCALER_NAME_MAX:
                errs.append(f"roles[{role.name}].name: would produce scaler name "
                            f"exceeding {SCALER_NAME_MAX} characters")

    # all-or-nothing replicas rule over non-External roles (CEL rule,
    # disaggregatedset_types.go:102)
    static_replicas = [r.spec.replicas or 0 for r in roles
                       if dsapi.role_scaling_mode(r) != dsapi.RoleScalingMode.External]
    if static_replicas and not (all(v == 0 for v in static_replicas)
                                or all(v > 0 for v in static_replicas)):
        errs.append("spec.roles: replicas must be zero for all non-External "
                    "roles or non-zero for all non-External roles")

    if has_external and slices > 1:
        errs.append("spec.slices: spec.slices > 1 is not supported while any "
                    "role has scaling.mode: External (alpha restriction)")

    # generated-name DNS-1035 math (disaggregatedset_webhook.go:117-154)
    slice_digits = len(str(slices - 1))
    for role in roles:
        lws_name_len = len(ds.metadata.name) + SEPARATORS + slice_digits + \
            REVISION_LEN + len(role.name)
        if lws_name_len + SERVICE_SUFFIX_LEN > DNS1035_MAX:
            errs.append(f"roles[{role.name}].name: the generated service name "
                        f"would exceed the DNS-1035 limit of {DNS1035_MAX} "
                        "characters; reduce the DisaggregatedSet name and/or "
                        "role name")

    # placement policy (disaggregatedset_webhook.go:160-185)
    policy = ds.spec.placement_policy
    if policy is not None and policy.type not in ("", dsapi.PlacementType.NoneType):
        if not policy.topology:
            errs.append("spec.placementPolicy.topology: topology is required "
                        "when type is not None")
        for role in roles:
            key = _role_exclusive_topology_annotation(role)
            if key is not None:
                errs.append(f"roles[{role.name}]: the {key!r} annotation must "
                            "not be combined with a non-None "
                            f"spec.placementPolicy.type ({policy.type})")

    if errs:
        raise InvalidError("; ".join(errs))


def default_ds(ds: DisaggregatedSet) -> None:
    if ds.spec.slices is None:
        ds.spec.slices = 1
    for role in ds.spec.roles or []:
        # default the embedded LWS template spec the way the LWS webhook would
        from .leaderworkerset_webhook import default_lws
        from ..api.leaderworkerset import LeaderWorkerSet

        shim = LeaderWorkerSet(spec=role.spec)
        default_lws(shim)
        role.spec = shim.spec


def register(store) -> None:
    store.add_mutator(dsapi.KIND, default_ds)
    store.add_validator(dsapi.KIND, validate_ds)
