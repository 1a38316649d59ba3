"""LeaderWorkerSet defaulting + validating admission.

Behavioral port of reference pkg/webhooks/leaderworkerset_webhook.go
(Default :52-85, generalValidate :123-190, ValidateUpdate :98-116,
validateUpdateSubGroupPolicy :242-261).
"""
from __future__ import annotations

import re
from typing import Optional

from ..api import leaderworkerset as lwsapi
from ..api.leaderworkerset import (LeaderWorkerSet, NetworkConfig,
                                   RollingUpdateConfiguration)
from ..api.meta import get_int_or_percent, is_percent
from ..cluster.store import InvalidError

DNS1035_RE = re.compile(r"^[a-z]([-a-z0-9]*[a-z0-9])?$")


class ValidationError(InvalidError):
    pass


def default_lws(lws: LeaderWorkerSet) -> None:
    """leaderworkerset_webhook.go:52-85."""
    spec = lws.spec
    t = spec.leader_worker_template
    if spec.replicas is None:
        spec.replicas = 1
    if t.size is None:
        t.size = 1
    if not t.restart_policy:
        t.restart_policy = lwsapi.RestartPolicyType.RecreateGroupOnPodRestart
    if t.restart_policy == lwsapi.RestartPolicyType.DeprecatedDefault:
        t.restart_policy = lwsapi.RestartPolicyType.NoneRestart
    if not spec.startup_policy:
        spec.startup_policy = lwsapi.StartupPolicyType.LeaderCreated
    if not spec.rollout_strategy.type:
        spec.rollout_strategy.type = lwsapi.RolloutStrategyType.RollingUpdate
    if (spec.rollout_strategy.type == lwsapi.RolloutStrategyType.RollingUpdate
            and spec.rollout_strategy.rolling_update_configuration is None):
        spec.rollout_strategy.rolling_update_configuration = \
            RollingUpdateConfiguration(partition=0, max_unavailable=1, max_surge=0)
    ruc = spec.rollout_strategy.rolling_update_configuration
    if ruc is not None:
        if ruc.partition is None:
            ruc.partition = 0
        if ruc.max_unavailable is None:
            ruc.max_unavailable = 1
        if ruc.max_surge is None:
            ruc.max_surge = 0
    if spec.network_config is None:
        spec.network_config = NetworkConfig(
            subdomain_policy=lwsapi.SubdomainPolicy.Shared)
    elif spec.network_config.subdomain_policy is None:
        spec.network_config.subdomain_policy = lwsapi.SubdomainPolicy.Shared
    if t.sub_group_policy is not None and t.sub_group_policy.type is None:
        t.sub_group_policy.type = lwsapi.SubGroupPolicyType.LeaderWorker


def _validate_int_or_percent(v, path: str, errs: list[str]) -> None:
    if isinstance(v, int):
        if v < 0:
            errs.append(f"{path}: must be greater than or equal to 0")
    elif isinstance(v, str):
        s = v.strip()
        if not re.fullmatch(r"[0-9]+%", s):
            errs.append(f"{path}: a valid percent string must be a numeric string "
                        "followed by an ending '%'")
        elif int(s[:-1]) > 100:
            errs.append(f"{path}: must not be greater than 100%")
    else:
        errs.append(f"{path}: must be an integer or percentage")


def general_validate(lws: LeaderWorkerSet) -> list[str]:
    errs: list[str] = []
    name = lws.metadata.name
    if not name or len(name) > 63 or not DNS1035_RE.match(name):
        errs.append("metadata.name: must be a DNS-1035 label")
    spec = lws.spec
    replicas = spec.replicas if spec.replicas is not None else 1
    size = spec.leader_worker_template.size \
        if spec.leader_worker_template.size is not None else 1
    if replicas < 0:
        errs.append("spec.replicas: replicas must be equal or greater than 0")
    if replicas > lwsapi.MAX_REPLICAS:
        errs.append("spec.replicas: replicas must be equal or less than 1000000")
    if size < 1:
        errs.append("spec.leaderWorkerTemplate.size: size must be equal or "
                    "greater than 1")
    if replicas * size > lwsapi.MAX_INT32:
        errs.append("spec.replicas: the product of replicas and worker replicas "
                    f"must not exceed {lwsapi.MAX_INT32}")
    ruc = spec.rollout_strategy.rolling_update_configuration
    if ruc is not None:
        _validate_int_or_percent(ruc.max_unavailable,
                                 "spec.rolloutStrategy.rollingUpdateConfiguration"
                                 ".maxUnavailable", errs)
        _validate_int_or_percent(ruc.max_surge,
                                 "spec.rolloutStrategy.rollingUpdateConfiguration"
                                 ".maxSurge", errs)
        if ruc.partition is not None and ruc.partition < 0:
            errs.append("spec.rolloutStrategy.rollingUpdateConfiguration"
                        ".partition: must be greater than or equal to 0")
        if not errs:
            mu = get_int_or_percent(ruc.max_unavailable, replicas, False)
            ms = get_int_or_percent(ruc.max_surge, replicas, True)
            if mu == 0 and ms == 0 and replicas != 0:
                errs.append("spec.rolloutStrategy.rollingUpdateConfiguration"
                            ".maxUnavailable: must not be 0 when `maxSurge` is 0")
    sgp = spec.leader_worker_template.sub_group_policy
    if sgp is not None:
        sgs = sgp.sub_group_size or 0
        if sgs < 1:
            errs.append("spec.leaderWorkerTemplate.SubGroupPolicy.subGroupSize: "
                        "subGroupSize must be equal or greater than 1")
        else:
            if size % sgs != 0 and (size - 1) % sgs != 0:
                errs.append("spec.leaderWorkerTemplate.SubGroupPolicy"
                            ".subGroupSize: size or size - 1 must be divisible "
                            "by subGroupSize")
            if size < sgs:
                errs.append("spec.leaderWorkerTemplate.SubGroupPolicy"
                            ".subGroupSize: subGroupSize cannot be larger than size")
            if sgp.type == lwsapi.SubGroupPolicyType.LeaderExcluded and \
                    (size - 1) % sgs != 0:
                errs.append("spec.leaderWorkerTemplate.SubGroupPolicy"
                            ".subGroupSize: size-1 must be divisible by "
                            "subGroupSize when using LeaderExcluded")
    else:
        if lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY in (lws.metadata.annotations or {}):
            errs.append(f"metadata.annotations.{lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY}: "
                        "cannot have subgroup-exclusive-topology without "
                        "subGroupSize set")
    return errs


def validate_lws(lws: LeaderWorkerSet, old: Optional[LeaderWorkerSet]) -> None:
    errs = general_validate(lws)
    if old is not None:
        new_sgp = lws.spec.leader_worker_template.sub_group_policy
        old_sgp = old.spec.leader_worker_template.sub_group_policy
        if new_sgp is not None and old_sgp is not None and \
                new_sgp.sub_group_size != old_sgp.sub_group_size:
            errs.append("spec.leaderWorkerTemplate.SubGroupPolicy.subGroupSize: "
                        "field is immutable")
        if new_sgp is not None and old_sgp is None:
            errs.append("cannot enable subGroupSize after the lws is already created")
        if new_sgp is None and old_sgp is not None:
            errs.append("cannot remove subGroupSize after enabled")
        if lws.spec.network_config is not None and \
                lws.spec.network_config.subdomain_policy is None:
            errs.append("networkConfig.subdomainPolicy: cannot set subdomainPolicy "
                        "as null")
    if errs:
        raise ValidationError("; ".join(errs))


def register(store) -> None:
    store.add_mutator(lwsapi.KIND, default_lws)
    store.add_validator(lwsapi.KIND, validate_lws)
