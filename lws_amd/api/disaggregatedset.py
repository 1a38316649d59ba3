"""DisaggregatedSet v1 + DisaggregatedSetRoleScaler v1 API types.

Mirrors the reference group `disaggregatedset.x-k8s.io/v1`
(/root/reference/api/disaggregatedset/v1/disaggregatedset_types.go and
 disaggregatedsetrolescaler_types.go).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from .leaderworkerset import LeaderWorkerSetSpec
from .meta import Condition, ObjectMeta

GROUP = "disaggregatedset.x-k8s.io"
VERSION = "v1"
API_VERSION = f"{GROUP}/{VERSION}"
KIND = "DisaggregatedSet"
SCALER_KIND = "DisaggregatedSetRoleScaler"

# --- well-known keys (disaggregatedset_types.go:25-42) ---
SET_NAME_LABEL_KEY = "disaggregatedset.x-k8s.io/name"
ROLE_LABEL_KEY = "disaggregatedset.x-k8s.io/role"
SLICE_LABEL_KEY = "disaggregatedset.x-k8s.io/slice"
REVISION_LABEL_KEY = "disaggregatedset.x-k8s.io/revision"
INITIAL_REPLICAS_ANNOTATION_KEY = "disaggregatedset.x-k8s.io/initial-replicas"

MIN_ROLES = 2
MAX_ROLES = 10
MIN_SLICES = 1
MAX_SLICES = 100


class RoleScalingMode:
    Static = "Static"
    External = "External"


class PlacementType:
    NoneType = "None"
    ExclusiveSlice = "ExclusiveSlice"
    ExclusiveTopology = "ExclusiveTopology"


class DisaggregatedSetConditionType:
    Available = "Available"
    Progressing = "Progressing"


DISAGGREGATED_SET_ROLE_SCALER_READY = "Ready"


@dataclass
class RoleScaling:
    mode: str = ""  # RoleScalingMode; default Static


@dataclass
class DisaggregatedRoleSpec:
    """disaggregatedset_types.go:75-94 — embeds LeaderWorkerSetTemplateSpec
    inline (metadata + spec)."""

    name: str = ""
    scaling: Optional[RoleScaling] = None
    # inline LeaderWorkerSetTemplateSpec:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: LeaderWorkerSetSpec = field(default_factory=LeaderWorkerSetSpec)


@dataclass
class PlacementPolicy:
    type: str = ""  # PlacementType; default None
    topology: str = ""


@dataclass
class DisaggregatedSetSpec:
    roles: list[DisaggregatedRoleSpec] = field(default_factory=list)
    slices: Optional[int] = None  # default 1
    placement_policy: Optional[PlacementPolicy] = None


@dataclass
class RoleStatus:
    name: str = ""
    replicas: int = 0
    ready_replicas: int = 0
    updated_replicas: int = 0


@dataclass
class DisaggregatedSetStatus:
    observed_generation: int = 0
    role_statuses: list[RoleStatus] = field(default_factory=list)
    conditions: list[Condition] = field(default_factory=list)


@dataclass
class DisaggregatedSet:
    api_version: str = API_VERSION
    kind: str = KIND
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: DisaggregatedSetSpec = field(default_factory=DisaggregatedSetSpec)
    status: DisaggregatedSetStatus = field(default_factory=DisaggregatedSetStatus)


@dataclass
class DisaggregatedSetRoleScalerSpec:
    # Non-pointer with default 0 for the /scale handler
    # (disaggregatedsetrolescaler_types.go:31-45).
    replicas: int = 0


@dataclass
class DisaggregatedSetRoleScalerStatus:
    replicas: int = 0
    selector: str = ""
    observed_generation: int = 0
    conditions: list[Condition] = field(default_factory=list)


@dataclass
class DisaggregatedSetRoleScaler:
    api_version: str = API_VERSION
    kind: str = SCALER_KIND
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: DisaggregatedSetRoleScalerSpec = field(
        default_factory=DisaggregatedSetRoleScalerSpec)
    status: DisaggregatedSetRoleScalerStatus = field(
        default_factory=DisaggregatedSetRoleScalerStatus)


def role_scaling_mode(role: DisaggregatedRoleSpec) -> str:
    if role.scaling is not None and role.scaling.mode == RoleScalingMode.External:
        return RoleScalingMode.External
    return RoleScalingMode.Static
