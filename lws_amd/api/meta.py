"""metav1-shaped metadata types for the lws_amd object model.

These mirror the subset of k8s.io/apimachinery that the reference
controllers rely on (ObjectMeta, OwnerReference, Condition, label
selectors), re-expressed as plain dataclasses over the in-repo cluster
substrate.  Reference: api/leaderworkerset/v1/leaderworkerset_types.go
uses metav1.ObjectMeta / metav1.Condition throughout.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Optional, Union

# IntOrString: k8s intstr.IntOrString equivalent ("30%" or 3)
IntOrString = Union[int, str]


@dataclass
class OwnerReference:
    api_version: str = ""
    kind: str = ""
    name: str = ""
    uid: str = ""
    controller: Optional[bool] = None
    block_owner_deletion: Optional[bool] = None


@dataclass
class ObjectMeta:
    name: str = ""
    namespace: str = ""
    uid: str = ""
    resource_version: str = ""
    generation: int = 0
    creation_timestamp: Optional[float] = None
    deletion_timestamp: Optional[float] = None
    labels: dict[str, str] = field(default_factory=dict)
    annotations: dict[str, str] = field(default_factory=dict)
    owner_references: list[OwnerReference] = field(default_factory=list)
    finalizers: list[str] = field(default_factory=list)


@dataclass
class Condition:
    """metav1.Condition equivalent."""

    type: str = ""
    status: str = ""  # "True" | "False" | "Unknown"
    observed_generation: int = 0
    last_transition_time: Optional[float] = None
    reason: str = ""
    message: str = ""


def new_condition(type_: str, status: str, reason: str, message: str,
                  observed_generation: int = 0) -> Condition:
    return Condition(type=type_, status=status, reason=reason, message=message,
                     observed_generation=observed_generation,
                     last_transition_time=time.time())


@dataclass
class LabelSelectorRequirement:
    key: str = ""
    operator: str = ""  # In | NotIn | Exists | DoesNotExist
    values: list[str] = field(default_factory=list)


@dataclass
class LabelSelector:
    match_labels: dict[str, str] = field(default_factory=dict)
    match_expressions: list[LabelSelectorRequirement] = field(default_factory=list)


def selector_matches(sel: Optional[LabelSelector], labels: dict[str, str]) -> bool:
    if sel is None:
        return True
    for k, v in (sel.match_labels or {}).items():
        if labels.get(k) != v:
            return False
    for req in sel.match_expressions or []:
        val = labels.get(req.key)
        if req.operator == "In":
            if val is None or val not in req.values:
                return False
        elif req.operator == "NotIn":
            if val is not None and val in req.values:
                return False
        elif req.operator == "Exists":
            if req.key not in labels:
                return False
        elif req.operator == "DoesNotExist":
            if req.key in labels:
                return False
        else:
            return False
    return True


def format_label_selector(sel: LabelSelector) -> str:
    """Render a LabelSelector the way metav1.FormatLabelSelector does
    (used for status.hpaPodSelector)."""
    parts = [f"{k}={v}" for k, v in sorted((sel.match_labels or {}).items())]
    for req in sel.match_expressions or []:
        if req.operator == "In":
            parts.append(f"{req.key} in ({','.join(sorted(req.values))})")
        elif req.operator == "NotIn":
            parts.append(f"{req.key} notin ({','.join(sorted(req.values))})")
        elif req.operator == "Exists":
            parts.append(req.key)
        elif req.operator == "DoesNotExist":
            parts.append(f"!{req.key}")
    return ",".join(parts)


def get_int_or_percent(v: Optional[IntOrString], total: int, round_up: bool) -> int:
    """intstr.GetScaledValueFromIntOrPercent equivalent."""
    if v is None:
        return 0
    if isinstance(v, int):
        return v
    s = str(v).strip()
    if s.endswith("%"):
        pct = int(s[:-1])
        scaled = pct * total / 100.0
        if round_up:
            return int(-(-scaled // 1))
        return int(scaled // 1)
    return int(s)


def is_percent(v: Optional[IntOrString]) -> bool:
    return isinstance(v, str) and str(v).strip().endswith("%")
