"""Hashing helpers shared by revisions, group keys and naming.

Mirrors the reference's hashing surface:
 - FNV-32a + safe-encode for ControllerRevision names
   (pkg/utils/revision/revision_utils.go:333-342, k8s rand.SafeEncodeString)
 - sha1 hex for group unique keys (pkg/webhooks/pod_webhook.go:180-182)
 - sha256[:8] for DisaggregatedSet revisions
   (pkg/utils/disaggregatedset/utils.go:142-169)
"""
from __future__ import annotations

import hashlib
import json
from typing import Any

# k8s.io/apimachinery/pkg/util/rand alphanums with ambiguous chars removed
# (rand.SafeEncodeString maps arbitrary bytes onto this alphabet)
_SAFE_ALPHABET = "bcdfghjklmnpqrstvwxz2456789"


def fnv32a(data: bytes) -> int:
    h = 2166136261
    for b in data:
        h ^= b
        h = (h * 16777619) & 0xFFFFFFFF
    return h


def safe_encode_uint32(v: int) -> str:
    """Encode a uint32 like k8s's fmt.Sprint + rand.SafeEncodeString of the
    decimal digits (stable, collision-preserving)."""
    s = str(v)
    return "".join(_SAFE_ALPHABET[ord(c) % len(_SAFE_ALPHABET)] for c in s)


def canonical_json(obj: Any) -> str:
    return json.dumps(obj, sort_keys=True, separators=(",", ":"))


def hash_object(obj: Any) -> str:
    """Stable short hash of a JSON-able object (controller-revision style)."""
    return safe_encode_uint32(fnv32a(canonical_json(obj).encode()))


def sha1_hex(s: str) -> str:
    return hashlib.sha1(s.encode()).hexdigest()


def sha256_short(s: str, n: int = 8) -> str:
    return hashlib.sha256(s.encode()).hexdigest()[:n]
