"""KV-cache handoff between disaggregated prefill and decode roles.

The DS prefill/decode split (reference disaggregatedset_types.go roles)
needs the prefilled context moved from the prefill role's engine to the
decode role's engine.  Inside one ExclusiveSlice domain the roles sit on
the same xGMI island, so the transfer is a point-to-point send over the
process group (RCCL on GPU, gloo on CPU): one metadata object + one
contiguous [L, 2, T, Hkv, D] tensor.

In-process handoff (roles co-hosted, e.g. the 1-GPU DS bench) is a
direct `Engine.export_kv` -> `Engine.import_kv` copy.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from .engine import Engine, SamplingParams


def handoff_local(src: Engine, dst: Engine, seq_id: int,
                  sampling: Optional[SamplingParams] = None) -> int:
    """Move a prefilled sequence between co-hosted engines."""
    token_ids, num_cached, layers = src.export_kv(seq_id)
    sid = dst.import_kv(token_ids, num_cached, layers, sampling)
    src.finish(seq_id)
    return sid


def _pack(layers: list) -> torch.Tensor:
    # [L, 2, T, Hkv, D] contiguous, one send instead of 2L
    return torch.stack([torch.stack([k, v]) for k, v in layers]).contiguous()


def send_kv(engine: Engine, seq_id: int, dst: int, group=None) -> None:
    """Send a prefilled sequence to the decode role at rank `dst`."""
    token_ids, num_cached, layers = engine.export_kv(seq_id)
    packed = _pack(layers)
    backend = dist.get_backend(group) if group is not None \
        else dist.get_backend()
    if backend == "gloo":
        packed = packed.cpu()
    meta = {"token_ids": token_ids, "num_cached": num_cached,
            "shape": list(packed.shape), "dtype": str(packed.dtype)}
    dist.send_object_list([meta], dst=dst, group=group)
    dist.send(packed, dst=dst, group=group)
    engine.finish(seq_id)


def recv_kv(engine: Engine, src: int, group=None,
            sampling: Optional[SamplingParams] = None) -> int:
    """Receive a prefilled sequence from the prefill role at rank `src`
    and adopt it into this engine's KV pools."""
    buf: list = [None]
    dist.recv_object_list(buf, src=src, group=group)
    meta = buf[0]
    backend = dist.get_backend(group) if group is not None \
        else dist.get_backend()
    dtype = getattr(torch, meta["dtype"].split(".")[-1])
    device = "cpu" if backend == "gloo" else engine.device
    packed = torch.empty(meta["shape"], dtype=dtype, device=device)
    dist.recv(packed, src=src, group=group)
    layers = [(packed[li, 0], packed[li, 1])
              for li in range(packed.size(0))]
    return engine.import_kv(meta["token_ids"], meta["num_cached"], layers,
                            sampling)
