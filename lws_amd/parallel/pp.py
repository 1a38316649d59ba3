"""Pipeline parallelism: point-to-point activation transfer between
stages (SURVEY.md §2.9 — the reference's PP example splits one group's
pods into pipeline stages over the leader-address rendezvous; here the
stages are the engine processes themselves, hidden states ride RCCL p2p
over xGMI on GPU and gloo on CPU tests).

Scope: TP x PP grids are not combined (assert one of them is 1); the PP
group is the default process group, stage r == global rank r.
"""
from __future__ import annotations

import torch
import torch.distributed as dist


def _wire(t: torch.Tensor) -> torch.Tensor:
    """gloo cannot move device tensors; nccl(RCCL) requires them."""
    if dist.get_backend() == "gloo":
        return t.cpu()
    return t


def send_stage(state: torch.Tensor, dst: int) -> None:
    dist.send(_wire(state.contiguous()), dst=dst)


def recv_stage(shape, dtype, device, src: int) -> torch.Tensor:
    backend = dist.get_backend()
    buf = torch.empty(shape, dtype=dtype,
                      device="cpu" if backend == "gloo" else device)
    dist.recv(buf, src=src)
    return buf.to(device)


def broadcast_tokens(tokens: torch.Tensor, src: int) -> torch.Tensor:
    """Sampled tokens travel from the LAST stage to every stage so
    sequence bookkeeping stays identical everywhere."""
    obj = [tokens.tolist()]
    dist.broadcast_object_list(obj, src=src)
    return torch.tensor(obj[0], dtype=tokens.dtype, device=tokens.device)
