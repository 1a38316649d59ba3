"""Plain-PyTorch fp32 reference implementations of every lws_amd op.

These exist for numerics tests (HIP kernel vs fp32 reference on the same
device) and CPU-only development of the model code.  They are NEVER used
as a GPU serving path.
"""
from __future__ import annotations

import torch


def rmsnorm_ref(x: torch.Tensor, weight: torch.Tensor,
                eps: float = 1e-5) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(x.dtype)


def fused_add_rmsnorm_ref(x: torch.Tensor, residual: torch.Tensor,
                          weight: torch.Tensor,
                          eps: float = 1e-5) -> tuple[torch.Tensor, torch.Tensor]:
    new_residual = (residual.float() + x.float()).to(x.dtype)
    return rmsnorm_ref(new_residual, weight, eps), new_residual


def silu_mul_ref(gateup: torch.Tensor) -> torch.Tensor:
    g, u = gateup.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(g) * u).to(gateup.dtype)


def rope_ref(q: torch.Tensor, k: torch.Tensor, cos_sin: torch.Tensor,
             positions: torch.Tensor, num_q_heads: int,
             num_kv_heads: int) -> tuple[torch.Tensor, torch.Tensor]:
    """q [T, Hq*D], k [T, Hkv*D]; neox-style pairs (i, i+D/2)."""
    D = cos_sin.size(1)
    half = D // 2
    cos = cos_sin[positions.long(), :half]      # [T, half]
    sin = cos_sin[positions.long(), half:]

    def rot(x, H):
        T = x.size(0)
        xf = x.float().view(T, H, D)
        x1, x2 = xf[..., :half], xf[..., half:]
        c = cos[:, None, :]
        s = sin[:, None, :]
        o1 = x1 * c - x2 * s
        o2 = x2 * c + x1 * s
        return torch.cat([o1, o2], dim=-1).view(T, H * D).to(x.dtype)

    return rot(q, num_q_heads), rot(k, num_kv_heads)


def paged_attention_decode_ref(q: torch.Tensor, k_cache: torch.Tensor,
                               v_cache: torch.Tensor,
                               block_tables: torch.Tensor,
                               seq_lens: torch.Tensor,
                               scale: float) -> torch.Tensor:
    """q [B, Hq, D]; caches [pages, Hkv, page, D]."""
    B, Hq, D = q.shape
    Hkv, page = k_cache.size(1), k_cache.size(2)
    G = Hq // Hkv
    out = torch.empty_like(q)
    for b in range(B):
        L = int(seq_lens[b])
        npages = (L + page - 1) // page
        pages = block_tables[b, :npages].long()
        k = k_cache[pages].float()      # [npages, Hkv, page, D]
        v = v_cache[pages].float()
        k = k.permute(1, 0, 2, 3).reshape(Hkv, npages * page, D)[:, :L]
        v = v.permute(1, 0, 2, 3).reshape(Hkv, npages * page, D)[:, :L]
        for h in range(Hq):
            hk = h // G
            scores = (k[hk] @ q[b, h].float()) * scale       # [L]
            p = torch.softmax(scores, dim=-1)
            out[b, h] = (p @ v[hk]).to(q.dtype)
    return out


def sdpa_prefill_ref(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     scale: float, causal: bool = True) -> torch.Tensor:
    """q [B, Hq, S, D], k/v [B, Hkv, S, D] -> [B, Hq, S, D] (fp32 math)."""
    B, Hq, S, D = q.shape
    Hkv = k.size(1)
    G = Hq // Hkv
    kf = k.float().repeat_interleave(G, dim=1)
    vf = v.float().repeat_interleave(G, dim=1)
    scores = q.float() @ kf.transpose(-1, -2) * scale
    if causal:
        mask = torch.full((S, S), float("-inf"), device=q.device).triu(1)
        scores = scores + mask
    p = torch.softmax(scores, dim=-1)
    return (p @ vf).to(q.dtype)
