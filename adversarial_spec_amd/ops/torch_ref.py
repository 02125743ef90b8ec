"""Plain-PyTorch fp32 reference implementations of every engine op.

These are the numerics oracle for the CDNA4 HIP kernels (tests compare the
HIP path against this fp32 reference, see tests/test_ops_gpu.py) and the
CPU execution path for plumbing tests and BASELINE config 1.

All functions accept/return torch tensors and compute in fp32 regardless of
input dtype (casting back at the end), matching the kernels' fp32
accumulation.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    """y = x / rms(x) * w, rowwise over the last dim."""
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * w.float()).to(x.dtype)


def add_rmsnorm(
    resid: torch.Tensor, delta: torch.Tensor, w: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Fused residual add + RMSNorm: resid' = resid + delta; y = rmsnorm(resid').

    On GPU this is one kernel (one read + one write of the residual stream
    instead of two passes over HBM).
    """
    r = (resid.float() + delta.float())
    inv = torch.rsqrt(r.pow(2).mean(-1, keepdim=True) + eps)
    y = r * inv * w.float()
    return r.to(resid.dtype), y.to(resid.dtype)


def rope_tables(
    head_dim: int, max_seq: int, theta: float, device, dtype=torch.float32
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Precomputed cos/sin tables [max_seq, head_dim/2] (host side, once).

    On-device trig per element would turn the memory-bound RoPE kernel into
    a VALU-bound one (cdna_hip_programming.md Appendix B), so tables are
    precomputed.
    """
    half = head_dim // 2
    freqs = 1.0 / (theta ** (torch.arange(half, device=device, dtype=torch.float64) / half))
    pos = torch.arange(max_seq, device=device, dtype=torch.float64)
    ang = torch.outer(pos, freqs)
    return ang.cos().to(dtype), ang.sin().to(dtype)


def rope(
    q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, pos0: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Rotary embedding, interleaved-pair convention.

    q: [t, n_heads, hd], k: [t, n_kv_heads, hd]; rotates pairs
    (x[2i], x[2i+1]) by the angle for absolute position pos0+row.
    (HF-format weights are permuted at load time to this convention.)
    """
    t = q.shape[0]
    c = cos[pos0 : pos0 + t].unsqueeze(1)  # [t,1,hd/2]
    s = sin[pos0 : pos0 + t].unsqueeze(1)

    def rot(x: torch.Tensor) -> torch.Tensor:
        xf = x.float()
        ev = xf[..., 0::2]
        od = xf[..., 1::2]
        out = torch.empty_like(xf)
        out[..., 0::2] = ev * c - od * s
        out[..., 1::2] = ev * s + od * c
        return out.to(x.dtype)

    return rot(q), rot(k)


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up, fused elementwise."""
    g = gate.float()
    return (g * torch.sigmoid(g) * up.float()).to(gate.dtype)


def attn_prefill(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    scale: Optional[float] = None,
    causal: bool = True,
    kv_offset: int = 0,
) -> torch.Tensor:
    """Full prefill attention with GQA.

    q: [tq, n_heads, hd]; k, v: [tk, n_kv_heads, hd]. Causal mask aligns
    query row i with absolute position kv_offset + i (for chunked prefill
    tk >= tq and earlier keys are always visible).
    Returns [tq, n_heads, hd].
    """
    tq, h, hd = q.shape
    tk, kh, _ = k.shape
    if scale is None:
        scale = 1.0 / math.sqrt(hd)
    group = h // kh
    qf = q.float().permute(1, 0, 2)  # [h, tq, hd]
    kf = k.float().permute(1, 0, 2)  # [kh, tk, hd]
    vf = v.float().permute(1, 0, 2)
    kf = kf.repeat_interleave(group, dim=0)  # [h, tk, hd]
    vf = vf.repeat_interleave(group, dim=0)
    scores = torch.bmm(qf, kf.transpose(1, 2)) * scale  # [h, tq, tk]
    if causal:
        qpos = torch.arange(tq, device=q.device).unsqueeze(1) + kv_offset
        kpos = torch.arange(tk, device=q.device).unsqueeze(0)
        mask = kpos > qpos  # future keys
        scores = scores.masked_fill(mask.unsqueeze(0), float("-inf"))
    p = torch.softmax(scores, dim=-1)
    out = torch.bmm(p, vf)  # [h, tq, hd]
    return out.permute(1, 0, 2).contiguous().to(q.dtype)


def attn_decode_paged(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    page_table: torch.Tensor,
    seq_len: int,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Single-token decode attention over a paged KV cache.

    q: [n_heads, hd]; k_cache/v_cache: [n_pages, page_size, n_kv_heads, hd];
    page_table: int32 [n_used_pages] mapping logical page -> physical page.
    Attends to positions [0, seq_len). Returns [n_heads, hd].
    """
    h, hd = q.shape
    npg, ps, kh, _ = k_cache.shape
    if scale is None:
        scale = 1.0 / math.sqrt(hd)
    n_used = (seq_len + ps - 1) // ps
    phys = page_table[:n_used].long()
    k = k_cache[phys].reshape(n_used * ps, kh, hd)[:seq_len]  # [t, kh, hd]
    v = v_cache[phys].reshape(n_used * ps, kh, hd)[:seq_len]
    group = h // kh
    qf = q.float()  # [h, hd]
    kf = k.float().repeat_interleave(group, dim=1).permute(1, 0, 2)  # [h, t, hd]
    vf = v.float().repeat_interleave(group, dim=1).permute(1, 0, 2)
    scores = torch.einsum("hd,htd->ht", qf, kf) * scale
    p = torch.softmax(scores, dim=-1)
    out = torch.einsum("ht,htd->hd", p, vf)
    return out.to(q.dtype)


def kv_write(
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    page_table: torch.Tensor,
    pos0: int,
    k: torch.Tensor,
    v: torch.Tensor,
) -> None:
    """Scatter t new K/V rows into the paged cache at positions pos0..pos0+t-1."""
    npg, ps, kh, hd = k_cache.shape
    t = k.shape[0]
    pos = torch.arange(pos0, pos0 + t, device=k.device)
    pages = page_table[(pos // ps).long()].long()
    flat = pages * ps + (pos % ps)
    k_cache.view(npg * ps, kh, hd)[flat] = k.to(k_cache.dtype)
    v_cache.view(npg * ps, kh, hd)[flat] = v.to(v_cache.dtype)


def sample(
    logits: torch.Tensor,
    temperature: float = 0.7,
    top_p: float = 1.0,
    generator: Optional[torch.Generator] = None,
) -> int:
    """Temperature + nucleus sampling over a [vocab] logits row."""
    lf = logits.float()
    if temperature <= 0.0:
        return int(lf.argmax().item())
    probs = torch.softmax(lf / temperature, dim=-1)
    if top_p < 1.0:
        sp, idx = probs.sort(descending=True)
        cum = sp.cumsum(-1)
        keep = cum - sp < top_p  # keep tokens whose prefix-before is < top_p
        sp = sp * keep
        sp = sp / sp.sum()
        choice = torch.multinomial(sp, 1, generator=generator)
        return int(idx[choice].item())
    choice = torch.multinomial(probs, 1, generator=generator)
    return int(choice.item())
