"""Tensor parallelism: head-sharded attention + column/row-parallel MLP.

MI355X mapping: TP=8 across the xGMI hive for the
Llama-3-70B opponent. Sharding plan:

  - attention is sharded BY HEADS (Ulysses-style): each rank owns
    n_heads/tp query heads and n_kv_heads/tp KV heads, so RoPE, the KV
    cache and the attention kernels stay entirely GPU-local — no ring
    attention is needed at 32k when 288 GB HBM3E holds the full local KV;
  - wqkv is column-parallel (per-head column blocks), wo row-parallel with
    one RCCL all-reduce per layer;
  - gate/up are column-parallel, down row-parallel with the second
    all-reduce;
  - embeddings, norms, and lm_head are replicated (1 GB at 70B — cheap in
    288 GB, and it keeps decode lockstep: every rank samples the same token
    from identical logits, so the decode loop needs no extra collectives).

Ring all-reduce over xGMI is per-link bound (7 p2p links x ~153 GB/s);
messages here are [t, dim] bf16 activations — at 70B/32k prefill ~0.5 GB
per layer boundary, solidly bandwidth-bound, so RCCL's multi-ring default
is the right transport (no custom collective).
"""

from __future__ import annotations

import dataclasses
from typing import Optional

import torch
import torch.distributed as dist

from ..models.config import LlamaConfig


def shard_config(config: LlamaConfig, tp: int) -> LlamaConfig:
    """Per-rank local config: heads and FFN shrink by tp; dim/vocab stay."""
    if tp == 1:
        return config
    if config.n_heads % tp or config.n_kv_heads % tp or config.ffn_dim % tp:
        raise ValueError(
            f"{config.name}: heads ({config.n_heads}/{config.n_kv_heads}) and "
            f"ffn ({config.ffn_dim}) must divide tp={tp}"
        )
    return dataclasses.replace(
        config,
        n_heads=config.n_heads // tp,
        n_kv_heads=config.n_kv_heads // tp,
        ffn_dim=config.ffn_dim // tp,
        head_dim_override=config.head_dim,
    )


def shard_qkv(wqkv: torch.Tensor, config: LlamaConfig, tp: int, rank: int) -> torch.Tensor:
    """Output-shard the fused [(h+2kh)*hd, d] QKV weight by heads
    (weights are row-major [out, in])."""
    h, kh, hd = config.n_heads, config.n_kv_heads, config.head_dim
    hl, khl = h // tp, kh // tp
    q = wqkv[: h * hd]
    k = wqkv[h * hd : (h + kh) * hd]
    v = wqkv[(h + kh) * hd :]
    return torch.cat(
        [
            q[rank * hl * hd : (rank + 1) * hl * hd],
            k[rank * khl * hd : (rank + 1) * khl * hd],
            v[rank * khl * hd : (rank + 1) * khl * hd],
        ],
        dim=0,
    ).contiguous()


def shard_o(wo: torch.Tensor, config: LlamaConfig, tp: int, rank: int) -> torch.Tensor:
    """Input-shard [d, h*hd] by this rank's head block (row-parallel: the
    partial outputs are summed by the all-reduce)."""
    h, hd = config.n_heads, config.head_dim
    hl = h // tp
    return wo[:, rank * hl * hd : (rank + 1) * hl * hd].contiguous()


def shard_gate_up(w: torch.Tensor, config: LlamaConfig, tp: int, rank: int) -> torch.Tensor:
    """Output-shard the fused [2*ffn, d] gate/up weight per half."""
    f = config.ffn_dim
    fl = f // tp
    gate = w[:f][rank * fl : (rank + 1) * fl]
    up = w[f:][rank * fl : (rank + 1) * fl]
    return torch.cat([gate, up], dim=0).contiguous()


def shard_down(w: torch.Tensor, config: LlamaConfig, tp: int, rank: int) -> torch.Tensor:
    """Input-shard [d, ffn] (row-parallel)."""
    f = config.ffn_dim
    fl = f // tp
    return w[:, rank * fl : (rank + 1) * fl].contiguous()


class TPContext:
    """Carries the process group + rank geometry into the model."""

    def __init__(self, tp_size: int, tp_rank: int,
                 group: Optional[dist.ProcessGroup] = None) -> None:
        self.size = tp_size
        self.rank = tp_rank
        self.group = group

    def all_reduce_(self, x: torch.Tensor) -> torch.Tensor:
        """In-place sum all-reduce of partial activations (RCCL on GPU,
        gloo in CPU tests). Gloo lacks bf16 and CUDA buffers: round-trip
        through fp32 on host there (the gloo path exists only so TP
        numerics are testable without N GPUs — RCCL needs one device per
        rank, so GPU-compute-with-gloo-collectives is the 1-GPU TP lane)."""
        if self.size == 1:
            return x
        backend = dist.get_backend(self.group)
        if backend == "gloo" and (x.dtype == torch.bfloat16 or x.is_cuda):
            xf = x.float().cpu()
            dist.all_reduce(xf, group=self.group)
            x.copy_(xf.to(x.dtype).to(x.device))
        else:
            dist.all_reduce(x, group=self.group)
        return x

    def any_flag(self, flag: bool) -> bool:
        """Logical OR of a host flag across the TP group (MAX reduce on a
        1-element tensor): coordinates decode stop decisions so no rank
        exits a collective loop early. CPU tensor on gloo, device tensor
        on nccl."""
        if self.size == 1:
            return flag
        backend = dist.get_backend(self.group)
        dev = "cpu" if backend == "gloo" else "cuda"
        t = torch.tensor([1 if flag else 0], dtype=torch.int32, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX, group=self.group)
        return bool(t.item())

    @classmethod
    def from_default_group(cls) -> "TPContext":
        return cls(dist.get_world_size(), dist.get_rank(), None)
