"""Pure-PyTorch reference implementations of every engine op.

These define the *semantics* each HIP kernel must reproduce:
  - used directly on CPU (tests, tiny models, gloo multi-process tests)
  - GPU numerics tests compare the HIP kernels against these in fp32
    (tests/test_ops_gpu.py)

Shapes use B=batch, S=sequence, H/Hq/Hkv=heads, D=head_dim, C=hidden.
KV caches are contiguous ``[B, Hkv, max_seq, D]``.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


# -- normalisation -----------------------------------------------------------

def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    """RMSNorm over the last dim; accumulate in fp32, output in x.dtype."""
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(x.dtype)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5
) -> Tuple[torch.Tensor, torch.Tensor]:
    """residual = residual + x (in fp32, rounded back); out = rmsnorm(residual).
    Returns (out, new_residual)."""
    new_res = (residual.float() + x.float()).to(x.dtype)
    return rmsnorm(new_res, weight, eps), new_res


# -- rotary embeddings -------------------------------------------------------

def rope_table(max_seq: int, dim: int, theta: float = 500000.0,
               device=None) -> torch.Tensor:
    """[max_seq, dim/2, 2] fp32 cos/sin table (half-split RoPE: pairs are
    (x[i], x[i + dim/2]) — Llama convention)."""
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, dtype=torch.float32,
                                        device=device) / dim))
    t = torch.arange(max_seq, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv)                      # [max_seq, dim/2]
    return torch.stack([freqs.cos(), freqs.sin()], dim=-1).contiguous()


def apply_rope(x: torch.Tensor, positions: torch.Tensor,
               table: torch.Tensor) -> torch.Tensor:
    """x: [..., S, H, D] or [..., H, D]; positions: [...,(S)] int.
    Half-split pairing: (x[:d/2], x[d/2:])."""
    D = x.shape[-1]
    cs = table[positions]                            # [..., D/2, 2]
    cos, sin = cs[..., 0], cs[..., 1]                # [..., D/2]
    cos = cos.unsqueeze(-2)                          # broadcast over heads
    sin = sin.unsqueeze(-2)
    x1 = x[..., : D // 2].float()
    x2 = x[..., D // 2:].float()
    o1 = x1 * cos - x2 * sin
    o2 = x2 * cos + x1 * sin
    return torch.cat([o1, o2], dim=-1).to(x.dtype)


def rope_kv_decode(
    q: torch.Tensor,            # [B, Hq, D] (modified logically; returns new)
    k: torch.Tensor,            # [B, Hkv, D]
    v: torch.Tensor,            # [B, Hkv, D]
    k_cache: torch.Tensor,      # [B, Hkv, max_seq, D]
    v_cache: torch.Tensor,
    pos: torch.Tensor,          # [B] int32 — write position (= current length)
    table: torch.Tensor,
) -> torch.Tensor:
    """Apply RoPE to q,k at pos; write k,v into the caches. Returns roped q."""
    q_out = apply_rope(q, pos, table)                # positions broadcast per batch
    k_rot = apply_rope(k, pos, table)
    B = q.shape[0]
    for b in range(B):
        p = int(pos[b])
        k_cache[b, :, p, :] = k_rot[b]
        v_cache[b, :, p, :] = v[b]
    return q_out


def rope_kv_prefill(
    q: torch.Tensor,            # [B, S, Hq, D]
    k: torch.Tensor,            # [B, S, Hkv, D]
    v: torch.Tensor,            # [B, S, Hkv, D]
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    pos0: torch.Tensor,         # [B] int32 — first absolute position
    table: torch.Tensor,
) -> torch.Tensor:
    B, S = q.shape[0], q.shape[1]
    positions = pos0.view(B, 1).long() + torch.arange(S, device=q.device).view(1, S)
    q_out = apply_rope(q, positions, table)
    k_rot = apply_rope(k, positions, table)
    for b in range(B):
        p = int(pos0[b])
        k_cache[b, :, p:p + S, :] = k_rot[b].transpose(0, 1)
        v_cache[b, :, p:p + S, :] = v[b].transpose(0, 1)
    return q_out


# -- attention ---------------------------------------------------------------

def attn_decode(
    q: torch.Tensor,            # [B, Hq, D] (already roped)
    k_cache: torch.Tensor,      # [B, Hkv, max_seq, D]
    v_cache: torch.Tensor,
    seqlen: torch.Tensor,       # [B] int32 — number of valid cache entries
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Single-token GQA attention over the cache. Returns [B, Hq, D]."""
    B, Hq, D = q.shape
    Hkv = k_cache.shape[1]
    G = Hq // Hkv
    scale = scale or 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(B):
        n = int(seqlen[b])
        k = k_cache[b, :, :n, :].float()             # [Hkv, n, D]
        v = v_cache[b, :, :n, :].float()
        qb = q[b].float().view(Hkv, G, D)            # [Hkv, G, D]
        s = torch.einsum("hgd,hnd->hgn", qb, k) * scale
        p = torch.softmax(s, dim=-1)
        o = torch.einsum("hgn,hnd->hgd", p, v)
        out[b] = o.reshape(Hq, D).to(q.dtype)
    return out


def attn_prefill(
    q: torch.Tensor,            # [B, S, Hq, D] (already roped)
    k_cache: torch.Tensor,      # [B, Hkv, max_seq, D] (k/v already appended)
    v_cache: torch.Tensor,
    pos0: torch.Tensor,         # [B] int32 — q token s sits at pos0+s
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Causal GQA prefill attention over the cache. q token s attends cache
    positions [0, pos0+s]. Returns [B, S, Hq, D]."""
    B, S, Hq, D = q.shape
    Hkv = k_cache.shape[1]
    G = Hq // Hkv
    scale = scale or 1.0 / math.sqrt(D)
    out = torch.empty_like(q)
    for b in range(B):
        p0 = int(pos0[b])
        n = p0 + S
        k = k_cache[b, :, :n, :].float()             # [Hkv, n, D]
        v = v_cache[b, :, :n, :].float()
        qb = q[b].float().view(S, Hkv, G, D)
        s = torch.einsum("shgd,hnd->hgsn", qb, k) * scale
        key_pos = torch.arange(n, device=q.device).view(1, 1, 1, n)
        q_pos = (p0 + torch.arange(S, device=q.device)).view(1, 1, S, 1)
        s = s.masked_fill(key_pos > q_pos, float("-inf"))
        p = torch.softmax(s, dim=-1)
        o = torch.einsum("hgsn,hnd->shgd", p, v)
        out[b] = o.reshape(S, Hq, D).to(q.dtype)
    return out


# -- encoder ops (BERT/bge architecture) --------------------------------------

def layernorm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
              eps: float = 1e-5,
              residual: Optional[torch.Tensor] = None) -> torch.Tensor:
    """(x [+ residual]) -> LayerNorm over the last dim; f32 stats."""
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
    mean = xf.mean(dim=-1, keepdim=True)
    var = (xf - mean).pow(2).mean(dim=-1, keepdim=True)
    out = (xf - mean) * torch.rsqrt(var + eps) * weight.float() + bias.float()
    return out.to(x.dtype)


def gelu(x: torch.Tensor) -> torch.Tensor:
    """Exact (erf) GELU."""
    return torch.nn.functional.gelu(x.float()).to(x.dtype)


# -- MLP ---------------------------------------------------------------------

def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    """gate_up: [..., 2*inter] (gate first half, up second). silu(g)*u."""
    inter = gate_up.shape[-1] // 2
    g = gate_up[..., :inter].float()
    u = gate_up[..., inter:].float()
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


# -- sampling ----------------------------------------------------------------

def argmax_sample(logits: torch.Tensor) -> torch.Tensor:
    """Greedy: [B, V] -> [B] int32."""
    return logits.float().argmax(dim=-1).to(torch.int32)


def gumbel_sample(logits: torch.Tensor, temperature: float,
                  generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Exact categorical sampling via the Gumbel-max trick:
    argmax(logits/T + G) with G ~ Gumbel(0,1)."""
    if temperature <= 0:
        return argmax_sample(logits)
    u = torch.rand(logits.shape, device=logits.device, dtype=torch.float32,
                   generator=generator).clamp_(1e-10, 1.0)
    g = -torch.log(-torch.log(u))
    return (logits.float() / temperature + g).argmax(dim=-1).to(torch.int32)


def topk_mask(logits: torch.Tensor, k: int) -> torch.Tensor:
    """Mask all but the top-k logits to -inf."""
    if k <= 0 or k >= logits.shape[-1]:
        return logits
    kth = torch.topk(logits.float(), k, dim=-1).values[..., -1:]
    return logits.masked_fill(logits.float() < kth, float("-inf"))


def topp_mask(logits: torch.Tensor, p: float) -> torch.Tensor:
    """Nucleus mask: keep the smallest prefix of sorted probs with mass >= p."""
    if p >= 1.0:
        return logits
    sorted_logits, idx = torch.sort(logits.float(), descending=True, dim=-1)
    probs = torch.softmax(sorted_logits, dim=-1)
    cum = probs.cumsum(dim=-1)
    keep = cum - probs < p                           # always keep the first
    masked = sorted_logits.masked_fill(~keep, float("-inf"))
    out = torch.full_like(masked, float("-inf"))
    out.scatter_(-1, idx, masked)
    return out.to(logits.dtype)


# -- fp8 weight quantization (OCP e4m3fn; decode-path serving mode) ----------

def quant_fp8(w: torch.Tensor):
    """Per-output-row absmax/448 scaling + e4m3fn encode.
    Returns (w8 uint8 [N,K], scales fp32 [N])."""
    wf = w.float()
    amax = wf.abs().amax(dim=-1).clamp_min(1e-12)
    scales = amax / 448.0
    q = (wf / scales.unsqueeze(-1)).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scales


def dequant_fp8(w8: torch.Tensor, scales: torch.Tensor) -> torch.Tensor:
    return w8.view(torch.float8_e4m3fn).float() * scales.unsqueeze(-1).float()


def hash_gumbel(B: int, v_local: int, v_offset: int, seed: int,
                step: int) -> torch.Tensor:
    """CPU replica of the GPU sampler's per-element Gumbel noise
    (fei_common.h hash_uniform: splitmix64 -> 24-bit uniform in (0,1]),
    keyed by (seed, step, batch row, GLOBAL vocab index) — so the
    tensor-parallel shard sampler matches the full sampler bit-for-bit
    in its noise stream."""
    import numpy as np

    idx = np.arange(v_offset, v_offset + v_local, dtype=np.uint64)
    out = np.empty((B, v_local), dtype=np.float32)
    with np.errstate(over="ignore"):
        for b in range(B):
            x = (np.uint64(seed)
                 ^ (np.uint64(step) * np.uint64(0x51ED27F1))
                 ^ (np.uint64(b) << np.uint64(40))
                 ^ idx)
            x = x + np.uint64(0x9E3779B97F4A7C15)
            x = (x ^ (x >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
            x = (x ^ (x >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
            x = x ^ (x >> np.uint64(31))
            m = (x >> np.uint64(40)).astype(np.float32)
            out[b] = (m + 1.0) * (1.0 / 16777216.0)
    u = torch.from_numpy(out)
    return -torch.log(-torch.log(u))
