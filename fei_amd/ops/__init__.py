"""fei_amd.ops — engine ops with a single dispatch rule:

  - CUDA (= ROCm/HIP) tensors -> the in-tree gfx950 kernel library,
    loaded via ctypes. If the library is missing on a GPU machine, ops
    FAIL LOUDLY — there is no silent eager fallback on GPU.
  - CPU tensors -> the fp32-accurate torch reference implementations
    (fei_amd/ops/reference.py), used by CPU tests and tiny models.

All kernels launch on torch's current CUDA stream, so the decode step can
be captured into a hipGraph (torch.cuda.CUDAGraph) — see engine/engine.py.
"""

from __future__ import annotations

import ctypes
import math
import os
from typing import Optional, Tuple

import torch

from fei_amd.ops import reference as ref

_LIB: Optional[ctypes.CDLL] = None
_LIB_ERR: Optional[str] = None
_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                         "libfei_kernels.so")

_c = ctypes
_vp, _i, _f, _u64, _l = _c.c_void_p, _c.c_int, _c.c_float, _c.c_ulonglong, _c.c_long


def _try_load() -> None:
    global _LIB, _LIB_ERR
    if _LIB is not None or _LIB_ERR is not None:
        return
    try:
        lib = ctypes.CDLL(_LIB_PATH)
    except OSError as e:
        _LIB_ERR = str(e)
        return
    lib.fei_rmsnorm.argtypes = [_vp, _vp, _vp, _i, _i, _f, _vp]
    lib.fei_add_rmsnorm.argtypes = [_vp, _vp, _vp, _vp, _i, _i, _f, _vp]
    lib.fei_rope_kv_decode.argtypes = [_vp, _vp, _vp, _vp, _vp, _vp, _vp,
                                       _i, _i, _i, _i, _i, _l, _l, _vp]
    lib.fei_rope_kv_prefill.argtypes = [_vp, _vp, _vp, _vp, _vp, _vp, _vp,
                                        _i, _i, _i, _i, _i, _i, _l, _l, _vp]
    lib.fei_attn_decode.argtypes = [_vp, _vp, _vp, _vp, _vp, _vp,
                                    _i, _i, _i, _i, _i, _i, _f, _l,
                                    _vp, _vp, _vp, _l, _vp, _vp, _i, _vp]
    lib.fei_attn_decode_combine.argtypes = [_vp, _vp, _vp, _i, _i, _i, _i, _vp]
    lib.fei_swiglu.argtypes = [_vp, _vp, _l, _i, _vp]
    lib.fei_sample.argtypes = [_vp, _vp, _vp, _vp, _vp, _i, _i, _i, _f, _u64,
                               _i, _vp]
    lib.fei_advance.argtypes = [_vp, _vp, _i, _i, _vp]
    lib.fei_attn_prefill.argtypes = [_vp, _vp, _vp, _vp, _vp, _vp,
                                     _i, _i, _i, _i, _i, _i, _f, _i, _l, _vp]
    lib.fei_mfma_probe.argtypes = [_vp, _vp, _vp, _vp]
    lib.fei_gemv.argtypes = [_vp, _vp, _vp, _i, _i, _i, _i, _vp]
    lib.fei_gemm_m8.argtypes = [_vp, _vp, _vp, _i, _i, _i, _i, _vp]
    lib.fei_attn_decode_fused.argtypes = [_vp, _vp, _vp, _vp, _vp, _vp, _vp,
                                          _vp, _i, _i, _i, _i, _i, _f,
                                          _l, _l, _vp]
    lib.fei_sample_onepass.argtypes = [_vp, _vp, _vp, _vp, _i, _i, _f, _u64,
                                       _i, _vp]
    lib.fei_sample_shard.argtypes = [_vp, _vp, _vp, _i, _i, _i, _f, _u64, _vp]
    lib.fei_stream_layer_check.argtypes = [_i, _i, _i, _i, _i]
    lib.fei_stream_layer_check.restype = _i
    lib.fei_stream_layer.argtypes = [_vp, _vp, _vp, _vp, _vp, _vp, _vp, _vp,
                                     _vp, _vp, _vp, _vp, _vp, _vp, _vp, _vp,
                                     _vp, _vp, _vp, _i, _i, _i, _i, _i,
                                     _i, _i, _f, _f, _vp]
    lib.fei_attn_decode_paged.argtypes = [_vp, _vp, _vp, _vp, _vp, _vp, _vp,
                                          _i, _i, _i, _i, _i, _i, _i, _f,
                                          _l, _vp, _vp, _vp, _l, _vp]
    lib.fei_add_layernorm.argtypes = [_vp, _vp, _vp, _vp, _vp, _i, _i, _f,
                                      _i, _vp]
    lib.fei_gelu.argtypes = [_vp, _vp, _l, _vp]
    lib.fei_gemv_norm_fp8.argtypes = [_vp, _vp, _vp, _vp, _vp, _i, _i, _i,
                                      _f, _vp]
    lib.fei_gemv_res_fp8.argtypes = [_vp, _vp, _vp, _vp, _i, _i, _i, _vp]
    lib.fei_gemv_swiglu_norm_fp8.argtypes = [_vp, _vp, _vp, _vp, _vp, _i, _i,
                                             _i, _f, _vp]
    lib.fei_quant_fp8_rows.argtypes = [_vp, _vp, _vp, _i, _i, _vp]
    lib.fei_gemv_swiglu.argtypes = [_vp, _vp, _vp, _i, _i, _i, _i, _vp]
    lib.fei_prefetch.argtypes = [_vp, _l, _vp, _i, _vp]
    lib.fei_gemv_res.argtypes = [_vp, _vp, _vp, _i, _i, _i, _i, _vp, _vp]
    lib.fei_gemv_norm.argtypes = [_vp, _vp, _vp, _vp, _i, _i, _i, _f, _i, _vp,
                                  _vp]
    lib.fei_gemv_swiglu_norm.argtypes = [_vp, _vp, _vp, _vp, _i, _i, _i, _f,
                                         _i, _vp, _vp]
    _LIB = lib


def kernels_available() -> bool:
    _try_load()
    return _LIB is not None


def require_lib() -> ctypes.CDLL:
    _try_load()
    if _LIB is None:
        raise RuntimeError(
            "fei_amd HIP kernel library not found at "
            f"{_LIB_PATH} ({_LIB_ERR}). Build it with "
            "`python -m fei_amd.ops.build` — GPU execution refuses to fall "
            "back to eager PyTorch.")
    return _LIB


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _ptr(t: torch.Tensor) -> int:
    return t.data_ptr()


# ---------------------------------------------------------------------------


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-5,
            out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """RMSNorm over the last dim. x: [..., C] bf16 (GPU) / any float (CPU)."""
    if not x.is_cuda:
        return ref.rmsnorm(x, weight, eps)
    lib = require_lib()
    x2 = x.contiguous()
    if out is None:
        out = torch.empty_like(x2)
    rows = x2.numel() // x2.shape[-1]
    lib.fei_rmsnorm(_ptr(out), _ptr(x2), _ptr(weight), rows, x2.shape[-1],
                    eps, _stream())
    return out


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float = 1e-5,
                      out: Optional[torch.Tensor] = None
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """residual += x (in place on GPU); out = rmsnorm(residual)."""
    if not x.is_cuda:
        o, res = ref.fused_add_rmsnorm(x, residual, weight, eps)
        residual.copy_(res)
        return o, residual
    lib = require_lib()
    x2 = x.contiguous()
    if out is None:
        out = torch.empty_like(x2)
    rows = x2.numel() // x2.shape[-1]
    lib.fei_add_rmsnorm(_ptr(out), _ptr(residual), _ptr(x2), _ptr(weight),
                        rows, x2.shape[-1], eps, _stream())
    return out, residual


def rope_kv_decode(q, k, v, k_cache, v_cache, pos, table) -> torch.Tensor:
    """In-place RoPE on q; k roped + v appended into caches at pos[b].
    q [B,Hq,D], k/v [B,Hkv,D], caches [B,Hkv,max_seq,D], pos [B] int32."""
    if not q.is_cuda:
        q_out = ref.rope_kv_decode(q, k, v, k_cache, v_cache, pos, table)
        q.copy_(q_out)
        return q
    lib = require_lib()
    B, Hq, D = q.shape
    Hkv = k.shape[1]
    assert q.stride(2) == 1 and q.stride(1) == D, "head rows must be contiguous"
    assert k.stride(1) == D and v.stride(1) == D
    assert k.stride(0) == v.stride(0)
    lib.fei_rope_kv_decode(_ptr(q), _ptr(k), _ptr(v), _ptr(k_cache),
                           _ptr(v_cache), _ptr(table), _ptr(pos),
                           B, Hq, Hkv, D, k_cache.shape[2],
                           q.stride(0), k.stride(0), _stream())
    return q


def rope_kv_prefill(q, k, v, k_cache, v_cache, pos0, table) -> torch.Tensor:
    """In-place RoPE on q [B,S,Hq,D]; k roped + v appended at pos0[b]+s."""
    if not q.is_cuda:
        q_out = ref.rope_kv_prefill(q, k, v, k_cache, v_cache, pos0, table)
        q.copy_(q_out)
        return q
    lib = require_lib()
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    assert q.stride(3) == 1 and q.stride(2) == D
    assert q.stride(0) == S * q.stride(1), "token rows must be uniform"
    assert k.stride(2) == D and k.stride(0) == S * k.stride(1)
    assert k.stride(1) == v.stride(1)
    lib.fei_rope_kv_prefill(_ptr(q), _ptr(k), _ptr(v), _ptr(k_cache),
                            _ptr(v_cache), _ptr(table), _ptr(pos0),
                            B, S, Hq, Hkv, D, k_cache.shape[2],
                            q.stride(1), k.stride(1), _stream())
    return q


def attn_decode(q, k_cache, v_cache, pos, splits: int = 32,
                scale: Optional[float] = None,
                workspace: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
                out: Optional[torch.Tensor] = None,
                k: Optional[torch.Tensor] = None,
                v: Optional[torch.Tensor] = None,
                table: Optional[torch.Tensor] = None,
                layer: int = 0) -> torch.Tensor:
    """Single-token GQA attention over n = pos[b]+1 cache entries.
    q [B,Hq,D] -> out [B,Hq,D]. The length is read on DEVICE (hipGraph).
    When (k, v, table) are given, the kernel also fuses the step's RoPE +
    KV-append: q is the RAW qkv view and cache[pos] is written in-kernel."""
    B, Hq, D = q.shape
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    if not q.is_cuda:
        if table is not None:
            rope_kv_decode(q, k, v, k_cache, v_cache, pos, table)
        seqlen = pos + 1
        return ref.attn_decode(q, k_cache, v_cache, seqlen, scale)
    lib = require_lib()
    arrive = None
    if workspace is None:
        part_o = torch.empty(B, Hq, splits, D, dtype=torch.float32, device=q.device)
        part_ml = torch.empty(B, Hq, splits, 2, dtype=torch.float32, device=q.device)
    else:
        part_o, part_ml = workspace[0], workspace[1]
        # a 3rd workspace element (zeroed int32 [B*Hkv*splits] per-split
        # done-flag buffer) selects the FUSED combine: split 0 of each
        # (b, hkv) pair polls the flags and reduces the partials in-kernel,
        # and the separate combine launch is skipped (FEI_FUSED_CMB=0 opts
        # out for A/B; pass `layer` so the flag tags stay unique across the
        # layers of one step).
        if len(workspace) >= 3 and _FUSED_CMB:
            arrive = workspace[2]
        assert part_o.shape[2] == splits and part_ml.shape[2] == splits, \
            f"workspace sized for {part_o.shape[2]} splits, kernel asked {splits}"
    if out is None:
        out = torch.empty_like(q)
    Hkv = k_cache.shape[1]
    assert q.stride(2) == 1 and q.stride(1) == D
    # the kernel's V lane maps tile 256 threads by D/2 and D/8 — both must
    # divide the block (holds for the llama head dims 64/128; D=96 would
    # silently under-cover a tile)
    assert D in (64, 128), f"decode attention supports D in (64,128), got {D}"
    assert splits <= 64, "combine kernel stages at most 64 split partials"
    if table is not None:
        assert k is not None and v is not None
        assert k.stride(1) == D and k.stride(0) == v.stride(0)
        kin, vin, cs, kv_bs = _ptr(k), _ptr(v), _ptr(table), k.stride(0)
    else:
        kin = vin = cs = None
        kv_bs = 0
    if arrive is not None:
        assert arrive.dtype == torch.int32 and \
            arrive.numel() >= B * Hkv * splits
    lib.fei_attn_decode(_ptr(q), _ptr(k_cache), _ptr(v_cache), _ptr(part_o),
                        _ptr(part_ml), _ptr(pos), B, Hq, Hkv, D,
                        k_cache.shape[2], splits, scale, q.stride(0),
                        kin, vin, cs, kv_bs,
                        _ptr(out) if arrive is not None else None,
                        _ptr(arrive) if arrive is not None else None,
                        layer, _stream())
    if arrive is None:
        lib.fei_attn_decode_combine(_ptr(out), _ptr(part_o), _ptr(part_ml),
                                    B, Hq, D, splits, _stream())
    return out


def attn_decode_fused(q, k, v, k_cache, v_cache, pos, table,
                      scale: Optional[float] = None,
                      out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Fused decode attention: RoPE(q,k) + KV-append + GQA attention in one
    kernel (grid Hkv x B, no split partials). For agent-length sequences;
    long-context uses rope_kv_decode + attn_decode (split-K)."""
    B, Hq, D = q.shape
    Hkv = k.shape[1]
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    if not q.is_cuda:
        rope_kv_decode(q, k, v, k_cache, v_cache, pos, table)
        return ref.attn_decode(q, k_cache, v_cache, pos + 1, scale)
    lib = require_lib()
    assert q.stride(2) == 1 and q.stride(1) == D
    assert k.stride(1) == D and k.stride(0) == v.stride(0)
    if out is None:
        out = torch.empty(B, Hq, D, dtype=torch.bfloat16, device=q.device)
    lib.fei_attn_decode_fused(_ptr(q), _ptr(k), _ptr(v), _ptr(k_cache),
                              _ptr(v_cache), _ptr(out), _ptr(table), _ptr(pos),
                              B, Hq, Hkv, D, k_cache.shape[2], scale,
                              q.stride(0), k.stride(0), _stream())
    return out


def attn_prefill(q, k_cache, v_cache, pos0, scale: Optional[float] = None,
                 causal: bool = True, kv_len: Optional[torch.Tensor] = None,
                 out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Causal (or bidirectional) GQA attention of q [B,S,Hq,D] over the
    caches. q token s attends [0, pos0[b]+s] when causal, else [0, kv_len[b])."""
    B, S, Hq, D = q.shape
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    if not q.is_cuda:
        if causal:
            return ref.attn_prefill(q, k_cache, v_cache, pos0, scale)
        # bidirectional reference: softmax over [0, kv_len)
        return _ref_bidir(q, k_cache, v_cache, kv_len, scale)
    lib = require_lib()
    if out is None:
        out = torch.empty_like(q)
    if kv_len is None:
        kv_len = pos0  # unused when causal
    Hkv = k_cache.shape[1]
    assert q.stride(3) == 1 and q.stride(2) == D
    assert q.stride(0) == S * q.stride(1)
    assert out.is_contiguous()
    lib.fei_attn_prefill(_ptr(q), _ptr(k_cache), _ptr(v_cache), _ptr(out),
                         _ptr(pos0), _ptr(kv_len), B, S, Hq, Hkv, D,
                         k_cache.shape[2], scale, 1 if causal else 0,
                         q.stride(1), _stream())
    return out


def _ref_bidir(q, k_cache, v_cache, kv_len, scale):
    B, S, Hq, D = q.shape
    Hkv = k_cache.shape[1]
    G = Hq // Hkv
    out = torch.empty_like(q)
    for b in range(B):
        n = int(kv_len[b])
        k = k_cache[b, :, :n, :].float()
        v = v_cache[b, :, :n, :].float()
        qb = q[b].float().view(S, Hkv, G, D)
        s = torch.einsum("shgd,hnd->hgsn", qb, k) * scale
        p = torch.softmax(s, dim=-1)
        o = torch.einsum("hgsn,hnd->shgd", p, v)
        out[b] = o.reshape(S, Hq, D).to(q.dtype)
    return out


def swiglu(gate_up: torch.Tensor,
           out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """gate_up [..., 2*inter] -> [..., inter]: silu(gate) * up."""
    if not gate_up.is_cuda:
        return ref.swiglu(gate_up)
    lib = require_lib()
    gu = gate_up.contiguous()
    inter = gu.shape[-1] // 2
    rows = gu.numel() // gu.shape[-1]
    if out is None:
        out = torch.empty(*gu.shape[:-1], inter, dtype=gu.dtype, device=gu.device)
    lib.fei_swiglu(_ptr(out), _ptr(gu), rows, inter, _stream())
    return out


def sample(logits: torch.Tensor, token: torch.Tensor,
           step: torch.Tensor, workspace: torch.Tensor,
           out_tokens: Optional[torch.Tensor] = None,
           temperature: float = 0.0, seed: int = 0,
           nchunks: int = 64) -> torch.Tensor:
    """Greedy (T<=0) or Gumbel-max categorical (T>0) sampling.
    logits [B,V] bf16 -> token [B] int32 (written in place; also appended to
    out_tokens[b, step] when given). All state on device (hipGraph-safe)."""
    B, V = logits.shape
    if not logits.is_cuda:
        if temperature <= 0:
            t = ref.argmax_sample(logits)
        else:
            # deterministic by (seed, step) like the GPU kernel's splitmix
            # hash (not bit-identical, but the same reproducibility
            # contract on both paths)
            g = torch.Generator().manual_seed(
                (int(seed) * 1000003 + int(step)) & 0x7FFFFFFF)
            t = ref.gumbel_sample(logits, temperature, g)
        token.copy_(t)
        if out_tokens is not None:
            st = int(step)
            for b in range(B):
                if st < out_tokens.shape[1]:
                    out_tokens[b, st] = t[b]
        return token
    lib = require_lib()
    max_new = out_tokens.shape[1] if out_tokens is not None else 0
    out_ptr = _ptr(out_tokens) if out_tokens is not None else None
    if B <= 8:
        lib.fei_sample_onepass(_ptr(logits), _ptr(token), out_ptr, _ptr(step),
                               B, V, temperature, seed, max_new, _stream())
    else:
        lib.fei_sample(_ptr(logits), _ptr(token), out_ptr, _ptr(step),
                       _ptr(workspace), B, V, nchunks, temperature, seed,
                       max_new, _stream())
    return token


def sample_shard(logits: torch.Tensor, step: torch.Tensor,
                 v_offset: int, out: torch.Tensor,
                 temperature: float = 0.0, seed: int = 0) -> torch.Tensor:
    """Tensor-parallel shard sampler: reduce THIS rank's logits shard
    [B, V_local] (bf16 on GPU) to (best value, best GLOBAL index) per
    sequence, written into ``out`` [B, 2] f32 (index stored as int bits).
    The ranks then all-gather 8 bytes/seq instead of the vocab row.
    Gumbel noise is keyed by the global index (same splitmix hash as the
    full sampler), so shard+combine is bit-identical to sampling the
    gathered logits. CPU path mirrors the hash exactly."""
    B, Vl = logits.shape
    if not logits.is_cuda:
        vals = logits.float()
        if temperature > 0:
            vals = vals / temperature + ref.hash_gumbel(
                B, Vl, v_offset, int(seed), int(step))
        best = vals.max(dim=-1)
        out[:, 0] = best.values
        out[:, 1].view(torch.int32).copy_(
            (best.indices + v_offset).to(torch.int32))
        return out
    lib = require_lib()
    lib.fei_sample_shard(_ptr(logits), _ptr(out), _ptr(step), B, Vl,
                         int(v_offset), float(temperature), int(seed),
                         _stream())
    return out


# -- persistent weight-streaming decode layer (csrc/stream_layer.hip) -------

STREAM_NSPLIT = 32


def stream_layer_check(C: int, Hq: int, Hkv: int, D: int, I: int) -> int:
    """Residency + shape guard for the persistent layer engine. 0 = OK;
    negative = unsupported (caller uses the launch path). GPU only."""
    lib = require_lib()
    return int(lib.fei_stream_layer_check(C, Hq, Hkv, D, I))


def stream_workspace(spec, device) -> dict:
    """Granule buffers (u64 {tag, payload}) + fail word for the stream
    engine. ~1.3 MB for llama3-8b; zeroed by the engine whenever pos can
    move backwards (prefill) — tags are pos-keyed otherwise."""
    C, D = spec.hidden_size, spec.head_dim
    Hq, Hkv, I = spec.num_heads, spec.num_kv_heads, spec.intermediate_size
    mk = lambda n: torch.zeros(n, dtype=torch.int64, device=device)
    return {
        "g_qkv": mk((Hq + 2 * Hkv) * D),
        "g_att": mk(Hq * D // 2),
        "g_h2": mk(C // 2),
        "g_act": mk(I),
        "g_done": mk(6 * 256),        # [stage, NWG] producer-done granules
        "fail": torch.zeros(1, dtype=torch.int32, device=device),
    }


def stream_layer(x_in: torch.Tensor, h_out: torch.Tensor, lw, spec,
                 k_cache: torch.Tensor, v_cache: torch.Tensor,
                 cos_sin: torch.Tensor, pos: torch.Tensor, ws: dict,
                 layer: int, scale: float) -> None:
    """ONE persistent launch = one decode layer at batch 1 (S1 qkv GEMV ->
    attention -> combine -> O GEMV -> SwiGLU GEMV -> down GEMV with the
    LDS-DMA loader streaming weights ahead across every stage edge).
    Replaces six launch-path kernels; see csrc/stream_layer.hip."""
    lib = require_lib()
    lib.fei_stream_layer(
        _ptr(x_in), _ptr(h_out), _ptr(lw.wqkv), _ptr(lw.wo), _ptr(lw.wgu),
        _ptr(lw.wdown), _ptr(lw.norm_attn), _ptr(lw.norm_mlp),
        _ptr(k_cache), _ptr(v_cache), _ptr(cos_sin), _ptr(pos),
        _ptr(ws["g_qkv"]), _ptr(ws["g_att"]),
        _ptr(ws["g_h2"]), _ptr(ws["g_act"]), _ptr(ws["g_done"]),
        _ptr(ws["dbg"]) if "dbg" in ws else None,
        _ptr(ws["fail"]),
        spec.hidden_size, spec.num_heads, spec.num_kv_heads, spec.head_dim,
        spec.intermediate_size, k_cache.shape[-2], layer, spec.norm_eps,
        scale, _stream())


def advance(pos: torch.Tensor, step: torch.Tensor,
            max_pos: int = 1 << 30) -> None:
    """pos[b] = min(pos[b]+1, max_pos); step += 1 (device-side,
    graph-capturable). max_pos guards the RoPE table / KV cache against
    out-of-bounds indexing when a caller decodes at capacity."""
    if not pos.is_cuda:
        pos.clamp_(max=max_pos - 1)
        pos += 1
        step += 1
        return
    lib = require_lib()
    lib.fei_advance(_ptr(pos), _ptr(step), pos.shape[0], max_pos, _stream())


def mfma_probe(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """GPU-only: 16x32 @ 32x16 through one MFMA with the kernels' fragment
    maps; falsifies layout assumptions (tests/test_ops_gpu.py)."""
    lib = require_lib()
    C = torch.empty(16, 16, dtype=torch.float32, device=A.device)
    lib.fei_mfma_probe(_ptr(A.contiguous()), _ptr(B.contiguous()), _ptr(C),
                       _stream())
    return C


# -- decode GEMV dispatch ----------------------------------------------------

# Nontemporal weight loads for ALL decode GEMVs: in the real decode loop
# every weight byte is cache-cold (the 15 GB per-step cycle flushes the
# 256 MB L3), so NT never hurts and avoids polluting L2/L3 for the hot
# activations — measured 257.2 vs 255.6 tok/s on the 8B headline. The
# microbench that suggested a 512 MB threshold was L3-warm across runs.
# FEI_GEMV_NT_MIN (bytes) restores a threshold.
_GEMV_NT_MIN_BYTES = int(os.environ.get("FEI_GEMV_NT_MIN", 0))
# Fused split-K combine in decode attention (per-head reducer splits poll
# sc1 done-flags and reduce the partials in-kernel; needs the 3-element
# workspace). Verified token-identical and measured at PARITY with the
# separate k_attn_decode_combine launch (275.5 vs 275.8 tok/s on the 8B
# chain — in a captured graph the combine launch already overlaps the
# attention grid's drain, so there was no boundary to win; the full
# 4-protocol measurement ladder is profiles/r02_fused_combine.md).
# Opt-in via FEI_FUSED_CMB=1; the launch form stays the default.
_FUSED_CMB = os.environ.get("FEI_FUSED_CMB", "0") == "1"


def _gemm_m8_ok(M: int, N: int, K: int) -> bool:
    # MFMA GEMM tile: A-fragment rows clamp to M-1 so M <= 16 is CORRECT,
    # but at M = 16 hipBLASLt measured faster (serving-16: 2473 vs 2337
    # tok/s) — route only the 4..8 range where the tile wins 1.27-1.65x
    return 4 <= M <= 8 and K % 1024 == 0 and N % 16 == 0


def _gemv_ok(M: int, K: int) -> bool:
    return M in (1, 2, 4, 8) and K % 8 == 0


def linear_decode(x: torch.Tensor, w: torch.Tensor,
                  out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """F.linear for skinny M: hand-rolled streaming GEMV on GPU
    (profiles/r01: rocBLAS reaches ~4 TB/s on M=1; streaming reaches the
    flat-read ceiling). Falls back to torch for unsupported shapes/CPU."""
    M = x.numel() // x.shape[-1]
    K = x.shape[-1]
    N = w.shape[0]
    if not x.is_cuda or not (_gemv_ok(M, K) or _gemm_m8_ok(M, N, K)):
        return torch.nn.functional.linear(x, w)
    lib = require_lib()
    x2 = x.contiguous().view(M, K)
    if out is None:
        out = torch.empty(*x.shape[:-1], N, dtype=x.dtype, device=x.device)
    nt = 1 if (N * K * 2 >= _GEMV_NT_MIN_BYTES) else 0
    if _gemm_m8_ok(M, N, K):
        # MFMA tile: beats the VALU GEMV at every llama batch shape
        # (1.27-1.65x — profiles/r02_batch_attention.md addendum).
        # ALWAYS plain loads: the nt hint on the 16 B/lane staged W
        # fetch measured 6.50 vs 5.01 ms/step at batch 8.
        lib.fei_gemm_m8(_ptr(out), _ptr(x2), _ptr(w), M, N, K, 0,
                        _stream())
        return out
    lib.fei_gemv(_ptr(out), _ptr(x2), _ptr(w), M, N, K, nt, _stream())
    return out


def gemv_swiglu(x: torch.Tensor, wgu: torch.Tensor,
                out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Fused gate/up GEMV + SwiGLU: x [M,K] @ [2I,K]^T -> silu(g)*u [M,I]."""
    M = x.numel() // x.shape[-1]
    K = x.shape[-1]
    I = wgu.shape[0] // 2
    if not x.is_cuda or not (_gemv_ok(M, K) or _gemm_m8_ok(M, 2 * I, K)):
        return swiglu(torch.nn.functional.linear(x, wgu))
    lib = require_lib()
    x2 = x.contiguous().view(M, K)
    if out is None:
        out = torch.empty(*x.shape[:-1], I, dtype=x.dtype, device=x.device)
    nt = 1 if (2 * I * K * 2 >= _GEMV_NT_MIN_BYTES) else 0
    if _gemm_m8_ok(M, 2 * I, K):
        # MFMA over the stacked [2I, K] weight (its [M, 2I] output is
        # k_swiglu's packed input): 68.3 -> ~47 us on the 8B gate/up at
        # batch 8 vs the fused VALU kernel
        gu = torch.empty(M, 2 * I, dtype=x.dtype, device=x.device)
        lib.fei_gemm_m8(_ptr(gu), _ptr(x2), _ptr(wgu), M, 2 * I, K, 0,
                        _stream())
        lib.fei_swiglu(_ptr(out), _ptr(gu), M, I, _stream())
        return out
    lib.fei_gemv_swiglu(_ptr(out), _ptr(x2), _ptr(wgu), M, I, K, nt,
                        _stream())
    return out


def gemv_res(x: torch.Tensor, w: torch.Tensor, res: torch.Tensor,
             ssq_out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """res += x @ w^T (epilogue residual add; in place on res [M,N]).

    ``ssq_out`` [M] f32, pre-zeroed: the kernel also accumulates the
    sum-of-squares of the updated residual rows so a following
    gemv_norm/gemv_swiglu_norm can skip its sumsq prologue pass (pass the
    same tensor as their ``ssq``). Ignored on the CPU fallback — consumers
    recompute exactly there."""
    M = x.numel() // x.shape[-1]
    K = x.shape[-1]
    N = w.shape[0]
    if not x.is_cuda or not _gemv_ok(M, K):
        lin = torch.nn.functional.linear(x, w)
        res.copy_((res.float() + lin.float().view_as(res)).to(res.dtype))
        return res
    lib = require_lib()
    x2 = x.contiguous().view(M, K)
    nt = 1 if (N * K * 2 >= _GEMV_NT_MIN_BYTES) else 0
    lib.fei_gemv_res(_ptr(res), _ptr(x2), _ptr(w), M, N, K, nt,
                     _ptr(ssq_out) if ssq_out is not None else None,
                     _stream())
    return res


def gemv_norm(res: torch.Tensor, wnorm: torch.Tensor, w: torch.Tensor,
              eps: float = 1e-5,
              out: Optional[torch.Tensor] = None,
              ssq: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out = rmsnorm(res)*wnorm @ w^T (norm-prologue GEMV).

    ``ssq`` [M] f32: precomputed sum-of-squares of res rows (from a prior
    gemv_res ``ssq_out``); skips the prologue pass over res."""
    M = res.numel() // res.shape[-1]
    K = res.shape[-1]
    N = w.shape[0]
    if not res.is_cuda or not _gemv_ok(M, K):
        return torch.nn.functional.linear(rmsnorm(res, wnorm, eps), w)
    lib = require_lib()
    r2 = res.contiguous().view(M, K)
    if out is None:
        out = torch.empty(*res.shape[:-1], N, dtype=res.dtype, device=res.device)
    nt = 1 if (N * K * 2 >= _GEMV_NT_MIN_BYTES) else 0
    lib.fei_gemv_norm(_ptr(out), _ptr(r2), _ptr(wnorm), _ptr(w), M, N, K,
                      eps, nt, _ptr(ssq) if ssq is not None else None,
                      _stream())
    return out


def gemv_swiglu_norm(res: torch.Tensor, wnorm: torch.Tensor,
                     wgu: torch.Tensor, eps: float = 1e-5,
                     out: Optional[torch.Tensor] = None,
                     ssq: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out = silu(g)*u where [g;u] = rmsnorm(res)*wnorm @ wgu^T.
    ``ssq``: see gemv_norm."""
    M = res.numel() // res.shape[-1]
    K = res.shape[-1]
    I = wgu.shape[0] // 2
    if not res.is_cuda or not _gemv_ok(M, K):
        return swiglu(torch.nn.functional.linear(rmsnorm(res, wnorm, eps), wgu))
    lib = require_lib()
    r2 = res.contiguous().view(M, K)
    if out is None:
        out = torch.empty(*res.shape[:-1], I, dtype=res.dtype, device=res.device)
    nt = 1 if (wgu.shape[0] * K * 2 >= _GEMV_NT_MIN_BYTES) else 0
    lib.fei_gemv_swiglu_norm(_ptr(out), _ptr(r2), _ptr(wnorm), _ptr(wgu),
                             M, I, K, eps, nt,
                             _ptr(ssq) if ssq is not None else None,
                             _stream())
    return out


def prefetch(w: torch.Tensor, max_bytes: int, sink: torch.Tensor,
             n_blocks: int = 512, stream=None) -> None:
    """Stream the first ``max_bytes`` of ``w`` through cache-filling loads
    (L2/L3 prefill for a later kernel's reads). GPU-only no-op helper for
    the decode prefetch experiment; launched on a side stream."""
    if not w.is_cuda:
        return
    lib = require_lib()
    bytes_ = min(max_bytes, w.numel() * w.element_size())
    sp = stream if stream is not None else _stream()
    lib.fei_prefetch(_ptr(w), bytes_, _ptr(sink), n_blocks, sp)


def attn_decode_paged(q, k_pool, v_pool, block_table, pos, splits: int = 32,
                      scale: Optional[float] = None,
                      workspace: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
                      out: Optional[torch.Tensor] = None,
                      k: Optional[torch.Tensor] = None,
                      v: Optional[torch.Tensor] = None,
                      table: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Decode attention over a PAGED KV pool (engine/kv_cache.PagedKVPool):
    pools [num_blocks, Hkv, BS, D], block_table [B, max_blocks] int32 maps
    logical key blocks to physical pool blocks. n = pos[b]+1 keys.
    When (k, v, table) are given the kernel also fuses the step's RoPE +
    paged KV-append: q is the RAW qkv view and the pool row for position
    pos[b] is written in-kernel (mirrors attn_decode's fused form)."""
    B, Hq, D = q.shape
    n_blocks, Hkv, BS, _ = k_pool.shape
    assert BS & (BS - 1) == 0, "block size must be a power of two"
    bs_log = BS.bit_length() - 1
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    if not q.is_cuda:
        if table is not None:
            # reference path: rope q/k and append into the POOL rows
            q = ref.apply_rope(q, pos.long(), table).to(q.dtype)
            k_r = ref.apply_rope(k, pos.long(), table).to(k.dtype)
            for b in range(B):
                p = int(pos[b])
                blk = int(block_table[b, p >> bs_log])
                k_pool[blk, :, p & (BS - 1), :] = k_r[b]
                v_pool[blk, :, p & (BS - 1), :] = v[b]
        # gather logical order into a contiguous cache, then reference
        max_len = block_table.shape[1] * BS
        kc = torch.zeros(B, Hkv, max_len, D, dtype=k_pool.dtype)
        vc = torch.zeros_like(kc)
        for b in range(B):
            for j, blk in enumerate(block_table[b].tolist()):
                if blk < 0:
                    continue
                kc[b, :, j * BS:(j + 1) * BS] = k_pool[blk]
                vc[b, :, j * BS:(j + 1) * BS] = v_pool[blk]
        return ref.attn_decode(q, kc, vc, pos + 1, scale)
    lib = require_lib()
    if workspace is None:
        part_o = torch.empty(B, Hq, splits, D, dtype=torch.float32, device=q.device)
        part_ml = torch.empty(B, Hq, splits, 2, dtype=torch.float32, device=q.device)
    else:
        part_o, part_ml = workspace
        assert part_o.shape[2] == splits and part_ml.shape[2] == splits
    if out is None:
        out = torch.empty_like(q)
    assert q.stride(2) == 1 and q.stride(1) == D
    assert D in (64, 128), f"paged attention supports D in (64,128), got {D}"
    assert splits <= 64
    if table is not None:
        assert k is not None and v is not None
        assert k.stride(1) == D and k.stride(0) == v.stride(0)
        kin, vin, cs, kv_bs = _ptr(k), _ptr(v), _ptr(table), k.stride(0)
    else:
        kin = vin = cs = None
        kv_bs = 0
    lib.fei_attn_decode_paged(_ptr(q), _ptr(k_pool), _ptr(v_pool),
                              _ptr(block_table), _ptr(part_o), _ptr(part_ml),
                              _ptr(pos), B, Hq, Hkv, D, bs_log,
                              block_table.shape[1], splits, scale,
                              q.stride(0), kin, vin, cs, kv_bs, _stream())
    lib.fei_attn_decode_combine(_ptr(out), _ptr(part_o), _ptr(part_ml),
                                B, Hq, D, splits, _stream())
    return out


def layernorm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
              eps: float = 1e-5, residual: Optional[torch.Tensor] = None,
              out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """LayerNorm of (x [+ residual]) over the last dim (encoder blocks)."""
    if not x.is_cuda:
        return ref.layernorm(x, weight, bias, eps, residual)
    lib = require_lib()
    x2 = x.contiguous()
    rows = x2.numel() // x2.shape[-1]
    if out is None:
        out = torch.empty_like(x2)
    res_ptr = _ptr(residual.contiguous()) if residual is not None else _ptr(x2)
    lib.fei_add_layernorm(_ptr(out), _ptr(x2), res_ptr, _ptr(weight),
                          _ptr(bias), rows, x2.shape[-1], eps,
                          1 if residual is not None else 0, _stream())
    return out


def gelu(x: torch.Tensor, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    if not x.is_cuda:
        return ref.gelu(x)
    lib = require_lib()
    x2 = x.contiguous()
    if out is None:
        out = torch.empty_like(x2)
    lib.fei_gelu(_ptr(out), _ptr(x2), x2.numel(), _stream())
    return out


# -- fp8 weight-quantized decode path (OCP e4m3fn; serving mode) -------------

def quant_fp8(w: torch.Tensor):
    """Quantize a weight matrix [N,K] to e4m3fn + per-row fp32 scales."""
    if not w.is_cuda:
        return ref.quant_fp8(w)
    lib = require_lib()
    N, K = w.shape
    w8 = torch.empty(N, K, dtype=torch.uint8, device=w.device)
    scales = torch.empty(N, dtype=torch.float32, device=w.device)
    lib.fei_quant_fp8_rows(_ptr(w8), _ptr(scales), _ptr(w.contiguous()),
                           N, K, _stream())
    return w8, scales


def gemv_norm_fp8(res, wnorm, w8, wscale, eps: float = 1e-5,
                  out: Optional[torch.Tensor] = None) -> torch.Tensor:
    M = res.numel() // res.shape[-1]
    K = res.shape[-1]
    N = w8.shape[0]
    if not res.is_cuda:
        w = ref.dequant_fp8(w8, wscale).to(res.dtype)
        return torch.nn.functional.linear(rmsnorm(res, wnorm, eps), w)
    assert M * K * 2 <= 64 * 1024, \
        "fp8 norm-GEMV stages M*K bf16 activations in LDS (<=64 KB)"
    lib = require_lib()
    r2 = res.contiguous().view(M, K)
    if out is None:
        out = torch.empty(*res.shape[:-1], N, dtype=res.dtype, device=res.device)
    lib.fei_gemv_norm_fp8(_ptr(out), _ptr(r2), _ptr(wnorm), _ptr(w8),
                          _ptr(wscale), M, N, K, eps, _stream())
    return out


def gemv_res_fp8(x, w8, wscale, res) -> torch.Tensor:
    M = x.numel() // x.shape[-1]
    K = x.shape[-1]
    N = w8.shape[0]
    if not x.is_cuda:
        w = ref.dequant_fp8(w8, wscale).to(x.dtype)
        lin = torch.nn.functional.linear(x, w)
        res.copy_((res.float() + lin.float().view_as(res)).to(res.dtype))
        return res
    lib = require_lib()
    x2 = x.contiguous().view(M, K)
    lib.fei_gemv_res_fp8(_ptr(res), _ptr(x2), _ptr(w8), _ptr(wscale),
                         M, N, K, _stream())
    return res


def gemv_swiglu_norm_fp8(res, wnorm, w8, wscale, eps: float = 1e-5,
                         out: Optional[torch.Tensor] = None) -> torch.Tensor:
    M = res.numel() // res.shape[-1]
    K = res.shape[-1]
    I = w8.shape[0] // 2
    if not res.is_cuda:
        w = ref.dequant_fp8(w8, wscale).to(res.dtype)
        return swiglu(torch.nn.functional.linear(rmsnorm(res, wnorm, eps), w))
    assert M * K * 2 <= 64 * 1024
    lib = require_lib()
    r2 = res.contiguous().view(M, K)
    if out is None:
        out = torch.empty(*res.shape[:-1], I, dtype=res.dtype, device=res.device)
    lib.fei_gemv_swiglu_norm_fp8(_ptr(out), _ptr(r2), _ptr(wnorm), _ptr(w8),
                                 _ptr(wscale), M, I, K, eps, _stream())
    return out
