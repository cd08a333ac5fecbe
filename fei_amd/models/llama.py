"""Llama-3 style decoder built on fei_amd.ops.

Not an nn.Module: weights are plain tensors, the forward passes are
allocation-light functions designed around the MI355X execution model —
fused HIP kernels for norm/rope/attention/activation/sampling, rocBLAS
(hipBLASLt) for the plain GEMMs, one hipGraph-capturable decode step.

Tensor parallel (TP): column-parallel QKV and gate/up, row-parallel O and
down with one all-reduce each per layer; lm_head column-parallel over the
vocab with an all-gather of logits.
"""

from __future__ import annotations

import math
import os
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.nn.functional as F

from fei_amd import ops
from fei_amd.engine.config import ModelSpec
from fei_amd.ops import reference as ref
from fei_amd.parallel.pg import ParallelContext, all_gather_cat, all_reduce_sum


@dataclass
class LayerWeights:
    norm_attn: torch.Tensor       # [C]
    wqkv: torch.Tensor            # [(Hq_l + 2*Hkv_l)*D, C]
    wo: torch.Tensor              # [C, Hq_l*D]
    norm_mlp: torch.Tensor        # [C]
    wgu: torch.Tensor             # [2*I_l, C]   (gate rows then up rows)
    wdown: torch.Tensor           # [C, I_l]


class LlamaModel:
    def __init__(
        self,
        spec: ModelSpec,
        device: torch.device,
        dtype: torch.dtype = torch.bfloat16,
        tp: Optional[ParallelContext] = None,
        seed: int = 1234,
        max_seq_len: Optional[int] = None,
    ):
        self.spec = spec
        self.device = device
        self.dtype = dtype
        self.tp = tp or ParallelContext()
        self.max_seq_len = max_seq_len or spec.max_seq_len
        tp_size = self.tp.world_size if self.tp.is_distributed else 1
        if spec.num_kv_heads % tp_size or spec.num_heads % tp_size or \
                spec.intermediate_size % tp_size or spec.vocab_size % tp_size:
            raise ValueError(f"model {spec.name} not divisible by tp={tp_size}")
        self.tp_size = tp_size
        self.tp_rank = self.tp.rank if self.tp.is_distributed else 0
        self.hq_l = spec.num_heads // tp_size
        self.hkv_l = spec.num_kv_heads // tp_size
        self.inter_l = spec.intermediate_size // tp_size
        self.vocab_l = spec.vocab_size // tp_size
        self.D = spec.head_dim
        self._init_weights(seed)
        self.rope = ref.rope_table(self.max_seq_len, self.D, spec.rope_theta,
                                   device=device)
        # FEI_PREFETCH=<MB>: during each decode layer's attention window
        # (low HBM traffic) a side-stream kernel pre-reads the first MB of
        # that layer's gate/up weights into L2/L3 so the big GEMV partially
        # hits cache (experiment; 0 = off)
        self.prefetch_mb = int(os.environ.get("FEI_PREFETCH", "0") or 0)
        self._pf_stream = None
        self._pf_sink = None
        # FEI_NORM_PRECOMP=1: precompute each norm's sumsq in the producing
        # residual epilogue. Measured NEGATIVE on MI355X (242.5 vs 247.5
        # tok/s, 8B decode): the norm prologue overlaps the previous
        # kernel's wave drain and is effectively free, while the epilogue
        # atomics + block barrier + per-step zeroing cost ~84 us/step.
        # Kept opt-in for re-measurement on future silicon.
        self.norm_precomp = os.environ.get("FEI_NORM_PRECOMP", "0") == "1"

    # -- weights -------------------------------------------------------------

    def _rand(self, gen, *shape, scale: float) -> torch.Tensor:
        w = torch.randn(*shape, generator=gen, device=self.device,
                        dtype=torch.float32) * scale
        return w.to(self.dtype)

    def _init_weights(self, seed: int) -> None:
        """Random init (synthetic weights — BASELINE.json: no network for
        checkpoints). Each TP rank seeds identically and slices its shard so
        ranks are consistent without communication."""
        s = self.spec
        gen = torch.Generator(device=self.device)
        gen.manual_seed(seed)
        C = s.hidden_size
        sc = 1.0 / math.sqrt(C)
        # embedding: replicated
        self.emb = self._rand(gen, s.vocab_size, C, scale=0.02)
        self.layers: List[LayerWeights] = []
        r = self.tp_rank
        for _ in range(s.num_layers):
            wq = self._rand(gen, s.num_heads * self.D, C, scale=sc)
            wk = self._rand(gen, s.num_kv_heads * self.D, C, scale=sc)
            wv = self._rand(gen, s.num_kv_heads * self.D, C, scale=sc)
            wo = self._rand(gen, C, s.num_heads * self.D, scale=sc / math.sqrt(2 * s.num_layers))
            wg = self._rand(gen, s.intermediate_size, C, scale=sc)
            wu = self._rand(gen, s.intermediate_size, C, scale=sc)
            wd = self._rand(gen, C, s.intermediate_size,
                            scale=1.0 / math.sqrt(s.intermediate_size) / math.sqrt(2 * s.num_layers))
            # shard
            wq = wq.view(s.num_heads, self.D, C)[r * self.hq_l:(r + 1) * self.hq_l].reshape(-1, C)
            wk = wk.view(s.num_kv_heads, self.D, C)[r * self.hkv_l:(r + 1) * self.hkv_l].reshape(-1, C)
            wv = wv.view(s.num_kv_heads, self.D, C)[r * self.hkv_l:(r + 1) * self.hkv_l].reshape(-1, C)
            wo = wo.view(C, s.num_heads, self.D)[:, r * self.hq_l:(r + 1) * self.hq_l].reshape(C, -1)
            wg = wg[r * self.inter_l:(r + 1) * self.inter_l]
            wu = wu[r * self.inter_l:(r + 1) * self.inter_l]
            wd = wd[:, r * self.inter_l:(r + 1) * self.inter_l]
            self.layers.append(LayerWeights(
                norm_attn=torch.ones(C, device=self.device, dtype=self.dtype),
                wqkv=torch.cat([wq, wk, wv], dim=0).contiguous(),
                wo=wo.contiguous(),
                norm_mlp=torch.ones(C, device=self.device, dtype=self.dtype),
                wgu=torch.cat([wg, wu], dim=0).contiguous(),
                wdown=wd.contiguous(),
            ))
        self.norm_f = torch.ones(C, device=self.device, dtype=self.dtype)
        lm = self._rand(gen, s.vocab_size, C, scale=sc)
        self.lm_head = lm[r * self.vocab_l:(r + 1) * self.vocab_l].contiguous()
        self.vocab_offset = r * self.vocab_l   # first GLOBAL vocab id of
                                               # this rank's lm_head shard

    def quantize_fp8(self) -> None:
        """Quantize the DECODE-path weights to e4m3fn (per-row scales).
        bf16 weights stay for prefill; the fused-chain decode then streams
        half the bytes per token (serving mode; bench label fp8-weight)."""
        self.fp8 = []
        for lw in self.layers:
            self.fp8.append({
                "wqkv": ops.quant_fp8(lw.wqkv),
                "wo": ops.quant_fp8(lw.wo),
                "wgu": ops.quant_fp8(lw.wgu),
                "wdown": ops.quant_fp8(lw.wdown),
            })
        self.fp8_lm_head = ops.quant_fp8(self.lm_head)

    def param_bytes(self) -> int:
        total = self.emb.numel() + self.norm_f.numel() + self.lm_head.numel()
        for lw in self.layers:
            for t in (lw.norm_attn, lw.wqkv, lw.wo, lw.norm_mlp, lw.wgu, lw.wdown):
                total += t.numel()
        return total * self.emb.element_size()

    # -- kv cache ------------------------------------------------------------

    def new_kv_cache(self, batch: int, max_seq: Optional[int] = None):
        max_seq = max_seq or self.max_seq_len
        shape = (batch, self.hkv_l, max_seq, self.D)
        return (
            [torch.zeros(shape, device=self.device, dtype=self.dtype)
             for _ in range(self.spec.num_layers)],
            [torch.zeros(shape, device=self.device, dtype=self.dtype)
             for _ in range(self.spec.num_layers)],
        )

    # -- qkv views -----------------------------------------------------------

    def _qkv_views(self, qkv: torch.Tensor, B: int, S: Optional[int] = None):
        """Strided views into the fused qkv output (no copies; the HIP
        kernels take batch/token strides)."""
        D = self.D
        W = qkv.shape[-1]
        off_k = self.hq_l * D
        off_v = off_k + self.hkv_l * D
        if S is None:
            q = qkv.as_strided((B, self.hq_l, D), (W, D, 1))
            k = qkv.as_strided((B, self.hkv_l, D), (W, D, 1), storage_offset=qkv.storage_offset() + off_k)
            v = qkv.as_strided((B, self.hkv_l, D), (W, D, 1), storage_offset=qkv.storage_offset() + off_v)
        else:
            q = qkv.as_strided((B, S, self.hq_l, D), (S * W, W, D, 1))
            k = qkv.as_strided((B, S, self.hkv_l, D), (S * W, W, D, 1), storage_offset=qkv.storage_offset() + off_k)
            v = qkv.as_strided((B, S, self.hkv_l, D), (S * W, W, D, 1), storage_offset=qkv.storage_offset() + off_v)
        return q, k, v

    # -- forward passes ------------------------------------------------------

    def forward_decode(
        self,
        token: torch.Tensor,          # [B] int32/int64
        pos: torch.Tensor,            # [B] int32 (current length; kv written here)
        k_caches: List[torch.Tensor],
        v_caches: List[torch.Tensor],
        attn_splits: int = 32,
        workspace=None,
        fused_attn: bool = False,
        attn_out: Optional[torch.Tensor] = None,
        fused_norm: bool = False,
        gather_logits: bool = True,
    ) -> torch.Tensor:
        """One decode step -> logits [B, vocab] (gathered across TP;
        ``gather_logits=False`` returns THIS rank's [B, vocab/W] shard —
        the TP hot loop samples on shards and gathers 8 bytes/seq instead,
        engine._decode_step_tp).
        ``fused_norm`` (tp=1 only): the residual stream stays inside the
        GEMV kernels — norm-prologue QKV/gate-up, residual-epilogue O/down;
        5 kernels per layer instead of 9."""
        # the fused-norm chain's per-wave norm prologue scales with B
        # (recomputes all B rows' sumsq per wave): batch 1 only — B=8
        # measured 572 vs 1130 tok/s on the plain path (docs/BENCHMARKS.md).
        # The r02 byte-form of that gate (B*C*2 <= 8 KB) also excluded
        # hidden > 4096 at B=1, silently dropping 70B fp8 decode to the
        # bf16 plain chain (measured 40.4 == bf16 vs r01's 41.3 with fp8
        # engaged); the batch gate is B==1 with a separate LDS-budget cap.
        if fused_norm and self.tp_size == 1 and token.shape[0] == 1 and \
                self.spec.hidden_size * 2 <= 32 * 1024:
            return self._forward_decode_fused_norm(
                token, pos, k_caches, v_caches, attn_splits, workspace,
                fused_attn, attn_out)
        s = self.spec
        B = token.shape[0]
        h = F.embedding(token.long(), self.emb)          # residual stream
        scale = 1.0 / math.sqrt(self.D)
        n_layers = len(self.layers)
        x = ops.rmsnorm(h, self.layers[0].norm_attn, s.norm_eps)
        for li, lw in enumerate(self.layers):
            qkv = ops.linear_decode(x, lw.wqkv)
            q, k, v = self._qkv_views(qkv, B)
            if fused_attn:
                att = ops.attn_decode_fused(q, k, v, k_caches[li],
                                            v_caches[li], pos, self.rope,
                                            scale=scale, out=attn_out)
            else:
                # split-K attention with fused in-kernel RoPE + KV-append
                att = ops.attn_decode(q, k_caches[li], v_caches[li], pos,
                                      splits=attn_splits, scale=scale,
                                      workspace=workspace, out=attn_out,
                                      k=k, v=v, table=self.rope, layer=li)
            o = ops.linear_decode(att.reshape(B, -1), lw.wo)
            all_reduce_sum(o, self.tp)
            x, h = ops.fused_add_rmsnorm(o, h, lw.norm_mlp, s.norm_eps)
            act = ops.gemv_swiglu(x, lw.wgu)
            d = ops.linear_decode(act, lw.wdown)
            all_reduce_sum(d, self.tp)
            next_norm = (self.layers[li + 1].norm_attn if li + 1 < n_layers
                         else self.norm_f)
            x, h = ops.fused_add_rmsnorm(d, h, next_norm, s.norm_eps)
        logits = ops.linear_decode(x, self.lm_head)
        if gather_logits:
            logits = all_gather_cat(logits, dim=-1, ctx=self.tp)
        return logits

    def forward_decode_stream(self, token, pos, k_caches, v_caches,
                              ws: dict, bufs) -> torch.Tensor:
        """One decode step through the persistent weight-streaming layer
        engine (ops.stream_layer: ONE launch per layer instead of six;
        csrc/stream_layer.hip). Batch 1, tp=1, bf16 only — the engine
        guards with ops.stream_layer_check before enabling. ``bufs`` is a
        persistent pair of [1, C] bf16 residual buffers (stable addresses
        for hipGraph capture)."""
        s = self.spec
        h = F.embedding(token.long(), self.emb)
        bufs[0].copy_(h.to(bufs[0].dtype))
        scale = 1.0 / math.sqrt(self.D)
        cur, nxt = bufs
        for li, lw in enumerate(self.layers):
            ops.stream_layer(cur, nxt, lw, s, k_caches[li], v_caches[li],
                             self.rope, pos, ws, li, scale)
            cur, nxt = nxt, cur
        return ops.gemv_norm(cur, self.norm_f, self.lm_head, s.norm_eps)

    def _ssq_slots(self, B: int, device) -> torch.Tensor:
        """[2L, B] f32 sum-of-squares slots for the norm-precompute chain:
        slot 2i  = residual after layer i's O projection (feeds the MLP norm),
        slot 2i+1 = residual after layer i's down projection (feeds layer
        i+1's attention norm, or the final norm). Zeroed every step."""
        buf = getattr(self, "_ssq", None)
        if buf is None or buf.shape[1] != B or buf.device != device:
            buf = torch.zeros(2 * len(self.layers), B, dtype=torch.float32,
                              device=device)
            self._ssq = buf
        return buf

    def _forward_decode_fused_norm(self, token, pos, k_caches, v_caches,
                                   attn_splits, workspace, fused_attn,
                                   attn_out):
        s = self.spec
        B = token.shape[0]
        fp8 = getattr(self, "fp8", None)
        # fp8 norm-GEMVs stage B*C bf16 activations in LDS and recompute
        # all B norms per wave — measured slower than bf16 already at B=2
        # (386 vs 410 tok/s) and 3.5x slower at B=8. fp8 pays for B=1
        # agent decode only (any hidden size whose row fits the 32 KB
        # dynamic-LDS budget — the old 8 KB byte gate wrongly excluded
        # 70B's hidden=8192 at B=1).
        if fp8 is not None and (B != 1 or s.hidden_size * 2 > 32 * 1024):
            fp8 = None
        h = F.embedding(token.long(), self.emb).contiguous()
        scale = 1.0 / math.sqrt(self.D)
        # norm-precompute chain (opt-in FEI_NORM_PRECOMP=1; measured loss,
        # see __init__): each residual GEMV's epilogue accumulates the next
        # norm's sumsq so the norm prologues collapse to one scalar read.
        # The layer-0 attention norm has no producing epilogue (embedding)
        # and keeps the full prologue.
        ssq = (self._ssq_slots(B, h.device)
               if (self.norm_precomp and fp8 is None and h.is_cuda) else None)
        if ssq is not None:
            ssq.zero_()
        pf = self.prefetch_mb > 0 and fp8 is None and h.is_cuda
        if pf and self._pf_stream is None:
            self._pf_stream = torch.cuda.Stream()
            self._pf_sink = torch.zeros(1, dtype=torch.float32,
                                        device=h.device)
        for li, lw in enumerate(self.layers):
            if fp8 is not None:
                q8 = fp8[li]
                qkv = ops.gemv_norm_fp8(h, lw.norm_attn, *q8["wqkv"],
                                        s.norm_eps)
            else:
                qkv = ops.gemv_norm(
                    h, lw.norm_attn, lw.wqkv, s.norm_eps,
                    ssq=ssq[2 * li - 1] if (ssq is not None and li > 0) else None)
            if pf:
                # warm L2/L3 with this layer's gate/up weights while the
                # attention chain (low HBM traffic) runs on the main stream
                ev = torch.cuda.Event()
                ev.record()
                with torch.cuda.stream(self._pf_stream):
                    self._pf_stream.wait_event(ev)
                    ops.prefetch(lw.wgu, self.prefetch_mb << 20,
                                 self._pf_sink)
            q, k, v = self._qkv_views(qkv, B)
            if fused_attn:
                att = ops.attn_decode_fused(q, k, v, k_caches[li],
                                            v_caches[li], pos, self.rope,
                                            scale=scale, out=attn_out)
            else:
                att = ops.attn_decode(q, k_caches[li], v_caches[li], pos,
                                      splits=attn_splits, scale=scale,
                                      workspace=workspace, out=attn_out,
                                      k=k, v=v, table=self.rope, layer=li)
            if fp8 is not None:
                ops.gemv_res_fp8(att.reshape(B, -1), *q8["wo"], h)
                act = ops.gemv_swiglu_norm_fp8(h, lw.norm_mlp, *q8["wgu"],
                                               s.norm_eps)
                ops.gemv_res_fp8(act, *q8["wdown"], h)
            else:
                ops.gemv_res(att.reshape(B, -1), lw.wo, h,
                             ssq_out=ssq[2 * li] if ssq is not None else None)
                act = ops.gemv_swiglu_norm(
                    h, lw.norm_mlp, lw.wgu, s.norm_eps,
                    ssq=ssq[2 * li] if ssq is not None else None)
                ops.gemv_res(act, lw.wdown, h,
                             ssq_out=ssq[2 * li + 1] if ssq is not None else None)
        if pf:
            # rejoin the prefetch stream so graph capture sees all work
            # ordered (pure cache warming: no data dependency otherwise)
            torch.cuda.current_stream().wait_stream(self._pf_stream)
        if fp8 is not None:
            return ops.gemv_norm_fp8(h, self.norm_f, *self.fp8_lm_head,
                                     s.norm_eps)
        return ops.gemv_norm(h, self.norm_f, self.lm_head, s.norm_eps,
                             ssq=ssq[-1] if ssq is not None else None)

    def forward_prefill(
        self,
        tokens: torch.Tensor,         # [B, S]
        pos0: torch.Tensor,           # [B] int32 (first absolute position)
        k_caches: List[torch.Tensor],
        v_caches: List[torch.Tensor],
        all_positions: bool = False,
    ) -> torch.Tensor:
        """Prefill S tokens -> logits of the LAST position [B, vocab]
        (or of every position [B, S, vocab] when ``all_positions`` — the
        ragged-batch path gathers each row's own last real token)."""
        s = self.spec
        B, S = tokens.shape
        h = F.embedding(tokens.long(), self.emb)          # [B,S,C] residual
        scale = 1.0 / math.sqrt(self.D)
        n_layers = len(self.layers)
        x = ops.rmsnorm(h, self.layers[0].norm_attn, s.norm_eps)
        for li, lw in enumerate(self.layers):
            qkv = F.linear(x, lw.wqkv)                    # [B,S,W]
            qkv2 = qkv.view(B * S, -1)
            q, k, v = self._qkv_views(qkv2, B, S)
            ops.rope_kv_prefill(q, k, v, k_caches[li], v_caches[li], pos0, self.rope)
            att = ops.attn_prefill(q, k_caches[li], v_caches[li], pos0, scale=scale)
            o = F.linear(att.reshape(B, S, -1), lw.wo)
            all_reduce_sum(o, self.tp)
            x, h = ops.fused_add_rmsnorm(o, h, lw.norm_mlp, s.norm_eps)
            gu = F.linear(x, lw.wgu)
            act = ops.swiglu(gu)
            d = F.linear(act, lw.wdown)
            all_reduce_sum(d, self.tp)
            next_norm = (self.layers[li + 1].norm_attn if li + 1 < n_layers
                         else self.norm_f)
            x, h = ops.fused_add_rmsnorm(d, h, next_norm, s.norm_eps)
        last = x if all_positions else x[:, -1, :]
        logits = F.linear(last, self.lm_head)
        logits = all_gather_cat(logits, dim=-1, ctx=self.tp)
        return logits

    # -- checkpointing (safetensors) -----------------------------------------

    def state_dict_full(self) -> Dict[str, torch.Tensor]:
        """Unsharded state dict (tp=1 only): HF-free flat naming."""
        if self.tp_size != 1:
            raise RuntimeError("state_dict_full requires tp=1 (save from an "
                               "unsharded model; any-tp models can LOAD it)")
        s = self.spec
        out: Dict[str, torch.Tensor] = {
            "emb": self.emb, "norm_f": self.norm_f, "lm_head": self.lm_head,
        }
        for i, lw in enumerate(self.layers):
            qd = s.num_heads * self.D
            kvd = s.num_kv_heads * self.D
            out[f"layers.{i}.wq"] = lw.wqkv[:qd]
            out[f"layers.{i}.wk"] = lw.wqkv[qd:qd + kvd]
            out[f"layers.{i}.wv"] = lw.wqkv[qd + kvd:]
            out[f"layers.{i}.wo"] = lw.wo
            out[f"layers.{i}.wg"] = lw.wgu[:s.intermediate_size]
            out[f"layers.{i}.wu"] = lw.wgu[s.intermediate_size:]
            out[f"layers.{i}.wd"] = lw.wdown
            out[f"layers.{i}.norm_attn"] = lw.norm_attn
            out[f"layers.{i}.norm_mlp"] = lw.norm_mlp
        return out

    def save_weights(self, path: str) -> None:
        """Save an unsharded checkpoint (tp=1) as safetensors + spec json."""
        import json as _json
        import os as _os
        from dataclasses import asdict
        from safetensors.torch import save_file

        _os.makedirs(path, exist_ok=True)
        tensors = {k: v.contiguous().cpu() for k, v in self.state_dict_full().items()}
        save_file(tensors, _os.path.join(path, "model.safetensors"))
        with open(_os.path.join(path, "config.json"), "w") as f:
            _json.dump(asdict(self.spec), f, indent=2)

    def load_weights(self, path: str) -> None:
        """Load an unsharded checkpoint, slicing this rank's TP shard
        (safetensors -> per-rank shards)."""
        import os as _os
        from safetensors.torch import load_file

        s = self.spec
        full = load_file(_os.path.join(path, "model.safetensors"))
        r = self.tp_rank
        dev, dt = self.device, self.dtype

        def to_dev(t):
            return t.to(device=dev, dtype=dt)

        self.emb = to_dev(full["emb"])
        self.norm_f = to_dev(full["norm_f"])
        lm = full["lm_head"]
        self.lm_head = to_dev(lm[r * self.vocab_l:(r + 1) * self.vocab_l]).contiguous()
        C = s.hidden_size
        for i, lw in enumerate(self.layers):
            wq = full[f"layers.{i}.wq"].view(s.num_heads, self.D, C)
            wk = full[f"layers.{i}.wk"].view(s.num_kv_heads, self.D, C)
            wv = full[f"layers.{i}.wv"].view(s.num_kv_heads, self.D, C)
            wq = wq[r * self.hq_l:(r + 1) * self.hq_l].reshape(-1, C)
            wk = wk[r * self.hkv_l:(r + 1) * self.hkv_l].reshape(-1, C)
            wv = wv[r * self.hkv_l:(r + 1) * self.hkv_l].reshape(-1, C)
            lw.wqkv = to_dev(torch.cat([wq, wk, wv], dim=0)).contiguous()
            wo = full[f"layers.{i}.wo"].view(C, s.num_heads, self.D)
            lw.wo = to_dev(wo[:, r * self.hq_l:(r + 1) * self.hq_l].reshape(C, -1)).contiguous()
            wg = full[f"layers.{i}.wg"][r * self.inter_l:(r + 1) * self.inter_l]
            wu = full[f"layers.{i}.wu"][r * self.inter_l:(r + 1) * self.inter_l]
            lw.wgu = to_dev(torch.cat([wg, wu], dim=0)).contiguous()
            lw.wdown = to_dev(full[f"layers.{i}.wd"][:, r * self.inter_l:(r + 1) * self.inter_l]).contiguous()
            lw.norm_attn = to_dev(full[f"layers.{i}.norm_attn"])
            lw.norm_mlp = to_dev(full[f"layers.{i}.norm_mlp"])
