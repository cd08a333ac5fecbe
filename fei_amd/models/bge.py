"""Embedding encoder for memdir semantic search — bge-base architecture
(BERT-style: 12 layers, 768 hidden, 12 heads, post-LayerNorm residual
blocks, GELU MLP, biased linears).

The compute path is the point (BASELINE.json configs[3]: "bge-base
embeddings on MFMA"): GEMMs on hipBLASLt/MFMA, bidirectional attention on
the HIP flash kernel (causal=0), fused (residual-add+)LayerNorm and GELU
HIP kernels. Weights are random-init — there is no network for
checkpoints — but the architecture matches bge-base so real weights could
be loaded via the same tensors.

Output: mean-pooled last hidden state, L2-normalised — [B, hidden].
"""

from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.nn.functional as F

from fei_amd import ops
from fei_amd.engine.config import ModelSpec, get_spec


class BgeEncoder:
    def __init__(self, spec: Optional[ModelSpec] = None,
                 device: Optional[torch.device] = None,
                 dtype: Optional[torch.dtype] = None, seed: int = 42,
                 max_seq_len: int = 512):
        self.spec = spec or get_spec("bge-base")
        if device is None:
            device = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
        self.device = device
        if dtype is None:
            dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
        self.dtype = dtype
        self.max_seq_len = min(max_seq_len, self.spec.max_seq_len)
        self.eps = 1e-12 if device.type == "cpu" else 1e-6   # BERT uses 1e-12
        self._init_weights(seed)

    def _init_weights(self, seed: int) -> None:
        s = self.spec
        # CPU generator: identical weights on every device for the same seed
        # (the encoder is small; cross-device reproducibility matters for
        # the GPU-vs-CPU numerics test and shared federation indexes)
        g = torch.Generator().manual_seed(seed)
        C, I = s.hidden_size, s.intermediate_size
        sc = 1.0 / math.sqrt(C)

        def rand(*shape, scale):
            w = torch.randn(*shape, generator=g, dtype=torch.float32) * scale
            return w.to(device=self.device, dtype=self.dtype)

        def zeros(*shape):
            return torch.zeros(*shape, device=self.device, dtype=self.dtype)

        def ones(*shape):
            return torch.ones(*shape, device=self.device, dtype=self.dtype)

        self.tok_emb = rand(s.vocab_size, C, scale=0.02)
        self.pos_emb = rand(self.max_seq_len, C, scale=0.02)
        self.emb_ln_w, self.emb_ln_b = ones(C), zeros(C)
        self.layers = []
        for _ in range(s.num_layers):
            self.layers.append({
                "wqkv": rand(3 * s.num_heads * s.head_dim, C, scale=sc),
                "bqkv": zeros(3 * s.num_heads * s.head_dim),
                "wo": rand(C, s.num_heads * s.head_dim, scale=sc),
                "bo": zeros(C),
                "ln1_w": ones(C), "ln1_b": zeros(C),
                "wi": rand(I, C, scale=sc),
                "bi": zeros(I),
                "wo2": rand(C, I, scale=1.0 / math.sqrt(I)),
                "bo2": zeros(C),
                "ln2_w": ones(C), "ln2_b": zeros(C),
            })

    @torch.no_grad()
    def encode_ids(self, token_ids: torch.Tensor,
                   lengths: Optional[torch.Tensor] = None) -> torch.Tensor:
        """token_ids [B, S] -> embeddings [B, C] (fp32, L2-normalised).
        BERT block: h = LN(h + Attn(h)); h = LN(h + GELU(h Wi + bi) Wo2 + bo2)."""
        s = self.spec
        B, S = token_ids.shape
        H, D = s.num_heads, s.head_dim
        h = F.embedding(token_ids.long(), self.tok_emb) + \
            self.pos_emb[:S].unsqueeze(0)
        h = ops.layernorm(h.to(self.dtype), self.emb_ln_w, self.emb_ln_b,
                          self.eps)
        scale = 1.0 / math.sqrt(D)
        kv_len = (lengths.to(torch.int32) if lengths is not None
                  else torch.full((B,), S, dtype=torch.int32, device=self.device))
        pos0 = torch.zeros(B, dtype=torch.int32, device=self.device)
        for lw in self.layers:
            qkv = (F.linear(h, lw["wqkv"]) + lw["bqkv"]).view(B, S, 3, H, D)
            q = qkv[:, :, 0].contiguous()
            k = qkv[:, :, 1].transpose(1, 2).contiguous()   # [B,H,S,D] "cache"
            v = qkv[:, :, 2].transpose(1, 2).contiguous()
            att = ops.attn_prefill(q.to(self.dtype), k.to(self.dtype),
                                   v.to(self.dtype), pos0, scale=scale,
                                   causal=False, kv_len=kv_len)
            o = F.linear(att.reshape(B, S, H * D), lw["wo"]) + lw["bo"]
            h = ops.layernorm(o.to(self.dtype), lw["ln1_w"], lw["ln1_b"],
                              self.eps, residual=h)
            inner = ops.gelu((F.linear(h, lw["wi"]) + lw["bi"]).to(self.dtype))
            o2 = F.linear(inner, lw["wo2"]) + lw["bo2"]
            h = ops.layernorm(o2.to(self.dtype), lw["ln2_w"], lw["ln2_b"],
                              self.eps, residual=h)
        hf = h.float()
        if lengths is not None:
            mask = (torch.arange(S, device=self.device).unsqueeze(0)
                    < lengths.unsqueeze(1)).float().unsqueeze(-1)
            pooled = (hf * mask).sum(1) / mask.sum(1).clamp_min(1.0)
        else:
            pooled = hf.mean(dim=1)
        return F.normalize(pooled, dim=-1)

    @torch.no_grad()
    def encode_texts(self, texts: List[str], tokenizer=None,
                     batch_size: int = 64) -> torch.Tensor:
        """Convenience: byte-tokenize + pad + encode in batches."""
        from fei_amd.engine.tokenizer import ByteTokenizer
        tok = tokenizer or ByteTokenizer()
        out = []
        for i in range(0, len(texts), batch_size):
            chunk = texts[i:i + batch_size]
            ids = [tok.encode(t)[: self.max_seq_len] for t in chunk]
            lens = torch.tensor([len(x) for x in ids], device=self.device)
            S = max(int(lens.max()), 1)
            padded = torch.full((len(ids), S), tok.pad_id, dtype=torch.long,
                                device=self.device)
            for j, row in enumerate(ids):
                padded[j, :len(row)] = torch.tensor(row, device=self.device)
            out.append(self.encode_ids(padded, lens))
        return torch.cat(out, dim=0)
