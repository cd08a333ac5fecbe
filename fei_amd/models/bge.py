"""Embedding encoder for memdir semantic search (bge-base-class shape:
12 layers, 768 hidden, 12 heads, D=64).

The compute path is the point (BASELINE.json configs[3]: "bge-base
embeddings on MFMA"): GEMMs on hipBLASLt/MFMA, bidirectional attention on
the HIP flash kernel (causal=0), fused RMSNorm + SwiGLU kernels. Weights
are random-init (no network for checkpoints), so the block uses the same
pre-norm/SwiGLU idiom as the rest of the engine rather than replicating
BERT's LayerNorm/GELU bit-for-bit — the shapes and FLOPs match bge-base.

Output: mean-pooled last hidden state, L2-normalised — [B, hidden].
"""

from __future__ import annotations

import math
from dataclasses import dataclass
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
        self._init_weights(seed)

    def _init_weights(self, seed: int) -> None:
        s = self.spec
        g = torch.Generator(device=self.device).manual_seed(seed)
        C, I = s.hidden_size, s.intermediate_size
        sc = 1.0 / math.sqrt(C)

        def rand(*shape, scale):
            return (torch.randn(*shape, generator=g, device=self.device,
                                dtype=torch.float32) * scale).to(self.dtype)

        self.tok_emb = rand(s.vocab_size, C, scale=0.02)
        self.pos_emb = rand(self.max_seq_len, C, scale=0.02)
        self.layers = []
        for _ in range(s.num_layers):
            self.layers.append({
                "norm1": torch.ones(C, device=self.device, dtype=self.dtype),
                "wqkv": rand(3 * s.num_heads * s.head_dim, C, scale=sc),
                "wo": rand(C, s.num_heads * s.head_dim, scale=sc),
                "norm2": torch.ones(C, device=self.device, dtype=self.dtype),
                "wgu": rand(2 * I, C, scale=sc),
                "wdown": rand(C, I, scale=1.0 / math.sqrt(I)),
            })
        self.norm_f = torch.ones(C, device=self.device, dtype=self.dtype)

    @torch.no_grad()
    def encode_ids(self, token_ids: torch.Tensor,
                   lengths: Optional[torch.Tensor] = None) -> torch.Tensor:
        """token_ids [B, S] -> embeddings [B, C] (fp32, L2-normalised)."""
        s = self.spec
        B, S = token_ids.shape
        H, D = s.num_heads, s.head_dim
        h = F.embedding(token_ids.long(), self.tok_emb) + \
            self.pos_emb[:S].unsqueeze(0)
        h = h.to(self.dtype)
        scale = 1.0 / math.sqrt(D)
        kv_len = (lengths.to(torch.int32) if lengths is not None
                  else torch.full((B,), S, dtype=torch.int32, device=self.device))
        pos0 = torch.zeros(B, dtype=torch.int32, device=self.device)
        for lw in self.layers:
            x = ops.rmsnorm(h, lw["norm1"], s.norm_eps)
            qkv = F.linear(x, lw["wqkv"]).view(B, S, 3, H, D)
            q = qkv[:, :, 0].contiguous()
            k = qkv[:, :, 1].transpose(1, 2).contiguous()   # [B,H,S,D] "cache"
            v = qkv[:, :, 2].transpose(1, 2).contiguous()
            att = ops.attn_prefill(q, k, v, pos0, scale=scale, causal=False,
                                   kv_len=kv_len)
            h = h + F.linear(att.reshape(B, S, H * D), lw["wo"])
            x = ops.rmsnorm(h, lw["norm2"], s.norm_eps)
            act = ops.swiglu(F.linear(x, lw["wgu"]))
            h = h + F.linear(act, lw["wdown"])
        h = ops.rmsnorm(h, self.norm_f, s.norm_eps).float()
        if lengths is not None:
            mask = (torch.arange(S, device=self.device).unsqueeze(0)
                    < lengths.unsqueeze(1)).float().unsqueeze(-1)
            pooled = (h * mask).sum(1) / mask.sum(1).clamp_min(1.0)
        else:
            pooled = h.mean(dim=1)
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
