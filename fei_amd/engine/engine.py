"""LocalEngine: the MI355X inference engine behind the Assistant.

The decode step (the agent-tok/s hot path) is fully device-resident.
The bf16 default is the PLAIN chain (r02: non-temporal weight loads made
it +2% over the fused-norm chain, profiles/r02_decode_nt.md):
  rmsnorm -> QKV GEMV -> split-K decode attention (in-kernel RoPE +
  KV-append) -> combine -> O GEMV -> fused add+rmsnorm -> gate/up GEMV
  with fused SwiGLU -> down GEMV -> fused add+rmsnorm
(9 nt-streaming kernels/layer; FEI_FUSED_NORM=1 restores the 5-kernel
norm-prologue chain, which remains the fp8 decode path), then the
lm_head GEMV, the sampling kernel (writes the next input token + the
out_tokens ring ON DEVICE) and the position advance —
the whole step is captured once into a hipGraph (torch.cuda.CUDAGraph)
and replayed with zero host work per token. EOS is checked every
`eos_check_every` replays (one device->host copy per chunk, not per
token). Prefix caching: prefill can start from any cache position, so
agent continuation rounds prefill only the delta.

Replaces the reference's remote-API call site
(fei/core/assistant.py:527-530) per BASELINE.json's north star.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional, Union

import torch

from fei_amd import ops
from fei_amd.engine.config import ModelSpec, get_spec
from fei_amd.engine.tokenizer import ByteTokenizer
from fei_amd.models.llama import LlamaModel
from fei_amd.parallel.pg import ParallelContext
from fei_amd.utils.logging import get_logger

logger = get_logger("engine.engine")

SAMPLE_CHUNKS = 64       # stage-1 blocks for the sampling kernel


class LocalEngine:
    def __init__(
        self,
        spec: ModelSpec,
        device: Optional[torch.device] = None,
        batch_size: int = 1,
        max_seq_len: Union[int, str, None] = None,   # int | "hbm" | None
        use_hip_graph: Optional[bool] = None,
        tp: Optional[ParallelContext] = None,
        seed: int = 1234,
        attn_splits: int = 32,
        dtype: Optional[torch.dtype] = None,
        weight_quant: Optional[str] = None,    # None | "fp8" (decode path)
        tokenizer=None,                        # default ByteTokenizer; or
                                               # SpmTokenizer(model_path)
    ):
        self.spec = spec
        if device is None:
            device = torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")
        self.device = device
        self.is_gpu = device.type == "cuda"
        if dtype is None:
            dtype = torch.bfloat16 if self.is_gpu else torch.float32
        self.dtype = dtype
        self.B = batch_size
        # max_seq_len="hbm": size the KV caches toward the 288 GB of HBM3E
        # instead of the model's nominal window (SURVEY §5 long-context
        # note) — resolved after model init, when free memory is known
        self._hbm_seq = (max_seq_len == "hbm")
        if self._hbm_seq:
            max_seq_len = spec.max_seq_len          # placeholder until init
        self.max_seq_len = max_seq_len or spec.max_seq_len
        self.tp = tp or ParallelContext()
        # hipGraph decode is on under TP too: RCCL collectives are
        # graph-capturable on ROCm (ensure_graph falls back to eager if
        # capture fails) — VERDICT r01 missing #1/weak #3
        if use_hip_graph is None:
            use_hip_graph = self.is_gpu
        self.use_graph = use_hip_graph and self.is_gpu
        import os as _os
        # split-K decode attention: splits x Hkv x B workgroups. 4 splits
        # (32 WGs on a 256-CU chip) starved KV bandwidth at agent context
        # lengths (r01: 4 -> 32 measured +41% at seq 3400). Two r02
        # post-nt sweeps (r2c27, r2c32): 16 vs 32 differ only within
        # run-to-run noise (±2%) below ~4k context, while 32 is robustly
        # ahead at 8k (249 vs 217 tok/s) — 32 stays the B<=2 default.
        # Batches multiply the grid by B, so B>=4 halves the split count
        # (batch-8: splits 32 -> 16 measured 6.34 -> 5.85 ms/step; 4/8/16
        # within noise — the grid already fills the chip at 16).
        # The combine kernel now stages up to 64 splits for experiments.
        env_splits = _os.environ.get("FEI_ATTN_SPLITS")
        if env_splits:
            self.attn_splits = int(env_splits)
        elif attn_splits == 32 and batch_size >= 4:
            self.attn_splits = 16
        else:
            self.attn_splits = attn_splits
        # Fused single-pass attention (rope+append+attn in one kernel) runs
        # grid (Hkv x B) — too few workgroups to pull KV bandwidth: 194 vs
        # 244 tok/s at B=1 (r01), and the r01 "pays when B*Hkv >= 64"
        # heuristic was also wrong — 64 workgroups still leaves 3/4 of the
        # 256-CU chip idle (batch-8: split-K 1363-1368 tok/s vs fused 1159,
        # r02 late sweep). Split-K is the default at EVERY batch; the fused
        # kernel stays available via FEI_FUSED_ATTN=1.
        import os as _os
        env = _os.environ.get("FEI_FUSED_ATTN")
        self.fused_attn = env not in ("0", "false") if env is not None \
            else False
        # fused-norm chain default flipped in r02: after the non-temporal
        # weight fix the PLAIN path (separate rmsnorm + nt GEMVs) measured
        # FASTER (271.8 vs 266.4 tok/s, same-box A/B r2c30) — the norm
        # prologue's per-element VALU chain throttles the GEMV stream to
        # ~3-5 TB/s vs 6.6 plain. The fused chain stays for fp8 decode
        # (its kernels are norm-fused) and via FEI_FUSED_NORM=1.
        self._fused_norm_env = _os.environ.get("FEI_FUSED_NORM")
        self.seed = seed
        self.temperature = 0.0       # graph-captured; set before capture
        if tokenizer is None:
            tok_path = _os.environ.get("FEI_TOKENIZER")
            if tok_path == "byte":
                tokenizer = ByteTokenizer()
            elif tok_path:
                from fei_amd.engine.tokenizer import SpmTokenizer
                tokenizer = SpmTokenizer(tok_path)
            else:
                # default: the packaged 16k BPE model (trained offline,
                # scripts/train_tokenizer.py) whenever it fits the model's
                # vocab — agent-turn prompts then cost ~1/3 the tokens of
                # the byte fallback (VERDICT r01 missing #5). llama3-tiny
                # (vocab 512) keeps the byte tokenizer.
                default = _os.path.join(_os.path.dirname(__file__),
                                        "fei16k.model")
                if spec.vocab_size >= 16000 and _os.path.exists(default):
                    from fei_amd.engine.tokenizer import SpmTokenizer
                    tokenizer = SpmTokenizer(default)
        self.tokenizer = tokenizer or ByteTokenizer()
        if self.tokenizer.vocab_size > spec.vocab_size:
            raise ValueError(
                f"tokenizer vocab {self.tokenizer.vocab_size} exceeds model "
                f"vocab {spec.vocab_size}")

        if self.is_gpu and not ops.kernels_available():
            ops.require_lib()        # fail loudly: no eager fallback on GPU

        if not self.is_gpu and spec.hidden_size >= 2048:
            logger.warning(
                "building %s on CPU: ~%d GB of fp32 host RAM and minutes of "
                "init — use llama3-tiny for CPU work", spec.name,
                2 * spec.num_layers * spec.hidden_size *
                spec.intermediate_size * 12 // 10 ** 9)
        t0 = time.perf_counter()
        self.model = LlamaModel(spec, device, dtype, tp=self.tp, seed=seed,
                                max_seq_len=self.max_seq_len)
        w_path = _os.environ.get("FEI_WEIGHTS")
        if w_path:
            t1 = time.perf_counter()
            self.model.load_weights(w_path)
            logger.info("loaded weights from %s in %.1fs", w_path,
                        time.perf_counter() - t1)
        if self._hbm_seq:
            self.max_seq_len = self._resolve_hbm_seq()
            # RoPE table must cover the enlarged window
            from fei_amd.ops import reference as _ref
            self.model.max_seq_len = self.max_seq_len
            self.model.rope = _ref.rope_table(
                self.max_seq_len, self.model.D, spec.rope_theta,
                device=device)
            logger.info("max_seq_len='hbm' -> %d tokens (%.1f GB KV)",
                        self.max_seq_len,
                        self._kv_bytes_per_token() * self.B *
                        self.max_seq_len / 2 ** 30)
        self.k_caches, self.v_caches = self.model.new_kv_cache(self.B, self.max_seq_len)
        logger.info("model %s init in %.1fs (%.2f GB params)", spec.name,
                    time.perf_counter() - t0, self.model.param_bytes() / 2**30)
        self.weight_quant = weight_quant or _os.environ.get("FEI_WEIGHT_QUANT")
        if self._fused_norm_env is not None:
            self.fused_norm = self._fused_norm_env not in ("0", "false")
        else:
            self.fused_norm = (self.is_gpu and not self.tp.is_distributed
                               and self.weight_quant == "fp8")
        if self.weight_quant == "fp8":
            t0 = time.perf_counter()
            self.model.quantize_fp8()
            logger.info("fp8 weight quantization in %.1fs",
                        time.perf_counter() - t0)

        # static decode state (device-resident; graph-stable)
        self.token = torch.zeros(self.B, dtype=torch.int32, device=device)
        self.pos = torch.zeros(self.B, dtype=torch.int32, device=device)
        self.step = torch.zeros(1, dtype=torch.int32, device=device)
        self.out_tokens = torch.zeros(self.B, self.max_seq_len,
                                      dtype=torch.int32, device=device)
        self.sample_ws = torch.zeros(self.B, SAMPLE_CHUNKS, 2,
                                     dtype=torch.float32, device=device)
        Hq_l, D = self.model.hq_l, self.model.D
        # sized from self.attn_splits (NOT the ctor arg: FEI_ATTN_SPLITS
        # overrides it, and an undersized workspace is an out-of-bounds
        # write in the split kernel)
        # 3rd element: zeroed per-split done-flag buffer [B*Hkv_l*splits]
        # selecting the FUSED split-K combine (split 0 of each (b, hkv)
        # pair polls the flags and reduces the partials in-kernel — no
        # separate combine launch per layer).
        self.attn_ws = (
            torch.zeros(self.B, Hq_l, self.attn_splits, D, dtype=torch.float32, device=device),
            torch.zeros(self.B, Hq_l, self.attn_splits, 2, dtype=torch.float32, device=device),
            torch.zeros(self.B * self.model.hkv_l * self.attn_splits,
                        dtype=torch.int32, device=device),
        )
        self.attn_out = torch.zeros(self.B, Hq_l, D, dtype=self.dtype,
                                    device=device)
        # Persistent weight-streaming decode engine (FEI_STREAM_DECODE=1):
        # one launch per layer on the LDS-DMA loader/consumer structure
        # (ops/csrc/stream_layer.hip) instead of six. Guarded by the
        # residency/shape check; batch-1 bf16 tp=1 only. The launch path
        # stays the default and the correctness oracle.
        self.stream_decode = False
        self._stream_ws = None
        if (self.is_gpu and _os.environ.get("FEI_STREAM_DECODE", "0") == "1"
                and self.B == 1 and not self.tp.is_distributed
                and self.weight_quant is None and spec.arch == "llama"):
            rc = ops.stream_layer_check(
                spec.hidden_size, spec.num_heads, spec.num_kv_heads,
                spec.head_dim, spec.intermediate_size)
            if rc == 0:
                self._stream_ws = ops.stream_workspace(spec, device)
                self._stream_bufs = (
                    torch.zeros(1, spec.hidden_size, dtype=self.dtype,
                                device=device),
                    torch.zeros(1, spec.hidden_size, dtype=self.dtype,
                                device=device),
                )
                self.stream_decode = True
                logger.info("persistent stream-decode engine enabled")
            else:
                logger.warning("stream-decode refused (rc=%d); launch path",
                               rc)
        self._graph: Optional[torch.cuda.CUDAGraph] = None
        if self.tp.is_distributed:
            # TP decode samples on each rank's logits SHARD and gathers
            # (value, global index) — 8 bytes/seq — instead of the vocab
            # row (513 KB at Llama-3 vocab). Persistent buffers: the
            # collective is allocation-free and graph-capturable.
            self._tp_local = torch.zeros(self.B, 2, dtype=torch.float32,
                                         device=device)
            self._tp_all = torch.zeros(self.tp.world_size, self.B, 2,
                                       dtype=torch.float32, device=device)
            self._tp_all_i32 = self._tp_all.view(torch.int32)  # same storage
        self.last_metrics: Dict[str, float] = {}
        if self.is_gpu:
            # one-time hipBLASLt init off the first prefill's critical path
            a = torch.zeros(8, 64, device=device, dtype=self.dtype)
            w = torch.zeros(64, 64, device=device, dtype=self.dtype)
            torch.nn.functional.linear(a, w)
            torch.cuda.synchronize(device)

    # -- construction helpers ------------------------------------------------

    def _kv_bytes_per_token(self) -> int:
        s = self.spec
        el = 2 if self.dtype == torch.bfloat16 else 4
        return (s.num_layers * (s.num_kv_heads //
                                (self.tp.world_size
                                 if self.tp.is_distributed else 1))
                * s.head_dim * 2 * el)

    def _resolve_hbm_seq(self) -> int:
        """Largest KV window that fits the remaining device memory with
        20% headroom (activations, graph pool, workspace), in 1k steps,
        capped at 256k (the RoPE table and position math stay exact far
        beyond any trained window; weights here are synthetic anyway)."""
        if not self.is_gpu:
            return self.spec.max_seq_len
        free, _total = torch.cuda.mem_get_info(self.device)
        budget = int(free * 0.8)
        per_tok = self._kv_bytes_per_token() * self.B
        seq = min(budget // per_tok, 262144)
        seq = max((seq // 1024) * 1024, self.spec.max_seq_len)
        return int(seq)

    @classmethod
    def create(cls, model: Union[str, ModelSpec] = "llama3-8b", **kwargs) -> "LocalEngine":
        spec = get_spec(model) if isinstance(model, str) else model
        return cls(spec, **kwargs)

    # -- decode step ---------------------------------------------------------

    def _decode_step(self) -> None:
        if self.tp.is_distributed:
            return self._decode_step_tp()
        if self.stream_decode:
            logits = self.model.forward_decode_stream(
                self.token, self.pos, self.k_caches, self.v_caches,
                self._stream_ws, self._stream_bufs)
            ops.sample(logits, self.token, self.step,
                       self.sample_ws.view(self.B, -1),
                       out_tokens=self.out_tokens,
                       temperature=self.temperature, seed=self.seed,
                       nchunks=SAMPLE_CHUNKS)
            ops.advance(self.pos, self.step, max_pos=self.max_seq_len - 1)
            return
        logits = self.model.forward_decode(
            self.token, self.pos, self.k_caches, self.v_caches,
            attn_splits=self.attn_splits, workspace=self.attn_ws,
            fused_attn=self.fused_attn, attn_out=self.attn_out,
            fused_norm=self.fused_norm)
        ops.sample(logits, self.token, self.step, self.sample_ws.view(self.B, -1),
                   out_tokens=self.out_tokens, temperature=self.temperature,
                   seed=self.seed, nchunks=SAMPLE_CHUNKS)
        ops.advance(self.pos, self.step, max_pos=self.max_seq_len - 1)

    def _decode_step_tp(self) -> None:
        """TP decode step: per-rank logits SHARD -> shard sampler (value,
        global idx) -> 8-byte/seq all-gather -> identical winner math on
        every rank. Replaces the per-step vocab-width logits all-gather
        (VERDICT r01 weak #3). Every op is device-resident on stable
        buffers, so the step is hipGraph-capturable including the RCCL
        collectives."""
        from fei_amd.parallel import pg as _pg

        logits = self.model.forward_decode(
            self.token, self.pos, self.k_caches, self.v_caches,
            attn_splits=self.attn_splits, workspace=self.attn_ws,
            fused_attn=self.fused_attn, attn_out=self.attn_out,
            fused_norm=False, gather_logits=False)
        ops.sample_shard(logits, self.step, self.model.vocab_offset,
                         self._tp_local, temperature=self.temperature,
                         seed=self.seed)
        _pg.all_gather_into(self._tp_all, self._tp_local, self.tp)
        vals = self._tp_all[:, :, 0]                       # [W, B]
        idx = self._tp_all_i32[:, :, 1]                    # [W, B] int32 view
        # first max along ranks == lowest global index on ties (shards are
        # rank-ordered), matching the full sampler's tie-break
        win = vals.argmax(dim=0, keepdim=True)             # [1, B]
        tok = idx.gather(0, win).squeeze(0)                # [B] int32
        self.token.copy_(tok)
        st = self.step.clamp(max=self.out_tokens.shape[1] - 1).to(torch.int64)
        self.out_tokens.scatter_(1, st.view(1, 1).expand(self.B, 1),
                                 tok.view(self.B, 1))
        ops.advance(self.pos, self.step, max_pos=self.max_seq_len - 1)

    def _capture_graph(self) -> None:
        assert self.is_gpu
        # The two warmup decode steps and the capture itself mutate real
        # engine state: KV rows at the warmup positions, token/pos/step and
        # out_tokens[:, :2]. ensure_graph() can run AFTER a prefill (e.g. a
        # recapture when sampling params change between generate() calls, or
        # first capture on a chunked >PREFILL_CHUNK prompt), so snapshot and
        # restore everything the warmup touches — otherwise the cached
        # prefix is silently corrupted.
        saved = {
            "token": self.token.clone(),
            "pos": self.pos.clone(),
            "step": self.step.clone(),
            "out": self.out_tokens[:, :4].clone(),
        }
        # warmup at positions 1..2: snapshot those KV rows in every layer
        kv_rows = slice(1, 4)
        saved_k = [k[:, :, kv_rows].clone() for k in self.k_caches]
        saved_v = [v[:, :, kv_rows].clone() for v in self.v_caches]
        # state must be valid during warmup/capture: pretend one token exists
        self.pos.fill_(1)
        self.step.zero_()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._decode_step()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._decode_step()
        torch.cuda.synchronize()
        # restore the clobbered state
        self.token.copy_(saved["token"])
        self.pos.copy_(saved["pos"])
        self.step.copy_(saved["step"])
        self.out_tokens[:, :4].copy_(saved["out"])
        for k, sk in zip(self.k_caches, saved_k):
            k[:, :, kv_rows].copy_(sk)
        for v, sv in zip(self.v_caches, saved_v):
            v[:, :, kv_rows].copy_(sv)
        # fused-combine done-flags are (pos, layer)-tagged: the warmup ran
        # at pos 1..2, and the first real replay may run at those same
        # positions — stale matching tags would let the in-kernel reducer
        # read partials before the current step's splits wrote them.
        if len(self.attn_ws) >= 3:
            self.attn_ws[2].zero_()
        torch.cuda.synchronize()
        self._graph = g
        self._graph_params = (self.temperature, self.seed)
        logger.info("decode step captured into hipGraph")

    def ensure_graph(self) -> None:
        if not self.use_graph:
            return
        # temperature and seed are kernel ARGUMENTS baked into the capture:
        # a later generate() with different sampling params must recapture
        # or the graph would silently sample with the old ones
        if self._graph is not None and \
                getattr(self, "_graph_params", None) != (self.temperature,
                                                         self.seed):
            logger.info("sampling params changed; recapturing decode graph")
            self._graph = None
        if self._graph is None:
            t0 = time.perf_counter()
            try:
                self._capture_graph()
            except Exception as e:
                if not self.tp.is_distributed:
                    raise
                # RCCL graph capture is version-dependent; a TP engine
                # falls back to eager decode rather than failing the job.
                # (Deterministic across ranks: same code path everywhere.)
                logger.warning("hipGraph capture failed under TP (%s); "
                               "falling back to eager decode", e)
                self.use_graph = False
                self._graph = None
                return
            logger.info("graph capture took %.2fs", time.perf_counter() - t0)

    # -- public API ----------------------------------------------------------

    PREFILL_CHUNK = 2048      # bounds activation memory for long prompts

    def prefill(self, token_ids: List[int], from_pos: int = 0) -> None:
        """Prefill ``token_ids`` starting at cache position ``from_pos``
        (0 = fresh conversation; >0 = prefix-cache extension: the first
        ``from_pos`` tokens are already in the KV caches) and sample the
        first new token. Long prompts are processed in PREFILL_CHUNK
        slices (each slice extends the KV caches; only the last slice's
        logits are sampled)."""
        if self.stream_decode:
            # pos can move backwards across prefills; the stream engine's
            # granule tags are pos-keyed, so stale tags must be cleared
            for t in self._stream_ws.values():
                t.zero_()
        # same pos-keyed-tag rule for the fused-combine done flags: a new
        # request replays earlier positions, and a stale matching tag would
        # release the in-kernel reducer before this step's splits published
        if self.is_gpu and len(self.attn_ws) >= 3:
            self.attn_ws[2].zero_()
        # Cap to the context budget BEFORE chunking (mirrors _prep_prompt's
        # cap for the from_pos>0 prefix-cached path, which bypasses it):
        # keep the TAIL of the new tokens — a mid-loop break would silently
        # drop a chunk from the MIDDLE and prefill the remainder at the
        # dropped chunk's positions.
        budget = self.max_seq_len - 1 - from_pos
        if budget <= 0:
            from_pos = 0
            budget = self.max_seq_len - 1
        if len(token_ids) > budget:
            token_ids = token_ids[-budget:]
        while len(token_ids) > self.PREFILL_CHUNK:
            head, token_ids = (token_ids[:self.PREFILL_CHUNK],
                               token_ids[self.PREFILL_CHUNK:])
            tokens = torch.tensor([head] * self.B, dtype=torch.int64,
                                  device=self.device)
            pos0 = torch.full((self.B,), from_pos, dtype=torch.int32,
                              device=self.device)
            self.model.forward_prefill(tokens, pos0, self.k_caches,
                                       self.v_caches)
            from_pos += len(head)
        S = len(token_ids)
        self.ensure_graph()
        tokens = torch.tensor([token_ids] * self.B, dtype=torch.int64,
                              device=self.device)
        pos0 = torch.full((self.B,), from_pos, dtype=torch.int32,
                          device=self.device)
        logits = self.model.forward_prefill(tokens, pos0, self.k_caches,
                                            self.v_caches)
        self.step.zero_()
        self.out_tokens.zero_()
        ops.sample(logits.to(self.dtype), self.token, self.step,
                   self.sample_ws.view(self.B, -1), out_tokens=self.out_tokens,
                   temperature=self.temperature, seed=self.seed,
                   nchunks=SAMPLE_CHUNKS)
        self.pos.fill_(from_pos + S)
        if self.is_gpu:
            torch.cuda.synchronize(self.device)
        else:
            self.step += 1   # reference sample() does not advance step
            return
        # step was advanced by... (sample kernel does not touch step); advance
        # host-side: first generated token occupies out_tokens[:,0]
        self.step.fill_(1)

    def decode(self, n_tokens: int, eos_check_every: int = 32,
               stop_on_eos: bool = True,
               stop: Optional[List[str]] = None) -> List[List[int]]:
        """Generate up to n_tokens per sequence; returns new token ids
        (including the one sampled by prefill). ``stop``: stop strings —
        generation halts once any appears in the decoded text (checked at
        chunk boundaries; the match itself is trimmed by the caller)."""
        done = 1                      # prefill already produced token 0
        eos = self.tokenizer.eos_id
        while done < n_tokens:
            chunk = min(eos_check_every, n_tokens - done)
            if self._graph is not None:
                for _ in range(chunk):
                    self._graph.replay()
            else:
                for _ in range(chunk):
                    self._decode_step()
            done += chunk
            if int(self.pos.max()) >= self.max_seq_len - 1:
                break                      # cache capacity reached
            if stop_on_eos or stop:
                toks = self.out_tokens[:, :done].tolist()
                if stop_on_eos and all(eos in row for row in toks):
                    break
                if stop and all(
                        any(ss in self.tokenizer.decode(row) for ss in stop)
                        for row in toks):
                    break
        if self.is_gpu:
            torch.cuda.synchronize(self.device)
        self._check_stream_fail()
        rows = self.out_tokens[:, :done].tolist()
        if stop_on_eos:
            rows = [row[: row.index(eos) + 1] if eos in row else row
                    for row in rows]
        return rows

    def _check_stream_fail(self) -> None:
        """Stream-engine give-up detection: a bounded spin that timed out
        stamps the fail word instead of hanging the GPU; surface it."""
        if self.stream_decode and int(self._stream_ws["fail"][0]):
            code = int(self._stream_ws["fail"][0])
            self._stream_ws["fail"].zero_()
            raise RuntimeError(
                f"stream-decode engine gave up (code {code}); outputs are "
                "poisoned — rerun with FEI_STREAM_DECODE=0")

    def decode_speculative(self, n_tokens: int, ctx_tokens: List[int],
                           spec_k: int = 8, stop_on_eos: bool = True,
                           ) -> List[List[int]]:
        """Greedy decode with prompt-lookup speculation (engine/speculative.py):
        propose up to ``spec_k`` tokens from the context's own repeats and
        verify the block with one prefill forward (all-positions logits).
        Token-identical to decode() at temperature 0; call after prefill().
        Batch 1 only."""
        from fei_amd.engine.speculative import NgramIndex

        assert self.B == 1, "speculative decode is a batch-1 agent path"
        assert self.temperature == 0.0, "speculative decode is greedy-only"
        eos = self.tokenizer.eos_id
        first = int(self.token[0])            # token sampled by prefill
        index = NgramIndex()
        index.extend(list(ctx_tokens) + [first])
        pos_h = int(self.pos[0])              # host mirror of the position
        done = 1
        n_blocks = n_proposed = n_accepted = 0
        while done < n_tokens and not (stop_on_eos and index.ctx[-1] == eos):
            props = index.propose(spec_k)
            block = [index.ctx[-1]] + props
            L = len(block)
            if pos_h + L >= self.max_seq_len:
                block = block[: self.max_seq_len - 1 - pos_h]
                props = block[1:]
                L = len(block)
                if L == 0:
                    break
            tokens = torch.tensor([block], dtype=torch.int64, device=self.device)
            pos0 = torch.full((1,), pos_h, dtype=torch.int32, device=self.device)
            logits = self.model.forward_prefill(tokens, pos0, self.k_caches,
                                                self.v_caches,
                                                all_positions=True)
            greedy = logits[0].argmax(dim=-1).tolist()        # len L
            n_acc = 0
            while n_acc < len(props) and props[n_acc] == greedy[n_acc]:
                n_acc += 1
            emitted = props[:n_acc] + [greedy[n_acc]]
            emitted = emitted[: n_tokens - done]
            if stop_on_eos and eos in emitted:
                emitted = emitted[: emitted.index(eos) + 1]
            n_blocks += 1
            n_proposed += len(props)
            n_accepted += n_acc
            index.extend(emitted)
            self.out_tokens[0, done: done + len(emitted)] = torch.tensor(
                emitted, dtype=torch.int32, device=self.device)
            done += len(emitted)
            pos_h += len(emitted)
        # re-sync the device-resident decode state so plain decode()/graph
        # replay can continue from here
        self.pos.fill_(pos_h)
        self.step.fill_(done)
        self.token.fill_(index.ctx[-1])
        if self.is_gpu:
            torch.cuda.synchronize(self.device)
        self.last_metrics.update({
            "spec_blocks": n_blocks,
            "spec_proposed": n_proposed,
            "spec_accepted": n_accepted,
            "spec_acceptance": n_accepted / max(n_proposed, 1),
            "spec_tokens_per_block": done / max(n_blocks, 1),
        })
        rows = self.out_tokens[:, :done].tolist()
        if stop_on_eos:
            rows = [row[: row.index(eos) + 1] if eos in row else row
                    for row in rows]
        return rows

    def generate(self, prompt: Union[str, List[int]], max_new_tokens: int = 256,
                 temperature: float = 0.0, stop_on_eos: bool = True,
                 from_pos: int = 0,
                 speculative: Optional[bool] = None,
                 stop: Optional[List[str]] = None) -> Dict[str, object]:
        """Prefill + decode; returns text and timing metrics. ``from_pos``
        enables prefix caching: the prompt's first ``from_pos`` tokens are
        already in the KV caches and only the remainder is prefilled.
        ``speculative`` (default: FEI_SPEC_DECODE env) uses prompt-lookup
        speculative decoding — greedy/batch-1 only, token-identical output;
        pays on contexts that repeat themselves (tool output, code)."""
        prompt_ids, new_ids, from_pos, max_new_tokens = self._prep_prompt(
            prompt, max_new_tokens, from_pos)
        self.temperature = temperature
        if self.is_gpu:
            torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        self.prefill(new_ids, from_pos=from_pos)
        t1 = time.perf_counter()
        if speculative is None:
            import os as _os
            speculative = _os.environ.get("FEI_SPEC_DECODE", "0") == "1"
        spec_metrics: Dict[str, object] = {}
        if speculative and self.B == 1 and temperature == 0.0:
            ctx = prompt_ids[:from_pos] + new_ids
            self.last_metrics = {}
            rows = self.decode_speculative(max_new_tokens, ctx,
                                           stop_on_eos=stop_on_eos)
            spec_metrics = dict(self.last_metrics)
        else:
            rows = self.decode(max_new_tokens, stop_on_eos=stop_on_eos,
                               stop=stop)
        t2 = time.perf_counter()
        new_tokens = len(rows[0])
        decode_s = t2 - t1
        self.last_metrics = {
            **spec_metrics,
            "prompt_tokens": len(prompt_ids),
            "cached_prefix": from_pos,
            "new_tokens": new_tokens,
            "prefill_s": t1 - t0,
            "decode_s": decode_s,
            "prefill_tok_s": len(new_ids) / max(t1 - t0, 1e-9),
            "decode_tok_s": (max(new_tokens - 1, 0) * self.B) / max(decode_s, 1e-9),
        }
        text = self.tokenizer.decode(rows[0])
        finish = "stop" if (stop_on_eos and rows[0] and
                            rows[0][-1] == self.tokenizer.eos_id) else "length"
        if stop:
            cut = min((text.find(ss) for ss in stop if ss in text),
                      default=-1)
            if cut >= 0:
                text = text[:cut]
                finish = "stop"
        return {
            "text": text,
            "token_ids": rows[0],
            "finish_reason": finish,
            **self.last_metrics,
        }

    def _prep_prompt(self, prompt: Union[str, List[int]],
                     max_new_tokens: int, from_pos: int):
        """Shared prompt handling for generate()/generate_stream(): encode,
        validate the prefix-cache offset, and truncate over-long fresh
        prompts while reserving the decode budget."""
        if isinstance(prompt, str):
            prompt_ids = self.tokenizer.encode(prompt)
        else:
            prompt_ids = list(prompt)
        if from_pos > len(prompt_ids):
            from_pos = 0
        new_ids = prompt_ids[from_pos:]
        if not new_ids:                      # identical prompt: redo last token
            from_pos = max(0, len(prompt_ids) - 1)
            new_ids = prompt_ids[from_pos:]
        # over-long fresh prompts: keep the tail, reserving the requested
        # decode budget (prefix-cached prompts are left alone)
        if from_pos == 0:
            cap = max(1, self.max_seq_len - 1 - max_new_tokens)
            if len(new_ids) > cap:
                new_ids = new_ids[-cap:]
        eff_len = min(from_pos + len(new_ids), self.max_seq_len - 1)
        max_new_tokens = min(max_new_tokens, self.max_seq_len - eff_len - 1)
        if max_new_tokens < 1:
            max_new_tokens = 1
        return prompt_ids, new_ids, from_pos, max_new_tokens

    def generate_stream(self, prompt: Union[str, List[int]],
                        max_new_tokens: int = 256,
                        temperature: float = 0.0, stop_on_eos: bool = True,
                        from_pos: int = 0, chunk: int = 16,
                        stop: Optional[List[str]] = None):
        """Streaming generate (batch 1): yields a dict per decoded chunk —
        {"new_token_ids", "text", "done"} where ``text`` is the cumulative
        decode (byte tokens can split multi-byte characters, so deltas are
        only stable on the cumulative string). The final chunk carries the
        full metrics of generate()."""
        assert self.B == 1, "streaming is a batch-1 interactive path"
        prompt_ids, new_ids, from_pos, max_new_tokens = self._prep_prompt(
            prompt, max_new_tokens, from_pos)
        self.temperature = temperature
        if self.is_gpu:
            torch.cuda.synchronize(self.device)
        t0 = time.perf_counter()
        self.prefill(new_ids, from_pos=from_pos)
        t1 = time.perf_counter()
        eos = self.tokenizer.eos_id
        done = 1
        emitted = 0
        finished = False
        while not finished:
            n_chunk = min(chunk, max_new_tokens - done)
            if self._graph is not None:
                for _ in range(n_chunk):
                    self._graph.replay()
            else:
                for _ in range(n_chunk):
                    self._decode_step()
            done += n_chunk
            if self.is_gpu:
                torch.cuda.synchronize(self.device)
            self._check_stream_fail()
            row = self.out_tokens[0, :done].tolist()
            if stop_on_eos and eos in row:
                row = row[: row.index(eos) + 1]
                finished = True
            if done >= max_new_tokens:
                finished = True
            text = self.tokenizer.decode(row)
            if stop:
                cut = min((text.find(ss) for ss in stop if ss in text),
                          default=-1)
                if cut >= 0:
                    text = text[:cut]
                    finished = True
            new = row[emitted:]
            emitted = len(row)
            out = {"new_token_ids": new,
                   "text": text,
                   "done": finished}
            if finished:
                decode_s = time.perf_counter() - t1
                self.last_metrics = {
                    "prompt_tokens": len(prompt_ids),
                    "cached_prefix": from_pos,
                    "new_tokens": len(row),
                    "prefill_s": t1 - t0,
                    "decode_s": decode_s,
                    "prefill_tok_s": len(new_ids) / max(t1 - t0, 1e-9),
                    "decode_tok_s": max(len(row) - 1, 0) / max(decode_s, 1e-9),
                }
                out["token_ids"] = row
                out.update(self.last_metrics)
            yield out

    @torch.no_grad()
    def loglikelihood(self, context: Union[str, List[int]],
                      continuation: Union[str, List[int]]
                      ) -> Dict[str, object]:
        """Teacher-forced log P(continuation | context) — the evaluation
        primitive (lm-eval style): ONE prefill forward over
        context+continuation with all-positions logits; returns the summed
        logprob, per-token logprobs, and whether the continuation is the
        greedy decode."""
        ctx = (self.tokenizer.encode(context) if isinstance(context, str)
               else list(context))
        cont = (self.tokenizer.encode(continuation, add_bos=False)
                if isinstance(continuation, str) else list(continuation))
        if not cont:
            return {"logprob": 0.0, "token_logprobs": [], "is_greedy": True}
        ids = (ctx + cont)[-(self.max_seq_len - 1):]
        n_cont = min(len(cont), len(ids) - 1)
        tokens = torch.tensor([ids], dtype=torch.int64, device=self.device)
        pos0 = torch.zeros(1, dtype=torch.int32, device=self.device)
        logits = self.model.forward_prefill(tokens, pos0, self.k_caches,
                                            self.v_caches,
                                            all_positions=True)[0].float()
        # logits[i] predicts ids[i+1]
        lp = torch.log_softmax(logits[:-1], dim=-1)
        start = len(ids) - 1 - n_cont
        tgt = torch.tensor(ids[start + 1:], device=self.device)
        rows = lp[start:]
        tok_lp = rows.gather(1, tgt.unsqueeze(1)).squeeze(1)
        greedy = bool((rows.argmax(dim=-1) == tgt).all())
        return {"logprob": float(tok_lp.sum()),
                "token_logprobs": [float(x) for x in tok_lp],
                "is_greedy": greedy}

    def shutdown(self) -> None:
        self._graph = None
        self.k_caches = self.v_caches = None  # release HBM
        if self.is_gpu:
            torch.cuda.empty_cache()

    # -- eager sampling variants (top-k / top-p) -----------------------------

    def _decode_step_eager_sampled(self, top_k: int = 0, top_p: float = 1.0,
                                   generator=None) -> None:
        """Non-graph decode step with top-k / nucleus filtering (the graph
        path covers greedy + plain Gumbel; filtered sampling runs eagerly)."""
        from fei_amd.ops import reference as ref

        logits = self.model.forward_decode(
            self.token, self.pos, self.k_caches, self.v_caches,
            attn_splits=self.attn_splits, workspace=self.attn_ws,
            fused_attn=self.fused_attn, attn_out=self.attn_out,
            fused_norm=self.fused_norm)
        masked = logits.float()
        if top_k:
            masked = ref.topk_mask(masked, top_k)
        if top_p < 1.0:
            masked = ref.topp_mask(masked, top_p)
        nxt = ref.gumbel_sample(masked, self.temperature, generator)
        self.token.copy_(nxt)
        st = int(self.step)
        if st < self.out_tokens.shape[1]:
            self.out_tokens[:, st] = nxt
        self.pos += 1
        self.step += 1

    def generate_sampled(self, prompt, max_new_tokens: int = 256,
                         temperature: float = 0.8, top_k: int = 0,
                         top_p: float = 1.0, seed: int = 0):
        """generate() with top-k / top-p filtering (eager decode path)."""
        if isinstance(prompt, str):
            prompt_ids = self.tokenizer.encode(prompt)
        else:
            prompt_ids = list(prompt)
        self.temperature = temperature
        gen = torch.Generator(device=self.device)
        gen.manual_seed(seed or self.seed)
        torch.manual_seed(seed or self.seed)   # prefill's sampler (CPU path)
        self.prefill(prompt_ids)
        n = min(max_new_tokens, self.max_seq_len - len(prompt_ids) - 1)
        for _ in range(max(n - 1, 0)):
            self._decode_step_eager_sampled(top_k, top_p, gen)
        if self.is_gpu:
            torch.cuda.synchronize(self.device)
        rows = self.out_tokens[:, :n].tolist()
        eos = self.tokenizer.eos_id
        rows = [row[: row.index(eos) + 1] if eos in row else row for row in rows]
        return {"text": self.tokenizer.decode(rows[0]), "token_ids": rows[0],
                "new_tokens": len(rows[0])}

    # -- ragged multi-session batch ------------------------------------------

    def prefill_ragged(self, prompts: List[List[int]]) -> None:
        """Prefill B DIFFERENT prompts (padded to the longest; each row's
        cache rows past its real length are overwritten by decode before
        they are ever attended). Samples each row's first token from the
        logits at its own last real position."""
        assert len(prompts) == self.B, f"need exactly {self.B} prompts"
        lens = [min(len(p), self.max_seq_len - 1) for p in prompts]
        prompts = [p[-n:] if n < len(p) else p for p, n in zip(prompts, lens)]
        S = max(lens)
        self.ensure_graph()
        tok = self.tokenizer
        padded = torch.full((self.B, S), tok.pad_id, dtype=torch.int64,
                            device=self.device)
        for b, p in enumerate(prompts):
            padded[b, :lens[b]] = torch.tensor(p, dtype=torch.int64,
                                               device=self.device)
        pos0 = torch.zeros(self.B, dtype=torch.int32, device=self.device)
        all_logits = self.model.forward_prefill(
            padded, pos0, self.k_caches, self.v_caches, all_positions=True)
        idx = torch.tensor([n - 1 for n in lens], device=self.device)
        logits = all_logits[torch.arange(self.B, device=self.device), idx]
        self.step.zero_()
        self.out_tokens.zero_()
        ops.sample(logits.to(self.dtype).contiguous(), self.token, self.step,
                   self.sample_ws.view(self.B, -1), out_tokens=self.out_tokens,
                   temperature=self.temperature, seed=self.seed,
                   nchunks=SAMPLE_CHUNKS)
        self.pos.copy_(torch.tensor(lens, dtype=torch.int32,
                                    device=self.device))
        if self.is_gpu:
            torch.cuda.synchronize(self.device)
            self.step.fill_(1)
        else:
            self.step += 1

    def generate_batch(self, prompts, max_new_tokens: int = 256,
                       temperature: float = 0.0, stop_on_eos: bool = True):
        """Independent agent sessions in one batch: B different prompts ->
        B completions, decoded together (per-row positions/EOS)."""
        ids = [self.tokenizer.encode(p) if isinstance(p, str) else list(p)
               for p in prompts]
        budget = self.max_seq_len - max(len(p) for p in ids) - 1
        max_new_tokens = max(1, min(max_new_tokens, budget))
        self.temperature = temperature
        self.prefill_ragged(ids)
        rows = self.decode(max_new_tokens, stop_on_eos=stop_on_eos)
        return [{"text": self.tokenizer.decode(r), "token_ids": r}
                for r in rows]
