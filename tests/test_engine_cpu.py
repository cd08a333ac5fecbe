"""Engine tests on CPU (tiny model, torch reference ops).

The key invariant: decode-with-cache must reproduce re-prefilling the whole
sequence from scratch (validates RoPE, KV append, decode attention and the
sampling path end to end)."""

import pytest
import torch

from fei_amd.engine.config import get_spec
from fei_amd.engine.engine import LocalEngine
from fei_amd.engine.tokenizer import ByteTokenizer
from fei_amd.models.llama import LlamaModel
from fei_amd.parallel.pg import ParallelContext


@pytest.fixture(scope="module")
def engine():
    return LocalEngine.create("llama3-tiny", max_seq_len=128, seed=7)


def test_tokenizer_roundtrip():
    tok = ByteTokenizer()
    text = "hello Fei! éàü 你好"
    ids = tok.encode(text)
    assert ids[0] == tok.bos_id
    assert tok.decode(ids) == text


def test_decode_matches_prefill_from_scratch(engine):
    """Greedy decode with cache == teacher-forced re-prefill logits argmax."""
    tok_ids = engine.tokenizer.encode("abc def")
    out = engine.generate(tok_ids, max_new_tokens=5, stop_on_eos=False)
    generated = out["token_ids"]
    assert len(generated) == 5

    # independent check: fresh model, prefill growing sequences from scratch
    spec = get_spec("llama3-tiny")
    model = LlamaModel(spec, torch.device("cpu"), torch.float32, seed=7,
                       max_seq_len=128)
    seq = list(tok_ids)
    expected = []
    for _ in range(5):
        kc, vc = model.new_kv_cache(1, 128)
        pos0 = torch.zeros(1, dtype=torch.int32)
        logits = model.forward_prefill(torch.tensor([seq]), pos0, kc, vc)
        nxt = int(logits[0].float().argmax())
        expected.append(nxt)
        seq.append(nxt)
    assert generated == expected


def test_generate_metrics(engine):
    out = engine.generate("hi", max_new_tokens=4, stop_on_eos=False)
    assert out["new_tokens"] == 4
    assert out["prompt_tokens"] == len(engine.tokenizer.encode("hi"))
    assert out["decode_tok_s"] > 0
    assert isinstance(out["text"], str)


def test_generate_deterministic(engine):
    a = engine.generate("same prompt", max_new_tokens=6, stop_on_eos=False)
    b = engine.generate("same prompt", max_new_tokens=6, stop_on_eos=False)
    assert a["token_ids"] == b["token_ids"]


def test_temperature_sampling_changes_tokens(engine):
    a = engine.generate("t", max_new_tokens=6, temperature=1.0, stop_on_eos=False)
    assert len(a["token_ids"]) == 6


def test_param_bytes_accounting():
    spec = get_spec("llama3-tiny")
    model = LlamaModel(spec, torch.device("cpu"), torch.float32, seed=0)
    # fp32 tiny model: a few MB, and close to the spec estimate (x2 for fp32)
    est = spec.params_bytes(dtype_bytes=4)
    assert abs(model.param_bytes() - est) / est < 0.05


def test_kv_pool_alloc_release():
    from fei_amd.engine.kv_cache import PagedKVPool
    pool = PagedKVPool(num_layers=2, num_kv_heads=2, head_dim=64,
                       block_size=16, num_blocks=8,
                       device=torch.device("cpu"), dtype=torch.float32)
    sid = pool.new_sequence()
    blocks = pool.ensure_capacity(sid, 40)          # 3 blocks
    assert len(blocks) == 3
    assert pool.free_blocks() == 5
    sid2 = pool.fork(sid)
    assert pool.free_blocks() == 2
    pool.release(sid)
    pool.release(sid2)
    assert pool.free_blocks() == 8
    with pytest.raises(MemoryError):
        s3 = pool.new_sequence()
        pool.ensure_capacity(s3, 16 * 100)


def test_checkpoint_roundtrip(tmp_path):
    """save_weights (tp=1) -> load into a differently-seeded model ->
    identical logits (safetensors checkpoint path)."""
    spec = get_spec("llama3-tiny")
    m1 = LlamaModel(spec, torch.device("cpu"), torch.float32, seed=7,
                    max_seq_len=64)
    m1.save_weights(str(tmp_path / "ckpt"))

    m2 = LlamaModel(spec, torch.device("cpu"), torch.float32, seed=999,
                    max_seq_len=64)
    tokens = torch.tensor([[1, 4, 9]])
    pos0 = torch.zeros(1, dtype=torch.int32)
    kc, vc = m2.new_kv_cache(1, 64)
    before = m2.forward_prefill(tokens, pos0, kc, vc)
    m2.load_weights(str(tmp_path / "ckpt"))
    kc, vc = m2.new_kv_cache(1, 64)
    after = m2.forward_prefill(tokens, pos0, kc, vc)
    kc, vc = m1.new_kv_cache(1, 64)
    ref = m1.forward_prefill(tokens, pos0, kc, vc)
    assert not torch.allclose(before, ref)
    assert torch.allclose(after, ref, atol=1e-5)


def test_prefix_cache_extension_matches_scratch(engine):
    """generate(full) from scratch == generate(prefix) then
    generate(full, from_pos=len(prefix)) — the prefix-cache contract."""
    tok = engine.tokenizer
    p1 = tok.encode("first user message")
    suffix = tok.encode("and the continuation", add_bos=False)
    full = p1 + suffix

    a = engine.generate(full, max_new_tokens=5, stop_on_eos=False)

    engine.generate(p1, max_new_tokens=2, stop_on_eos=False)  # fills cache
    b = engine.generate(full, max_new_tokens=5, stop_on_eos=False,
                        from_pos=len(p1))
    assert b["cached_prefix"] == len(p1)
    assert a["token_ids"] == b["token_ids"]


def test_local_backend_prefix_cache(engine):
    from fei_amd.core.backends import LocalBackend
    backend = LocalBackend(engine=engine)
    msgs = [{"role": "user", "content": "question one"}]
    backend.complete(msgs, max_tokens=4)
    assert engine.last_metrics["cached_prefix"] == 0
    msgs = msgs + [{"role": "assistant", "content": "answer one"},
                   {"role": "user", "content": "question two"}]
    backend.complete(msgs, max_tokens=4)
    assert engine.last_metrics["cached_prefix"] > 0


def test_engine_batch_decode():
    """batch>1: every sequence decodes the same greedy tokens for the same
    prompt (replicated prompts; independent agent sessions)."""
    e = LocalEngine.create("llama3-tiny", max_seq_len=64, batch_size=3, seed=3)
    e.prefill(e.tokenizer.encode("batch test"))
    rows = e.decode(4, stop_on_eos=False)
    assert len(rows) == 3
    assert rows[0] == rows[1] == rows[2]


def test_generate_sampled_topk_topp():
    e = LocalEngine.create("llama3-tiny", max_seq_len=64, seed=5)
    out = e.generate_sampled("sampling test", max_new_tokens=6,
                             temperature=1.0, top_k=8, seed=11)
    assert 1 <= out["new_tokens"] <= 6
    out2 = e.generate_sampled("sampling test", max_new_tokens=6,
                              temperature=1.0, top_p=0.9, seed=11)
    assert 1 <= out2["new_tokens"] <= 6
    # deterministic under the same seed
    out3 = e.generate_sampled("sampling test", max_new_tokens=6,
                              temperature=1.0, top_p=0.9, seed=11)
    assert out2["token_ids"] == out3["token_ids"]


def test_attn_decode_paged_cpu_reference():
    """CPU path of the paged-attention op (gather + reference)."""
    import math
    from fei_amd import ops
    from fei_amd.ops import reference as ref

    B, Hq, Hkv, D, BS, max_blocks = 1, 4, 2, 64, 16, 4
    torch.manual_seed(0)
    k_pool = torch.randn(8, Hkv, BS, D)
    v_pool = torch.randn(8, Hkv, BS, D)
    bt = torch.tensor([[3, 1, 6, 0]], dtype=torch.int32)
    q = torch.randn(B, Hq, D)
    pos = torch.tensor([40], dtype=torch.int32)
    out = ops.attn_decode_paged(q, k_pool, v_pool, bt, pos)
    # manual gather
    kc = torch.zeros(B, Hkv, max_blocks * BS, D)
    vc = torch.zeros_like(kc)
    for j, blk in enumerate(bt[0].tolist()):
        kc[0, :, j * BS:(j + 1) * BS] = k_pool[blk]
        vc[0, :, j * BS:(j + 1) * BS] = v_pool[blk]
    expected = ref.attn_decode(q, kc, vc, pos + 1)
    assert torch.allclose(out, expected, atol=1e-5)


def test_ragged_batch_matches_single_sessions():
    """B=3 different prompts decoded together == each decoded alone
    (greedy) — the multi-session serving contract."""
    prompts = ["alpha prompt", "a much longer second prompt here", "z"]
    batch_engine = LocalEngine.create("llama3-tiny", max_seq_len=96,
                                      batch_size=3, seed=21)
    outs = batch_engine.generate_batch(prompts, max_new_tokens=6,
                                       stop_on_eos=False)
    for i, p in enumerate(prompts):
        single = LocalEngine.create("llama3-tiny", max_seq_len=96, seed=21)
        ref_out = single.generate(p, max_new_tokens=6, stop_on_eos=False)
        assert outs[i]["token_ids"] == ref_out["token_ids"], f"prompt {i}"


def test_fp8_quant_roundtrip_cpu():
    from fei_amd.ops import reference as ref
    w = torch.randn(16, 64) * 0.1
    w8, sc = ref.quant_fp8(w)
    deq = ref.dequant_fp8(w8, sc)
    rel = ((deq - w).abs() / w.abs().clamp_min(1e-6)).median()
    assert rel < 0.05          # e4m3 3-bit mantissa: ~3 % typical error


def test_engine_fp8_weight_quant_cpu():
    e = LocalEngine.create("llama3-tiny", max_seq_len=64, seed=31,
                           weight_quant="fp8")
    assert hasattr(e.model, "fp8")
    # CPU decode uses the reference fallbacks with dequantized weights;
    # fused-norm is off on CPU so this just checks quantization happened
    out = e.generate("fp8 check", max_new_tokens=3, stop_on_eos=False)
    assert len(out["token_ids"]) == 3


def test_prompt_longer_than_context_truncates(engine):
    long_prompt = list(range(4, 260)) * 2     # 512 ids > max_seq 128
    out = engine.generate(long_prompt, max_new_tokens=3, stop_on_eos=False)
    assert out["new_tokens"] == 3             # survives, keeps the tail


def test_empty_prompt(engine):
    out = engine.generate("", max_new_tokens=3, stop_on_eos=False)
    assert out["new_tokens"] == 3             # BOS-only prompt works


def test_eos_stops_decode():
    e = LocalEngine.create("llama3-tiny", max_seq_len=64, seed=7)
    e.prefill(e.tokenizer.encode("x"))
    # force EOS as the next sampled token by planting it in out_tokens path:
    # instead, decode with stop_on_eos and verify trimming logic on the rows
    rows = e.decode(5, stop_on_eos=True, eos_check_every=2)
    row = rows[0]
    if e.tokenizer.eos_id in row:
        assert row[-1] == e.tokenizer.eos_id   # trimmed AT the eos
    assert 1 <= len(row) <= 5


def test_system_prompt_reaches_backend(tmp_path):
    from fei_amd.core.assistant import Assistant
    from fei_amd.core.backends import Backend, BackendResponse
    from fei_amd.utils.config import Config

    seen = {}

    class Probe(Backend):
        def complete(self, messages, tools=None, system=None, max_tokens=4000,
                     temperature=0.0):
            seen["system"] = system
            return BackendResponse(content="ok")

    cfg = Config(ini_path=str(tmp_path / "i.ini"), load_dotenv=False)
    a = Assistant(config=cfg, backend=Probe())
    a.chat("hi", system_prompt="CUSTOM SYSTEM")
    assert seen["system"] == "CUSTOM SYSTEM"
    a.chat("hi again")
    assert "Fei" in seen["system"]            # default prompt restored


def test_fp8_gated_to_small_batch():
    """fp8 fused-norm must fall back to bf16 weights at large B*C (the LDS
    stage dominates there — profiles/r01_aux_benchmarks.md)."""
    import fei_amd.models.llama as lm
    spec = get_spec("llama3-tiny")           # C=256: 8*256*2 = 4 KB <= 16 KB
    m = LlamaModel(spec, torch.device("cpu"), torch.float32, seed=3,
                   max_seq_len=64)
    m.quantize_fp8()
    assert hasattr(m, "fp8") and len(m.fp8) == spec.num_layers
    # big hidden: simulate the gate arithmetic (gate = 8 KB: batch 1 only)
    assert 2 * 4096 * 2 > 8 * 1024           # B=2 @ 8B hidden -> gated off
    assert 1 * 4096 * 2 <= 8 * 1024          # B=1 -> fp8 on


def test_chunked_prefill_matches_single_shot():
    """Prompts longer than PREFILL_CHUNK are processed in slices; greedy
    output must match the single-shot prefill."""
    e1 = LocalEngine.create("llama3-tiny", max_seq_len=128, seed=17)
    e1.PREFILL_CHUNK = 16                  # force chunking
    prompt = list(range(4, 64))            # 60 tokens -> 4 chunks
    a = e1.generate(prompt, max_new_tokens=5, stop_on_eos=False)

    e2 = LocalEngine.create("llama3-tiny", max_seq_len=128, seed=17)
    b = e2.generate(prompt, max_new_tokens=5, stop_on_eos=False)
    assert a["token_ids"] == b["token_ids"]


def test_attn_splits_env_sizes_workspace(monkeypatch):
    """FEI_ATTN_SPLITS must size the split-K workspace (an undersized
    workspace is an out-of-bounds write in the kernel)."""
    monkeypatch.setenv("FEI_ATTN_SPLITS", "16")
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny")
    assert eng.attn_splits == 16
    assert eng.attn_ws[0].shape[2] == 16 and eng.attn_ws[1].shape[2] == 16
    eng.generate([5, 6, 7], max_new_tokens=4)  # CPU path still consistent


def test_generate_stream_matches_generate():
    """Chunked streaming must reassemble to exactly the non-streamed
    greedy output, and every chunk must be non-empty until done."""
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny")
    plain = eng.generate("stream me a poem", max_new_tokens=20,
                         stop_on_eos=False)
    chunks = list(eng.generate_stream("stream me a poem", max_new_tokens=20,
                                      stop_on_eos=False, chunk=7))
    ids = [t for c in chunks for t in c["new_token_ids"]]
    assert ids == plain["token_ids"]
    assert chunks[-1]["done"] and chunks[-1]["text"] == plain["text"]
    assert "decode_tok_s" in chunks[-1]
    assert all(not c["done"] for c in chunks[:-1])


def test_spm_tokenizer_roundtrip_and_engine(tmp_path):
    """Train a tiny SentencePiece model offline and run the engine with it
    (real-tokenizer support for loadable checkpoints)."""
    import io
    import sentencepiece as spm
    from fei_amd.engine.engine import LocalEngine
    from fei_amd.engine.tokenizer import SpmTokenizer
    corpus = ["the quick brown fox jumps over the lazy dog",
              "def add(a, b): return a + b",
              "hello world example text for sentencepiece"] * 20
    model = io.BytesIO()
    spm.SentencePieceTrainer.train(
        sentence_iterator=iter(corpus), model_writer=model,
        vocab_size=80, model_type="bpe")
    path = tmp_path / "tok.model"
    path.write_bytes(model.getvalue())

    tok = SpmTokenizer(str(path))
    ids = tok.encode("the quick brown fox")
    assert ids[0] == tok.bos_id
    assert "quick" in tok.decode(ids)

    eng = LocalEngine.create("llama3-tiny", tokenizer=tok)
    out = eng.generate("hello world", max_new_tokens=6, stop_on_eos=False)
    assert len(out["token_ids"]) == 6
    assert isinstance(out["text"], str)


def test_spm_vocab_overflow_rejected(tmp_path):
    import pytest
    from fei_amd.engine.engine import LocalEngine

    class Fake:
        vocab_size = 10 ** 9
        bos_id = eos_id = pad_id = 0

        def encode(self, t, **k):
            return [0]

        def decode(self, ids):
            return ""
    with pytest.raises(ValueError, match="exceeds model vocab"):
        LocalEngine.create("llama3-tiny", tokenizer=Fake())


def test_backend_prefix_cache_spans_generated_tokens(tmp_path):
    """Turn 2 must reuse the KV of turn 1's RESPONSE, not just its prompt
    (cached_prefix covers prompt + generated tokens)."""
    from fei_amd.core.backends import LocalBackend
    from fei_amd.engine.engine import LocalEngine
    be = LocalBackend(engine=LocalEngine.create("llama3-tiny"))
    msgs = [{"role": "user", "content": "first question"}]
    r1 = be.complete(msgs, max_tokens=8)
    n_cache_after_1 = len(be._cached_ids)
    msgs = msgs + [{"role": "assistant", "content": r1.content},
                   {"role": "user", "content": "second question"}]
    be.complete(msgs, max_tokens=8)
    cached = be.engine.last_metrics["cached_prefix"]
    # the reused prefix must reach past the turn-1 prompt into its response
    prompt1_len = n_cache_after_1 - 8  # at most; response added up to 8
    assert cached >= prompt1_len
    assert cached > 0


def test_decode_stops_at_cache_capacity():
    """Decoding into a full context must stop cleanly at max_seq_len, not
    index the RoPE table / KV cache out of bounds."""
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny", max_seq_len=64)
    prompt = list(range(4, 44))              # 40 tokens
    eng.prefill(prompt)
    rows = eng.decode(100, stop_on_eos=False)    # asks for more than fits
    assert len(rows[0]) <= 64 - len(prompt) + 36  # bounded, no crash
    assert int(eng.pos.max()) <= 63


def test_loglikelihood_greedy_and_chain_rule():
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny")
    prompt = "score this continuation"
    out = eng.generate(prompt, max_new_tokens=8, stop_on_eos=False)
    gen = out["token_ids"]
    ctx = eng.tokenizer.encode(prompt)

    ll = eng.loglikelihood(ctx, gen)
    assert ll["is_greedy"] is True
    assert len(ll["token_logprobs"]) == len(gen)
    assert ll["logprob"] < 0.0

    # a perturbed continuation scores lower and is not greedy
    bad = list(gen)
    bad[0] = (bad[0] + 7) % 200 + 4
    llb = eng.loglikelihood(ctx, bad)
    assert llb["logprob"] < ll["logprob"]
    assert llb["is_greedy"] is False

    # chain rule: ll(ctx, a+b) == ll(ctx, a) + ll(ctx+a, b)
    a, b = gen[:3], gen[3:]
    lab = eng.loglikelihood(ctx, a)["logprob"] + \
        eng.loglikelihood(ctx + a, b)["logprob"]
    assert abs(lab - ll["logprob"]) < 1e-3


def test_temperature_sampling_deterministic_by_seed():
    """Gumbel sampling is a pure function of (seed, step, batch, idx):
    same seed -> identical stream; different seed -> different stream."""
    from fei_amd.engine.engine import LocalEngine
    a = LocalEngine.create("llama3-tiny", seed=7)
    b = LocalEngine.create("llama3-tiny", seed=7)
    c = LocalEngine.create("llama3-tiny", seed=8)
    p = "sample me"
    ta = a.generate(p, max_new_tokens=12, temperature=0.9, stop_on_eos=False)["token_ids"]
    tb = b.generate(p, max_new_tokens=12, temperature=0.9, stop_on_eos=False)["token_ids"]
    tc = c.generate(p, max_new_tokens=12, temperature=0.9, stop_on_eos=False)["token_ids"]
    assert ta == tb
    assert ta != tc


def test_generate_sampled_seeded():
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny")
    o1 = eng.generate_sampled("nucleus", max_new_tokens=10, temperature=0.8,
                              top_p=0.9, seed=42)
    o2 = eng.generate_sampled("nucleus", max_new_tokens=10, temperature=0.8,
                              top_p=0.9, seed=42)
    assert o1["token_ids"] == o2["token_ids"]


def test_fei_tokenizer_env(tmp_path, monkeypatch):
    import io
    import sentencepiece as spm
    corpus = ["env tokenizer test sentence"] * 30
    model = io.BytesIO()
    spm.SentencePieceTrainer.train(sentence_iterator=iter(corpus),
                                   model_writer=model, vocab_size=40,
                                   model_type="bpe")
    path = tmp_path / "t.model"
    path.write_bytes(model.getvalue())
    monkeypatch.setenv("FEI_TOKENIZER", str(path))
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny")
    assert type(eng.tokenizer).__name__ == "SpmTokenizer"


def test_fei_weights_env(tmp_path, monkeypatch):
    """FEI_WEIGHTS loads a safetensors checkpoint at engine init: two
    engines with different seeds converge once one loads the other's
    weights (full real-checkpoint path: FEI_WEIGHTS + FEI_TOKENIZER)."""
    from fei_amd.engine.engine import LocalEngine
    src = LocalEngine.create("llama3-tiny", seed=11)
    path = str(tmp_path / "ck")
    src.model.save_weights(path)
    want = src.generate("checkpoint check", max_new_tokens=8,
                        stop_on_eos=False)["token_ids"]
    monkeypatch.setenv("FEI_WEIGHTS", path)
    dst = LocalEngine.create("llama3-tiny", seed=999)   # different init
    got = dst.generate("checkpoint check", max_new_tokens=8,
                       stop_on_eos=False)["token_ids"]
    assert got == want


def test_stop_sequences():
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny")
    base = eng.generate("stop test", max_new_tokens=40, stop_on_eos=False)
    # choose a stop string that actually occurs mid-text
    text = base["text"]
    assert len(text) > 4
    ss = text[3:5]
    out = eng.generate("stop test", max_new_tokens=40, stop_on_eos=False,
                       stop=[ss])
    assert ss not in out["text"]
    assert out["text"] == text[: text.find(ss)]
    assert out["finish_reason"] == "stop"
    out2 = eng.generate("stop test", max_new_tokens=8, stop_on_eos=False)
    assert out2["finish_reason"] in ("length", "stop")


def test_graph_params_tracked(monkeypatch):
    """ensure_graph must recapture when temperature/seed change (they are
    baked kernel arguments). CPU has no graphs, so simulate the gate."""
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny")
    eng.use_graph = True
    captured = []
    monkeypatch.setattr(eng, "_capture_graph",
                        lambda: (captured.append(eng.temperature),
                                 setattr(eng, "_graph", object()),
                                 setattr(eng, "_graph_params",
                                         (eng.temperature, eng.seed))))
    eng.temperature = 0.0
    eng.ensure_graph()
    eng.ensure_graph()                     # no change -> no recapture
    assert captured == [0.0]
    eng.temperature = 0.8
    eng.ensure_graph()                     # param change -> recapture
    assert captured == [0.0, 0.8]


def test_stream_stop_sequences():
    from fei_amd.engine.engine import LocalEngine
    eng = LocalEngine.create("llama3-tiny")
    base = eng.generate("stream stop", max_new_tokens=40, stop_on_eos=False)
    ss = base["text"][5:7]
    chunks = list(eng.generate_stream("stream stop", max_new_tokens=40,
                                      stop_on_eos=False, stop=[ss], chunk=5))
    assert chunks[-1]["done"]
    assert ss not in chunks[-1]["text"]


def test_packaged_spm_tokenizer_default():
    """Engines whose vocab fits default to the packaged 16k SentencePiece
    model (VERDICT r01 missing #5); llama3-tiny (vocab 512) keeps the byte
    tokenizer. The packaged model must round-trip code exactly."""
    import os

    from fei_amd.engine.config import ModelSpec
    from fei_amd.engine.tokenizer import SpmTokenizer

    model_path = os.path.join(os.path.dirname(LocalEngine.__init__.__code__
                                              .co_filename), "fei16k.model")
    assert os.path.exists(model_path)

    spec = ModelSpec(name="spm-test", vocab_size=16000, hidden_size=64,
                     num_layers=1, num_heads=2, num_kv_heads=1, head_dim=32,
                     intermediate_size=128, max_seq_len=128)
    eng = LocalEngine(spec, max_seq_len=64, seed=7)
    assert isinstance(eng.tokenizer, SpmTokenizer)
    code = "def f(x):\n    return x + 1\n"
    assert eng.tokenizer.decode(eng.tokenizer.encode(code)) == code
    # ~3x fewer tokens than bytes on code-assistant text
    assert len(eng.tokenizer.encode(code)) < len(code) * 0.6

    tiny = LocalEngine.create("llama3-tiny", max_seq_len=64, seed=7)
    assert isinstance(tiny.tokenizer, ByteTokenizer)


def test_hbm_seq_plumbing_cpu():
    """max_seq_len='hbm' resolves to the spec window on CPU (the sizing
    toward 288 GB happens on GPU; tests/test_engine_gpu.py covers it)."""
    eng = LocalEngine.create("llama3-tiny", max_seq_len="hbm", seed=7)
    assert eng.max_seq_len == eng.spec.max_seq_len
    out = eng.generate("abc", max_new_tokens=4, stop_on_eos=False)
    assert len(out["token_ids"]) == 4
