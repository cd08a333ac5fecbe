"""Persistent weight-streaming decode engine (ops/csrc/stream_layer.hip)
vs the launch path — the launch kernels are the correctness oracle
(docs/DESIGN_persistent_decode.md: validate each fused stage against its
unfused kernel). 8B shapes (the engine is shape-guarded to them)."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng8b():
    from fei_amd.engine.engine import LocalEngine

    return LocalEngine.create("llama3-8b", max_seq_len=512, seed=7,
                              use_hip_graph=False)


def test_stream_layer_check_accepts_8b():
    from fei_amd import ops
    from fei_amd.engine.config import get_spec

    s = get_spec("llama3-8b")
    assert ops.stream_layer_check(s.hidden_size, s.num_heads,
                                  s.num_kv_heads, s.head_dim,
                                  s.intermediate_size) == 0
    t = get_spec("llama3-tiny")
    assert ops.stream_layer_check(t.hidden_size, t.num_heads,
                                  t.num_kv_heads, t.head_dim,
                                  t.intermediate_size) != 0


def test_stream_step_matches_launch_path(eng8b):
    """One decode step: stream-engine logits vs launch-path logits from
    IDENTICAL state (cloned KV caches), plus the appended KV rows."""
    from fei_amd import ops

    eng = eng8b
    prompt = eng.tokenizer.encode("the quick brown fox jumps over")
    eng.temperature = 0.0
    eng.prefill(prompt)
    model = eng.model
    spec = eng.spec

    kc2 = [k.clone() for k in eng.k_caches]
    vc2 = [v.clone() for v in eng.v_caches]
    ws = ops.stream_workspace(spec, eng.device)
    bufs = (torch.zeros(1, spec.hidden_size, dtype=eng.dtype,
                        device=eng.device),
            torch.zeros(1, spec.hidden_size, dtype=eng.dtype,
                        device=eng.device))

    tok = eng.token.clone()
    pos = eng.pos.clone()
    logits_stream = model.forward_decode_stream(tok, pos, kc2, vc2, ws,
                                                bufs).float()
    torch.cuda.synchronize()
    assert int(ws["fail"][0]) == 0, f"stream gave up: {int(ws['fail'][0])}"

    logits_launch = model.forward_decode(
        eng.token.clone(), eng.pos.clone(), eng.k_caches, eng.v_caches,
        attn_splits=eng.attn_splits, workspace=eng.attn_ws,
        fused_attn=False, attn_out=eng.attn_out, fused_norm=True).float()
    torch.cuda.synchronize()

    # both are bf16 chains with different reduction orders; compare tight
    cos = torch.nn.functional.cosine_similarity(
        logits_stream.flatten(), logits_launch.flatten(), dim=0)
    diff = (logits_stream - logits_launch).abs().max()
    scale = logits_launch.abs().max().clamp_min(1e-6)
    assert float(cos) > 0.9995, float(cos)
    assert float(diff / scale) < 0.05, (float(diff), float(scale))
    assert int(logits_stream.argmax()) == int(logits_launch.argmax())

    # the appended KV row at position pos must match the launch path's
    p = int(eng.pos[0])      # un-advanced here (advance happens in sampler)
    # layer 0: same inputs -> tight; layer 31: 31 layers of accumulated
    # bf16 reduce-order drift -> compare by cosine (r2c3 measured ~1-2%
    # elementwise drift at matched cos>0.9999)
    a = kc2[0][0, :, p, :].float()
    b = eng.k_caches[0][0, :, p, :].float()
    assert torch.allclose(a, b, atol=2e-2, rtol=2e-2)
    li = len(model.layers) - 1
    a = kc2[li][0, :, p, :].float().flatten()
    b = eng.k_caches[li][0, :, p, :].float().flatten()
    cos_kv = torch.nn.functional.cosine_similarity(a, b, dim=0)
    assert float(cos_kv) > 0.999, float(cos_kv)
    av = vc2[0][0, :, p, :].float()
    bv = eng.v_caches[0][0, :, p, :].float()
    assert torch.equal(av, bv)


def test_stream_engine_generates(eng8b):
    """End-to-end: a stream-decode engine (env-gated path incl. hipGraph
    capture) generates a full greedy sequence and the native kernels ran
    (fail word clean, pos advanced)."""
    from fei_amd.engine.engine import LocalEngine

    os.environ["FEI_STREAM_DECODE"] = "1"
    try:
        eng = LocalEngine.create("llama3-8b", max_seq_len=512, seed=7,
                                 use_hip_graph=True)
        assert eng.stream_decode
        out = eng.generate("def main():", max_new_tokens=24,
                           stop_on_eos=False)
        assert len(out["token_ids"]) == 24
        assert int(eng._stream_ws["fail"][0]) == 0
        # same prompt on the launch path: greedy tokens should agree for
        # at least the first steps (bf16 reduce-order differences can
        # eventually diverge a random-init model's near-uniform logits)
        ref = eng8b.generate("def main():", max_new_tokens=24,
                             stop_on_eos=False)
        assert out["token_ids"][:4] == ref["token_ids"][:4]
        eng.shutdown()
    finally:
        os.environ.pop("FEI_STREAM_DECODE", None)
