"""Tensor-parallel correctness on CPU: TP=2 over gloo must reproduce the
TP=1 logits (same seed => identical weights, sharded)."""

import json
import os
import subprocess
import sys

import pytest
import torch

WORKER = r"""
import os, json, torch
from fei_amd.engine.config import get_spec
from fei_amd.models.llama import LlamaModel
from fei_amd.parallel.pg import init_from_env

ctx = init_from_env(backend="gloo")
spec = get_spec("llama3-tiny")
model = LlamaModel(spec, torch.device("cpu"), torch.float32, tp=ctx, seed=11,
                   max_seq_len=64)
kc, vc = model.new_kv_cache(1, 64)
tokens = torch.tensor([[1, 5, 9, 13, 21]])
pos0 = torch.zeros(1, dtype=torch.int32)
logits = model.forward_prefill(tokens, pos0, kc, vc)
# then one decode step
tok = torch.tensor([int(logits[0].argmax())], dtype=torch.int32)
pos = torch.full((1,), tokens.shape[1], dtype=torch.int32)
logits2 = model.forward_decode(tok, pos, kc, vc, attn_splits=2)
if ctx.rank == 0:
    out = {"prefill": logits[0, :8].tolist(),
           "argmax": int(logits[0].argmax()),
           "decode": logits2[0, :8].tolist(),
           "argmax2": int(logits2[0].argmax())}
    with open(os.environ["TP_OUT"], "w") as f:
        json.dump(out, f)
"""


def _single_process_reference():
    from fei_amd.engine.config import get_spec
    from fei_amd.models.llama import LlamaModel

    spec = get_spec("llama3-tiny")
    model = LlamaModel(spec, torch.device("cpu"), torch.float32, seed=11,
                       max_seq_len=64)
    kc, vc = model.new_kv_cache(1, 64)
    tokens = torch.tensor([[1, 5, 9, 13, 21]])
    pos0 = torch.zeros(1, dtype=torch.int32)
    logits = model.forward_prefill(tokens, pos0, kc, vc)
    tok = torch.tensor([int(logits[0].argmax())], dtype=torch.int32)
    pos = torch.full((1,), tokens.shape[1], dtype=torch.int32)
    logits2 = model.forward_decode(tok, pos, kc, vc, attn_splits=2)
    return logits, logits2


def test_tp2_matches_tp1(tmp_path):
    out_file = tmp_path / "tp_out.json"
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env["TP_OUT"] = str(out_file)
    env["MASTER_ADDR"] = "127.0.0.1"
    env.setdefault("PYTHONPATH", os.getcwd())
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", str(script)],
        env=env, capture_output=True, text=True, timeout=240,
    )
    assert proc.returncode == 0, proc.stderr[-3000:]
    data = json.loads(out_file.read_text())

    logits, logits2 = _single_process_reference()
    ref_prefill = logits[0, :8].tolist()
    ref_decode = logits2[0, :8].tolist()
    for a, b in zip(data["prefill"], ref_prefill):
        assert abs(a - b) < 1e-3, (a, b)
    for a, b in zip(data["decode"], ref_decode):
        assert abs(a - b) < 1e-3, (a, b)
    assert data["argmax"] == int(logits[0].argmax())
    assert data["argmax2"] == int(logits2[0].argmax())
