"""RCCL smoke on one GPU: world-size-1 process group over the nccl(=RCCL)
backend, collectives inside a hipGraph capture — the documented evidence
 that the TP engine's graph-captured collective path
is exercisable the day a multi-GPU node appears."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_rccl_world1_collectives_graph_capture():
    import torch.distributed as dist

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29551")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        dev = torch.device("cuda:0")
        torch.cuda.set_device(dev)
        x = torch.ones(4096, device=dev)
        shard = torch.full((8, 2), 3.0, device=dev)
        gathered = torch.zeros(1, 8, 2, device=dev)
        # warmup (RCCL communicator init outside capture)
        dist.all_reduce(x)
        dist.all_gather(list(gathered.unbind(0)), shard)
        torch.cuda.synchronize()

        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            dist.all_reduce(x)
            dist.all_gather(list(gathered.unbind(0)), shard)
        x.fill_(2.0)
        shard.fill_(7.0)
        g.replay()
        torch.cuda.synchronize()
        assert float(x[0]) == 2.0          # world 1: all_reduce = identity
        assert float(gathered[0, 0, 0]) == 7.0
    finally:
        dist.destroy_process_group()


def test_tp_context_world1_engine_eager():
    """A LocalEngine handed a world-size-1 context takes the plain path
    (is_distributed False) — the TP branch only activates at world>1."""
    from fei_amd.engine.engine import LocalEngine
    from fei_amd.parallel.pg import ParallelContext

    eng = LocalEngine.create("llama3-tiny", max_seq_len=128, seed=7,
                             tp=ParallelContext())
    out = eng.generate("abc", max_new_tokens=4, stop_on_eos=False)
    assert len(out["token_ids"]) == 4
