"""Gradient-training task on GPU: the full MapReduce loop (host-tier
scheduler, GPU model compute) + the RCCL-path gradsum utility."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_train_digits_on_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import importlib

    import mapreduce_amd.examples.train_digits as td
    from mapreduce_amd import job as jobmod, run_local
    importlib.reload(td)
    jobmod._module_cache.clear()
    jobmod._inited.clear()
    fns = {r: td for r in ("taskfn", "mapfn", "partitionfn", "reducefn",
                           "combinerfn", "finalfn")}
    srv = run_local({"fns": fns, "verbose": False,
                     "init_args": {"shards": 3, "iters": 2, "lr": 0.05,
                                   "device": "cuda:0"}},
                    nworkers=2)
    assert srv.finished
    assert len(td.STATE["losses"]) == 2
    assert all(x == x for x in td.STATE["losses"])
    # the model really lives on GPU
    assert next(td.STATE["model"].parameters()).is_cuda


def test_allreduce_gradients_single_rank_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from mapreduce_amd.gpu.gradsum import allreduce_gradients
    g = {"a": torch.randn(100, device="cuda"),
         "b": torch.randn(3, 7, device="cuda")}
    out = allreduce_gradients(g)
    for k in g:
        assert torch.equal(out[k], g[k])
        assert out[k].shape == g[k].shape
