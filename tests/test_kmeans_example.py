"""examples/kmeans: iterative MR clustering (the train_digits loop pattern
on a second model family).  Inertia must be non-increasing (fixed per-shard
data + exact global centroid update), and centroids must land near the
true blob centers."""

import importlib

import pytest
import torch

import mapreduce_amd.examples.kmeans as km
from mapreduce_amd.runner import run_local


def _fresh(cfg=None):
    importlib.reload(km)
    km.init(dict({"shards": 4, "k": 6, "dims": 8, "points": 2000,
                  "iters": 6}, **(cfg or {})))
    return km


def _fns(mod):
    return {"fns": {r: mod for r in ("taskfn", "mapfn", "partitionfn",
                                     "reducefn", "combinerfn", "finalfn")}}


def _match_centers(got: torch.Tensor, true: torch.Tensor) -> float:
    """Greedy max distance from each true center to its nearest learned
    centroid (k small; assignment-free check)."""
    d = torch.cdist(true, got)
    return float(d.min(dim=1).values.max())


def test_kmeans_loop_converges():
    mod = _fresh()
    run_local(_fns(mod))
    assert mod.STATE["iteration"] == 6
    assert len(mod.STATE["inertia"]) == 6
    # non-increasing inertia (allow fp noise)
    for a, b in zip(mod.STATE["inertia"], mod.STATE["inertia"][1:]):
        assert b <= a * (1 + 1e-6)
    # clusters found: every true center has a learned centroid nearby
    # (blobs have unit noise; centers are spread with sigma 5)
    assert _match_centers(mod.STATE["centroids"].cpu(),
                          mod.true_centers()) < 1.0
    # substantial improvement over the Forgy start
    assert mod.STATE["inertia"][-1] < 0.5 * mod.STATE["inertia"][0]


def test_kmeans_worker_count_invariance():
    mod = _fresh()
    run_local(_fns(mod), nworkers=1)
    one = [round(x, 3) for x in mod.STATE["inertia"]]
    c1 = mod.STATE["centroids"].clone()
    mod = _fresh()
    run_local(_fns(mod), nworkers=4)
    four = [round(x, 3) for x in mod.STATE["inertia"]]
    assert one == four
    assert torch.allclose(c1, mod.STATE["centroids"], atol=1e-4)


@pytest.mark.gpu
def test_kmeans_on_gpu_device():
    """Device-aware mapfn: distance argmin + partials on cuda:0; results
    must match the CPU run bit-for-bit at fp32 tolerance."""
    mod = _fresh()
    run_local(_fns(mod))
    cpu_inertia = mod.STATE["inertia"]
    cpu_c = mod.STATE["centroids"].cpu()
    mod = _fresh({"device": "cuda"})
    run_local(_fns(mod))
    assert mod.STATE["centroids"].device.type == "cuda"
    for a, b in zip(cpu_inertia, mod.STATE["inertia"]):
        assert abs(a - b) / max(abs(a), 1) < 1e-4
    assert torch.allclose(cpu_c, mod.STATE["centroids"].cpu(), atol=1e-3)
