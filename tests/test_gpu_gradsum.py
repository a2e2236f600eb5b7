"""K6 grad_colsum numerics (GPU) vs fp32 torch reference."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from mapreduce_amd import ops
    ops.require_gpu_ext()
    return torch.device("cuda:0")


@pytest.mark.parametrize("G,D", [(1, 4), (8, 1024), (32, 100_003),
                                 (5, 7), (64, 65536)])
@pytest.mark.parametrize("mfma", [False, True])
def test_grad_colsum(dev, G, D, mfma):
    from mapreduce_amd import ops
    torch.manual_seed(G * 1000 + D)
    grads = torch.randn(G, D, device=dev, dtype=torch.float32)
    out = ops.ext().grad_colsum(grads, mfma)
    ref = grads.sum(0)
    # both paths are f32 sums in row order; tolerance for assoc order diff
    assert torch.allclose(out, ref, rtol=1e-5, atol=1e-5), (G, D, mfma)


def test_grad_colsum_mfma_exactness_vs_sequential(dev):
    """The f32-in MFMA is a k-ordered fmaf chain (guide §3): summing in
    row order must be bitwise-equal to a sequential f32 accumulation."""
    from mapreduce_amd import ops
    torch.manual_seed(7)
    G, D = 16, 256
    grads = torch.randn(G, D, device=dev, dtype=torch.float32)
    out = ops.ext().grad_colsum(grads, True)
    acc = torch.zeros(D, device=dev)
    for g in range(G):
        acc = acc + grads[g]
    assert torch.equal(out, acc)


def test_local_shard_sum_wrapper(dev):
    from mapreduce_amd.gpu.gradsum import local_shard_sum
    grads = torch.randn(12, 4096, device=dev)
    assert torch.allclose(local_shard_sum(grads), grads.sum(0), atol=1e-5)
