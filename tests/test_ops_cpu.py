"""CPU numerics tests for the reference op implementations.

These define the semantics the HIP kernels are tested against in
tests/test_ops_gpu.py."""

import numpy as np
import pytest
import torch

from distegnn_amd.ops import reference as R


def manual_segment(data, row, n, mean):
    out = torch.zeros(n, data.size(1), dtype=data.dtype)
    cnt = torch.zeros(n)
    for e in range(row.numel()):
        out[row[e]] += data[e]
        cnt[row[e]] += 1
    if mean:
        out = out / cnt.clamp(min=1).unsqueeze(-1)
    return out


@pytest.mark.parametrize("mean", [False, True])
def test_segment_reduce(mean):
    g = torch.Generator().manual_seed(0)
    m, n, f = 500, 40, 7
    row = torch.randint(0, n, (m,), generator=g)
    data = torch.randn(m, f, generator=g)
    fn = R.segment_mean if mean else R.segment_sum
    out = fn(data, row, n)
    ref = manual_segment(data, row, n, mean)
    assert torch.allclose(out, ref, atol=1e-5)


def test_segment_empty_segments_zero():
    data = torch.ones(2, 3)
    row = torch.tensor([1, 1])
    out = R.segment_mean(data, row, 4)
    assert torch.equal(out[0], torch.zeros(3))
    assert torch.allclose(out[1], torch.ones(3))


@pytest.mark.parametrize("mean", [False, True])
def test_graph_pool(mean):
    g = torch.Generator().manual_seed(1)
    n, b, f = 100, 4, 5
    sizes = [20, 30, 25, 25]
    batch = torch.repeat_interleave(torch.arange(b), torch.tensor(sizes))
    x = torch.randn(n, f, generator=g)
    fn = R.graph_mean_pool if mean else R.graph_sum_pool
    out = fn(x, batch, b)
    for i in range(b):
        blk = x[batch == i]
        ref = blk.mean(0) if mean else blk.sum(0)
        assert torch.allclose(out[i], ref, atol=1e-5)


def test_radius_graph_vs_bruteforce():
    g = torch.Generator().manual_seed(2)
    pos = torch.rand(200, 3, generator=g)
    r = 0.2
    ei = R.radius_graph(pos, r)
    # brute force
    d = torch.cdist(pos, pos)
    mask = (d <= r) & ~torch.eye(200, dtype=torch.bool)
    expect = mask.nonzero().T
    got = set(map(tuple, ei.T.tolist()))
    want = set(map(tuple, expect.T.tolist()))
    assert got == want
    # row-sorted
    assert torch.all(ei[0][1:] >= ei[0][:-1])


def test_radius_graph_full():
    pos = torch.rand(10, 3)
    ei = R.radius_graph(pos, -1)
    assert ei.size(1) == 90
    assert torch.all(ei[0] != ei[1])


def test_spmm_adj_matches_dense():
    """CSR SpMM composition == dense adjacency matmul (K3), fwd + grad."""
    from distegnn_amd import ops

    torch.manual_seed(0)
    n, m, f = 30, 120, 5
    ei = torch.randint(0, n, (2, m))
    dense = torch.randn(n, f, requires_grad=True)
    out = ops.spmm_adj(ei, n, dense)
    adj = torch.zeros(n, n)
    adj.index_put_((ei[0], ei[1]), torch.ones(m), accumulate=True)
    want = adj @ dense
    assert torch.allclose(out, want, atol=1e-5)
    g = torch.randn(n, f)
    out.backward(g)
    dense2 = dense.detach().clone().requires_grad_(True)
    (adj @ dense2).backward(g)
    assert torch.allclose(dense.grad, dense2.grad, atol=1e-5)
