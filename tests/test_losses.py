"""MMD loss: the vectorized device implementation vs a direct per-graph
loop transcription of the reference formula (utils/train.py:119-147)."""

import torch

from distegnn_amd.data.graph import collate
from distegnn_amd.data.synthetic import make_cutoff_dataset
from distegnn_amd.runtime.losses import mmd_loss, sample_nodes_per_graph


def loop_mmd(vloc_bc3, target, batch, sigma, samples_per_channel,
             sampled_idx):
    """Reference math with externally fixed sample indices."""
    b, c, _ = vloc_bc3.shape
    num_sample = samples_per_channel * c
    l_vv = l_rv = 0.0
    for i in range(b):
        v = vloc_bc3[i]
        s = target[sampled_idx[i]]
        d_vv = torch.cdist(v, v, p=2)
        d_rv = torch.cdist(s, v, p=2)
        l_vv = l_vv + torch.exp(-d_vv / (2 * sigma * sigma)).sum()
        l_rv = l_rv + torch.exp(-d_rv / (2 * sigma * sigma)).sum()
    l_vv = l_vv / b / c / c
    l_rv = 2 * l_rv / b / num_sample / c
    return l_vv - l_rv


def test_sampling_valid():
    b = collate(make_cutoff_dataset("nbody_100", 4, seed=0))
    torch.manual_seed(0)
    samples, valid = sample_nodes_per_graph(b.target, b.batch, b.ptr,
                                            b.counts, 9)
    assert samples.shape == (4, 9, 3)
    assert valid.all()  # 100 nodes >= 9 samples


def test_sampling_without_replacement_and_in_graph():
    b = collate(make_cutoff_dataset("nbody_100", 3, seed=1))
    torch.manual_seed(1)
    n_s = 50
    keys = torch.rand(b.num_nodes) + 2.0 * b.batch.float()
    perm = torch.argsort(keys)
    # the permutation groups nodes by graph
    assert torch.equal(b.batch[perm],
                       torch.repeat_interleave(torch.arange(3),
                                               torch.tensor([100] * 3)))


def test_mmd_matches_loop():
    """Same distribution machinery: fix the sampled indices and compare the
    vectorized kernel against the loop."""
    b = collate(make_cutoff_dataset("nbody_100", 4, seed=2))
    torch.manual_seed(3)
    vloc = torch.randn(4, 3, 3)  # [B, C, 3]
    sigma, spc = 1.5, 3
    # monkey-path-free: compute vectorized with a fixed torch seed, then
    # reconstruct the same sample set by replaying the keyed argsort
    torch.manual_seed(42)
    got = mmd_loss(vloc, b.target, b.batch, b.ptr, b.counts, sigma, spc)
    torch.manual_seed(42)
    keys = torch.rand(b.num_nodes) + 2.0 * b.batch.float()
    perm = torch.argsort(keys)
    num_sample = spc * vloc.size(1)
    sampled_idx = [perm[b.ptr[i]:b.ptr[i] + num_sample] for i in range(4)]
    want = loop_mmd(vloc, b.target, b.batch, sigma, spc, sampled_idx)
    assert torch.allclose(got, want, atol=1e-6)


def test_mmd_small_graph_truncation():
    """Graphs smaller than num_sample contribute all their nodes."""
    from distegnn_amd.data.graph import Data

    pos = torch.randn(5, 3)
    d = Data(x=torch.randn(5, 2), pos=pos, vel=torch.randn(5, 3),
             attr=torch.randn(5, 1), target=pos.clone(),
             loc_mean=pos.mean(0, keepdim=True),
             edge_index=torch.tensor([[0, 1], [1, 0]]),
             edge_attr=torch.zeros(2, 2))
    b = collate([d])
    vloc = torch.randn(1, 3, 3)
    out = mmd_loss(vloc, b.target, b.batch, b.ptr, b.counts, 1.0, 10)
    assert torch.isfinite(out)
