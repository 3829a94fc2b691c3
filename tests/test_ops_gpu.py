"""GPU numerics tests: HIP/CDNA4 kernels vs the plain fp32 PyTorch
reference (distegnn_amd/ops/reference.py)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from distegnn_amd import ops
    from distegnn_amd.ops import reference as R
else:  # collected but skipped on CPU boxes
    pytest.skip("requires GPU", allow_module_level=True)


def dev():
    return torch.device("cuda:0")


def test_extension_is_loaded():
    """On a GPU box the HIP extension must be the executing path."""
    ext = ops.hip_ext()
    assert ext is not None, "HIP extension missing on a GPU box"
    assert "_hip_ext" in ext.__file__


def make_csr(m, n, seed=0):
    g = torch.Generator().manual_seed(seed)
    row = torch.sort(torch.randint(0, n, (m,), generator=g)).values
    rowptr = torch.cat([torch.zeros(1, dtype=torch.long),
                        torch.cumsum(torch.bincount(row, minlength=n), 0)])
    return row, rowptr


@pytest.mark.parametrize("f", [3, 15, 64, 320])
@pytest.mark.parametrize("mean", [False, True])
def test_segment_reduce_fp32(f, mean):
    m, n = 20000, 1500
    row, rowptr = make_csr(m, n)
    data = torch.randn(m, f)
    want = (R.segment_mean if mean else R.segment_sum)(data, row, n)
    ext = ops.hip_ext()
    got = ext.segment_reduce_csr(data.to(dev()), rowptr.to(dev()), mean)
    assert torch.allclose(got.cpu(), want, atol=1e-4, rtol=1e-4)


def test_segment_reduce_bf16():
    m, n, f = 5000, 400, 64
    row, rowptr = make_csr(m, n, seed=1)
    data = torch.randn(m, f)
    want = R.segment_mean(data, row, n)
    got = ops.hip_ext().segment_reduce_csr(
        data.to(dev()).bfloat16(), rowptr.to(dev()), True)
    assert got.dtype == torch.bfloat16
    assert torch.allclose(got.float().cpu(), want, atol=0.05, rtol=0.05)


def test_segment_reduce_empty_segments():
    data = torch.ones(2, 8)
    rowptr = torch.tensor([0, 0, 2, 2, 2])
    got = ops.hip_ext().segment_reduce_csr(data.to(dev()), rowptr.to(dev()),
                                           True)
    assert torch.equal(got[0].cpu(), torch.zeros(8))
    assert torch.allclose(got[1].cpu(), torch.ones(8))
    assert torch.equal(got[3].cpu(), torch.zeros(8))


@pytest.mark.parametrize("f", [3, 64, 320])
def test_chunked_pool_matches(f):
    """Two-stage huge-segment path == reference pooling (113K-node graph)."""
    n, b = 50000, 2
    ptr = torch.tensor([0, 30000, 50000])
    batch = torch.repeat_interleave(torch.arange(b),
                                    torch.tensor([30000, 20000]))
    x = torch.randn(n, f)
    want = R.graph_mean_pool(x, batch, b)
    chunk = 256
    cb, ce, scp = [], [], [0]
    for i in range(b):
        for k in range(int(ptr[i]), int(ptr[i + 1]), chunk):
            cb.append(k)
            ce.append(min(k + chunk, int(ptr[i + 1])))
        scp.append(len(cb))
    got = ops.hip_ext().segment_reduce_chunked(
        x.to(dev()), ptr.to(dev()),
        torch.tensor(cb, dtype=torch.long, device=dev()),
        torch.tensor(ce, dtype=torch.long, device=dev()),
        torch.tensor(scp, dtype=torch.long, device=dev()), True)
    assert torch.allclose(got.cpu(), want, atol=1e-4, rtol=1e-4)


def test_chunked_pool_deterministic():
    n, f = 120000, 64
    ptr = torch.tensor([0, n])
    x = torch.randn(n, f)
    cb = list(range(0, n, 2048))
    ce = [min(k + 2048, n) for k in cb]
    scp = [0, len(cb)]
    args = (x.to(dev()), ptr.to(dev()),
            torch.tensor(cb, device=dev()), torch.tensor(ce, device=dev()),
            torch.tensor(scp, device=dev()))
    a = ops.hip_ext().segment_reduce_chunked(*args, True)
    b2 = ops.hip_ext().segment_reduce_chunked(*args, True)
    assert torch.equal(a, b2)


@pytest.mark.parametrize("n,r", [(1000, 0.15), (20000, 0.06)])
def test_radius_graph_matches_cpu(n, r):
    g = torch.Generator().manual_seed(2)
    pos = torch.rand(n, 3, generator=g)
    want = R.radius_graph(pos, r)
    ei, rowptr = ops.hip_ext().radius_graph_gpu(pos.to(dev()), r)
    got = ei.cpu()
    assert got.size(1) == want.size(1), "edge count mismatch"
    # same edge SET (order within a row may differ)
    gs = set(map(tuple, got.T.tolist()))
    ws = set(map(tuple, want.T.tolist()))
    assert gs == ws
    # row-sorted + consistent rowptr
    assert torch.all(got[0][1:] >= got[0][:-1])
    deg = torch.bincount(got[0], minlength=n)
    assert torch.equal(rowptr.cpu()[1:] - rowptr.cpu()[:-1], deg)


def test_radius_graph_dispatch_path():
    pos = torch.rand(500, 3, device=dev())
    ei = ops.radius_graph(pos, 0.2)
    assert ei.is_cuda and ei.size(0) == 2


def test_gather_rows_backward_row_sorted():
    """gather_rows backward (CSR segment sum) == index_select backward."""
    m, n, f = 4000, 300, 48
    row, rowptr = make_csr(m, n, seed=4)
    x = torch.randn(n, f, requires_grad=True)
    xg = x.detach().clone().to(dev()).requires_grad_(True)
    y = x.index_select(0, row)
    (y * torch.linspace(0.5, 1.5, m).unsqueeze(-1)).sum().backward()
    yg = ops.gather_rows(xg, row.to(dev()), rowptr.to(dev()))
    (yg * torch.linspace(0.5, 1.5, m, device=dev()).unsqueeze(-1)).sum() \
        .backward()
    assert torch.allclose(yg.cpu(), y, atol=1e-5)
    assert torch.allclose(xg.grad.cpu(), x.grad, atol=1e-4, rtol=1e-4)


def test_gather_rows_backward_unsorted_with_perm():
    m, n, f = 4000, 300, 16
    g = torch.Generator().manual_seed(5)
    col = torch.randint(0, n, (m,), generator=g)
    perm = torch.argsort(col, stable=True)
    colptr = torch.cat([torch.zeros(1, dtype=torch.long),
                        torch.cumsum(torch.bincount(col, minlength=n), 0)])
    x = torch.randn(n, f, requires_grad=True)
    xg = x.detach().clone().to(dev()).requires_grad_(True)
    w = torch.randn(m, 1, generator=g)
    (x.index_select(0, col) * w).sum().backward()
    (ops.gather_rows(xg, col.to(dev()), colptr.to(dev()),
                     perm=perm.to(dev())) * w.to(dev())).sum().backward()
    assert torch.allclose(xg.grad.cpu(), x.grad, atol=1e-4, rtol=1e-4)


def test_segment_autograd_gpu():
    """ops.segment_mean custom Function: forward HIP, backward gather."""
    m, n, f = 3000, 200, 32
    row, rowptr = make_csr(m, n, seed=3)
    data = torch.randn(m, f, requires_grad=True)
    data_g = data.detach().clone().to(dev()).requires_grad_(True)
    out = R.segment_mean(data, row, n)
    out.pow(2).sum().backward()
    out_g = ops.segment_mean(data_g, row.to(dev()), n, rowptr=rowptr.to(dev()))
    out_g.pow(2).sum().backward()
    assert torch.allclose(out_g.cpu(), out, atol=1e-4, rtol=1e-4)
    assert torch.allclose(data_g.grad.cpu(), data.grad, atol=1e-4, rtol=1e-4)


def test_mid_reduce_expand():
    from distegnn_amd.ops import mid_mean, hip_ext
    x = torch.randn(500, 5, 64, device="cuda:0", dtype=torch.bfloat16,
                    requires_grad=True)
    out = mid_mean(x)
    ref = x.detach().float().mean(dim=1).to(torch.bfloat16)
    assert torch.allclose(out.float(), ref.float(), atol=2e-2)
    g = torch.randn_like(out)
    out.backward(g)
    gref = (g.float() / 5).unsqueeze(1).expand(-1, 5, -1)
    assert torch.allclose(x.grad.float(), gref, atol=2e-2)
    ext = hip_ext()
    y = torch.randn(300, 3, 3, device="cuda:0")
    assert torch.allclose(ext.mid_reduce(y, -1.0), -y.sum(1), atol=1e-5)


def test_gather_rows_fast_matches_index_select():
    from distegnn_amd.ops import hip_ext
    ext = hip_ext()
    for dtype, f in [(torch.bfloat16, 64), (torch.float32, 3),
                     (torch.float32, 15)]:
        x = torch.randn(1000, f, device="cuda:0", dtype=dtype)
        idx = torch.randint(0, 1000, (5000,), device="cuda:0")
        assert torch.equal(ext.gather_rows_fast(x, idx),
                           x.index_select(0, idx))


def test_segment_reduce_perm():
    from distegnn_amd.ops import hip_ext
    ext = hip_ext()
    m, n, f = 4000, 100, 64
    data = torch.randn(m, f, device="cuda:0", dtype=torch.bfloat16)
    seg = torch.sort(torch.randint(0, n, (m,), device="cuda:0")).values
    rowptr = torch.searchsorted(
        seg, torch.arange(n + 1, device="cuda:0"), right=False)
    rowptr[-1] = m
    perm = torch.randperm(m, device="cuda:0")
    out = ext.segment_reduce_csr_perm(data, rowptr, perm, False)
    ref = ext.segment_reduce_csr(data.index_select(0, perm), rowptr, False)
    assert torch.allclose(out.float(), ref.float(), atol=1e-2, rtol=1e-2)


def test_coord_update_matches_eager():
    """Fused coord+agg+trans_v+phiv*vel == the eager chain, fwd+bwd."""
    torch.manual_seed(0)
    n = 5000
    dev = "cuda:0"
    mk = lambda *s: torch.randn(*s, device=dev)

    coord, agg, tv, vel = mk(n, 3), mk(n, 3), mk(n, 3), mk(n, 3)
    phiv = mk(n, 1)
    leaves_f = [t.clone().requires_grad_(True)
                for t in (coord, agg, tv, phiv)]
    out_f = ops.coord_update(leaves_f[0], leaves_f[1], leaves_f[2],
                             leaves_f[3], vel)
    leaves_e = [t.clone().requires_grad_(True)
                for t in (coord, agg, tv, phiv)]
    out_e = (leaves_e[0] + leaves_e[1] + leaves_e[2]
             + leaves_e[3] * vel)
    assert torch.allclose(out_f, out_e, atol=1e-6)
    g = mk(n, 3)
    out_f.backward(g)
    out_e.backward(g)
    for a, b in zip(leaves_f, leaves_e):
        assert torch.allclose(a.grad, b.grad, atol=1e-6)


@pytest.mark.parametrize("f", [64, 128])
def test_cfconv_fused_matches_eager(f):
    """Fused CFConv messages (smearing+filter MLP+cutoff+gather mul) ==
    the eager composition, fwd + bwd."""
    import math

    import torch.nn.functional as F

    torch.manual_seed(0)
    dev_ = "cuda:0"
    n, m, g = 500, 4000, 50
    cutoff = 10.0
    x = (torch.randn(n, f, device=dev_) * 0.5).bfloat16()
    row = torch.sort(torch.randint(0, n, (m,), device=dev_)).values
    col = torch.randint(0, n, (m,), device=dev_)
    dist = torch.rand(m, device=dev_) * cutoff
    # CSR metadata
    rowptr = torch.zeros(n + 1, dtype=torch.long, device=dev_)
    rowptr.scatter_add_(0, row + 1, torch.ones_like(row))
    rowptr = rowptr.cumsum(0)
    col_perm = torch.argsort(col, stable=True)
    colptr = torch.zeros(n + 1, dtype=torch.long, device=dev_)
    colptr.scatter_add_(0, col[col_perm] + 1, torch.ones_like(col))
    colptr = colptr.cumsum(0)
    offsets = torch.linspace(0.0, cutoff, g, device=dev_)
    coeff = -0.5 / float(offsets[1] - offsets[0]) ** 2
    w1 = torch.randn(f, g, device=dev_) * 0.2
    b1 = torch.randn(f, device=dev_) * 0.1
    w2 = torch.randn(f, f, device=dev_) * 0.1
    b2 = torch.randn(f, device=dev_) * 0.1

    def eager(leaves):
        lx, lw1, lb1, lw2, lb2 = leaves
        d = dist.view(-1, 1) - offsets.view(1, -1)
        gauss = torch.exp(coeff * d.pow(2)).bfloat16()
        c = 0.5 * (torch.cos(dist * math.pi / cutoff) + 1.0)
        z1 = (F.softplus(F.linear(gauss, lw1.bfloat16(), lb1.bfloat16()))
              - math.log(2.0))
        w = F.linear(z1.bfloat16(), lw2.bfloat16(), lb2.bfloat16())
        xj = ops.gather_rows(lx, col, colptr, col_perm)
        return xj * w * c.view(-1, 1).bfloat16()

    leaves_e = [t.clone().requires_grad_(True) for t in (x, w1, b1, w2, b2)]
    msg_e = eager(leaves_e)
    leaves_f = [t.clone().requires_grad_(True) for t in (x, w1, b1, w2, b2)]
    msg_f = ops.cfconv_msg(leaves_f[0], dist, row, col, colptr, col_perm,
                           leaves_f[1], leaves_f[2], leaves_f[3],
                           leaves_f[4], offsets, coeff, cutoff)
    assert msg_f.shape == (m, f)
    assert torch.allclose(msg_f.float(), msg_e.float(), atol=0.05,
                          rtol=0.05), (msg_f.float() - msg_e.float()
                                       ).abs().max()
    gout = torch.randn_like(msg_e)
    msg_e.backward(gout)
    msg_f.backward(gout)
    for i, (a, b) in enumerate(zip(leaves_f, leaves_e)):
        assert torch.allclose(a.grad.float(), b.grad.float(), atol=0.1,
                              rtol=0.1), (i, (a.grad.float()
                                              - b.grad.float()).abs().max())


def test_spmm_adj_gpu_matches_dense():
    """GPU CSR SpMM path (sort + gather + deterministic segment sum)."""
    torch.manual_seed(0)
    n, m, f = 300, 5000, 16
    dev_ = "cuda:0"
    ei = torch.randint(0, n, (2, m), device=dev_)
    dense = torch.randn(n, f, device=dev_, requires_grad=True)
    out = ops.spmm_adj(ei, n, dense)
    adj = torch.zeros(n, n, device=dev_)
    adj.index_put_((ei[0], ei[1]), torch.ones(m, device=dev_),
                   accumulate=True)
    want = adj @ dense
    assert torch.allclose(out, want, atol=1e-3, rtol=1e-4)
    g = torch.randn(n, f, device=dev_)
    out.backward(g)
    assert torch.allclose(dense.grad, adj.t() @ g, atol=1e-3, rtol=1e-4)


@pytest.mark.parametrize("h", [1, 4, 8])
def test_edge_softmax_fused_matches_composition(h):
    """Fused CSR edge_softmax == the scatter/exp/segment composition,
    fwd + bwd (SE(3) attention)."""
    from distegnn_amd.models.se3.graph import EdgeGraph
    from distegnn_amd.models.se3.modules import edge_softmax

    torch.manual_seed(0)
    dev_ = "cuda:0"
    n, m = 300, 4000
    ei = torch.stack([torch.randint(0, n, (m,), device=dev_),
                      torch.randint(0, n, (m,), device=dev_)])
    G = EdgeGraph(ei, n)
    scores = (torch.randn(m, h, device=dev_) * 2).requires_grad_(True)
    attn_f = edge_softmax(scores, G.dst, n, csr=G.dst_csr())
    scores_e = scores.detach().clone().requires_grad_(True)
    attn_e = edge_softmax(scores_e, G.dst, n, csr=None)
    assert torch.allclose(attn_f, attn_e, atol=1e-5, rtol=1e-4), \
        (attn_f - attn_e).abs().max()
    # per-segment normalization
    seg = torch.zeros(n, h, device=dev_)
    seg.index_add_(0, G.dst, attn_f.detach())
    covered = torch.bincount(G.dst, minlength=n) > 0
    assert torch.allclose(seg[covered],
                          torch.ones_like(seg[covered]), atol=1e-4)
    g = torch.randn(m, h, device=dev_)
    attn_f.backward(g)
    attn_e.backward(g)
    assert torch.allclose(scores.grad, scores_e.grad, atol=1e-4,
                          rtol=1e-3), (scores.grad - scores_e.grad
                                       ).abs().max()
