"""Split-K aware Linear for tall-skinny activations.

hipBLASLt's heuristic picks a non-split-K kernel for weight-gradient GEMMs
of shape [64..144] x [M ~ 10^6] (K = M deep): ~6 workgroups on 256 CUs,
measured 2.9 ms where the roofline is ~0.1 ms. ``chunked_wgrad`` performs
the split-K manually as a batched GEMM over row chunks + a partial sum
(measured 0.23 ms on [1.65M,64]^T @ [1.65M,144]).

``SplitKLinear`` is a drop-in ``nn.Linear`` (same state-dict keys) whose
backward uses the chunked weight gradient for large inputs.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

_CHUNKS = 64
_MIN_ROWS = 1 << 16


def chunked_wgrad(g: torch.Tensor, x: torch.Tensor,
                  nc: int = _CHUNKS) -> torch.Tensor:
    """g [M, O], x [M, I] -> g^T @ x [O, I], split-K.

    Primary path: the hand-written MFMA split-K kernel (csrc/wgrad.hip) for
    the standard O=64 bf16 case. (A torch.bmm split-K was tried first and
    costs 2.6 ms of HOST time per call in hipBLASLt's batched heuristic.)"""
    from . import hip_ext

    ext = hip_ext()
    if (ext is not None and g.is_cuda and g.dtype == torch.bfloat16
            and g.size(1) == 64 and x.size(1) <= 208
            and x.dtype == torch.bfloat16):
        return ext.wgrad_splitk(g, x)
    if g.size(1) == 1:
        # head wgrad [1,M]x[M,I]: a weighted column sum — the library's
        # MT64x16x512 pick costs ~0.5 ms; mul+reduce is ~0.1 ms
        return (g * x).sum(0, keepdim=True, dtype=torch.float32).to(g.dtype)
    if g.size(1) < 16:
        return torch.mm(g.t(), x)
    m = g.size(0)
    mc = m // nc
    if mc == 0:
        return torch.mm(g.t(), x)
    main = nc * mc
    out = torch.bmm(g[:main].view(nc, mc, -1).transpose(1, 2),
                    x[:main].view(nc, mc, -1)).sum(0)
    if main < m:
        out = out + torch.mm(g[main:].t(), x[main:])
    return out


def _ext_ok(x, w):
    from . import hip_ext

    return (hip_ext() is not None and x.is_cuda
            and x.dtype == torch.bfloat16 and w.dtype == torch.bfloat16
            and w.size(0) <= 208 and w.size(1) <= 208)


def _cast(t, dt):
    """Compute-dtype cast, version-cached for bf16 weights (ops/prep.py):
    under captured hipGraphs the per-call .to(bf16) copies replay every
    step; the cache turns them into reads of refreshed buffers."""
    if t is None or dt is None or t.dtype == dt:
        return t
    if dt == torch.bfloat16 and t.is_cuda:
        from . import prep

        return prep.get(t, "bf16")
    return t.to(dt)


class _SplitKLinearFn(torch.autograd.Function):
    """Takes the RAW parameters (grads returned in parameter dtype); the
    compute-dtype cast happens inside so it can be version-cached."""

    @staticmethod
    def forward(ctx, x, w, b, dt):
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        ctx.dt = dt
        wc = _cast(w, dt)
        bc = _cast(b, dt)
        if _ext_ok(x, wc):
            from . import hip_ext

            return hip_ext().tall_linear(x, wc, bc, 0)
        return F.linear(x, wc, bc)

    @staticmethod
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        g = g.contiguous()
        if ctx.dt == torch.bfloat16 and w.is_cuda and w.dtype != ctx.dt:
            from . import prep

            wt = prep.get(w, "t_bf16")
        else:
            wt = w.to(g.dtype).t().contiguous()
        if _ext_ok(g, wt):
            from . import hip_ext

            gx = hip_ext().tall_linear(g, wt, None, 0)
        else:
            gx = g @ wt.t()
        gw = chunked_wgrad(g, x.to(g.dtype)).to(w.dtype)
        gb = g.sum(0).to(w.dtype) if ctx.has_bias else None
        return gx, gw, gb, None


class SplitKLinear(nn.Linear):
    """nn.Linear with manual split-K weight gradients for tall inputs."""

    def forward(self, x):
        if (x.is_cuda and x.dim() >= 2
                and x.numel() // x.size(-1) >= _MIN_ROWS):
            shape = x.shape
            x2 = x.reshape(-1, shape[-1])
            if torch.is_autocast_enabled():
                dt = torch.get_autocast_dtype("cuda")
                x2 = x2.to(dt)
            elif x2.dtype != self.weight.dtype and x2.dtype in (
                    torch.bfloat16, torch.float16):
                dt = x2.dtype
            else:
                dt = None
            out = _SplitKLinearFn.apply(x2.contiguous(), self.weight,
                                        self.bias, dt)
            return out.reshape(*shape[:-1], out.size(-1))
        return super().forward(x)
