"""Self-validating gate for hipGraph-captured training steps.

The captured-RCCL step (runtime/graphs.py + parallel/comm.py) was validated
end-to-end on a 1-rank RCCL group; the first true multi-rank xGMI execution
happens in the driver's scaling run with nobody watching. This gate makes
that run self-checking: before the measured/production phase, run a short
training trajectory twice from identical state — once eager, once through
the capture/replay machinery — and compare the per-step losses. Any replay
corruption (allocator aliasing, communicator mixing, watchdog interference —
all bisected failure modes from round 1, see profiles/README.md) shows up as
a trajectory divergence; the gate then disables capture COHERENTLY on every
rank (an all-reduced verdict), so no rank is left replaying while others run
eager on a different collective schedule.

Used by bench.py (--integrity auto|on|off) and runtime/trainer.py
(train.graph_integrity_check). Pure-python control flow; CPU-testable with a
mock graphed step (tests/test_integrity_cpu.py).
"""

from __future__ import annotations

import copy
from typing import Callable, List, Optional

import torch

from ..parallel import comm


def _snapshot_state(params: List[torch.nn.Parameter], optimizer):
    return (
        [p.detach().clone() for p in params],
        copy.deepcopy(optimizer.state_dict()) if optimizer is not None
        else None,
    )


def _restore_state(params, optimizer, snap):
    vals, opt_sd = snap
    with torch.no_grad():
        for p, v in zip(params, vals):
            p.copy_(v)
        # grad buffer ADDRESSES must stay stable (captured graphs reference
        # them) — zero in place, never rebind
        for p in params:
            if p.grad is not None:
                p.grad.zero_()
    if optimizer is not None and opt_sd is not None:
        optimizer.load_state_dict(copy.deepcopy(opt_sd))
    # restoring parameters bumps their version counters: cached weight
    # transforms (ops/prep.py) referenced by captured graphs must be
    # refreshed in the same region
    from ..ops import refresh_weight_prep

    refresh_weight_prep()


def run_capture_integrity_gate(
    graphed,
    run_step: Callable[[int], torch.Tensor],
    n_steps: int,
    params: List[torch.nn.Parameter],
    optimizer,
    seed: int = 1234,
    rtol: float = 1e-3,
    atol: float = 1e-6,
    verbose: bool = True,
    rank: int = 0,
) -> bool:
    """Run ``run_step(k)`` for k in [0, n_steps) twice from identical
    model/optimizer/RNG state: once with capture disabled, once enabled
    (capturing + replaying). Compare the returned per-step scalar losses.

    ``run_step`` must drive the full train step THROUGH ``graphed`` (so the
    second phase captures/replays) and be deterministic given the RNG seed
    and parameter state. ``n_steps`` should cover at least
    (warmup_occurrences + 1) occurrences of every batch shape so each graph
    is captured AND replayed at least once.

    On divergence: restores state, sets ``graphed.enabled = False`` on every
    rank (verdict all-reduced MIN over ranks, routed to the capture
    communicator — a default-group collective between replays is itself a
    bisected corruption), and returns False. On success restores state and
    returns True.
    """
    if not graphed.enabled:
        return True
    device = params[0].device if params else torch.device("cpu")

    def _seed():
        torch.manual_seed(seed)
        if torch.cuda.is_available():
            torch.cuda.manual_seed_all(seed)

    # grads must exist with stable addresses before any capture
    for p in params:
        if p.grad is None:
            p.grad = torch.zeros_like(p)
    snap = _snapshot_state(params, optimizer)
    # the gate reseeds and consumes RNG; restore the global streams at the
    # end so a gated run follows the SAME training trajectory as an
    # ungated one (trajectory parity is itself a test invariant)
    rng_cpu = torch.get_rng_state()
    rng_cuda = (torch.cuda.get_rng_state(device)
                if device.type == "cuda" else None)

    # Phase 1: eager reference trajectory (no graphs exist yet — default
    # stream/communicator are still safe here)
    graphed.enabled = False
    _seed()
    eager_losses = torch.stack([run_step(k).detach().float().reshape(())
                                for k in range(n_steps)])
    eager_vals = eager_losses.cpu()
    _restore_state(params, optimizer, snap)

    # Phase 2: same trajectory through capture/replay. Per-step losses are
    # copied in place into a buffer allocated BEFORE any capture — an eager
    # default-stream allocation between replays may alias graph-pool blocks
    # (bisected round-1 hazard).
    vals = torch.zeros(n_steps, device=device)
    graphed.enabled = True
    _seed()
    capture_error: Optional[str] = None
    try:
        for k in range(n_steps):
            vals[k].copy_(run_step(k).detach())
        graphed_vals = graphed.run_eager(lambda: vals.float().cpu())
    except Exception as exc:  # capture blew up outright
        capture_error = repr(exc)
        graphed_vals = None

    if graphed_vals is None:
        ok_local = False
        max_rel = float("inf")
    else:
        denom = eager_vals.abs().clamp(min=atol)
        max_rel = float(((graphed_vals - eager_vals).abs() / denom).max())
        ok_local = bool(
            torch.isfinite(graphed_vals).all()) and max_rel <= rtol

    # coherent verdict across ranks — on the capture communicator, on the
    # side stream (an eager default-group collective between replays is a
    # bisected corruption; so are default-stream allocations)
    if comm.is_distributed():
        def _verdict():
            flag = torch.tensor(
                0.0 if ok_local else 1.0,
                device=device if device.type == "cuda" else "cpu")
            with comm.capture_comm_fallback():
                torch.distributed.all_reduce(
                    flag, op=torch.distributed.ReduceOp.MAX,
                    group=comm._capture_group())
            return flag.item()
        ok = graphed.run_eager(_verdict) == 0.0
    else:
        ok = ok_local

    # final restore happens between replays — side stream discipline
    graphed.run_eager(lambda: _restore_state(params, optimizer, snap))

    torch.set_rng_state(rng_cpu)
    if rng_cuda is not None:
        torch.cuda.set_rng_state(rng_cuda, device)

    # the gate itself just did a pageable D2H (.cpu() of the loss buffer):
    # a device-wide sync garbles existing graph execs on this stack
    # (GraphedStep.invalidate docstring) — drop the gate's captures so
    # production replays use fresh ones. Callers holding prebuilt comm
    # graphs (GradBucket.graph_sync, CapturedAllReduce) must rebuild them
    # after the gate for the same reason.
    if hasattr(graphed, "invalidate"):
        graphed.invalidate("post-gate D2H sync")

    if not ok:
        graphed.enabled = False
        if verbose and rank == 0:
            print(f"[integrity] captured step diverged from eager "
                  f"(max rel err {max_rel:.3e}"
                  + (f", capture error {capture_error}" if capture_error
                     else "")
                  + ") — capture DISABLED on all ranks, continuing eager",
                  flush=True)
    elif verbose and rank == 0:
        print(f"[integrity] captured step matches eager trajectory over "
              f"{n_steps} steps (max rel err {max_rel:.3e})", flush=True)
    return ok
