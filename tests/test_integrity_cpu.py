"""CPU tests for the capture integrity gate (runtime/integrity.py), the
CapturedAllReduce gloo fallback, and the GraphedStep shape-key signature.

The gate's control flow (two trajectories from identical state, coherent
verdict, state restore, coherent disable) is device-independent — it is
exercised here with a mock "graphed" wrapper whose enabled-phase behavior we
control; the real capture/replay path is validated on the GPU box
(tests/test_graphs_gpu.py) and by the pre-flight gate inside bench.py.
"""

import os
from types import SimpleNamespace

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from distegnn_amd.runtime.integrity import run_capture_integrity_gate


class _MockGraphed:
    """Looks like GraphedStep to the gate: enabled flag + run_eager."""

    def __init__(self, corrupt=False):
        self.enabled = True
        self.warmup = 2
        self.corrupt = corrupt

    def run_eager(self, fn):
        return fn()


def _make_problem(seed=0):
    torch.manual_seed(seed)
    model = torch.nn.Linear(4, 1)
    opt = torch.optim.Adam(model.parameters(), lr=1e-2)
    xs = [torch.randn(8, 4) for _ in range(4)]
    return model, opt, xs


def _make_run_step(graphed, model, opt, xs, accum=2):
    def run_step(k):
        x = xs[k % len(xs)]
        noise = torch.rand(())          # consumes RNG: phases must reseed
        loss = (model(x).pow(2).mean() + 0.0 * noise)
        if graphed.corrupt and graphed.enabled:
            # emulate replay corruption: the "captured" phase silently
            # rescales the loss (the bisected round-1 failure mode)
            loss = loss * 1.5
        loss.backward()
        if (k + 1) % accum == 0:
            opt.step()
            opt.zero_grad(set_to_none=False)
        return loss.detach()

    return run_step


def test_gate_passes_on_identical_trajectories():
    graphed = _MockGraphed(corrupt=False)
    model, opt, xs = _make_problem()
    params = list(model.parameters())
    before = [p.detach().clone() for p in params]
    ok = run_capture_integrity_gate(
        graphed, _make_run_step(graphed, model, opt, xs), 8, params, opt,
        verbose=False)
    assert ok
    assert graphed.enabled
    # state restored to pre-gate values
    for p, b in zip(params, before):
        assert torch.equal(p.detach(), b)
    assert len(opt.state) == 0 or all(
        torch.all(s["exp_avg"] == 0) for s in opt.state.values()
        if "exp_avg" in s)


def test_gate_detects_divergence_and_disables():
    graphed = _MockGraphed(corrupt=True)
    model, opt, xs = _make_problem()
    params = list(model.parameters())
    before = [p.detach().clone() for p in params]
    ok = run_capture_integrity_gate(
        graphed, _make_run_step(graphed, model, opt, xs), 8, params, opt,
        verbose=False)
    assert not ok
    assert not graphed.enabled          # capture disabled
    for p, b in zip(params, before):
        assert torch.equal(p.detach(), b)


def test_gate_detects_phase2_exception():
    graphed = _MockGraphed(corrupt=False)
    model, opt, xs = _make_problem()

    base = _make_run_step(graphed, model, opt, xs)

    def run_step(k):
        if graphed.enabled and k == 3:
            raise RuntimeError("capture blew up")
        return base(k)

    ok = run_capture_integrity_gate(graphed, run_step, 6,
                                    list(model.parameters()), opt,
                                    verbose=False)
    assert not ok and not graphed.enabled


def test_gate_noop_when_disabled():
    graphed = _MockGraphed()
    graphed.enabled = False
    called = []

    def run_step(k):
        called.append(k)
        return torch.zeros(())

    assert run_capture_integrity_gate(graphed, run_step, 4, [], None,
                                      verbose=False)
    assert called == []


def test_graphedstep_key_distinguishes_optional_fields():
    from distegnn_amd.runtime.graphs import GraphedStep

    def mk(with_chunks, chunk_len=3):
        b = SimpleNamespace(num_nodes=10, num_edges=20, num_graphs=2,
                            x=torch.zeros(10, 4),
                            edge_index=torch.zeros(2, 20, dtype=torch.long))
        if with_chunks:
            b.pool_chunk_begin = torch.zeros(chunk_len, dtype=torch.long)
        return b

    k_plain = GraphedStep._key(mk(False))
    k_chunk3 = GraphedStep._key(mk(True, 3))
    k_chunk4 = GraphedStep._key(mk(True, 4))
    assert k_plain != k_chunk3          # field presence differs
    assert k_chunk3 != k_chunk4         # field shape differs
    assert k_chunk3 == GraphedStep._key(mk(True, 3))


def test_graphedstep_invalidate_drops_graphs_keeps_meta():
    """invalidate() (the device-sync replay-hazard mitigation) must drop
    every captured graph but keep shape statistics/out_meta, so each shape
    recaptures on its next occurrence without extra eager warmups."""
    from distegnn_amd.runtime.graphs import GraphedStep, _ShapeEntry

    g = GraphedStep(lambda b: (torch.zeros(()),), [], enabled=False)
    e1, e2, e3 = _ShapeEntry(), _ShapeEntry(), _ShapeEntry()
    e1.graph = object()
    e1.static = {"x": torch.zeros(1)}
    e1.outputs = (torch.zeros(()),)
    e1.seen = 5
    e1.out_meta = [((1,), torch.float32, "cpu")]
    e2.seen = 1                       # still warming up — untouched
    e3.disabled = True                # capture failed earlier — untouched
    g.entries = {"a": e1, "b": e2, "c": e3}

    assert g.invalidate("test") == 1
    assert e1.graph is None and e1.static is None and e1.outputs is None
    assert e1.seen == 5 and e1.out_meta is not None   # recapture-ready
    assert e2.seen == 1 and not e2.disabled
    assert e3.disabled
    assert g.invalidate() == 0        # idempotent


def test_captured_allreduce_rebuild_noop_without_entries():
    from distegnn_amd.parallel.comm import CapturedAllReduce

    red = CapturedAllReduce()
    red.rebuild()                     # no entries, no dist: must not throw
    assert red._entries == {}


# ---------------------------------------------------------------------------
# 2-rank gloo: gate verdict coherence + CapturedAllReduce eager fallback


def _rank_worker(rank, ws, init_file, result_dir):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=ws)
    try:
        from distegnn_amd.parallel.comm import CapturedAllReduce

        red = CapturedAllReduce()
        t = torch.tensor([float(rank + 1)])
        out = red(t, dist.ReduceOp.SUM)
        assert torch.allclose(out, torch.tensor([3.0]))
        mx = red(torch.tensor([float(rank)]), dist.ReduceOp.MAX)
        assert torch.allclose(mx, torch.tensor([1.0]))

        # gate: rank 1's "captured" phase corrupts -> BOTH ranks must
        # disable (coherent verdict through the all-reduced flag)
        graphed = _MockGraphed(corrupt=(rank == 1))
        model, opt, xs = _make_problem(seed=rank)
        ok = run_capture_integrity_gate(
            graphed, _make_run_step(graphed, model, opt, xs), 4,
            list(model.parameters()), opt, verbose=False, rank=rank)
        assert not ok
        assert not graphed.enabled
        torch.save({"ok": ok}, os.path.join(result_dir, f"r{rank}.pt"))
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_two_rank_coherent_disable(tmp_path):
    ws = 2
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_rank_worker,
                         args=(r, ws, str(tmp_path / "init"), str(tmp_path)))
             for r in range(ws)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    for r in range(ws):
        assert not torch.load(tmp_path / f"r{r}.pt",
                              weights_only=False)["ok"]
