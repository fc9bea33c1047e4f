"""GraphedStepper cycle bookkeeping on CPU (EagerBackend).

The hipGraph stepper's risk is not the captured kernels (replay runs the
identical sequence) but the host-side bookkeeping it takes over from the
engine: deferred ``_on_sync_done``, absolute accum-count state, cycle
position tracking, signature-change fallback.  The EagerBackend executes
the full Python path with ``engine.graph_mode`` set — exactly the deferred
flow — so training through it must be bit-identical to plain eager
training.  (The real capture backend is validated by the env-gated GPU
test in tests/test_gpu_e2e.py.)
"""

import os

import numpy as np
import torch
import torch.nn.functional as F

import adaptdl_amd.collective as collective

from conftest import elastic_multiprocessing


def _build(accum_steps):
    """Tiny deterministic training stack with forced accumulation."""
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.data import AdaptiveDataLoaderHelper

    # Force a fixed (atomic_bsz, accum_steps) choice instead of the
    # goodput model's, so eager and graphed runs see identical cycles.
    def fake_sync(self):
        self._state.current_local_bsz = 8
        self._state.accumulation_steps = accum_steps
        return 8

    AdaptiveDataLoaderHelper._sync_local_bsz = fake_sync

    torch.manual_seed(0)
    xs = torch.randn(48, 8)
    ys = torch.randint(0, 4, (48,))
    dataset = torch.utils.data.TensorDataset(xs, ys)

    torch.manual_seed(1)
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    optim = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    adp = adl.AdaptiveDataParallel(model, optim)
    loader = adl.AdaptiveDataLoader(dataset, batch_size=8, shuffle=True)
    return adl, model, optim, adp, loader


def _train(accum_steps, graphed, out_path):
    from adaptdl_amd.torch.graph_step import GraphedStepper, EagerBackend

    collective.initialize()
    adl, model, optim, adp, loader = _build(accum_steps)

    def fwd_bwd(x, y):
        optim.zero_grad()
        loss = F.cross_entropy(adp(x), y)
        loss.backward()
        return loss

    stepper = None
    if graphed:
        stepper = GraphedStepper(adp, optim, fwd_bwd,
                                 backend=EagerBackend(), warmup_cycles=1)

    for _epoch in adl.remaining_epochs_until(3):
        for x, y in loader:
            if stepper is not None:
                stepper.microbatch(x, y)
            else:
                fwd_bwd(x, y)
            optim.step()

    gns_state = optim.state["gns"]
    result = {
        "weights": [p.detach().clone() for p in model.parameters()],
        "sqr_avg": np.array(gns_state["sqr_avg"]),
        "var_avg": np.array(gns_state["var_avg"]),
        "progress": float(gns_state["progress"]),
        "gain": float(adp.gain),
        "stats": dict(stepper.stats) if stepper is not None else None,
    }
    torch.save(result, out_path)
    collective.teardown()
    return 0


@elastic_multiprocessing
def _train_child(accum_steps, graphed, out_path):
    return _train(accum_steps, graphed, out_path)


def _compare(tmp_path, accum_steps, expect_stats):
    eager_path = str(tmp_path / "eager.pt")
    graph_path = str(tmp_path / "graphed.pt")
    _train_child(accum_steps, False, eager_path)
    _train_child(accum_steps, True, graph_path)
    eager = torch.load(eager_path, weights_only=False)
    graphed = torch.load(graph_path, weights_only=False)

    for we, wg in zip(eager["weights"], graphed["weights"]):
        assert torch.equal(we, wg), "weights diverged"
    np.testing.assert_allclose(eager["sqr_avg"], graphed["sqr_avg"])
    np.testing.assert_allclose(eager["var_avg"], graphed["var_avg"])
    assert eager["progress"] == graphed["progress"]
    assert eager["gain"] == graphed["gain"]
    assert graphed["stats"] == expect_stats


def test_graphed_equals_eager_with_accumulation(tmp_path):
    """A=2 cycle: kinds first/mid/sync; bit-identical to eager.

    Expected stats over 3 epochs x 6 microbatches: mb1 runs before the
    loader is marked training (eager), which consumes one accumulation
    slot and shortens the first cycle — the stepper detects the desync
    at mb3 and falls back for that cycle (1 fallback, by design); mb4-6
    are the warmup cycle, mb7-9 capture the three kinds, mb10-18 replay.
    """
    _compare(tmp_path, accum_steps=2, expect_stats={
        "captures": 3, "replays": 9, "eager": 6, "fallbacks": 1})


def test_graphed_equals_eager_no_accumulation(tmp_path):
    """A=0 cycle: single "solo" kind (differenced GNS estimator path).

    mb1 eager (loader not yet marked training), mb2 warmup cycle, mb3
    captures, mb4-18 replay; the solo cycle never desyncs.
    """
    _compare(tmp_path, accum_steps=0, expect_stats={
        "captures": 1, "replays": 15, "eager": 2, "fallbacks": 0})


@elastic_multiprocessing
def _train_resize_child(out_path):
    """Batch-size change mid-training: graphs must drop and re-capture."""
    from adaptdl_amd.torch.graph_step import GraphedStepper, EagerBackend
    from adaptdl_amd.torch.data import AdaptiveDataLoaderHelper

    collective.initialize()
    adl, model, optim, adp, loader = _build(accum_steps=0)

    def fwd_bwd(x, y):
        optim.zero_grad()
        loss = F.cross_entropy(adp(x), y)
        loss.backward()
        return loss

    stepper = GraphedStepper(adp, optim, fwd_bwd,
                             backend=EagerBackend(), warmup_cycles=1)

    for epoch in adl.remaining_epochs_until(4):
        if epoch == 2:
            # Grow the atomic batch size: new shapes from epoch 2 on.
            def fake_sync(self):
                self._state.current_local_bsz = 16
                self._state.accumulation_steps = 0
                return 16
            AdaptiveDataLoaderHelper._sync_local_bsz = fake_sync
        for x, y in loader:
            stepper.microbatch(x, y)
            optim.step()
        for p in model.parameters():
            assert torch.isfinite(p).all()

    # Two signatures -> two capture generations, each preceded by one
    # warmup cycle; no desyncs.
    assert stepper.stats["captures"] == 2, stepper.stats
    assert stepper.stats["fallbacks"] == 0, stepper.stats
    assert stepper.stats["replays"] > 0
    torch.save({"ok": True}, out_path)
    collective.teardown()
    return 0


def test_graphed_recaptures_on_batch_size_change(tmp_path):
    out = str(tmp_path / "resize.pt")
    _train_resize_child(out)
    assert torch.load(out, weights_only=False)["ok"]


def test_env_gate_off_returns_none():
    """maybe_graphed_stepper is a no-op without ADAPTDL_HIPGRAPH=1."""
    from adaptdl_amd.torch.graph_step import maybe_graphed_stepper
    assert os.getenv("ADAPTDL_HIPGRAPH") != "1"
    assert maybe_graphed_stepper(None, None, None) is None


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@elastic_multiprocessing
def _train_two_replicas(graphed, out_path):
    """2-replica gloo training: the stepper's bookkeeping must hold with
    real collectives in the cycle (per-bucket all-reduce at the sync
    position, stats all-reduce, module-state broadcast)."""
    import adaptdl_amd.env as env
    from adaptdl_amd.torch.graph_step import GraphedStepper, EagerBackend

    collective.initialize()
    if env.num_restarts() == 0:
        collective.teardown()
        return 2
    torch.distributed.init_process_group(
        "gloo", init_method="tcp://127.0.0.1:{}".format(
            collective.broadcast(_free_port())),
        world_size=env.num_replicas(), rank=env.replica_rank())

    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.data import AdaptiveDataLoaderHelper

    def fake_sync(self):
        self._state.current_local_bsz = 8
        self._state.accumulation_steps = 2
        return 8

    AdaptiveDataLoaderHelper._sync_local_bsz = fake_sync

    torch.manual_seed(0)
    xs = torch.randn(48, 8)
    ys = torch.randint(0, 4, (48,))
    dataset = torch.utils.data.TensorDataset(xs, ys)
    torch.manual_seed(1 + env.replica_rank())  # broadcast must fix this
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    optim = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    adp = adl.AdaptiveDataParallel(model, optim)
    loader = adl.AdaptiveDataLoader(dataset, batch_size=16)

    def fwd_bwd(x, y):
        optim.zero_grad()
        loss = F.cross_entropy(adp(x), y)
        loss.backward()
        return loss

    stepper = GraphedStepper(adp, optim, fwd_bwd,
                             backend=EagerBackend(), warmup_cycles=1) \
        if graphed else None

    for _epoch in adl.remaining_epochs_until(4):
        for x, y in loader:
            if stepper is not None:
                stepper.microbatch(x, y)
            else:
                fwd_bwd(x, y)
            optim.step()

    if env.replica_rank() == 0:
        gns_state = optim.state["gns"]
        torch.save({
            "weights": [p.detach().clone() for p in model.parameters()],
            "sqr_avg": np.array(gns_state["sqr_avg"]),
            "var_avg": np.array(gns_state["var_avg"]),
            "stats": dict(stepper.stats) if stepper is not None else None,
        }, out_path)
    torch.distributed.destroy_process_group()
    collective.teardown()
    return 0


def test_graphed_equals_eager_two_replicas(tmp_path):
    eager_path = str(tmp_path / "eager2.pt")
    graph_path = str(tmp_path / "graphed2.pt")
    _train_two_replicas(False, eager_path)
    _train_two_replicas(True, graph_path)
    eager = torch.load(eager_path, weights_only=False)
    graphed = torch.load(graph_path, weights_only=False)
    for we, wg in zip(eager["weights"], graphed["weights"]):
        assert torch.equal(we, wg), "weights diverged"
    np.testing.assert_allclose(eager["sqr_avg"], graphed["sqr_avg"])
    np.testing.assert_allclose(eager["var_avg"], graphed["var_avg"])
    # 4 epochs x 3 microbatches/replica; mb1 untrained + one desync
    # cycle (see test_graphed_equals_eager_with_accumulation).
    assert graphed["stats"] == {"captures": 3, "replays": 3,
                                "eager": 6, "fallbacks": 1}


@elastic_multiprocessing
def _train_with_restart(graphed, out_path):
    """Checkpoint-restart elasticity x graph stepping: rescale 1 -> 2
    replicas at an epoch boundary; the fresh processes rebuild a fresh
    stepper (warm + recapture) while model/optimizer/GNS state resumes
    from the checkpoint.  Final weights must match the eager run with
    the identical restart schedule."""
    import adaptdl_amd.checkpoint as checkpoint
    import adaptdl_amd.env as env
    from adaptdl_amd.torch.graph_step import GraphedStepper, EagerBackend

    collective.initialize()
    torch.distributed.init_process_group(
        "gloo", init_method="tcp://127.0.0.1:{}".format(
            collective.broadcast(_free_port())),
        world_size=env.num_replicas(), rank=env.replica_rank())

    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.data import AdaptiveDataLoaderHelper

    def fake_sync(self):
        self._state.current_local_bsz = 8
        self._state.accumulation_steps = 2
        return 8

    AdaptiveDataLoaderHelper._sync_local_bsz = fake_sync

    torch.manual_seed(0)
    xs = torch.randn(48, 8)
    ys = torch.randint(0, 4, (48,))
    dataset = torch.utils.data.TensorDataset(xs, ys)
    torch.manual_seed(1 + env.replica_rank())
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    optim = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    adp = adl.AdaptiveDataParallel(model, optim)
    loader = adl.AdaptiveDataLoader(dataset, batch_size=16)

    def fwd_bwd(x, y):
        optim.zero_grad()
        loss = F.cross_entropy(adp(x), y)
        loss.backward()
        return loss

    stepper = GraphedStepper(adp, optim, fwd_bwd,
                             backend=EagerBackend(), warmup_cycles=1) \
        if graphed else None

    for epoch in adl.remaining_epochs_until(6):
        for x, y in loader:
            if stepper is not None:
                stepper.microbatch(x, y)
            else:
                fwd_bwd(x, y)
            optim.step()
        if env.num_restarts() == 0 and epoch == 1:
            checkpoint.save_all_states()
            torch.distributed.destroy_process_group()
            collective.teardown()
            return 2  # rescale to two replicas

    if env.replica_rank() == 0:
        torch.save({
            "weights": [p.detach().clone() for p in model.parameters()],
            "restarts": env.num_restarts(),
            "stats": dict(stepper.stats) if stepper is not None else None,
        }, out_path)
    torch.distributed.destroy_process_group()
    collective.teardown()
    return 0


def test_graphed_restart_rescale_matches_eager(tmp_path):
    eager_path = str(tmp_path / "re.pt")
    graph_path = str(tmp_path / "rg.pt")
    _train_with_restart(False, eager_path)
    _train_with_restart(True, graph_path)
    eager = torch.load(eager_path, weights_only=False)
    graphed = torch.load(graph_path, weights_only=False)
    assert eager["restarts"] == graphed["restarts"] == 1
    for we, wg in zip(eager["weights"], graphed["weights"]):
        assert torch.equal(we, wg), "weights diverged after rescale"
    # The post-restart processes re-captured from scratch.
    assert graphed["stats"]["captures"] == 3
    assert graphed["stats"]["replays"] > 0


def test_refuses_adam_preconditioned_gns(tmp_ckpt_env):
    """Adam-preconditioned statistics bake the per-step bias correction
    into captured kernels; construction must refuse (and the env-gated
    factory must decline gracefully)."""
    import pytest
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.graph_step import GraphedStepper, EagerBackend

    model = torch.nn.Linear(4, 2)
    optim = torch.optim.Adam(model.parameters(), lr=1e-3)
    adp = adl.AdaptiveDataParallel(model, optim, name="adam-graph-test")
    with pytest.raises(ValueError, match="Adam"):
        GraphedStepper(adp, optim, lambda *a: None,
                       backend=EagerBackend())


def test_refuses_mp_scaler(tmp_ckpt_env):
    import pytest
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.graph_step import GraphedStepper, EagerBackend

    model = torch.nn.Linear(4, 2)
    optim = torch.optim.SGD(model.parameters(), lr=0.1)
    scaler = torch.amp.GradScaler("cuda", enabled=False)
    adp = adl.AdaptiveDataParallel(model, optim, mp_scaler=scaler,
                                   name="scaler-graph-test")
    with pytest.raises(ValueError, match="mp_scaler"):
        GraphedStepper(adp, optim, lambda *a: None,
                       backend=EagerBackend())


def test_env_gate_on_without_gpu_returns_none(monkeypatch):
    """ADAPTDL_HIPGRAPH=1 on a CPU-only host declines gracefully."""
    from adaptdl_amd.torch.graph_step import maybe_graphed_stepper
    monkeypatch.setenv("ADAPTDL_HIPGRAPH", "1")
    if torch.cuda.is_available():  # pragma: no cover - CPU CI only
        import pytest
        pytest.skip("CPU-only path")
    assert maybe_graphed_stepper(None, None, None) is None


@elastic_multiprocessing
def _train_short_final_batch(out_path):
    """drop_last=False: the final batch of a data pass is smaller than the
    captured signature.  The stepper must detect the mid-cycle shape
    change, fall back to eager for that cycle, and keep training (the
    ADVICE r1 medium finding: previously this crashed in the static-
    buffer copy inside _replay)."""
    from adaptdl_amd.torch.graph_step import GraphedStepper, EagerBackend
    from adaptdl_amd.torch.data import AdaptiveDataLoaderHelper

    collective.initialize()
    import adaptdl_amd.torch as adl

    def fake_sync(self):
        self._state.current_local_bsz = 8
        self._state.accumulation_steps = 2
        return 8

    AdaptiveDataLoaderHelper._sync_local_bsz = fake_sync

    torch.manual_seed(0)
    # 44 samples at bsz 8 -> passes end with a short 4-sample batch.
    xs = torch.randn(44, 8)
    ys = torch.randint(0, 4, (44,))
    dataset = torch.utils.data.TensorDataset(xs, ys)
    torch.manual_seed(1)
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    optim = torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9)
    adp = adl.AdaptiveDataParallel(model, optim)
    loader = adl.AdaptiveDataLoader(dataset, batch_size=8, shuffle=True)

    def fwd_bwd(x, y):
        optim.zero_grad()
        loss = F.cross_entropy(adp(x), y)
        loss.backward()
        return loss

    stepper = GraphedStepper(adp, optim, fwd_bwd,
                             backend=EagerBackend(), warmup_cycles=1)
    for _epoch in adl.remaining_epochs_until(3):
        for x, y in loader:
            stepper.microbatch(x, y)
            optim.step()
    for p in model.parameters():
        assert torch.isfinite(p).all()
    # The short batch must have triggered shape-change fallbacks, while
    # full-size cycles kept being captured (each fallback re-warms, so
    # this schedule recaptures rather than replays).
    assert stepper.stats["fallbacks"] > 0, stepper.stats
    assert stepper.stats["captures"] > 0, stepper.stats
    torch.save({"ok": True, "stats": dict(stepper.stats)}, out_path)
    collective.teardown()
    return 0


def test_short_final_batch_falls_back(tmp_path):
    out = str(tmp_path / "short.pt")
    _train_short_final_batch(out)
    assert torch.load(out, weights_only=False)["ok"]


@elastic_multiprocessing
def _train_fused_adam(graphed, out_path):
    """FusedAdam + AdamScale + Adam-preconditioned GNS through the
    stepper: supported since the statistics moved to one graph-safe
    kernel per bucket with device-resident bias-correction scalars."""
    from adaptdl_amd.torch.graph_step import GraphedStepper, EagerBackend
    from adaptdl_amd.torch.data import AdaptiveDataLoaderHelper

    collective.initialize()
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.scaling_rules import AdamScale

    def fake_sync(self):
        self._state.current_local_bsz = 8
        self._state.accumulation_steps = 2
        return 8

    AdaptiveDataLoaderHelper._sync_local_bsz = fake_sync
    torch.manual_seed(0)
    xs = torch.randn(48, 8)
    ys = torch.randint(0, 4, (48,))
    dataset = torch.utils.data.TensorDataset(xs, ys)
    torch.manual_seed(1)
    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 4))
    optim = adl.FusedAdam(model.parameters(), lr=1e-3)
    adp = adl.AdaptiveDataParallel(model, optim)
    assert isinstance(adp.scaling_rule, AdamScale)
    loader = adl.AdaptiveDataLoader(dataset, batch_size=8, shuffle=True)

    def fwd_bwd(x, y):
        optim.zero_grad()
        loss = F.cross_entropy(adp(x), y)
        loss.backward()
        return loss

    stepper = GraphedStepper(adp, optim, fwd_bwd,
                             backend=EagerBackend(), warmup_cycles=1) \
        if graphed else None
    for _epoch in adl.remaining_epochs_until(4):
        for x, y in loader:
            if stepper is not None:
                stepper.microbatch(x, y)
            else:
                fwd_bwd(x, y)
            optim.step()

    gns_state = optim.state["gns"]
    torch.save({
        "weights": [p.detach().clone() for p in model.parameters()],
        "sqr_avg": np.array(gns_state["sqr_avg"]),
        "var_avg": np.array(gns_state["var_avg"]),
        "stats": dict(stepper.stats) if stepper is not None else None,
    }, out_path)
    collective.teardown()
    return 0


def test_graphed_fused_adam_equals_eager(tmp_path):
    eager_path = str(tmp_path / "adam_eager.pt")
    graph_path = str(tmp_path / "adam_graphed.pt")
    _train_fused_adam(False, eager_path)
    _train_fused_adam(True, graph_path)
    eager = torch.load(eager_path, weights_only=False)
    graphed = torch.load(graph_path, weights_only=False)
    for we, wg in zip(eager["weights"], graphed["weights"]):
        assert torch.equal(we, wg), "weights diverged"
    np.testing.assert_allclose(eager["sqr_avg"], graphed["sqr_avg"])
    np.testing.assert_allclose(eager["var_avg"], graphed["var_avg"])
    assert graphed["stats"]["captures"] > 0
    assert graphed["stats"]["replays"] > 0
