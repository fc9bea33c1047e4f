"""Elastic trial runner tests (Ray Tune counterpart, local form).

Two concurrent trials share the (CPU) node under one controller; both
train a tiny model elastically and report results; run_trials collects
them.
"""

import json
import os
import sys
import textwrap

from adaptdl_amd.tune import Trial, run_trials

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = textwrap.dedent("""
    import json, os, sys
    sys.path.insert(0, "@@REPO@@")
    import torch
    torch.set_num_threads(1)
    import adaptdl_amd.env as env
    import adaptdl_amd.torch as adl

    lr = float(sys.argv[1])
    adl.init_process_group("gloo")
    torch.manual_seed(0)
    xs = torch.randn(96, 2)
    ys = xs @ torch.tensor([[3.0], [4.0]])
    model = torch.nn.Linear(2, 1, bias=False)
    with torch.no_grad():
        model.weight.zero_()
    optim = torch.optim.SGD(model.parameters(), lr=lr)
    adp = adl.AdaptiveDataParallel(model, optim)
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(xs, ys), batch_size=16)
    loss = None
    for epoch in adl.remaining_epochs_until(10):
        for x, y in loader:
            optim.zero_grad()
            loss = ((adp(x) - y) ** 2).mean()
            loss.backward()
            optim.step()
    if env.replica_rank() == 0:
        with open(os.path.join(os.environ["ADAPTDL_SHARE_PATH"],
                               "result.json"), "w") as f:
            json.dump({"lr": lr, "final_loss": loss.item()}, f)
""")


def test_run_trials_concurrently(tmp_path):
    script = tmp_path / "trial.py"
    script.write_text(WORKER.replace("@@REPO@@", REPO))
    trials = [
        Trial(name="lr-0.05", argv=[sys.executable, str(script), "0.05"],
              min_replicas=1, max_replicas=2, gpus_per_replica=0),
        Trial(name="lr-0.001",
              argv=[sys.executable, str(script), "0.001"],
              min_replicas=1, max_replicas=2, gpus_per_replica=0),
    ]
    results = run_trials(trials, trial_dir=str(tmp_path / "trials"),
                         num_gpus=0, interval=3600, timeout=180)
    assert set(results) == {"lr-0.05", "lr-0.001"}
    for r in results.values():
        assert r.state == "Succeeded", r
        assert r.result is not None
    # the higher learning rate should fit the linear teacher better
    assert results["lr-0.05"].result["final_loss"] < \
        results["lr-0.001"].result["final_loss"]


def test_median_stopper_cancels_weak_trial(tmp_path):
    """Early stopping (the Tune-scheduler role): a trial whose reported
    metric trails the median of its peers is cancelled (state Stopped)
    while the strong trial completes normally."""
    import textwrap
    from adaptdl_amd.tune import MedianStopper

    worker = textwrap.dedent("""
        import json, os, sys, time
        acc = float(sys.argv[1])
        share = os.environ["ADAPTDL_SHARE_PATH"]
        for i in range(60):
            with open(os.path.join(share, "metrics.jsonl"), "a") as f:
                f.write(json.dumps({"acc": acc + i * 0.001}) + "\\n")
            time.sleep(0.1)
        with open(os.path.join(share, "result.json"), "w") as f:
            json.dump({"final_acc": acc}, f)
    """)
    script = tmp_path / "trial.py"
    script.write_text(worker)
    trials = [
        Trial(name="strong", argv=[sys.executable, str(script), "0.9"],
              max_replicas=1, gpus_per_replica=0),
        Trial(name="weak", argv=[sys.executable, str(script), "0.1"],
              max_replicas=1, gpus_per_replica=0),
        Trial(name="mid", argv=[sys.executable, str(script), "0.5"],
              max_replicas=1, gpus_per_replica=0),
    ]
    results = run_trials(trials, trial_dir=str(tmp_path / "trials"),
                         num_gpus=0, interval=3600, timeout=120,
                         stopper=MedianStopper("acc", grace=3))
    assert results["weak"].state == "Stopped"
    assert results["strong"].state == "Succeeded"
    assert results["strong"].result == {"final_acc": 0.9}


def test_median_stopper_rules():
    """MedianStopper unit semantics: grace period, median tie-keeping,
    and min-mode inversion."""
    from adaptdl_amd.tune import MedianStopper

    s = MedianStopper("acc", grace=2)
    hist = {
        "a": [{"acc": 0.9}, {"acc": 0.95}],
        "b": [{"acc": 0.5}, {"acc": 0.55}],
        "c": [{"acc": 0.7}, {"acc": 0.75}],
    }
    assert not s.should_stop("a", hist)
    assert s.should_stop("b", hist)       # below median of bests
    assert not s.should_stop("c", hist)   # the median itself is kept
    # grace: one report only -> never stopped
    hist["d"] = [{"acc": 0.01}]
    assert not s.should_stop("d", hist)
    # missing metric entries don't count toward grace
    hist["e"] = [{"loss": 1.0}, {"loss": 0.9}]
    assert not s.should_stop("e", hist)

    smin = MedianStopper("loss", mode="min", grace=1)
    hist2 = {"x": [{"loss": 0.2}], "y": [{"loss": 0.9}],
             "z": [{"loss": 0.5}]}
    assert smin.should_stop("y", hist2)
    assert not smin.should_stop("x", hist2)

    import pytest
    with pytest.raises(ValueError):
        MedianStopper("m", mode="best")
