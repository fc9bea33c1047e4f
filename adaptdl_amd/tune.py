"""Elastic hyperparameter trials on the local controller.

Counterpart of the reference's Ray Tune integration
(/root/reference/ray/adaptdl_ray/tune/adaptdl_trial_sched.py:71-100 and
adaptdl_trial.py): concurrent trials run as controller jobs sharing the
node's GPUs under ONE Pollux policy, which continuously re-allocates
replicas between them from their reported goodput hints; trials are
rescaled via the standard checkpoint-restart protocol (in-place for
scale-downs when enabled).

Usage:

    from adaptdl_amd.tune import Trial, run_trials
    trials = [Trial(name="lr-0.1", argv=[sys.executable, "train.py",
                                         "--lr", "0.1"]),
              Trial(name="lr-0.01", argv=[..., "--lr", "0.01"])]
    results = run_trials(trials, num_gpus=8, interval=60)

Each trial's worker may write a JSON dict to ``$ADAPTDL_SHARE_PATH/
result.json`` (rank 0); it is returned in its TrialResult.

Early stopping (the Tune-scheduler role): workers append intermediate
metric reports, one JSON object per line, to ``$ADAPTDL_SHARE_PATH/
metrics.jsonl`` (rank 0), e.g. ``{"acc": 0.71}``; pass
``stopper=MedianStopper("acc")`` to run_trials to cancel trials whose
best reported value falls below the median of their peers' bests
(final state "Stopped", GPUs immediately reusable by the survivors).
"""

import json
import logging
import os
import time

from adaptdl_amd.sched import JobSpec, LocalController

LOG = logging.getLogger(__name__)


class Trial(object):
    def __init__(self, name, argv, env=None, min_replicas=0,
                 max_replicas=8, gpus_per_replica=1,
                 inplace_scaledown=False):
        self.name = name
        self.argv = list(argv)
        self.env = dict(env or {})
        self.min_replicas = min_replicas
        self.max_replicas = max_replicas
        self.gpus_per_replica = gpus_per_replica
        self.inplace_scaledown = inplace_scaledown


class MedianStopper(object):
    """Median early-stopping rule (the classic Tune MedianStoppingRule):
    after ``grace`` reports, a trial stops when its best metric so far
    is strictly worse than the median of all trials' bests."""

    def __init__(self, metric, mode="max", grace=3):
        if mode not in ("max", "min"):
            raise ValueError("mode must be 'max' or 'min'")
        self.metric = metric
        self.mode = mode
        self.grace = grace

    def _best(self, history):
        vals = [h[self.metric] for h in history if self.metric in h]
        if not vals:
            return None
        return max(vals) if self.mode == "max" else min(vals)

    def should_stop(self, name, histories):
        """histories: {trial name: [report dict, ...]}."""
        mine = histories.get(name) or []
        if len([h for h in mine if self.metric in h]) < self.grace:
            return False
        bests = [b for b in (self._best(h) for h in histories.values())
                 if b is not None]
        if len(bests) < 2:
            return False
        bests_sorted = sorted(bests)
        median = bests_sorted[len(bests_sorted) // 2]
        my_best = self._best(mine)
        if self.mode == "max":
            return my_best < median
        return my_best > median


def _read_metrics(job_dir):
    path = os.path.join(job_dir, "metrics.jsonl")
    history = []
    try:
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                try:
                    history.append(json.loads(line))
                except ValueError:
                    pass
    except OSError:
        pass
    return history


class TrialResult(object):
    def __init__(self, name, state, restarts, result, job_dir):
        self.name = name
        self.state = state
        self.restarts = restarts
        self.result = result
        self.job_dir = job_dir

    def __repr__(self):
        return ("TrialResult(name={!r}, state={!r}, restarts={}, "
                "result={!r})".format(self.name, self.state,
                                      self.restarts, self.result))


def run_trials(trials, trial_dir=".adaptdl/trials", num_gpus=None,
               interval=60.0, timeout=None, controller=None,
               stopper=None):
    """Run all trials to completion under one Pollux allocation loop.

    ``stopper``: optional early-stopping rule (e.g. MedianStopper);
    consulted against the trials' metrics.jsonl reports, with stopped
    trials cancelled (final state "Stopped") so their GPUs return to
    the pool.  Returns {trial name: TrialResult}.
    """
    own_controller = controller is None
    if own_controller:
        controller = LocalController(num_gpus=num_gpus, interval=interval)
    trial_dir = os.path.abspath(trial_dir)
    try:
        for trial in trials:
            job_dir = os.path.join(trial_dir, trial.name)
            env = dict(trial.env)
            env.setdefault("ADAPTDL_SHARE_PATH", job_dir)
            controller.submit(JobSpec(
                trial.argv, name=trial.name, job_dir=job_dir,
                min_replicas=trial.min_replicas,
                max_replicas=trial.max_replicas,
                gpus_per_replica=trial.gpus_per_replica, env=env,
                inplace_scaledown=trial.inplace_scaledown))
        deadline = None if timeout is None else time.time() + timeout
        results = {}
        pending = {t.name for t in trials}
        while pending:
            if stopper is not None:
                histories = {t.name: _read_metrics(
                    os.path.join(trial_dir, t.name)) for t in trials}
                for name in list(pending):
                    if controller.status(name)["state"] == "Running" \
                            and stopper.should_stop(name, histories):
                        LOG.info("early-stopping trial %s", name)
                        controller.cancel(name)
            for name in list(pending):
                st = controller.status(name)
                if st["state"] in ("Succeeded", "Failed", "Stopped"):
                    job_dir = os.path.join(trial_dir, name)
                    result = None
                    rpath = os.path.join(job_dir, "result.json")
                    if os.path.exists(rpath):
                        try:
                            with open(rpath) as f:
                                result = json.load(f)
                        except ValueError:
                            LOG.warning("trial %s: bad result.json", name)
                    results[name] = TrialResult(
                        name, st["state"], st["restarts"], result,
                        job_dir)
                    pending.discard(name)
            if pending:
                if deadline is not None and time.time() > deadline:
                    raise TimeoutError(
                        "trials still running: {}".format(sorted(pending)))
                time.sleep(0.5)
        return results
    finally:
        if own_controller:
            controller.shutdown()
