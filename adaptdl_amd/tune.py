"""Elastic hyperparameter trials on the local controller.

Counterpart of the reference's Ray Tune integration
(/root/reference/ray/adaptdl_ray/tune/adaptdl_trial_sched.py:71-100 and
adaptdl_trial.py): concurrent trials run as controller jobs sharing the
node's GPUs under ONE Pollux policy, which continuously re-allocates
replicas between them from their reported goodput hints; trials are
rescaled via the standard checkpoint-restart protocol.

Usage:

    from adaptdl_amd.tune import Trial, run_trials
    trials = [Trial(name="lr-0.1", argv=[sys.executable, "train.py",
                                         "--lr", "0.1"]),
              Trial(name="lr-0.01", argv=[..., "--lr", "0.01"])]
    results = run_trials(trials, num_gpus=8, interval=60)

Each trial's worker may write a JSON dict to ``$ADAPTDL_SHARE_PATH/
result.json`` (rank 0); it is returned in its TrialResult.
"""

import json
import logging
import os
import time

from adaptdl_amd.sched import JobSpec, LocalController

LOG = logging.getLogger(__name__)


class Trial(object):
    def __init__(self, name, argv, env=None, min_replicas=0,
                 max_replicas=8, gpus_per_replica=1):
        self.name = name
        self.argv = list(argv)
        self.env = dict(env or {})
        self.min_replicas = min_replicas
        self.max_replicas = max_replicas
        self.gpus_per_replica = gpus_per_replica


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
               interval=60.0, timeout=None, controller=None):
    """Run all trials to completion under one Pollux allocation loop.

    Returns {trial name: TrialResult}.
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
                gpus_per_replica=trial.gpus_per_replica, env=env))
        deadline = None if timeout is None else time.time() + timeout
        results = {}
        pending = {t.name for t in trials}
        while pending:
            for name in list(pending):
                st = controller.status(name)
                if st["state"] in ("Succeeded", "Failed"):
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
