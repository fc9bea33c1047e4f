"""Local job controller: elastic checkpoint-restart on one MI355X node.

Single-node counterpart of the reference's Kubernetes controller
(/root/reference/sched/adaptdl_sched/controller.py:101-421): instead of
pods it manages one OS process per replica (one GPU each via
HIP_VISIBLE_DEVICES), injects the same ADAPTDL_* environment the
reference's controller injects, and drives the identical elasticity
protocol — SIGTERM -> trainer checkpoints -> exit(143) counts as
graceful preemption -> restart group with the new replica count and
ADAPTDL_NUM_RESTARTS+1.  The allocator (adaptdl_amd.sched.allocator)
periodically re-optimizes replica counts from the supervisor's sched
hints, exactly like the reference's 60 s _optimize_all loop.

State machine per job: PENDING -> STARTING -> RUNNING -> STOPPING ->
(PENDING | SUCCEEDED | FAILED), mirroring controller.py:101-184.
"""

import logging
import os
import shutil
import signal
import socket
import subprocess
import sys
import threading
import time

from adaptdl_amd.sched.allocator import (LocalAllocator, job_info_from_hints,
                                         GPU_RESOURCE)
from adaptdl_amd.sched.supervisor import Supervisor

LOG = logging.getLogger(__name__)

try:  # optional Prometheus metrics (reference controller.py:35-41)
    from prometheus_client import Counter, Gauge, start_http_server

    METRICS = {
        "submitted": Counter("adaptdl_jobs_submitted",
                             "Jobs submitted to the local controller"),
        "succeeded": Counter("adaptdl_jobs_succeeded", "Jobs succeeded"),
        "failed": Counter("adaptdl_jobs_failed", "Jobs failed"),
        "preemptions": Counter("adaptdl_job_preemptions",
                               "Graceful checkpoint-restart preemptions"),
        "inplace_rescales": Counter(
            "adaptdl_job_inplace_rescales",
            "In-place (no-restart) scale-downs started"),
        "replicas": Gauge("adaptdl_running_replicas",
                          "Currently running replica processes"),
        "desired_nodes": Gauge(
            "adaptdl_desired_nodes",
            "Pollux's desired node count (cluster-expander signal for "
            "an external provisioner)"),
    }
except ImportError:  # pragma: no cover - prometheus_client is optional
    METRICS = None

    def start_http_server(port):
        raise RuntimeError("prometheus_client is not installed")


def _metric(name, amount=1):
    if METRICS is not None:
        METRICS[name].inc(amount)


GRACEFUL_EXIT = 143  # SIGTERM-driven checkpoint exit (reference parity)

PENDING = "Pending"
STARTING = "Starting"
RUNNING = "Running"
STOPPING = "Stopping"
SUCCEEDED = "Succeeded"
FAILED = "Failed"
STOPPED = "Stopped"  # cancelled by the user / a trial scheduler


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _version():
    import adaptdl_amd
    return adaptdl_amd.__version__


class JobSpec(object):
    """What to run and how elastic it may be.

    Arguments:
        argv: worker command line, e.g. [sys.executable, "train.py"].
        name: unique job name.
        min_replicas / max_replicas: elasticity bounds.
        gpus_per_replica: 1 binds each replica to one GPU via
            HIP_VISIBLE_DEVICES; 0 runs CPU/gloo workers (tests).
        env: extra environment variables.
        workdir: working directory for workers (default: inherited).
        job_dir: checkpoint/log directory (required).
        restart_limit: max unexpected-failure restarts before FAILED.
        preemptible: may the allocator move/shrink it.
    """

    def __init__(self, argv, name, job_dir, min_replicas=0, max_replicas=8,
                 gpus_per_replica=1, env=None, workdir=None,
                 restart_limit=3, preemptible=True,
                 inplace_scaledown=False):
        # Admission validation (reference validator.py:70-101 enforces
        # these via a k8s webhook; locally they are constructor checks).
        if not argv:
            raise ValueError("argv must be a non-empty command line")
        if max_replicas <= 0 or max_replicas < min_replicas:
            raise ValueError("maxReplicas must be positive and >= "
                             "minReplicas")
        if gpus_per_replica < 0:
            raise ValueError("gpus_per_replica must be >= 0")
        self.argv = list(argv)
        self.name = name
        self.job_dir = job_dir
        self.min_replicas = min_replicas
        self.max_replicas = max_replicas
        self.gpus_per_replica = gpus_per_replica
        self.env = dict(env or {})
        self.workdir = workdir
        self.restart_limit = restart_limit
        self.preemptible = preemptible
        # Scale-downs rejoin in place (SIGUSR2 directive; survivors keep
        # all state in memory, leavers exit) instead of checkpoint-
        # restarting the whole group.  Requires the worker to train via
        # AdaptiveDataLoader or AdaptiveBPTTIterator (eval-only phases
        # fall back to the restart path via the controller's escalation
        # timeout).
        self.inplace_scaledown = inplace_scaledown
        self._frozen = False

    def freeze(self):
        """Make the spec immutable (called at submission).

        Mirrors the reference's update validator (sched/adaptdl_sched/
        validator.py:103-113): updates to a running job's spec are
        forbidden — restarts must always relaunch the admitted spec.
        """
        import types
        object.__setattr__(self, "argv", tuple(self.argv))
        object.__setattr__(self, "env",
                           types.MappingProxyType(dict(self.env)))
        object.__setattr__(self, "_frozen", True)

    def __setattr__(self, key, value):
        if getattr(self, "_frozen", False):
            raise AttributeError(
                "updates to a submitted JobSpec are forbidden "
                "(field {!r}); submit a new job instead".format(key))
        object.__setattr__(self, key, value)


class _Job(object):
    def __init__(self, spec, creation_timestamp):
        self.spec = spec
        self.creation_timestamp = creation_timestamp
        self.state = PENDING
        self.allocation = []      # list of node names (all "local")
        self.target_allocation = None
        self.procs = []           # list of subprocess.Popen
        self.gpus = []            # GPU indices assigned to current group
        self.num_restarts = 0
        self.failures = 0
        self.completion = None    # set when SUCCEEDED/FAILED
        self.warm_dir = None      # RAM-backed rescale checkpoint root
        self.inplace = None       # pending in-place scale-down state
        self.inplace_version = 0
        self.cancelled = False    # cancel() requested -> final Stopped

    @property
    def num_replicas(self):
        return len(self.allocation)


class LocalController(object):
    """Submit and elastically run training jobs on this node."""

    def __init__(self, num_gpus=None, allocator=None, interval=30.0,
                 poll_interval=0.2, metrics_port=None):
        if metrics_port is not None:
            start_http_server(metrics_port)
        if num_gpus is None:
            num_gpus = int(os.getenv("ADAPTDL_NUM_GPUS", "8"))
        self.num_gpus = num_gpus
        self.supervisor = Supervisor().start()
        self.allocator = allocator or LocalAllocator(num_gpus=num_gpus)
        self._jobs = {}
        self._lock = threading.RLock()
        self._interval = interval
        self._poll_interval = poll_interval
        self._stop = threading.Event()
        self._wake = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="adaptdl-controller")
        self._thread.start()

    # ---- public API ----------------------------------------------------

    def submit(self, spec):
        with self._lock:
            if spec.name in self._jobs:
                raise ValueError("job {} already exists; updates to a "
                                 "running job's spec are forbidden"
                                 .format(spec.name))
            os.makedirs(os.path.join(spec.job_dir, "logs"), exist_ok=True)
            spec.freeze()  # admitted specs are immutable from here on
            job = _Job(spec, time.time())
            self._jobs[spec.name] = job
        _metric("submitted")
        self._wake.set()
        return spec.name

    def status(self, name):
        with self._lock:
            job = self._jobs[name]
            return {"state": job.state, "replicas": job.num_replicas,
                    "restarts": job.num_restarts,
                    "allocation": list(job.allocation),
                    "inplace_rescale_pending":
                        None if job.inplace is None
                        else job.inplace["world"],
                    "job_dir": job.spec.job_dir}

    def jobs(self):
        with self._lock:
            return {name: self.status(name) for name in self._jobs}

    def desired_nodes(self):
        """Pollux's desired node count from the last allocator cycle
        (the reference cluster-expander's output, exposed for external
        provisioners; also a Prometheus gauge)."""
        return getattr(self.allocator, "desired_nodes", 1)

    def wait(self, name, timeout=None):
        """Block until the job completes; returns its final state."""
        deadline = None if timeout is None else time.time() + timeout
        while True:
            with self._lock:
                job = self._jobs[name]
                if job.state in (SUCCEEDED, FAILED, STOPPED):
                    return job.state
            if deadline is not None and time.time() > deadline:
                raise TimeoutError("job {} still {}".format(
                    name, self.status(name)["state"]))
            time.sleep(self._poll_interval)

    def rescale(self, name, num_replicas):
        """Manually request a replica count (overrides one cycle)."""
        with self._lock:
            job = self._jobs[name]
            job.target_allocation = ["local"] * num_replicas
        self._wake.set()

    def restart(self, name):
        """Force a checkpoint-restart at the current replica count."""
        with self._lock:
            job = self._jobs[name]
            if job.state == RUNNING:
                job.state = STOPPING
                self._signal_group(job, signal.SIGTERM)
        self._wake.set()

    def cancel(self, name):
        """Stop a job permanently (final state: Stopped).

        Used by trial schedulers (tune.run_trials early stopping) and
        operators; the group gets the graceful SIGTERM-checkpoint
        treatment but is not restarted."""
        with self._lock:
            job = self._jobs[name]
            if job.state in (SUCCEEDED, FAILED, STOPPED):
                return
            job.cancelled = True
            if job.procs:
                job.state = STOPPING
                self._signal_group(job, signal.SIGTERM)
            else:
                job.state = STOPPED
                job.completion = time.time()
                self._drop_warm_dir(job)
        self._wake.set()

    def reallocate(self):
        """Run one allocator cycle immediately."""
        self._optimize()
        self._wake.set()

    def shutdown(self):
        self._stop.set()
        self._wake.set()
        self._thread.join(timeout=10)
        with self._lock:
            jobs = list(self._jobs.values())
        for job in jobs:
            self._signal_group(job, signal.SIGKILL)
            self._drop_warm_dir(job)
        self.supervisor.stop()

    @staticmethod
    def _drop_warm_dir(job):
        if job.warm_dir is not None:
            shutil.rmtree(job.warm_dir, ignore_errors=True)
            job.warm_dir = None

    def log_paths(self, name):
        job = self._jobs[name]
        d = os.path.join(job.spec.job_dir, "logs")
        return sorted(os.path.join(d, f) for f in os.listdir(d))

    # ---- control loop --------------------------------------------------

    def _run(self):
        last_optimize = 0.0
        while not self._stop.is_set():
            try:
                now = time.time()
                if now - last_optimize >= self._interval:
                    self._optimize()
                    last_optimize = now
                self._sync_all()
            except Exception:
                LOG.exception("controller loop error")
            self._wake.wait(self._poll_interval)
            self._wake.clear()

    def _optimize(self):
        """Allocator cycle: hints -> JobInfos -> PolluxPolicy -> targets."""
        with self._lock:
            active = {name: job for name, job in self._jobs.items()
                      if job.state not in (SUCCEEDED, FAILED)}
        if not active:
            return
        jobs_info = {}
        base = {}
        for name, job in active.items():
            hints = self.supervisor.get_hints(name)
            jobs_info[name] = job_info_from_hints(
                hints, job.creation_timestamp,
                min_replicas=job.spec.min_replicas,
                max_replicas=job.spec.max_replicas,
                preemptible=job.spec.preemptible,
                resources={GPU_RESOURCE: job.spec.gpus_per_replica,
                           "pods": 1})
            base[name] = list(job.allocation)
        allocations = self.allocator.optimize(jobs_info, base)
        if METRICS is not None:
            METRICS["desired_nodes"].set(
                getattr(self.allocator, "desired_nodes", 1))
        with self._lock:
            for name, alloc in allocations.items():
                job = self._jobs.get(name)
                if job is None or job.state in (SUCCEEDED, FAILED):
                    continue
                if job.target_allocation is None and \
                        sorted(alloc) != sorted(job.allocation):
                    job.target_allocation = alloc

    def _sync_all(self):
        with self._lock:
            jobs = list(self._jobs.values())
        for job in jobs:
            self._sync_job(job)

    def _sync_job(self, job):
        with self._lock:
            if job.state == PENDING:
                target = job.target_allocation
                if target is None and job.allocation:
                    target = list(job.allocation)  # restart after preemption
                if target is None and job.num_restarts == 0 and \
                        not job.allocation:
                    # Newly submitted: first-fit like the reference's
                    # added-job watch loop (allocator.py:56-106).
                    info = job_info_from_hints(
                        None, job.creation_timestamp,
                        min_replicas=job.spec.min_replicas,
                        max_replicas=job.spec.max_replicas,
                        resources={GPU_RESOURCE:
                                   job.spec.gpus_per_replica, "pods": 1})
                    target = self.allocator.allocate_new(info)
                if target:
                    job.allocation = list(target)
                    job.target_allocation = None
                    self._start_group(job)
            elif job.state == RUNNING:
                if job.inplace is not None:
                    self._check_inplace(job)
                elif job.target_allocation is not None and \
                        sorted(job.target_allocation) != \
                        sorted(job.allocation):
                    n_target = len(job.target_allocation)
                    if job.spec.inplace_scaledown and \
                            0 < n_target < len(job.allocation):
                        self._start_inplace(job, n_target)
                    else:
                        job.state = STOPPING
                        self._signal_group(job, signal.SIGTERM)
                else:
                    self._check_group(job)
            elif job.state == STOPPING:
                if all(p.poll() is not None for p in job.procs):
                    self._reap_group(job, expect_preemption=True)

    # ---- process management --------------------------------------------

    def _assigned_gpus(self):
        with self._lock:
            busy = []
            for job in self._jobs.values():
                if job.state in (STARTING, RUNNING, STOPPING):
                    busy.extend(job.gpus)
        return busy

    @staticmethod
    def _make_warm_dir(name):
        """RAM-backed (tmpfs) per-job dir for the in-memory rescale
        checkpoint; /dev/shm on Linux, with a tempdir fallback."""
        import tempfile
        base = "/dev/shm" if os.access("/dev/shm", os.W_OK) \
            else tempfile.gettempdir()
        return tempfile.mkdtemp(prefix="adaptdl-warm-{}-".format(name),
                                dir=base)

    def _start_group(self, job):
        spec = job.spec
        n = len(job.allocation)
        job.state = STARTING
        master_port = _free_port()
        if job.warm_dir is None:
            job.warm_dir = self._make_warm_dir(spec.name)
        # Stale SIGUSR2-readiness markers from the previous group.
        for f in os.listdir(spec.job_dir):
            if f.startswith(".sigusr2-ready-"):
                try:
                    os.unlink(os.path.join(spec.job_dir, f))
                except OSError:
                    pass
        gpus = []
        if spec.gpus_per_replica > 0:
            busy = self._assigned_gpus()
            free = [g for g in range(self.num_gpus) if g not in busy]
            need = n * spec.gpus_per_replica
            if len(free) < need:
                LOG.warning("job %s needs %d GPUs, only %d free; deferring",
                            spec.name, need, len(free))
                job.state = PENDING
                return
            gpus = free[:need]
        job.gpus = gpus
        job.procs = []
        logdir = os.path.join(spec.job_dir, "logs")
        for rank in range(n):
            env = dict(os.environ)
            env.update(spec.env)
            env.update({
                "ADAPTDL_JOB_ID": spec.name,
                "ADAPTDL_CHECKPOINT_PATH": spec.job_dir,
                "ADAPTDL_WARM_CHECKPOINT_PATH": job.warm_dir,
                "ADAPTDL_MASTER_ADDR": "127.0.0.1",
                "ADAPTDL_MASTER_PORT": str(master_port),
                "MASTER_ADDR": "127.0.0.1",
                "ADAPTDL_REPLICA_RANK": str(rank),
                "ADAPTDL_NUM_REPLICAS": str(n),
                "ADAPTDL_NUM_NODES": "1",
                "ADAPTDL_NUM_RESTARTS": str(job.num_restarts),
                "ADAPTDL_SUPERVISOR_URL": self.supervisor.url,
                "ADAPTDL_SCHED_VERSION": _version(),
            })
            if spec.gpus_per_replica > 0:
                mine = gpus[rank * spec.gpus_per_replica:
                            (rank + 1) * spec.gpus_per_replica]
                env["HIP_VISIBLE_DEVICES"] = ",".join(map(str, mine))
            log = open(os.path.join(logdir, "restart-{}-rank-{}.log".format(
                job.num_restarts, rank)), "w")
            proc = subprocess.Popen(spec.argv, env=env, cwd=spec.workdir,
                                    stdout=log, stderr=subprocess.STDOUT,
                                    start_new_session=True)
            proc._adaptdl_log = log
            job.procs.append(proc)
        self.supervisor.set_endpoints(spec.name, job.num_restarts,
                                      ["127.0.0.1"] * n)
        if METRICS is not None:
            METRICS["replicas"].inc(n)
        job.state = RUNNING
        LOG.info("job %s group %d started with %d replicas (gpus=%s)",
                 spec.name, job.num_restarts, n, gpus)

    # ---- in-place scale-down (north star: in-memory rejoin) ----------

    INPLACE_TIMEOUT = 60.0

    def _start_inplace(self, job, world):
        """Write the rescale directive and signal the group; survivors
        rejoin at the next optimizer-cycle boundary, leavers exit(143).
        State never leaves the survivors' memory."""
        import json
        job.inplace_version += 1
        directive = {"version": job.inplace_version, "world": world,
                     "master_port": _free_port()}
        path = os.path.join(job.spec.job_dir, "rescale-inplace.json")
        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            json.dump(directive, f)
        os.replace(tmp, path)
        job.inplace = {"world": world, "since": time.time(),
                       "last_signal": 0.0}
        self._signal_inplace_ready(job)
        _metric("inplace_rescales")
        LOG.info("job %s: in-place scale-down %d -> %d (directive v%d)",
                 job.spec.name, len(job.allocation), world,
                 job.inplace_version)

    def _signal_inplace_ready(self, job):
        """SIGUSR2 only the workers that installed the handler (marker
        file) — the default disposition of SIGUSR2 terminates, so a
        directive racing worker startup must not kill the group.
        Idempotent; re-sent ~1/s while the directive is pending."""
        info = job.inplace
        now = time.time()
        if now - info["last_signal"] < 1.0:
            return
        info["last_signal"] = now
        for rank, p in enumerate(job.procs):
            if p.poll() is not None:
                continue
            marker = os.path.join(job.spec.job_dir,
                                  ".sigusr2-ready-{}".format(rank))
            if not os.path.exists(marker):
                continue
            try:
                os.killpg(p.pid, signal.SIGUSR2)
            except (ProcessLookupError, PermissionError):
                pass

    def _check_inplace(self, job):
        info = job.inplace
        world = info["world"]
        self._signal_inplace_ready(job)
        codes = [p.poll() for p in job.procs]
        survivors, leavers = codes[:world], codes[world:]
        if any(c is not None for c in survivors):
            # A survivor died mid-rescale: abandon the in-place path and
            # let the normal crash/preemption machinery take over.
            LOG.warning("job %s: survivor exited during in-place rescale "
                        "(codes=%s); falling back", job.spec.name, codes)
            job.inplace = None
            self._check_group(job)
            return
        if all(c in (0, GRACEFUL_EXIT) for c in leavers):
            # Leavers gone, survivors running: adopt the new size with
            # NO restart (num_restarts unchanged, same processes).
            for p in job.procs[world:]:
                try:
                    p._adaptdl_log.close()
                except Exception:  # noqa: BLE001
                    pass
            if METRICS is not None:
                METRICS["replicas"].dec(len(job.procs) - world)
            job.procs = job.procs[:world]
            job.gpus = job.gpus[:world * job.spec.gpus_per_replica]
            job.allocation = ["local"] * world
            job.target_allocation = None
            job.inplace = None
            self.supervisor.set_endpoints(job.spec.name, job.num_restarts,
                                          ["127.0.0.1"] * world)
            LOG.info("job %s: in-place scale-down complete (%d replicas, "
                     "restarts still %d)", job.spec.name, world,
                     job.num_restarts)
            return
        if time.time() - info["since"] > self.INPLACE_TIMEOUT:
            # Workers never reached a safe point (BPTT loop, eval
            # phase, hung): escalate to the checkpoint-restart path.
            LOG.warning("job %s: in-place rescale timed out; escalating "
                        "to SIGTERM restart", job.spec.name)
            job.inplace = None
            job.state = STOPPING
            self._signal_group(job, signal.SIGTERM)

    def _signal_group(self, job, sig):
        for p in job.procs:
            if p.poll() is None:
                try:
                    os.killpg(p.pid, sig)
                except (ProcessLookupError, PermissionError):
                    pass

    def _check_group(self, job):
        codes = [p.poll() for p in job.procs]
        if all(c is None for c in codes):
            return
        if any(c not in (None, 0, GRACEFUL_EXIT) for c in codes):
            # A replica crashed: stop the rest, count a failure.
            self._signal_group(job, signal.SIGTERM)
            if all(c is not None for c in codes):
                self._reap_group(job, expect_preemption=False)
            else:
                job.state = STOPPING
        elif all(c == 0 for c in codes):
            self._reap_group(job, expect_preemption=False)
        elif all(c is not None for c in codes):
            self._reap_group(job, expect_preemption=True)
        # else: some replicas still draining (e.g. rank0 checkpointing).

    def _reap_group(self, job, expect_preemption):
        codes = [p.poll() for p in job.procs]
        for p in job.procs:
            try:
                p._adaptdl_log.close()
            except Exception:
                pass
        if METRICS is not None:
            METRICS["replicas"].dec(len(job.procs))
        job.procs = []
        job.gpus = []
        self.supervisor.clear_job(job.spec.name)
        if job.cancelled:
            job.state = STOPPED
            job.completion = time.time()
            self._drop_warm_dir(job)
            LOG.info("job %s stopped (cancelled; codes=%s)",
                     job.spec.name, codes)
        elif all(c == 0 for c in codes):
            job.state = SUCCEEDED
            job.completion = time.time()
            self._drop_warm_dir(job)
            _metric("succeeded")
            LOG.info("job %s succeeded", job.spec.name)
        elif all(c in (0, GRACEFUL_EXIT) for c in codes) or \
                expect_preemption:
            # Graceful preemption: restart with the (new) allocation.
            job.num_restarts += 1
            if job.target_allocation is not None:
                job.allocation = list(job.target_allocation)
                job.target_allocation = None
            job.state = PENDING
            _metric("preemptions")
            LOG.info("job %s preempted (codes=%s); restart %d with %d "
                     "replicas", job.spec.name, codes, job.num_restarts,
                     len(job.allocation))
        else:
            job.failures += 1
            if job.failures > job.spec.restart_limit:
                job.state = FAILED
                job.completion = time.time()
                self._drop_warm_dir(job)
                _metric("failed")
                LOG.warning("job %s failed (codes=%s)", job.spec.name,
                            codes)
            else:
                job.num_restarts += 1
                job.state = PENDING
                LOG.warning("job %s crashed (codes=%s); retry %d/%d",
                            job.spec.name, codes, job.failures,
                            job.spec.restart_limit)
