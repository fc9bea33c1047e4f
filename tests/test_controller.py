"""LocalController elasticity: submit -> run -> rescale -> complete.

End-to-end local-cluster counterpart of the reference's workload shell
tests (/root/reference/tests/testworkload.sh and controller state machine
controller.py:101-318): a real training job (full adaptdl_amd stack,
gloo) is started as 1 worker process, rescaled to 2 mid-run via SIGTERM/
checkpoint/exit(143)/restart, and must finish with the learned weights,
having trained under both replica counts.
"""

import json
import os
import sys
import textwrap
import time

import pytest

from adaptdl_amd.sched import JobSpec, LocalController, Supervisor

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = textwrap.dedent("""
    import json, os, sys
    sys.path.insert(0, "@@REPO@@")
    import torch
    torch.set_num_threads(1)
    import adaptdl_amd.env as env
    import adaptdl_amd.torch as adl

    adl.init_process_group("gloo")
    torch.manual_seed(7)
    true_w = torch.tensor([[3.0], [4.0]])
    xs = torch.randn(160, 2)
    ys = xs @ true_w + 0.01 * torch.randn(160, 1)
    model = torch.nn.Linear(2, 1, bias=False)
    with torch.no_grad():
        model.weight.zero_()
    optim = torch.optim.SGD(model.parameters(), lr=0.05)
    adp = adl.AdaptiveDataParallel(model, optim)
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(xs, ys), batch_size=16)

    trace = os.path.join(env.checkpoint_path(), "trace.jsonl")
    for epoch in adl.remaining_epochs_until(30):
        for x, y in loader:
            optim.zero_grad()
            ((adp(x) - y) ** 2).mean().backward()
            optim.step()
        if env.replica_rank() == 0:
            with open(trace, "a") as f:
                rec = dict(epoch=epoch, replicas=env.num_replicas(),
                           restarts=env.num_restarts())
                f.write(json.dumps(rec) + "\\n")
        # Slow the job down a little so the rescale lands mid-run.
        import time as _t
        _t.sleep(0.05)
    if env.replica_rank() == 0:
        w = model.weight.detach().reshape(-1).tolist()
        with open(os.path.join(env.checkpoint_path(), "final.json"),
                  "w") as f:
            json.dump(w, f)
""")


@pytest.fixture
def controller():
    ctrl = LocalController(num_gpus=0, interval=3600)  # manual allocation
    yield ctrl
    ctrl.shutdown()


def test_job_runs_rescales_completes(tmp_path, controller):
    script = tmp_path / "worker.py"
    script.write_text(WORKER.replace("@@REPO@@", REPO))
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="lr-job",
                   job_dir=job_dir, min_replicas=1, max_replicas=2,
                   gpus_per_replica=0)
    controller.submit(spec)

    # wait until running, then rescale to 2 replicas mid-run
    deadline = time.time() + 60
    while controller.status("lr-job")["state"] != "Running":
        assert time.time() < deadline, controller.status("lr-job")
        time.sleep(0.1)
    trace_path = os.path.join(job_dir, "trace.jsonl")
    while not os.path.exists(trace_path):
        assert time.time() < deadline
        time.sleep(0.1)
    controller.rescale("lr-job", 2)

    state = controller.wait("lr-job", timeout=180)
    assert state == "Succeeded", controller.status("lr-job")

    trace = [json.loads(line) for line in open(trace_path)]
    epochs = [t["epoch"] for t in trace]
    assert sorted(set(epochs)) == list(range(30))  # every epoch ran once
    assert {t["replicas"] for t in trace} == {1, 2}  # both configs trained
    assert max(t["restarts"] for t in trace) >= 1
    assert controller.status("lr-job")["restarts"] >= 1

    final = json.load(open(os.path.join(job_dir, "final.json")))
    assert abs(final[0] - 3.0) < 0.1 and abs(final[1] - 4.0) < 0.1


def test_crash_restart_limit(tmp_path, controller):
    script = tmp_path / "crash.py"
    script.write_text("import sys; sys.exit(7)\n")
    job_dir = str(tmp_path / "job2")
    spec = JobSpec([sys.executable, str(script)], name="crash-job",
                   job_dir=job_dir, min_replicas=1, max_replicas=1,
                   gpus_per_replica=0, restart_limit=1)
    controller.submit(spec)
    assert controller.wait("crash-job", timeout=60) == "Failed"
    assert controller.status("crash-job")["restarts"] >= 1


def test_supervisor_http_roundtrip():
    sup = Supervisor().start()
    try:
        import urllib.request
        req = urllib.request.Request(
            sup.url + "/hints/myjob", method="PUT",
            data=json.dumps({"initBatchSize": 64}).encode())
        with urllib.request.urlopen(req) as resp:
            assert resp.status == 200
        assert sup.get_hints("myjob") == {"initBatchSize": 64}
        sup.set_endpoints("myjob", 0, ["127.0.0.1", "127.0.0.1"])
        with urllib.request.urlopen(sup.url + "/discover/myjob/0") as resp:
            assert json.loads(resp.read()) == ["127.0.0.1", "127.0.0.1"]
        with urllib.request.urlopen(sup.url + "/healthz") as resp:
            assert resp.status == 200
    finally:
        sup.stop()


def test_allocator_scales_up_from_hints(tmp_path):
    """The reference's 60 s _optimize_all loop equivalent: worker reports
    goodput hints -> allocator's Pollux cycle raises the replica count."""
    from adaptdl_amd.sched import LocalAllocator
    from adaptdl_amd.sched.policy import PolluxPolicy

    ctrl = LocalController(
        num_gpus=0, interval=6.0,
        allocator=LocalAllocator(
            num_gpus=0, policy=PolluxPolicy(seed=0, pop_size=30,
                                            generations=20)))
    try:
        script = tmp_path / "worker.py"
        # Deterministic per-batch compute (sleep) so the fitted perf
        # model is load-independent and the 2-replica speedup is solid.
        worker_src = WORKER.replace("@@REPO@@", REPO) \
            .replace("remaining_epochs_until(30)",
                     "remaining_epochs_until(150)") \
            .replace("((adp(x) - y) ** 2).mean().backward()",
                     "((adp(x) - y) ** 2).mean().backward(); "
                     "__import__('time').sleep(0.004)")
        script.write_text(worker_src)
        seen_hints = []
        ctrl.supervisor.register_hints_callback(
            lambda job, hints: seen_hints.append((job, hints)))
        job_dir = str(tmp_path / "job3")
        os.makedirs(job_dir)
        spec = JobSpec([sys.executable, str(script)], name="auto-job",
                       job_dir=job_dir, min_replicas=1, max_replicas=4,
                       gpus_per_replica=0,
                       env={"ADAPTDL_FIT_INTERVAL": "1"})
        ctrl.submit(spec)
        deadline = time.time() + 240
        saw_multi = False
        while time.time() < deadline:
            st = ctrl.status("auto-job")
            if st["replicas"] > 1:
                saw_multi = True
                break
            if st["state"] in ("Succeeded", "Failed"):
                break
            time.sleep(0.5)
        assert seen_hints, "worker never reported hints"
        assert saw_multi, "allocator never scaled the job past 1 replica"
        # Stop allocator churn once scale-up is proven: pin the
        # current allocation so completion time is bounded.
        ctrl.rescale("auto-job",
                     ctrl.status("auto-job")["replicas"] or 1)
        ctrl._interval = 3600
        assert ctrl.wait("auto-job", timeout=300) == "Succeeded"
    finally:
        ctrl.shutdown()


def test_jobspec_admission_validation(tmp_path):
    """Reference validator.py parity: invalid specs rejected at submit."""
    with pytest.raises(ValueError):
        JobSpec([], name="bad", job_dir=str(tmp_path))
    with pytest.raises(ValueError):
        JobSpec(["x"], name="bad", job_dir=str(tmp_path),
                min_replicas=4, max_replicas=2)
    with pytest.raises(ValueError):
        JobSpec(["x"], name="bad", job_dir=str(tmp_path),
                max_replicas=0)


def test_warm_rescale_no_disk_checkpoint(tmp_path, controller):
    """In-memory (tmpfs) rescale handoff: a 1 -> 2 -> 1 elastic drill
    must resume correctly from the controller-provisioned warm root
    with NO checkpoint-K directory ever written to the on-disk job_dir
    (the cold path stays reference-format for crash recovery).
    North star: 'elastic rescale via in-HBM checkpoint and rejoin'."""
    script = tmp_path / "worker.py"
    script.write_text(WORKER.replace("@@REPO@@", REPO))
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="warm-job",
                   job_dir=job_dir, min_replicas=1, max_replicas=2,
                   gpus_per_replica=0)
    controller.submit(spec)

    deadline = time.time() + 60
    trace_path = os.path.join(job_dir, "trace.jsonl")
    while not os.path.exists(trace_path):
        assert time.time() < deadline, controller.status("warm-job")
        time.sleep(0.1)
    controller.rescale("warm-job", 2)
    # Wait for the first restart to land, then scale back down.
    while controller.status("warm-job")["restarts"] < 1:
        assert time.time() < deadline, controller.status("warm-job")
        time.sleep(0.1)
    while not [ln for ln in open(trace_path)
               if json.loads(ln)["restarts"] >= 1]:
        assert time.time() < deadline
        time.sleep(0.1)
    controller.rescale("warm-job", 1)

    state = controller.wait("warm-job", timeout=180)
    assert state == "Succeeded", controller.status("warm-job")

    trace = [json.loads(line) for line in open(trace_path)]
    assert sorted(set(t["epoch"] for t in trace)) == list(range(30))
    assert {t["replicas"] for t in trace} == {1, 2}
    assert max(t["restarts"] for t in trace) >= 2
    # THE warm-path property: the elastic rescales round-tripped through
    # RAM; nothing wrote the on-disk checkpoint-K format.
    assert not any(n.startswith("checkpoint-")
                   for n in os.listdir(job_dir)), os.listdir(job_dir)
    final = json.load(open(os.path.join(job_dir, "final.json")))
    assert abs(final[0] - 3.0) < 0.1 and abs(final[1] - 4.0) < 0.1


def test_submitted_spec_is_immutable(tmp_path, controller):
    """Admitted specs are frozen (reference validator.py:103-113:
    'updates to job spec are forbidden'): daemon-side mutation of a
    running job's spec must raise, and re-submitting the same name is
    rejected."""
    script = tmp_path / "sleep.py"
    script.write_text("import time; time.sleep(30)\n")
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="frozen-job",
                   job_dir=job_dir, gpus_per_replica=0, max_replicas=1)
    spec.max_replicas = 2            # mutable before admission
    controller.submit(spec)
    with pytest.raises(AttributeError, match="forbidden"):
        spec.max_replicas = 4        # frozen after admission
    with pytest.raises(AttributeError, match="forbidden"):
        spec.argv = ["true"]
    with pytest.raises(TypeError):
        spec.env["INJECTED"] = "1"   # env is a read-only mapping
    with pytest.raises(ValueError, match="forbidden"):
        controller.submit(JobSpec(["true"], name="frozen-job",
                                  job_dir=job_dir, gpus_per_replica=0))


def test_external_sigterm_preemption(tmp_path, controller):
    """Spot-instance-style external preemption: SIGTERM the running
    replica group from OUTSIDE the controller (as a cloud provider
    would); workers must checkpoint and exit(143), and the controller
    must treat it as a graceful preemption and restart the group, with
    training completing correctly (reference analog:
    ray/adaptdl_ray/aws/worker.py:34-70 spot reclamation)."""
    import signal as _signal
    script = tmp_path / "worker.py"
    script.write_text(WORKER.replace("@@REPO@@", REPO))
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="spot-job",
                   job_dir=job_dir, min_replicas=1, max_replicas=2,
                   gpus_per_replica=0)
    controller.submit(spec)

    deadline = time.time() + 60
    trace_path = os.path.join(job_dir, "trace.jsonl")
    while not os.path.exists(trace_path):
        assert time.time() < deadline, controller.status("spot-job")
        time.sleep(0.1)
    # External reclamation: signal the worker process groups directly.
    with controller._lock:
        pids = [p.pid for p in controller._jobs["spot-job"].procs]
    assert pids
    for pid in pids:
        os.killpg(pid, _signal.SIGTERM)

    state = controller.wait("spot-job", timeout=180)
    assert state == "Succeeded", controller.status("spot-job")
    assert controller.status("spot-job")["restarts"] >= 1
    trace = [json.loads(line) for line in open(trace_path)]
    assert sorted(set(t["epoch"] for t in trace)) == list(range(30))
    assert max(t["restarts"] for t in trace) >= 1
    final = json.load(open(os.path.join(job_dir, "final.json")))
    assert abs(final[0] - 3.0) < 0.1 and abs(final[1] - 4.0) < 0.1


def test_inplace_scaledown_no_restart(tmp_path, controller):
    """In-place (no-restart) scale-down: 2 -> 1 replicas via the SIGUSR2
    directive.  The surviving process must keep training WITHOUT a
    checkpoint-restart (num_restarts stays 0, ADAPTDL_NUM_RESTARTS never
    changes, every epoch runs exactly once) — model/optimizer state
    never leaves the survivor's memory (north star: in-HBM rejoin)."""
    script = tmp_path / "worker.py"
    script.write_text(WORKER.replace("@@REPO@@", REPO))
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="inplace-job",
                   job_dir=job_dir, min_replicas=3, max_replicas=3,
                   gpus_per_replica=0, inplace_scaledown=True)
    controller.submit(spec)

    deadline = time.time() + 90
    trace_path = os.path.join(job_dir, "trace.jsonl")
    while not os.path.exists(trace_path):
        assert time.time() < deadline, controller.status("inplace-job")
        time.sleep(0.1)
    with controller._lock:
        survivor_pid = controller._jobs["inplace-job"].procs[0].pid

    # 3 -> 2: the two survivors must re-rendezvous with each other.
    controller.rescale("inplace-job", 2)
    while controller.status("inplace-job")["replicas"] != 2:
        assert time.time() < deadline, controller.status("inplace-job")
        assert controller.status("inplace-job")["restarts"] == 0
        time.sleep(0.1)
    # let it train a bit at 2 before shrinking again
    n_lines = len(open(trace_path).readlines())
    while len(open(trace_path).readlines()) < n_lines + 2:
        assert time.time() < deadline
        time.sleep(0.1)

    # 2 -> 1: down to a solo survivor.
    controller.rescale("inplace-job", 1)
    while controller.status("inplace-job")["replicas"] != 1:
        assert time.time() < deadline, controller.status("inplace-job")
        assert controller.status("inplace-job")["restarts"] == 0
        time.sleep(0.1)
    with controller._lock:
        assert controller._jobs["inplace-job"].procs[0].pid == survivor_pid

    state = controller.wait("inplace-job", timeout=180)
    assert state == "Succeeded", controller.status("inplace-job")
    assert controller.status("inplace-job")["restarts"] == 0

    trace = [json.loads(line) for line in open(trace_path)]
    assert sorted(set(t["epoch"] for t in trace)) == list(range(30))
    # every epoch ran exactly once: the in-place path replays nothing
    assert len(trace) == 30
    assert {t["replicas"] for t in trace} == {3, 2, 1}
    assert all(t["restarts"] == 0 for t in trace)
    # no checkpoint was written anywhere (neither disk nor warm root)
    assert not any(n.startswith("checkpoint-")
                   for n in os.listdir(job_dir))
    final = json.load(open(os.path.join(job_dir, "final.json")))
    assert abs(final[0] - 3.0) < 0.1 and abs(final[1] - 4.0) < 0.1


def test_inplace_rescale_escalates_on_timeout(tmp_path, controller, monkeypatch):
    """Workers that never reach a safe point (no AdaptiveDataLoader
    iteration) cannot act on the in-place directive; the controller must
    escalate to the SIGTERM checkpoint-restart path after its timeout."""
    monkeypatch.setattr(LocalController, "INPLACE_TIMEOUT", 2.0)
    script = tmp_path / "sleeper.py"
    # Installs the SIGUSR2/SIGTERM handlers (via init_process_group)
    # but never iterates an AdaptiveDataLoader, so the directive is
    # seen but never acted on; on the escalation SIGTERM it exits 143.
    script.write_text(
        "import sys, time\n"
        "sys.path.insert(0, {!r})\n"
        "import adaptdl_amd.torch as adl\n"
        "from adaptdl_amd._signal import get_exit_flag\n"
        "adl.init_process_group('gloo')\n"
        "while not get_exit_flag():\n"
        "    time.sleep(0.05)\n"
        "sys.exit(143)\n".format(REPO))
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="stuck-job",
                   job_dir=job_dir, min_replicas=2, max_replicas=2,
                   gpus_per_replica=0, inplace_scaledown=True)
    controller.submit(spec)
    deadline = time.time() + 60
    while controller.status("stuck-job")["state"] != "Running":
        assert time.time() < deadline, controller.status("stuck-job")
        time.sleep(0.1)
    controller.rescale("stuck-job", 1)
    # Escalation: directive times out -> SIGTERM restart at 1 replica.
    while not (controller.status("stuck-job")["replicas"] == 1
               and controller.status("stuck-job")["state"] == "Running"):
        assert time.time() < deadline, controller.status("stuck-job")
        time.sleep(0.2)
    assert controller.status("stuck-job")["restarts"] >= 1
    controller.rescale("stuck-job", 0)  # cleanup is shutdown's job


def test_mixed_inplace_down_then_respawn_up(tmp_path, controller):
    """Interplay of the two rescale mechanisms: 2 -> 1 in place (no
    restart), then 1 -> 2 via the warm-checkpoint respawn path; state
    must flow survivor-memory -> warm RAM checkpoint -> new group, with
    every epoch still running exactly once and nothing on disk."""
    script = tmp_path / "worker.py"
    script.write_text(WORKER.replace("@@REPO@@", REPO))
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="mixed-job",
                   job_dir=job_dir, min_replicas=1, max_replicas=2,
                   gpus_per_replica=0, inplace_scaledown=True)
    controller.submit(spec)
    deadline = time.time() + 90
    trace_path = os.path.join(job_dir, "trace.jsonl")
    while not os.path.exists(trace_path):
        assert time.time() < deadline, controller.status("mixed-job")
        time.sleep(0.1)

    # needs 2 replicas first (min_replicas=1 may start at 1)
    if controller.status("mixed-job")["replicas"] != 2:
        controller.rescale("mixed-job", 2)
        while controller.status("mixed-job")["replicas"] != 2:
            assert time.time() < deadline, controller.status("mixed-job")
            time.sleep(0.1)
    base_restarts = controller.status("mixed-job")["restarts"]

    controller.rescale("mixed-job", 1)   # in place
    while controller.status("mixed-job")["replicas"] != 1:
        assert time.time() < deadline, controller.status("mixed-job")
        time.sleep(0.1)
    assert controller.status("mixed-job")["restarts"] == base_restarts

    n_lines = len(open(trace_path).readlines())
    while len(open(trace_path).readlines()) < n_lines + 1:
        assert time.time() < deadline
        time.sleep(0.1)

    controller.rescale("mixed-job", 2)   # respawn via warm checkpoint
    while not (controller.status("mixed-job")["replicas"] == 2 and
               controller.status("mixed-job")["state"] == "Running"):
        assert time.time() < deadline, controller.status("mixed-job")
        time.sleep(0.1)
    assert controller.status("mixed-job")["restarts"] == base_restarts + 1

    state = controller.wait("mixed-job", timeout=180)
    assert state == "Succeeded", controller.status("mixed-job")
    trace = [json.loads(line) for line in open(trace_path)]
    assert sorted(set(t["epoch"] for t in trace)) == list(range(30))
    assert len(trace) == 30  # no epoch replayed
    assert not any(n.startswith("checkpoint-")
                   for n in os.listdir(job_dir))
    final = json.load(open(os.path.join(job_dir, "final.json")))
    assert abs(final[0] - 3.0) < 0.1 and abs(final[1] - 4.0) < 0.1


BPTT_WORKER = textwrap.dedent("""
    import json, os, sys
    sys.path.insert(0, "@@REPO@@")
    import torch
    torch.set_num_threads(1)
    import torch.nn.functional as F
    import adaptdl_amd.env as env
    import adaptdl_amd.torch as adl

    adl.init_process_group("gloo")
    torch.manual_seed(3)
    vocab = 40
    corpus = torch.randint(0, vocab, (3000,))

    class LM(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.emb = torch.nn.Embedding(vocab, 12)
            self.fc = torch.nn.Linear(12, vocab)

        def forward(self, x):
            return self.fc(self.emb(x))

    model = LM()
    optim = torch.optim.SGD(model.parameters(), lr=0.05)
    adp = adl.AdaptiveDataParallel(model, optim)
    it = adl.AdaptiveBPTTIterator(corpus, batch_size=4, bptt_len=10)
    trace = os.path.join(env.checkpoint_path(), "trace.jsonl")
    for epoch in adl.remaining_epochs_until(25):
        for text, target in it:
            optim.zero_grad()
            loss = F.cross_entropy(adp(text).view(-1, vocab),
                                   target.reshape(-1))
            loss.backward()
            optim.step()
        if env.replica_rank() == 0:
            with open(trace, "a") as f:
                f.write(json.dumps(dict(
                    epoch=epoch, replicas=env.num_replicas(),
                    restarts=env.num_restarts())) + "\\n")
        import time as _t
        _t.sleep(0.05)
""")


def test_inplace_scaledown_bptt(tmp_path, controller):
    """The BPTT iterator also supports in-place scale-downs: 2 -> 1 with
    the corpus position remapped in place (no restart, no replay)."""
    script = tmp_path / "bptt_worker.py"
    script.write_text(BPTT_WORKER.replace("@@REPO@@", REPO))
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="bptt-job",
                   job_dir=job_dir, min_replicas=2, max_replicas=2,
                   gpus_per_replica=0, inplace_scaledown=True)
    controller.submit(spec)
    deadline = time.time() + 90
    trace_path = os.path.join(job_dir, "trace.jsonl")
    while not os.path.exists(trace_path):
        assert time.time() < deadline, controller.status("bptt-job")
        time.sleep(0.1)
    controller.rescale("bptt-job", 1)
    while controller.status("bptt-job")["replicas"] != 1:
        assert time.time() < deadline, controller.status("bptt-job")
        assert controller.status("bptt-job")["restarts"] == 0
        time.sleep(0.1)
    state = controller.wait("bptt-job", timeout=120)
    assert state == "Succeeded", controller.status("bptt-job")
    assert controller.status("bptt-job")["restarts"] == 0
    trace = [json.loads(line) for line in open(trace_path)]
    assert sorted(set(t["epoch"] for t in trace)) == list(range(25))
    assert len(trace) == 25
    assert {t["replicas"] for t in trace} == {2, 1}


ACCUM_WORKER = textwrap.dedent("""
    import json, os, sys
    sys.path.insert(0, "@@REPO@@")
    import torch
    torch.set_num_threads(1)
    import torch.nn.functional as F
    import adaptdl_amd.env as env
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.data import AdaptiveDataLoaderHelper

    # Force a 3-microbatch accumulation cycle so the in-place rescale
    # must defer to the cycle boundary.
    def fake_sync(self):
        self._state.current_local_bsz = 4
        self._state.accumulation_steps = 2
        return 4

    AdaptiveDataLoaderHelper._sync_local_bsz = fake_sync

    adl.init_process_group("gloo")
    torch.manual_seed(11)
    xs = torch.randn(96, 6)
    ys = (xs @ torch.tensor([[1.0], [2.0], [0.5], [-1.0], [0.3], [2.5]]))
    model = torch.nn.Linear(6, 1, bias=False)
    optim = torch.optim.SGD(model.parameters(), lr=0.02)
    adp = adl.AdaptiveDataParallel(model, optim)
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(xs, ys), batch_size=8)

    trace = os.path.join(env.checkpoint_path(), "trace.jsonl")
    for epoch in adl.remaining_epochs_until(25):
        for x, y in loader:
            optim.zero_grad()
            ((adp(x) - y) ** 2).mean().backward()
            optim.step()
        if env.replica_rank() == 0:
            with open(trace, "a") as f:
                f.write(json.dumps(dict(
                    epoch=epoch, replicas=env.num_replicas(),
                    restarts=env.num_restarts(),
                    accum=adp.gns.accum_count)) + "\\n")
        import time as _t
        _t.sleep(0.05)
    if env.replica_rank() == 0:
        with open(os.path.join(env.checkpoint_path(), "final.json"),
                  "w") as f:
            json.dump(model.weight.detach().reshape(-1).tolist(), f)
""")


def test_inplace_scaledown_with_accumulation(tmp_path, controller):
    """In-place rescale under gradient accumulation: the rejoin must
    land on an optimizer-cycle boundary (no partial gradients lost),
    with training completing at 0 restarts and finite learned weights."""
    script = tmp_path / "accum_worker.py"
    script.write_text(ACCUM_WORKER.replace("@@REPO@@", REPO))
    job_dir = str(tmp_path / "job")
    os.makedirs(job_dir)
    spec = JobSpec([sys.executable, str(script)], name="accum-job",
                   job_dir=job_dir, min_replicas=2, max_replicas=2,
                   gpus_per_replica=0, inplace_scaledown=True)
    controller.submit(spec)
    deadline = time.time() + 90
    trace_path = os.path.join(job_dir, "trace.jsonl")
    while not os.path.exists(trace_path):
        assert time.time() < deadline, controller.status("accum-job")
        time.sleep(0.1)
    controller.rescale("accum-job", 1)
    while controller.status("accum-job")["replicas"] != 1:
        assert time.time() < deadline, controller.status("accum-job")
        assert controller.status("accum-job")["restarts"] == 0
        time.sleep(0.1)
    state = controller.wait("accum-job", timeout=120)
    assert state == "Succeeded", controller.status("accum-job")
    assert controller.status("accum-job")["restarts"] == 0
    trace = [json.loads(line) for line in open(trace_path)]
    assert sorted(set(t["epoch"] for t in trace)) == list(range(25))
    assert len(trace) == 25
    assert {t["replicas"] for t in trace} == {2, 1}
    final = json.load(open(os.path.join(job_dir, "final.json")))
    assert all(abs(v) < 10 for v in final)
