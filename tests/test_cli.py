"""CLI tests: daemon admin API round-trip and foreground run.

Counterpart coverage for the reference's adaptdl CLI surface
(/root/reference/cli/bin/adaptdl: submit/ls/logs) in local-node form.
"""

import json
import os
import subprocess
import sys
import time
import urllib.request

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _req(url, method="GET", body=None):
    data = json.dumps(body).encode() if body is not None else None
    req = urllib.request.Request(url, data=data, method=method)
    with urllib.request.urlopen(req, timeout=20) as resp:
        payload = resp.read()
    return json.loads(payload) if payload else None


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_daemon_submit_ls_logs(tmp_path):
    port = _free_port()
    env = dict(os.environ, PYTHONPATH=REPO)
    daemon = subprocess.Popen(
        [sys.executable, "-m", "adaptdl_amd.cli", "daemon",
         "--bind", "127.0.0.1:{}".format(port),
         "--state-dir", str(tmp_path / "state"),
         "--num-gpus", "0", "--interval", "3600"],
        env=env, cwd=str(tmp_path),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    url = "http://127.0.0.1:{}".format(port)
    try:
        deadline = time.time() + 30
        while True:
            try:
                _req(url + "/jobs")
                break
            except Exception:
                assert time.time() < deadline
                time.sleep(0.2)

        script = tmp_path / "hello.py"
        script.write_text("print('hello from worker')\n")
        out = _req(url + "/jobs", "POST", {
            "argv": [sys.executable, str(script)], "name": "hello",
            "min_replicas": 1, "max_replicas": 1, "gpus_per_replica": 0})
        assert out["name"] == "hello"

        deadline = time.time() + 60
        while True:
            st = _req(url + "/jobs/hello")
            if st["state"] in ("Succeeded", "Failed"):
                break
            assert time.time() < deadline, st
            time.sleep(0.3)
        assert st["state"] == "Succeeded"
        assert st["job_dir"]  # consumed by `adaptdl-amd tensorboard`

        jobs = _req(url + "/jobs")
        assert "hello" in jobs

        logs = _req(url + "/jobs/hello/logs")
        assert any("hello from worker" in v for v in logs.values())
    finally:
        try:
            _req(url + "/shutdown", "POST", {})
        except Exception:
            pass
        try:
            daemon.wait(timeout=20)
        except subprocess.TimeoutExpired:
            daemon.kill()


def test_cli_run_foreground(tmp_path):
    script = tmp_path / "ok.py"
    script.write_text("print('ran fine')\n")
    env = dict(os.environ, PYTHONPATH=REPO)
    p = subprocess.run(
        [sys.executable, "-m", "adaptdl_amd.cli", "run",
         "--name", "fg", "--job-dir", str(tmp_path / "fg"),
         "--num-gpus", "0", "--gpus-per-replica", "0",
         "--min-replicas", "1", "--max-replicas", "1", "--",
         sys.executable, str(script)],
        env=env, cwd=str(tmp_path), capture_output=True, text=True,
        timeout=90)
    assert p.returncode == 0, p.stdout + p.stderr
    assert "Succeeded" in p.stdout
    assert "ran fine" in p.stdout


def test_tensorboard_command_prints_fallback(tmp_path, capsys,
                                             monkeypatch):
    """`adaptdl-amd tensorboard --logdir D` without the tensorboard
    package prints the command instead of exec'ing it."""
    import shutil
    from adaptdl_amd import cli
    monkeypatch.setattr(shutil, "which", lambda name: None)
    cli.main(["tensorboard", "--logdir", str(tmp_path), "--port",
              "7007"])
    out = capsys.readouterr().out
    assert "tensorboard is not installed" in out
    assert "--logdir {}".format(tmp_path) in out
    assert "--port 7007" in out


def test_tensorboard_command_requires_name_or_logdir():
    from adaptdl_amd import cli
    with pytest.raises(SystemExit):
        cli.main(["tensorboard"])


def test_trace_gaps_union_math(tmp_path):
    """tools/trace_gaps.py: union-busy / idle accounting on a synthetic
    kernel trace (overlaps must not double-count)."""
    sys.path.insert(0, os.path.join(REPO, "tools"))
    import trace_gaps

    # Span 0..1000ns; kernels: [0,100), [50,200) (overlap), [300,400),
    # [900,1000).  Union busy = 200+100+100 = 400; idle = 600.
    csv_path = tmp_path / "t_kernel_trace.csv"
    csv_path.write_text(
        "Kernel_Name,Start_Timestamp,End_Timestamp\n"
        "a,0,100\nb,50,200\nc,300,400\nd,900,1000\n")
    out = trace_gaps.analyze(
        trace_gaps.load_intervals([str(csv_path)]), tail=1.0)
    assert out["kernels"] == 4
    assert abs(out["busy_ms"] - 400e-6) < 1e-12
    assert abs(out["idle_ms"] - 600e-6) < 1e-12
    assert abs(out["idle_pct"] - 60.0) < 1e-9
    # Largest gap is 500ns (400 -> 900).
    assert out["top_gaps_us"][0] == 0.5

    # Tail window 0.5: cut at 500; only [900,1000) remains -> idle 400.
    out = trace_gaps.analyze(
        trace_gaps.load_intervals([str(csv_path)]), tail=0.5)
    assert out["kernels"] == 1
    assert abs(out["idle_pct"] - 80.0) < 1e-9


def test_cli_cp(tmp_path):
    """`adaptdl-amd cp NAME:path dst` copies out of a job directory
    resolved from the daemon (reference `adaptdl cp` parity)."""
    port = _free_port()
    env = dict(os.environ, PYTHONPATH=REPO)
    daemon = subprocess.Popen(
        [sys.executable, "-m", "adaptdl_amd.cli", "daemon",
         "--bind", "127.0.0.1:{}".format(port),
         "--state-dir", str(tmp_path / "state"),
         "--num-gpus", "0", "--interval", "3600"],
        env=env, cwd=str(tmp_path),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    url = "http://127.0.0.1:{}".format(port)
    try:
        deadline = time.time() + 30
        while True:
            try:
                _req(url + "/jobs")
                break
            except Exception:
                assert time.time() < deadline
                time.sleep(0.2)
        script = tmp_path / "writer.py"
        script.write_text(
            "import os\n"
            "p = os.environ['ADAPTDL_CHECKPOINT_PATH']\n"
            "open(os.path.join(p, 'result.txt'), 'w').write('payload')\n")
        _req(url + "/jobs", "POST", {
            "argv": [sys.executable, str(script)], "name": "writer",
            "min_replicas": 1, "max_replicas": 1, "gpus_per_replica": 0})
        deadline = time.time() + 30
        while _req(url + "/jobs/writer")["state"] != "Succeeded":
            assert time.time() < deadline, _req(url + "/jobs/writer")
            time.sleep(0.2)
        dst = tmp_path / "out.txt"
        out = subprocess.run(
            [sys.executable, "-m", "adaptdl_amd.cli", "cp",
             "writer:result.txt", str(dst), "--url", url],
            env=env, capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, out.stderr
        assert dst.read_text() == "payload"
    finally:
        daemon.terminate()
        daemon.wait(timeout=10)


def test_daemon_cluster_endpoint(tmp_path):
    """GET /cluster exposes the Pollux desired-node count (the
    cluster-expander signal for external provisioners)."""
    port = _free_port()
    env = dict(os.environ, PYTHONPATH=REPO)
    daemon = subprocess.Popen(
        [sys.executable, "-m", "adaptdl_amd.cli", "daemon",
         "--bind", "127.0.0.1:{}".format(port),
         "--state-dir", str(tmp_path / "state"),
         "--num-gpus", "4", "--interval", "3600"],
        env=env, cwd=str(tmp_path),
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    url = "http://127.0.0.1:{}".format(port)
    try:
        deadline = time.time() + 30
        while True:
            try:
                out = _req(url + "/cluster")
                break
            except Exception:
                assert time.time() < deadline
                time.sleep(0.2)
        assert out["num_gpus"] == 4
        assert isinstance(out["desired_nodes"], int)
        assert out["desired_nodes"] >= 0
    finally:
        daemon.terminate()
        daemon.wait(timeout=10)
