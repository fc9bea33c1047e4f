"""bench.py driver-contract smoke: single process, CPU, tiny config.

The round driver runs `python bench.py` (N=1) and parses ONE JSON line
from rank 0; this guards the flag surface and the JSON schema against
regressions without a GPU.
"""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_single_process_json_contract(tmp_path):
    env = dict(os.environ)
    env["ADAPTDL_CHECKPOINT_PATH"] = str(tmp_path)
    env.pop("ADAPTDL_NUM_REPLICAS", None)
    env.pop("ADAPTDL_REPLICA_RANK", None)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "2",
         "--warmup", "5", "--max-batch", "256", "--dataset-size", "1024",
         "--pool", "128"],
        env=env, cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, out.stdout[-2000:]
    result = json.loads(lines[0])
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in result, field
    assert result["n_gpus"] == 1
    assert result["steps"] == 2
    assert result["scaling"] in ("weak", "strong")
    assert result["higher_is_better"] is True
    assert result["value"] > 0
    cfg = result["config"]
    for field in ("model", "global_batch", "parallelism"):
        assert field in cfg, field
    assert cfg["parallelism"] == "dp1"


def test_bench_two_rank_torchrun_contract(tmp_path):
    """The driver's N>1 launch: torch.distributed.run with 2 CPU/gloo
    ranks through the FULL bench (warmup, fit, probe, timed region,
    max-over-ranks aggregation, rank-0-only JSON)."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    env = dict(os.environ)
    env["ADAPTDL_CHECKPOINT_PATH"] = str(tmp_path)
    env["OMP_NUM_THREADS"] = "1"
    env.pop("ADAPTDL_NUM_REPLICAS", None)
    env.pop("ADAPTDL_REPLICA_RANK", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "5",
         "--init-batch", "64", "--max-batch", "128",
         "--dataset-size", "512", "--pool", "64"],
        env=env, cwd=REPO, capture_output=True, text=True, timeout=900)
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, out.stdout[-2000:]  # rank 0 only
    result = json.loads(lines[0])
    assert result["n_gpus"] == 2
    assert result["config"]["parallelism"] == "dp2"
    assert result["value"] > 0
    assert result["ms_per_step"] > 0


def _run_bench(tmp_path, extra_args, timeout=600):
    env = dict(os.environ)
    env["ADAPTDL_CHECKPOINT_PATH"] = str(tmp_path)
    env.pop("ADAPTDL_NUM_REPLICAS", None)
    env.pop("ADAPTDL_REPLICA_RANK", None)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py")] + extra_args,
        env=env, cwd=REPO, capture_output=True, text=True, timeout=timeout)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, out.stdout[-2000:]
    return json.loads(lines[0])


def test_bench_stat_efficiency_is_deterministic(tmp_path):
    """Two identical runs must quote the SAME efficiency factor (the
    fixed-seed init-weights probe)."""
    args = ["--steps", "1", "--warmup", "5", "--init-batch", "64",
            "--max-batch", "256", "--dataset-size", "1024", "--pool", "64"]
    r1 = _run_bench(tmp_path / "a", args)
    r2 = _run_bench(tmp_path / "b", args)
    assert r1["config"]["stat_efficiency"] == \
        r2["config"]["stat_efficiency"]
    assert r1["config"]["stat_efficiency_source"] == "fixed-seed-init-probe"


def test_bench_model_bert(tmp_path):
    """BERT bench config (BASELINE configs[3] shape): FusedAdam +
    AdamScale + accumulation through the same JSON contract."""
    r = _run_bench(tmp_path, ["--model", "bert-mini", "--steps", "1",
                              "--warmup", "5", "--eff-probe", "2"])
    assert r["config"]["model"] == "bert-mini"
    assert r["value"] > 0
    assert "BERT" in r["metric"]


def test_bench_model_transformer(tmp_path):
    """Transformer bench config (BASELINE configs[2] shape): BPTT
    iterator + adaptive batch through the same JSON contract."""
    r = _run_bench(tmp_path, ["--model", "transformer-wt2", "--steps", "1",
                              "--warmup", "5", "--eff-probe", "2",
                              "--init-batch", "8", "--max-batch", "32",
                              "--bounds", "4,16"])
    assert r["config"]["model"] == "transformer-wt2"
    assert r["config"]["seq_len"] == 35
    assert r["value"] > 0
