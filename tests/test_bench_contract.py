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
