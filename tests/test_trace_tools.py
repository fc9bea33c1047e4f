"""tools/trace_gaps.py + tools/xgmi_sweep.sh plumbing (CPU).

The first multi-GPU lease runs tools/xgmi_sweep.sh to tune the bucket
cap for xGMI in one shot; these tests keep the script and the overlap
analyzer runnable without hardware.
"""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(REPO, "tools"))


def test_trace_gaps_comm_overlap(tmp_path):
    """Union/idle math and the RCCL-overlap metric on a hand-built
    trace: comm [1500,2500]+[3100,3300], compute [1000,2000]+[2600,3000]
    -> overlap exactly [1500,2000] = 500ns = 41.7% of comm time."""
    trace = tmp_path / "t.csv"
    trace.write_text(
        '"Kernel_Name","Start_Timestamp","End_Timestamp"\n'
        '"k_conv3x3_wrw",1000,2000\n'
        '"ncclDevKernel_AllReduce_Sum_f32_RING",1500,2500\n'
        '"k_bn_fwd",2600,3000\n'
        '"ncclDevKernel_AllReduce_Sum_f32_RING",3100,3300\n')
    import trace_gaps
    out = trace_gaps.analyze(trace_gaps.load_intervals([str(trace)]),
                             tail=1.0)
    assert out["kernels"] == 4
    assert out["comm_kernels"] == 2
    assert abs(out["comm_overlap_ms"] - 500 / 1e6) < 1e-12
    assert abs(out["comm_overlap_pct"] - 100.0 * 500 / 1200) < 1e-9
    # union busy = 2100ns over span 2300ns
    assert abs(out["busy_ms"] - 2100 / 1e6) < 1e-12


def test_xgmi_sweep_dry_run(tmp_path):
    """DRY_RUN=1 exercises the full sweep plumbing (torchrun 2-rank CPU
    bench per cell, CSV output) without GPUs."""
    env = dict(os.environ)
    env["DRY_RUN"] = "1"
    env["OUT"] = str(tmp_path / "sweep.csv")
    env["ADAPTDL_CHECKPOINT_PATH"] = str(tmp_path)
    env["OMP_NUM_THREADS"] = "1"
    out = subprocess.run(["bash", os.path.join(REPO, "tools",
                                               "xgmi_sweep.sh"), "2"],
                         env=env, cwd=REPO, capture_output=True,
                         text=True, timeout=900)
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    rows = (tmp_path / "sweep.csv").read_text().strip().splitlines()
    assert rows[0].startswith("bucket_cap_mb")
    assert len(rows) == 3  # header + 2 dry-run cells
    for row in rows[1:]:
        assert "FAIL" not in row, rows
