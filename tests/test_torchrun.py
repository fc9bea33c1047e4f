"""torchrun-compatibility: the driver's exact launch path.

The scaling benchmark is launched as `python -m torch.distributed.run
--nnodes=1 --nproc-per-node N ... bench.py` — under torchelastic every
worker is a c10d-store CLIENT (TORCHELASTIC_USE_AGENT_STORE), so
init_process_group must use the env:// rendezvous there instead of
broadcasting a fresh tcp:// port (which would have no server and hang).
"""

import os
import socket
import subprocess
import sys
import textwrap

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SCRIPT = textwrap.dedent("""
    import sys
    sys.path.insert(0, {repo!r})
    import torch
    torch.set_num_threads(1)
    import adaptdl_amd.collective
    import adaptdl_amd.env as env
    import adaptdl_amd.torch as adl
    adl.init_process_group("gloo")
    t = torch.ones(4) * (env.replica_rank() + 1)
    torch.distributed.all_reduce(t)
    assert t[0].item() == 3.0, t  # 1 + 2
    v = adaptdl_amd.collective.broadcast("hello-{{}}".format(env.replica_rank()))
    assert v == "hello-0", v
    torch.distributed.barrier()
    if env.replica_rank() == 0:
        print("TORCHRUN_PATH_OK")
    torch.distributed.destroy_process_group()
""")


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_torchrun_init(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(SCRIPT.format(repo=REPO))
    p = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), str(script)],
        capture_output=True, text=True, timeout=180,
        env=dict(os.environ, PYTHONPATH=REPO))
    assert p.returncode == 0, p.stdout[-2000:] + p.stderr[-2000:]
    assert "TORCHRUN_PATH_OK" in p.stdout
