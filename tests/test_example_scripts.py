"""CPU smoke of every acceptance example script (tiny synthetic runs).

The underlying library paths have dedicated tests; these execute the
example PROGRAMS end-to-end (argparse surface, training loop wiring,
output line) so edits to an example can't regress silently until a GPU
round-end sweep.  pytorch-cifar has its own smoke in test_workloads.py.
"""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

_CASES = {
    "linear_regression": (["examples/linear_regression/main.py",
                           "--epochs", "3"], "epoch"),
    "transformer": (["examples/transformer/main.py", "--epochs", "1",
                     "--vocab", "500", "--bs", "16", "--bptt", "16"],
                    "ppl"),
    "bert": (["examples/BERT/main.py", "--config", "mini", "--epochs",
              "1", "--bs", "8"], "epoch"),
    "ncf": (["examples/NCF/main.py", "--epochs", "1", "--users", "100",
             "--items", "150", "--samples", "2000"], "epoch"),
    "dcgan": (["examples/dcgan/main.py", "--epochs", "1", "--bs", "32",
               "--samples", "256"], "epoch"),
    "mnist-tutorial": (["examples/tutorial/mnist.py", "--epochs", "1",
                        "--train-samples", "512", "--bs", "64"],
                       "epoch"),
}


@pytest.mark.parametrize("name", sorted(_CASES))
def test_example_script_cpu(name, tmp_path):
    argv, expect = _CASES[name]
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, argv[0])] + argv[1:],
        env=dict(os.environ, ADAPTDL_CHECKPOINT_PATH=str(tmp_path),
                 OMP_NUM_THREADS="2", PYTHONPATH=REPO),
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    assert expect in out.stdout, out.stdout[-500:]
