"""Workload-path tests: two ADP instances, AdamScale, mixed precision.

Covers the reference's example-exercised paths on CPU/gloo:
- DCGAN's dual named AdaptiveDataParallel instances
  (/root/reference/examples/dcgan/dcgan.py:500-501),
- BERT/NCF's Adam -> AdamScale + AdamGradientNoiseScale selection,
- the mp_scaler (GradScaler) statistics-unscaling path
  (/root/reference/adaptdl/adaptdl/torch/gradient_noise_scale.py:218-249).
"""

import numpy as np
import torch

import adaptdl_amd.collective
import adaptdl_amd.env
from conftest import elastic_multiprocessing


@elastic_multiprocessing
def _run_two_adp_instances():
    import adaptdl_amd.torch as adl
    adaptdl_amd.collective.initialize("127.0.0.1")
    torch.manual_seed(0)
    netG = torch.nn.Linear(4, 8)
    netD = torch.nn.Linear(8, 1)
    optG = torch.optim.Adam(netG.parameters(), lr=1e-3)
    optD = torch.optim.Adam(netD.parameters(), lr=1e-3)
    adpG = adl.AdaptiveDataParallel(netG, optG, name="netG")
    adpD = adl.AdaptiveDataParallel(netD, optD, name="netD")

    data = torch.randn(64, 8)
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(data), batch_size=16)
    for epoch in adl.remaining_epochs_until(2):
        for (real,) in loader:
            b = real.size(0)
            optD.zero_grad()
            fake = adpG(torch.randn(b, 4))
            loss_d = adpD(real).mean() - adpD(fake.detach()).mean()
            loss_d.backward()
            optD.step()
            optG.zero_grad()
            loss_g = adpD(fake).mean()
            loss_g.backward()
            optG.step()
    assert np.isfinite(adpG.gns.sqr_avg())
    assert np.isfinite(adpD.gns.sqr_avg())
    assert adpG.gain >= 0 and adpD.gain >= 0
    adaptdl_amd.collective.teardown()
    return 0


def test_two_adp_instances():
    _run_two_adp_instances()


@elastic_multiprocessing
def _run_adamscale_selection():
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.scaling_rules import AdamScale
    from adaptdl_amd.torch.gradient_noise_scale import \
        AdamGradientNoiseScale
    adaptdl_amd.collective.initialize("127.0.0.1")
    torch.manual_seed(1)
    model = torch.nn.Sequential(torch.nn.Linear(6, 12), torch.nn.GELU(),
                                torch.nn.Linear(12, 3))
    optim = torch.optim.Adam(model.parameters(), lr=1e-3)
    adp = adl.AdaptiveDataParallel(model, optim)
    assert isinstance(adp.scaling_rule, AdamScale)
    assert isinstance(adp.gns, AdamGradientNoiseScale)

    xs = torch.randn(96, 6)
    ys = torch.randint(0, 3, (96,))
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(xs, ys), batch_size=12)
    loader.autoscale_batch_size(96, local_bsz_bounds=(4, 48),
                                gradient_accumulation=True)
    for epoch in adl.remaining_epochs_until(3):
        for x, y in loader:
            optim.zero_grad()
            torch.nn.functional.cross_entropy(adp(x), y).backward()
            optim.step()
    assert np.isfinite(adp.gns.sqr_avg())
    assert np.isfinite(adp.gns.var_avg())
    # Adam preconditioning produced estimates after the first 5 steps.
    assert adp.gns.sqr_avg() >= 0
    adaptdl_amd.collective.teardown()
    return 0


def test_adamscale_selection_and_training():
    _run_adamscale_selection()


@elastic_multiprocessing
def _run_mp_scaler_path():
    import adaptdl_amd.torch as adl
    adaptdl_amd.collective.initialize("127.0.0.1")
    torch.manual_seed(2)
    model = torch.nn.Linear(5, 2)
    optim = torch.optim.SGD(model.parameters(), lr=0.05)
    scaler = torch.amp.GradScaler("cpu", init_scale=64.0, enabled=True)
    adp = adl.AdaptiveDataParallel(model, optim, mp_scaler=scaler)

    xs = torch.randn(64, 5)
    ys = torch.randint(0, 2, (64,))
    loader = adl.AdaptiveDataLoader(
        torch.utils.data.TensorDataset(xs, ys), batch_size=16)
    for epoch in adl.remaining_epochs_until(3):
        for x, y in loader:
            optim.zero_grad()
            loss = torch.nn.functional.cross_entropy(adp(x), y)
            scaler.scale(loss).backward()
            scaler.step(optim)
            scaler.update()
    # Statistics must be unscaled: gradient norms of this tiny problem
    # are O(1), nowhere near the 64^2 scale of the raw scaled grads.
    # (sqr_avg itself may sit at ~0: with random labels the true
    # gradient vanishes and noise dominates.)
    assert 0 <= adp.gns.sqr_avg() < 100.0
    assert 0 < adp.gns.var_avg() < 100.0
    adaptdl_amd.collective.teardown()
    return 0


def test_mp_scaler_statistics_unscaled():
    _run_mp_scaler_path()


def test_cifar_example_script_cpu_smoke(tmp_path):
    """Run examples/pytorch-cifar/main.py end-to-end on CPU (1 epoch,
    synthetic, no autoscale): guards the script itself — the underlying
    paths have their own tests, but edits to the example (e.g. the
    graph-stepper wiring) only showed up on GPU before this."""
    import os
    import subprocess
    import sys
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "examples", "pytorch-cifar",
                                      "main.py"),
         "--epochs", "1", "--samples", "128", "--bs", "64",
         "--max-bs", "0"],
        env=dict(os.environ, ADAPTDL_CHECKPOINT_PATH=str(tmp_path),
                 OMP_NUM_THREADS="2", PYTHONPATH=repo),
        cwd=repo, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stdout[-1500:] + out.stderr[-1500:]
    assert "epoch 0" in out.stdout
