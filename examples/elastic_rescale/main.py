"""Elastic rescale drill: ResNet-50 synthetic, 1 -> N -> N/2 replicas.

BASELINE.json config 5: "ResNet-50 ImageNet-shape synthetic, elastic
rescale via in-process Pollux allocator + checkpoint/restart".  This
script IS the driver: it starts a LocalController, submits the training
job at 1 replica, then forces the rescale sequence, demonstrating the
SIGTERM -> checkpoint -> exit(143) -> restart protocol with worker
processes bound to GPUs via HIP_VISIBLE_DEVICES.

Usage:  python main.py [--gpus N] [--phase-seconds S]
On a CPU-only host it runs the same drill on gloo workers.
"""

import argparse
import json
import os
import sys
import textwrap
import time

REPO = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))
sys.path.insert(0, REPO)

from adaptdl_amd.sched import JobSpec, LocalController  # noqa: E402

WORKER = textwrap.dedent("""
    import json, os, sys, time
    sys.path.insert(0, {repo!r})
    import torch
    import adaptdl_amd.env as env
    import adaptdl_amd.torch as adl
    import torch.nn.functional as F
    from adaptdl_amd.models import ResNet50, ResNet50Cifar

    use_gpu = torch.cuda.is_available()
    adl.init_process_group("nccl" if use_gpu else "gloo")
    device = torch.device("cuda" if use_gpu else "cpu")
    torch.manual_seed(7)
    torch.set_num_threads(2)
    # CPU drill uses the CIFAR-sized ResNet-50 so steps stay fast
    # enough for prompt SIGTERM/checkpoint turnaround.
    model = (ResNet50() if use_gpu else ResNet50Cifar(1000)).to(device)
    if use_gpu:
        model = model.to(memory_format=torch.channels_last)
    optim = adl.FusedSGD(model.parameters(), lr=0.1, momentum=0.9)
    adp = adl.AdaptiveDataParallel(model, optim)

    res = 32 if not use_gpu else 224
    bs = 2 if not use_gpu else 64
    pool = torch.randn(2 * bs, 3, res, res, device=device)
    if use_gpu:
        pool = pool.contiguous(memory_format=torch.channels_last)
    ys = torch.randint(0, 1000, (2 * bs,), device=device)
    nsteps = 32 if not use_gpu else 512
    dataset = torch.utils.data.TensorDataset(torch.arange(nsteps * bs))
    loader = adl.AdaptiveDataLoader(dataset, batch_size=bs,
                                    drop_last=True)

    trace = os.path.join(env.checkpoint_path(), "trace.jsonl")
    REPORT = 8
    for epoch in adl.remaining_epochs_until(10000):
        t0, steps = time.time(), 0
        for (idx,) in loader:
            sel = idx % (2 * bs)
            x, y = pool[sel], ys[sel]
            optim.zero_grad()
            if use_gpu:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    loss = F.cross_entropy(adp(x), y)
            else:
                loss = F.cross_entropy(adp(x), y)
            loss.backward()
            optim.step()
            steps += 1
            if steps % REPORT == 0 and env.replica_rank() == 0:
                if use_gpu:
                    torch.cuda.synchronize()
                dt = time.time() - t0
                gbs = loader.current_batch_size or (
                    bs * env.num_replicas())
                with open(trace, "a") as f:
                    f.write(json.dumps(dict(
                        epoch=epoch, replicas=env.num_replicas(),
                        restarts=env.num_restarts(),
                        images_per_sec=REPORT * gbs / dt)) + "\\n")
                t0 = time.time()
""")


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=None)
    parser.add_argument("--phase-seconds", type=float, default=30.0)
    parser.add_argument("--job-dir", default=None)
    args = parser.parse_args()

    try:
        import torch
        have = torch.cuda.device_count() if torch.cuda.is_available() \
            else 0
    except Exception:
        have = 0
    gpus = args.gpus if args.gpus is not None else have
    gpr = 1 if gpus > 0 else 0
    peak = gpus if gpus > 0 else 2   # CPU mode drills with 2 workers
    if peak < 2:
        print("note: only 1 GPU visible; the drill still exercises the "
              "checkpoint-restart protocol at 1 replica")
        phases = [1, "restart", 1]
    else:
        phases = [1, peak, max(peak // 2, 1)]
    print("rescale drill phases (replicas):", phases)

    job_dir = os.path.abspath(args.job_dir or ".adaptdl/elastic-drill")
    os.makedirs(job_dir, exist_ok=True)
    script = os.path.join(job_dir, "worker.py")
    with open(script, "w") as f:
        f.write(WORKER.format(repo=REPO))

    ctrl = LocalController(num_gpus=max(gpus, peak if gpr == 0 else 0),
                           interval=3600)
    spec = JobSpec([sys.executable, script], name="elastic-resnet50",
                   job_dir=job_dir, min_replicas=1, max_replicas=peak,
                   gpus_per_replica=gpr,
                   # Scale-downs (the N -> N/2 phase) rejoin in place:
                   # survivors keep model/optimizer state in HBM, no
                   # checkpoint-restart (scale-ups still respawn via the
                   # warm RAM checkpoint).
                   inplace_scaledown=True)
    ctrl.submit(spec)
    try:
        for target in phases:
            if target == "restart":
                ctrl.restart("elastic-resnet50")
                target = ctrl.status("elastic-resnet50")["replicas"]
            else:
                ctrl.rescale("elastic-resnet50", target)
            # Wait for the checkpoint-restart to land on the target...
            deadline = time.time() + 300
            while time.time() < deadline:
                st = ctrl.status("elastic-resnet50")
                if st["state"] == "Failed":
                    raise SystemExit("job failed: " + str(st))
                if st["state"] == "Running" and                         st["replicas"] == target:
                    break
                time.sleep(1.0)
            else:
                raise SystemExit("rescale to {} timed out: {}".format(
                    target, ctrl.status("elastic-resnet50")))
            print("[drill] running at {} replicas (restarts={})".format(
                target, st["restarts"]))
            # ...then hold the phase for steady-state measurements.
            time.sleep(args.phase_seconds)
    finally:
        ctrl.shutdown()

    trace_path = os.path.join(job_dir, "trace.jsonl")
    if os.path.exists(trace_path):
        by_replicas = {}
        for line in open(trace_path):
            rec = json.loads(line)
            by_replicas.setdefault(rec["replicas"], []).append(
                rec["images_per_sec"])
        print("throughput by replica count:")
        for r in sorted(by_replicas):
            xs = by_replicas[r]
            print("  {} replicas: {:.1f} images/s (median of {})".format(
                r, sorted(xs)[len(xs) // 2], len(xs)))


if __name__ == "__main__":
    main()
