"""adaptdl-amd CLI: submit and manage elastic jobs on this node.

Single-node counterpart of the reference's ``adaptdl`` CLI
(/root/reference/cli/bin/adaptdl:436-481 — submit/logs/ls/cp/
tensorboard against a Kubernetes cluster).  Here jobs run as local
worker processes under adaptdl_amd.sched.LocalController:

  adaptdl-amd run [opts] -- python train.py ...   # run in-process, wait
  adaptdl-amd daemon [opts]                       # persistent controller
  adaptdl-amd submit [opts] -- python train.py    # submit to the daemon
  adaptdl-amd ls                                  # list daemon jobs
  adaptdl-amd logs NAME [--rank R]                # print worker logs
  adaptdl-amd rescale NAME N                      # force a replica count
  adaptdl-amd stop                                # shut the daemon down
  adaptdl-amd tensorboard NAME|--logdir DIR       # TensorBoard on a job dir

The daemon exposes a small JSON/HTTP admin API (POST /jobs, GET /jobs,
GET /jobs/{name}, GET /jobs/{name}/logs, POST /jobs/{name}/rescale,
POST /shutdown), default address http://127.0.0.1:8077
(ADAPTDL_DAEMON_URL).
"""

import argparse
import json
import os
import signal
import sys
import threading
import time
import urllib.request
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

DEFAULT_URL = os.getenv("ADAPTDL_DAEMON_URL", "http://127.0.0.1:8077")


def _split_command(argv):
    if "--" in argv:
        i = argv.index("--")
        return argv[:i], argv[i + 1:]
    return argv, []


def _request(url, method="GET", body=None):
    data = json.dumps(body).encode() if body is not None else None
    req = urllib.request.Request(url, data=data, method=method,
                                 headers={"Content-Type":
                                          "application/json"})
    with urllib.request.urlopen(req, timeout=30) as resp:
        payload = resp.read()
    return json.loads(payload) if payload else None


class _AdminServer(object):
    def __init__(self, controller, host, port):
        self.controller = controller
        srv = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, fmt, *args):
                pass

            def _reply(self, code, obj=None):
                body = json.dumps(obj).encode() if obj is not None else b""
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                parts = [p for p in self.path.split("?")[0].split("/")
                         if p]
                try:
                    if parts == ["jobs"]:
                        self._reply(200, srv.controller.jobs())
                    elif parts == ["cluster"]:
                        self._reply(200, {
                            "num_gpus": srv.controller.num_gpus,
                            "desired_nodes":
                                srv.controller.desired_nodes()})
                    elif len(parts) == 2 and parts[0] == "jobs":
                        self._reply(200, srv.controller.status(parts[1]))
                    elif len(parts) == 3 and parts[0] == "jobs" and \
                            parts[2] == "logs":
                        out = {}
                        for p in srv.controller.log_paths(parts[1]):
                            with open(p) as f:
                                out[os.path.basename(p)] = f.read()
                        self._reply(200, out)
                    else:
                        self._reply(404)
                except KeyError:
                    self._reply(404, {"error": "no such job"})

            def do_POST(self):
                n = int(self.headers.get("Content-Length", "0"))
                body = json.loads(self.rfile.read(n) or b"{}")
                parts = [p for p in self.path.split("/") if p]
                try:
                    if parts == ["jobs"]:
                        from adaptdl_amd.sched import JobSpec
                        name = body["name"]
                        job_dir = body.get("job_dir") or os.path.join(
                            srv.state_dir, name)
                        spec = JobSpec(
                            body["argv"], name=name, job_dir=job_dir,
                            min_replicas=body.get("min_replicas", 0),
                            max_replicas=body.get("max_replicas", 8),
                            gpus_per_replica=body.get("gpus_per_replica",
                                                      1),
                            env=body.get("env") or {},
                            workdir=body.get("workdir"),
                            preemptible=body.get("preemptible", True),
                            inplace_scaledown=body.get(
                                "inplace_scaledown", False))
                        srv.controller.submit(spec)
                        self._reply(200, {"name": name,
                                          "job_dir": job_dir})
                    elif len(parts) == 3 and parts[0] == "jobs" and \
                            parts[2] == "rescale":
                        srv.controller.rescale(parts[1],
                                               int(body["replicas"]))
                        self._reply(200, {"status": "ok"})
                    elif parts == ["shutdown"]:
                        self._reply(200, {"status": "bye"})
                        threading.Thread(target=srv.stop_all,
                                         daemon=True).start()
                    else:
                        self._reply(404)
                except KeyError as e:
                    self._reply(400, {"error": str(e)})
                except ValueError as e:
                    # Admission rejection (invalid spec, or an attempt
                    # to re-submit/mutate an existing job's spec —
                    # reference validator.py:103-113 semantics).
                    self._reply(422, {"error": str(e)})

        self.state_dir = None
        self._server = ThreadingHTTPServer((host, port), Handler)
        self._server.daemon_threads = True
        self._done = threading.Event()

    def serve_forever(self):
        t = threading.Thread(target=self._server.serve_forever,
                             daemon=True)
        t.start()
        try:
            self._done.wait()
        except KeyboardInterrupt:
            pass
        self._server.shutdown()
        self.controller.shutdown()

    def stop_all(self):
        self._done.set()


def cmd_daemon(args):
    from adaptdl_amd.sched import LocalController
    state_dir = os.path.abspath(args.state_dir)
    os.makedirs(state_dir, exist_ok=True)
    controller = LocalController(num_gpus=args.num_gpus,
                                 interval=args.interval,
                                 metrics_port=args.metrics_port)
    host, port = args.bind.rsplit(":", 1)
    admin = _AdminServer(controller, host, int(port))
    admin.state_dir = state_dir
    print("adaptdl-amd daemon on http://{} (state: {}, gpus: {})".format(
        args.bind, state_dir, controller.num_gpus))
    admin.serve_forever()


def cmd_run(args, command):
    from adaptdl_amd.sched import JobSpec, LocalController
    if not command:
        sys.exit("run: missing command after --")
    name = args.name or "job-{}".format(int(time.time()))
    job_dir = os.path.abspath(args.job_dir or
                              os.path.join(".adaptdl", name))
    controller = LocalController(num_gpus=args.num_gpus,
                                 interval=args.interval)
    spec = JobSpec(command, name=name, job_dir=job_dir,
                   min_replicas=args.min_replicas,
                   max_replicas=args.max_replicas,
                   gpus_per_replica=args.gpus_per_replica,
                   inplace_scaledown=args.inplace_scaledown)
    controller.submit(spec)
    print("job {} -> {}".format(name, job_dir))

    def handle_sig(signum, frame):
        print("interrupt: stopping job")
        controller.shutdown()
        sys.exit(130)

    signal.signal(signal.SIGINT, handle_sig)
    last = None
    while True:
        st = controller.status(name)
        if (st["state"], st["replicas"]) != last:
            last = (st["state"], st["replicas"])
            print("[{}] {} replicas={} restarts={}".format(
                time.strftime("%H:%M:%S"), st["state"], st["replicas"],
                st["restarts"]))
        if st["state"] in ("Succeeded", "Failed"):
            break
        time.sleep(0.5)
    for p in controller.log_paths(name):
        print("==> {} <==".format(p))
        sys.stdout.write(open(p).read())
    controller.shutdown()
    sys.exit(0 if st["state"] == "Succeeded" else 1)


def cmd_submit(args, command):
    if not command:
        sys.exit("submit: missing command after --")
    name = args.name or "job-{}".format(int(time.time()))
    out = _request(args.url + "/jobs", "POST", {
        "argv": command, "name": name,
        "min_replicas": args.min_replicas,
        "max_replicas": args.max_replicas,
        "gpus_per_replica": args.gpus_per_replica,
        "inplace_scaledown": args.inplace_scaledown,
        "job_dir": args.job_dir,
        "workdir": os.getcwd()})
    print("submitted {} (job_dir {})".format(out["name"],
                                             out["job_dir"]))


def cmd_status(args):
    print(json.dumps(_request("{}/jobs/{}".format(args.url, args.name)),
                     indent=2))


def cmd_ls(args):
    jobs = _request(args.url + "/jobs")
    fmt = "{:<24} {:<10} {:>8} {:>8}"
    print(fmt.format("NAME", "STATE", "REPLICAS", "RESTARTS"))
    for name, st in sorted(jobs.items()):
        print(fmt.format(name, st["state"], st["replicas"],
                         st["restarts"]))


def cmd_logs(args):
    logs = _request("{}/jobs/{}/logs".format(args.url, args.name))
    for fname in sorted(logs):
        if args.rank is not None and \
                not fname.endswith("rank-{}.log".format(args.rank)):
            continue
        print("==> {} <==".format(fname))
        sys.stdout.write(logs[fname])


def cmd_rescale(args):
    _request("{}/jobs/{}/rescale".format(args.url, args.name), "POST",
             {"replicas": args.replicas})
    print("rescale {} -> {} requested".format(args.name, args.replicas))


def cmd_stop(args):
    _request(args.url + "/shutdown", "POST", {})
    print("daemon stopping")


def cmd_cp(args):
    """Copy files out of (or into) a job's directory.

    Counterpart of the reference's ``adaptdl cp`` (which proxies a PVC
    through a copy pod, cli/adaptdl_cli/proxy.py); locally the job dir
    is a directory on this node, so this resolves it from the daemon
    and copies with shutil.  ``NAME:PATH`` addresses inside the job
    dir; a bare path is local.
    """
    import shutil

    def resolve(spec):
        if ":" in spec:
            name, rel = spec.split(":", 1)
            st = _request("{}/jobs/{}".format(args.url, name))
            return os.path.join(st["job_dir"], rel.lstrip("/"))
        return spec

    src_path = resolve(args.src)
    dst_path = resolve(args.dst)
    if os.path.isdir(src_path):
        dst_dir = (os.path.join(dst_path, os.path.basename(src_path))
                   if os.path.isdir(dst_path) else dst_path)
        shutil.copytree(src_path, dst_dir, dirs_exist_ok=True)
    else:
        shutil.copy2(src_path, dst_path)
    print("copied {} -> {}".format(src_path, dst_path))


def cmd_tensorboard(args):
    """Point TensorBoard at a job's directory.

    Counterpart of the reference's ``adaptdl tensorboard`` (which
    manages an in-cluster TensorBoard Deployment per instance,
    cli/adaptdl_cli/tensorboard.py); locally the job_dir already holds
    the event files written by the ``to_tensorboard`` exporters, so
    this resolves the directory (from the daemon, or --logdir) and
    execs ``tensorboard`` against it — or prints the exact command if
    the tensorboard package is not installed.
    """
    import shutil
    if args.logdir:
        logdir = args.logdir
    else:
        if not args.name:
            raise SystemExit("tensorboard: a job NAME or --logdir "
                             "is required")
        logdir = _request("{}/jobs/{}".format(args.url,
                                              args.name))["job_dir"]
    argv = ["tensorboard", "--logdir", logdir,
            "--port", str(args.port)]
    exe = shutil.which("tensorboard")
    if exe is None:
        print("tensorboard is not installed; run:")
        print("  " + " ".join(argv))
        return
    os.execv(exe, argv)


def main(argv=None):
    argv, command = _split_command(list(argv
                                        if argv is not None
                                        else sys.argv[1:]))
    parser = argparse.ArgumentParser(prog="adaptdl-amd")
    sub = parser.add_subparsers(dest="cmd", required=True)

    def add_job_opts(p):
        p.add_argument("--name")
        p.add_argument("--job-dir")
        p.add_argument("--min-replicas", type=int, default=0)
        p.add_argument("--max-replicas", type=int, default=8)
        p.add_argument("--gpus-per-replica", type=int,
                       default=1 if _has_gpu() else 0)
        p.add_argument("--inplace-scaledown", action="store_true",
                       help="scale-downs rejoin in place (no restart; "
                            "survivors keep state in memory)")

    p = sub.add_parser("daemon", help="run the persistent controller")
    p.add_argument("--bind", default="127.0.0.1:8077")
    p.add_argument("--state-dir", default=".adaptdl")
    p.add_argument("--num-gpus", type=int, default=None)
    p.add_argument("--interval", type=float, default=30.0)
    p.add_argument("--metrics-port", type=int, default=None,
                   help="expose Prometheus metrics on this port")

    p = sub.add_parser("run", help="run a job in the foreground")
    add_job_opts(p)
    p.add_argument("--num-gpus", type=int, default=None)
    p.add_argument("--interval", type=float, default=30.0)

    p = sub.add_parser("submit", help="submit a job to the daemon")
    add_job_opts(p)
    p.add_argument("--url", default=DEFAULT_URL)

    p = sub.add_parser("ls", help="list daemon jobs")
    p.add_argument("--url", default=DEFAULT_URL)

    p = sub.add_parser("status", help="print one job's status JSON")
    p.add_argument("name")
    p.add_argument("--url", default=DEFAULT_URL)

    p = sub.add_parser("logs", help="print job logs")
    p.add_argument("name")
    p.add_argument("--rank", type=int, default=None)
    p.add_argument("--url", default=DEFAULT_URL)

    p = sub.add_parser("rescale", help="force a replica count")
    p.add_argument("name")
    p.add_argument("replicas", type=int)
    p.add_argument("--url", default=DEFAULT_URL)

    p = sub.add_parser("stop", help="shut the daemon down")
    p.add_argument("--url", default=DEFAULT_URL)

    p = sub.add_parser("cp", help="copy files from/to a job directory "
                                  "(NAME:PATH addresses inside it)")
    p.add_argument("src")
    p.add_argument("dst")
    p.add_argument("--url", default=DEFAULT_URL)

    p = sub.add_parser("tensorboard",
                       help="launch TensorBoard on a job's directory")
    p.add_argument("name", nargs="?")
    p.add_argument("--logdir", default=None,
                   help="explicit log directory (no daemon needed)")
    p.add_argument("--port", type=int, default=6006)
    p.add_argument("--url", default=DEFAULT_URL)

    args = parser.parse_args(argv)
    if args.cmd == "daemon":
        cmd_daemon(args)
    elif args.cmd == "run":
        cmd_run(args, command)
    elif args.cmd == "submit":
        cmd_submit(args, command)
    elif args.cmd == "ls":
        cmd_ls(args)
    elif args.cmd == "status":
        cmd_status(args)
    elif args.cmd == "cp":
        cmd_cp(args)
    elif args.cmd == "logs":
        cmd_logs(args)
    elif args.cmd == "rescale":
        cmd_rescale(args)
    elif args.cmd == "stop":
        cmd_stop(args)
    elif args.cmd == "tensorboard":
        cmd_tensorboard(args)


def _has_gpu():
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False


if __name__ == "__main__":
    main()
