"""Local supervisor: HTTP hints/discovery endpoint for worker processes.

Single-node counterpart of the reference's k8s supervisor REST service
(/root/reference/sched/adaptdl_sched/supervisor.py:45-99).  Worker
processes PUT their sched hints to ``/hints/{job}`` (sched_hints.py uses
ADAPTDL_SUPERVISOR_URL when set) and may GET ``/discover/{job}/{group}``
for the replica endpoints of their restart group.  Implemented on the
stdlib ThreadingHTTPServer — no aiohttp/k8s dependencies; runs as a
daemon thread inside the LocalController process.
"""

import json
import logging
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

LOG = logging.getLogger(__name__)


class Supervisor(object):
    def __init__(self, host="127.0.0.1", port=0):
        self._hints = {}
        self._endpoints = {}  # job -> (group, [addr, ...])
        self._lock = threading.Lock()
        self._callbacks = []
        sup = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, fmt, *args):
                LOG.debug("supervisor: " + fmt, *args)

            def _reply(self, code, obj=None):
                body = json.dumps(obj).encode() if obj is not None else b""
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                parts = [p for p in self.path.split("/") if p]
                if parts == ["healthz"]:
                    self._reply(200, {"status": "ok"})
                elif len(parts) >= 2 and parts[0] == "hints":
                    with sup._lock:
                        hints = sup._hints.get(parts[1])
                    self._reply(200 if hints else 404, hints)
                elif len(parts) >= 3 and parts[0] == "discover":
                    job, group = parts[1], parts[2]
                    with sup._lock:
                        entry = sup._endpoints.get(job)
                    if entry is not None and str(entry[0]) == group:
                        self._reply(200, entry[1])
                    else:
                        self._reply(404, [])
                else:
                    self._reply(404)

            def do_PUT(self):
                parts = [p for p in self.path.split("/") if p]
                if len(parts) >= 2 and parts[0] == "hints":
                    n = int(self.headers.get("Content-Length", "0"))
                    try:
                        hints = json.loads(self.rfile.read(n) or b"{}")
                    except ValueError:
                        self._reply(400, {"error": "bad json"})
                        return
                    sup.put_hints(parts[1], hints)
                    self._reply(200, {"status": "ok"})
                else:
                    self._reply(404)

        self._server = ThreadingHTTPServer((host, port), Handler)
        self._server.daemon_threads = True
        self._thread = threading.Thread(target=self._server.serve_forever,
                                        daemon=True,
                                        name="adaptdl-supervisor")

    @property
    def url(self):
        host, port = self._server.server_address[:2]
        return "http://{}:{}".format(host, port)

    def start(self):
        self._thread.start()
        return self

    def stop(self):
        self._server.shutdown()
        self._server.server_close()

    # ---- controller-side API -------------------------------------------

    def put_hints(self, job, hints):
        with self._lock:
            self._hints[job] = hints
            callbacks = list(self._callbacks)
        for fn in callbacks:
            try:
                fn(job, hints)
            except Exception:
                LOG.exception("hints callback failed")

    def get_hints(self, job=None):
        with self._lock:
            if job is None:
                return dict(self._hints)
            return self._hints.get(job)

    def register_hints_callback(self, fn):
        with self._lock:
            self._callbacks.append(fn)

    def set_endpoints(self, job, group, addrs):
        with self._lock:
            self._endpoints[job] = (group, list(addrs))

    def clear_job(self, job):
        with self._lock:
            self._hints.pop(job, None)
            self._endpoints.pop(job, None)
