"""Minimal training-status HTTP endpoint.

Parity target: reference ``veles/web_status.py`` (tornado dashboard of
running workflows).  MI355X redesign per SURVEY §2a: a dependency-free
``http.server`` thread serving the current status as JSON — enough for
fleet polling without the tornado/JS stack.
"""

import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class StatusServer(object):
    """Serves GET / -> JSON of everything publish()ed."""

    def __init__(self, host="127.0.0.1", port=8090):
        self.host = host
        self.port = port
        self._lock = threading.Lock()
        self._status = {"started": time.time()}
        self._httpd = None
        self._thread = None

    def publish(self, **fields):
        with self._lock:
            self._status.update(fields)
            self._status["updated"] = time.time()

    def snapshot(self):
        with self._lock:
            return dict(self._status)

    def start(self):
        server = self

        class Handler(BaseHTTPRequestHandler):
            def do_GET(self):
                body = json.dumps(server.snapshot()).encode()
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a):
                pass

        self._httpd = ThreadingHTTPServer((self.host, self.port), Handler)
        self.port = self._httpd.server_address[1]
        self._thread = threading.Thread(target=self._httpd.serve_forever,
                                        daemon=True)
        self._thread.start()
        return self

    def stop(self):
        if self._httpd is not None:
            self._httpd.shutdown()
            self._httpd.server_close()
            self._httpd = None


class WorkflowStatusReporter(object):
    """Glue: publish decision/loader progress each epoch."""

    def __init__(self, server, workflow):
        self.server = server
        self.workflow = workflow

    def report(self):
        wf = self.workflow
        d = getattr(wf, "decision", None)
        fields = {"workflow": type(wf).__name__,
                  "is_running": bool(getattr(wf, "is_running", False))}
        if d is not None:
            fields["epoch"] = int(d.epoch_number) if d.epoch_number \
                is not None else None
            if getattr(d, "epoch_n_err_pt", None) is not None:
                fields["err_pt"] = list(d.epoch_n_err_pt)
            if getattr(d, "epoch_metrics", None) is not None:
                fields["metrics"] = [m[0] if m else None
                                     for m in d.epoch_metrics]
        self.server.publish(**fields)
