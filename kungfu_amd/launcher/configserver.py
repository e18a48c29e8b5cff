"""Elastic cluster config server.

Reference parity: srcs/go/kungfu/elastic/configserver/configserver.go — an
HTTP endpoint holding the cluster JSON {"runners":[...],"workers":[...]}
with a version counter; GET returns it, PUT/POST replace it, DELETE clears.
Also embeddable into kungfu-run (builtin-config-server).
"""
import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class _State:
    def __init__(self):
        self.lock = threading.Lock()
        self.body = b""
        self.version = 0


def make_server(port, initial_cluster_json=None, host="0.0.0.0"):
    state = _State()
    if initial_cluster_json:
        state.body = initial_cluster_json.encode()
        state.version = 1

    class Handler(BaseHTTPRequestHandler):
        def log_message(self, *a):  # quiet
            pass

        def _respond(self, code, body=b"", extra=None):
            self.send_response(code)
            self.send_header("Content-Length", str(len(body)))
            self.send_header("Content-Type", "application/json")
            for k, v in (extra or {}).items():
                self.send_header(k, v)
            self.end_headers()
            if body:
                self.wfile.write(body)

        def do_GET(self):
            with state.lock:
                if not state.body:
                    self._respond(404)
                    return
                self._respond(200, state.body,
                              {"X-Kungfu-Version": str(state.version)})

        def _write(self):
            n = int(self.headers.get("Content-Length", 0))
            data = self.rfile.read(n)
            try:
                json.loads(data)
            except Exception:
                self._respond(400)
                return
            with state.lock:
                state.body = data
                state.version += 1
            self._respond(200)

        do_PUT = _write
        do_POST = _write

        def do_DELETE(self):
            with state.lock:
                state.body = b""
            self._respond(200)

    srv = ThreadingHTTPServer((host, port), Handler)
    srv.kungfu_state = state
    thread = threading.Thread(target=srv.serve_forever, daemon=True)
    return srv, thread


def main():
    import argparse

    p = argparse.ArgumentParser("kungfu-config-server")
    p.add_argument("-port", type=int, default=9100)
    args = p.parse_args()
    srv, t = make_server(args.port)
    print("config server on :%d" % args.port)
    t.start()
    t.join()


if __name__ == "__main__":
    main()
