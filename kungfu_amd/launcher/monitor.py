"""Failure detection + auto-recovery (the fork's marquee feature).

Reference parity: srcs/go/kungfu/runner/monitorserver/monitor.go +
runner/monitored.go + docs/monitor_proposal.md. Each host's runner serves
HTTP on :7756; workers POST heartbeats {"key":"begin:<rank>"} /
"end:<rank>" around every batch, "epoch:<rank>" per epoch and
"trainend:<rank>" at the end (kungfu_amd.cmd). A rank whose `begin` has no
matching `end` for longer than the grace period marks the machine down: the
runner broadcasts "otherdown:<min epoch>" to the other hosts' monitors,
kills its local workers, and relaunches them with `--n-epochs` reduced by
the epochs already completed and `--restart 1` appended so the training
script reloads its checkpoint.
"""
import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

MONITOR_PORT = 7756
GRACE_SECONDS = 10.0


class MonitorState:
    def __init__(self):
        self.lock = threading.Lock()
        self.begin = {}       # rank -> timestamp of pending begin
        self.epochs = {}      # rank -> completed epochs
        self.trainend = set()
        self.down = False
        self.other_down_epoch = None

    def handle(self, key):
        kind, _, rank = key.partition(":")
        with self.lock:
            if kind == "begin":
                self.begin[rank] = time.time()
            elif kind == "end":
                self.begin.pop(rank, None)
            elif kind == "epoch":
                self.epochs[rank] = self.epochs.get(rank, 0) + 1
                self.begin.pop(rank, None)
            elif kind == "trainend":
                self.trainend.add(rank)
                self.begin.pop(rank, None)
            elif kind == "otherdown":
                self.down = True
                try:
                    self.other_down_epoch = int(rank)
                except ValueError:
                    self.other_down_epoch = 0

    def check_down(self, grace):
        now = time.time()
        with self.lock:
            if self.down:
                return True
            for rank, t0 in self.begin.items():
                if now - t0 > grace:
                    self.down = True
                    return True
        return False

    def min_epoch(self):
        with self.lock:
            if self.other_down_epoch is not None:
                return self.other_down_epoch
            if not self.epochs:
                return 0
            return min(self.epochs.values())


def start_monitor_server(state, port=MONITOR_PORT, host="0.0.0.0"):
    class Handler(BaseHTTPRequestHandler):
        def log_message(self, *a):
            pass

        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            try:
                msg = json.loads(self.rfile.read(n))
                state.handle(str(msg.get("key", "")))
            except Exception:
                pass
            self.send_response(200)
            self.send_header("Content-Length", "0")
            self.end_headers()

    srv = ThreadingHTTPServer((host, port), Handler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    return srv


def _broadcast_otherdown(runner, min_epoch):
    import urllib.request

    for spec in runner.runners.split(","):
        ip = spec.rsplit(":", 1)[0]
        if ip == runner.args.self_ip:
            continue
        try:
            req = urllib.request.Request(
                "http://%s:%d/" % (ip, runner.args.monitor_port),
                data=json.dumps(
                    {"key": "otherdown:%d" % min_epoch}).encode(),
                method="POST")
            urllib.request.urlopen(req, timeout=2)
        except Exception:
            pass


def _adjust_prog(prog, completed_epochs):
    """Decrement --n-epochs by completed epochs, append --restart 1."""
    out = list(prog)
    for i, a in enumerate(out):
        if a in ("--n-epochs", "-n-epochs") and i + 1 < len(out):
            try:
                remaining = max(1, int(out[i + 1]) - completed_epochs)
                out[i + 1] = str(remaining)
            except ValueError:
                pass
    if "--restart" not in out:
        out += ["--restart", "1"]
    return out


def monitored_run(runner, grace=GRACE_SECONDS):
    """Run workers under heartbeat supervision; restart on failure
    (reference monitored.go:18-75)."""
    grace = grace or GRACE_SECONDS
    state = MonitorState()
    # bind the runner's own IP so one monitor per simulated host can
    # coexist on a single machine (loopback aliases)
    srv = start_monitor_server(state, runner.args.monitor_port,
                               runner.args.self_ip)
    world = len(runner.peers.split(","))
    # Workers on EVERY host heartbeat the FIRST host's monitor
    # (kungfu_amd.cmd._monitor_host), so trainend accumulates global ranks
    # there and never arrives on other hosts: the first host must compare
    # against the global world size (not its local worker count), and the
    # other hosts complete only via local process exit.
    first_ip = runner.runners.split(",")[0].rsplit(":", 1)[0]
    is_first_host = first_ip == runner.args.self_ip
    try:
        attempt = 0
        while True:
            with state.lock:
                state.down = False
                state.begin.clear()
            peers_csv = runner.peers
            for spec in runner.local_specs(peers_csv):
                runner.spawn(spec, peers_csv, runner.version)
            # supervise
            failed = False
            code = 0
            while True:
                with runner.lock:
                    live = list(runner.procs.items())
                all_exited = True
                for spec, proc in live:
                    rc = proc.popen.poll()
                    if rc is None:
                        all_exited = False
                    else:
                        with runner.lock:
                            runner.procs.pop(spec, None)
                        code = max(code, rc)
                with state.lock:
                    done = (is_first_host and
                            len(state.trainend) >= world)
                if done or (all_exited and not live):
                    return code
                if state.check_down(grace):
                    failed = True
                    break
                time.sleep(0.5)
            if failed:
                attempt += 1
                with state.lock:
                    locally_detected = state.other_down_epoch is None
                min_epoch = state.min_epoch()
                print("[kungfu-run] failure detected (attempt %d); "
                      "restarting from epoch %d" % (attempt, min_epoch),
                      flush=True)
                if locally_detected:
                    # only the detecting host fans out: echoing an
                    # otherdown back would ping-pong restarts
                    _broadcast_otherdown(runner, min_epoch)
                with runner.lock:
                    specs = list(runner.procs.keys())
                for s in specs:
                    runner.kill(s)
                runner.args.prog = _adjust_prog(runner.args.prog,
                                                min_epoch)
                with state.lock:
                    state.trainend.clear()
                    state.other_down_epoch = None
                time.sleep(1.0)
    finally:
        srv.shutdown()
