"""kungfu-run: the process launcher.

Reference parity: srcs/go/cmd/kungfu-run + srcs/go/kungfu/runner/
(flag surface flags.go:69-104; SimpleRun simple.go; WatchRun elastic mode
watch.go:42-104; MonitoredRun auto-recovery monitored.go). Spawns one
worker process per slot with the KUNGFU_* env protocol, streams their
output with colored per-worker prefixes, and in watch mode reacts to
control-plane "update" stages by starting/stopping local workers.

MI355X specifics: each local worker is pinned to one GPU via
HIP_VISIBLE_DEVICES=<slot>, so inside the worker the device is cuda:0 and
RCCL rides xGMI between the 8 GPUs of the node.
"""
import argparse
import json
import os
import signal
import subprocess
import sys
import threading
import time

COLORS = [31, 32, 33, 34, 35, 36, 91, 92, 93, 94, 95, 96]


def parse_args(argv=None):
    p = argparse.ArgumentParser(
        "kungfu-run", description="KungFu-AMD launcher")
    p.add_argument("-np", type=int, default=1, help="total workers")
    p.add_argument("-H", dest="hosts", default=None,
                   help="host list ip:slots[:pub],...")
    p.add_argument("-hostfile", default=None)
    p.add_argument("-self", dest="self_ip", default="127.0.0.1")
    p.add_argument("-nic", default=None, help="infer self IP from NIC")
    p.add_argument("-port", type=int, default=38080, help="runner port")
    p.add_argument("-port-range", dest="port_base", type=int, default=31100,
                   help="first worker port")
    p.add_argument("-P", dest="peers", default=None,
                   help="explicit worker peer list ip:port,... "
                        "(overrides -H/-np port allocation)")
    p.add_argument("-strategy", default="AUTO",
                   help="STAR|MULTI_STAR|RING|CLIQUE|TREE|BINARY_TREE|"
                        "BINARY_TREE_STAR|MULTI_BINARY_TREE_STAR|AUTO")
    p.add_argument("-w", dest="watch", action="store_true",
                   help="watch (elastic) mode")
    p.add_argument("-k", dest="keep", action="store_true",
                   help="keep runner alive after workers exit")
    p.add_argument("-init-version", type=int, default=0)
    p.add_argument("-config-server", default=None)
    p.add_argument("-builtin-config-port", type=int, default=0)
    p.add_argument("-auto-recover", default=None,
                   help="enable failure detection + restart, e.g. '10s'")
    p.add_argument("-monitor-port", type=int, default=7756,
                   help="heartbeat monitor server port (auto-recover)")
    p.add_argument("-timeout", default=None, help="job timeout, e.g. '120s'")
    p.add_argument("-q", dest="quiet", action="store_true")
    p.add_argument("-logdir", default=None)
    p.add_argument("-logfile", default=None)
    p.add_argument("-delay", default=None)
    p.add_argument("-t0", default=None,
                   help="job start timestamp override")
    p.add_argument("-u", dest="ssh_user", default=None,
                   help="ssh user (kungfu-distribute)")
    p.add_argument("-debug-port", type=int, default=0,
                   help="accepted for reference-CLI compatibility")
    p.add_argument("-allow-nvlink", action="store_true",
                   help="accepted for reference-CLI compatibility (no-op: "
                        "xGMI is always used on MI355X)")
    p.add_argument("prog", nargs=argparse.REMAINDER)
    args = p.parse_args(argv)
    if args.prog and args.prog[0] == "--":
        args.prog = args.prog[1:]
    if not args.prog:
        p.error("no program given")
    return args


def parse_hostfile(text):
    """MPI-style hostfile (reference plan/hostfile/hostfile.go): one
    'ip [slots=N] [public_addr=X]' per line, #-comments; bare 'ip:slots'
    lines are passed through."""
    specs = []
    for line in text.splitlines():
        line = line.split("#", 1)[0].strip()
        if not line:
            continue
        parts = line.split()
        ip = parts[0]
        if ":" in ip and len(parts) == 1:
            specs.append(ip)  # already an ip:slots[:pub] spec
            continue
        slots, pub = 1, None
        for kv in parts[1:]:
            k, _, v = kv.partition("=")
            if k == "slots":
                slots = int(v)
            elif k == "public_addr":
                pub = v
            else:
                raise ValueError("bad hostfile entry: %r" % line)
        spec = "%s:%d" % (ip, slots)
        if pub:
            spec += ":" + pub
        specs.append(spec)
    return ",".join(specs)


def parse_duration(s):
    if s is None:
        return None
    s = s.strip()
    mult = 1.0
    for suf, m in (("ms", 0.001), ("s", 1.0), ("m", 60.0), ("h", 3600.0)):
        if s.endswith(suf):
            return float(s[:-len(suf)]) * m
    return float(s) * mult


def infer_self_ip(nic):
    import socket
    import struct
    import fcntl

    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        return socket.inet_ntoa(
            fcntl.ioctl(s.fileno(), 0x8915,  # SIOCGIFADDR
                        struct.pack("256s", nic.encode()[:15]))[20:24])
    finally:
        s.close()


class Proc:
    def __init__(self, spec, popen, color):
        self.spec = spec
        self.popen = popen
        self.color = color
        self.threads = []


class Runner:
    def __init__(self, args):
        self.args = args
        self._logf = None
        if args.nic:
            args.self_ip = infer_self_ip(args.nic)
        if args.hostfile:
            with open(args.hostfile) as f:
                hosts = parse_hostfile(f.read())
        elif args.hosts:
            hosts = args.hosts
        else:
            hosts = "%s:%d" % (args.self_ip, args.np)
        self.hosts = hosts
        from kungfu_amd import _core

        self.core = _core
        if args.peers:
            self.peers = args.peers
        else:
            self.peers = _core.gen_peer_list(hosts, args.np,
                                             args.port_base)
        self.runners = _core.gen_runner_list(hosts, args.port)
        self.version = args.init_version
        self.procs = {}  # spec -> Proc
        self.lock = threading.Lock()
        self.stopped = False
        self.config_srv = None
        self.color_idx = 0

    # ---- env + spawn ----

    def worker_env(self, spec, peers_csv, version):
        env = dict(os.environ)
        # make the kungfu_amd package importable in workers regardless of
        # how the launcher itself was invoked
        pkg_root = os.path.dirname(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
        env["PYTHONPATH"] = pkg_root + os.pathsep + env.get("PYTHONPATH",
                                                            "")
        env["KUNGFU_SELF_SPEC"] = spec
        env["KUNGFU_INIT_PEERS"] = peers_csv
        env["KUNGFU_INIT_RUNNERS"] = ",".join(
            "%s" % r for r in self.runners.split(","))
        env["KUNGFU_INIT_CLUSTER_VERSION"] = str(version)
        env["KUNGFU_ALLREDUCE_STRATEGY"] = self.args.strategy
        if self.config_server_url():
            env["KUNGFU_CONFIG_SERVER"] = self.config_server_url()
        env["KUNGFU_JOB_START_TIMESTAMP"] = (
            self.args.t0 or str(int(time.time())))
        env["KUNGFU_MONITOR_PORT"] = str(self.args.monitor_port)
        # GPU slot assignment: local rank among this host's workers
        ip = spec.rsplit(":", 1)[0]
        local = [s for s in peers_csv.split(",")
                 if s.rsplit(":", 1)[0] == ip]
        if spec in local:
            slot = local.index(spec)
            for key in ("HIP_VISIBLE_DEVICES", "CUDA_VISIBLE_DEVICES"):
                if not env.get(key):  # empty or unset: assign our slot
                    env[key] = str(slot)
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        return env

    def spawn(self, spec, peers_csv, version):
        env = self.worker_env(spec, peers_csv, version)
        popen = subprocess.Popen(
            self.args.prog, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.PIPE, start_new_session=True)
        color = COLORS[self.color_idx % len(COLORS)]
        self.color_idx += 1
        proc = Proc(spec, popen, color)
        logdir = self.args.logdir
        logf = None
        if logdir:
            os.makedirs(logdir, exist_ok=True)
            logf = open(os.path.join(
                logdir, "%s@%d.log" % (spec.replace(":", "."), version)),
                "ab")
        elif self.args.logfile:
            logf = self._shared_logfile()

        def stream(src, dst):
            prefix = ("\x1b[%dm[%s]\x1b[0m " % (color, spec)).encode()
            for line in iter(src.readline, b""):
                if logf:
                    logf.write(line)
                    logf.flush()
                if not self.args.quiet:
                    dst.buffer.write(prefix + line)
                    dst.flush()

        for src, dst in ((popen.stdout, sys.stdout),
                         (popen.stderr, sys.stderr)):
            t = threading.Thread(target=stream, args=(src, dst),
                                 daemon=True)
            t.start()
            proc.threads.append(t)
        with self.lock:
            self.procs[spec] = proc
        return proc

    def _shared_logfile(self):
        if self._logf is None:
            self._logf = open(self.args.logfile, "ab")
        return self._logf

    def kill(self, spec):
        with self.lock:
            proc = self.procs.pop(spec, None)
        if proc is None:
            return
        try:
            os.killpg(proc.popen.pid, signal.SIGTERM)
        except ProcessLookupError:
            pass
        try:
            # short escalation: a worker blocked inside a collective never
            # handles SIGTERM (GIL released in C++), and a slow teardown
            # here can outlive a supervisor's own kill window, leaking
            # port-squatting orphans
            proc.popen.wait(timeout=3)
        except subprocess.TimeoutExpired:
            try:
                os.killpg(proc.popen.pid, signal.SIGKILL)
            except ProcessLookupError:
                pass
            try:
                proc.popen.wait(timeout=5)
            except subprocess.TimeoutExpired:
                pass

    # ---- config server ----

    def config_server_url(self):
        if self.args.config_server:
            return self.args.config_server
        if self.config_srv is not None:
            return "%s:%d" % (self.args.self_ip, self._builtin_port)
        return None

    def start_builtin_config_server(self):
        from kungfu_amd.launcher.configserver import make_server

        cluster = json.dumps({
            "runners": self.runners.split(","),
            "workers": self.peers.split(","),
        }, separators=(",", ":"))
        port = self.args.builtin_config_port or (self.args.port + 1)
        self._builtin_port = port
        self.config_srv, t = make_server(port, cluster)
        t.start()

    # ---- run modes ----

    def local_specs(self, peers_csv):
        ip = self.args.self_ip
        return [s for s in peers_csv.split(",")
                if s.rsplit(":", 1)[0] == ip]

    def simple_run(self):
        """Static mode (reference SimpleRun): spawn all local procs, wait."""
        peers_csv = self.peers
        for spec in self.local_specs(peers_csv):
            self.spawn(spec, peers_csv, self.version)
        return self.wait_all()

    def wait_all(self, timeout=None):
        deadline = time.time() + timeout if timeout else None
        code = 0
        while True:
            with self.lock:
                live = list(self.procs.items())
            if not live:
                break
            for spec, proc in live:
                rc = proc.popen.poll()
                if rc is not None:
                    with self.lock:
                        self.procs.pop(spec, None)
                    code = max(code, rc)
            if deadline and time.time() > deadline:
                sys.stderr.write("kungfu-run: timeout, killing workers\n")
                for spec, _ in live:
                    self.kill(spec)
                return 124
            time.sleep(0.1)
        return code

    def watch_run(self):
        """Elastic mode (reference WatchRun): listen for control 'update'
        stages from workers and reconcile the local process set."""
        from kungfu_amd import _core

        server = _core.RunnerServer(
            "%s:%d" % (self.args.self_ip, self.args.port), False)
        peers_csv = self.peers
        for spec in self.local_specs(peers_csv):
            self.spawn(spec, peers_csv, self.version)
        code = 0
        idle_since = None
        try:
            while True:
                msg = server.poll(200)
                if msg is not None:
                    name, payload = msg
                    if name == "update":
                        stage = json.loads(payload.decode())
                        self.apply_stage(stage)
                        idle_since = None
                # reap finished procs
                with self.lock:
                    live = list(self.procs.items())
                for spec, proc in live:
                    rc = proc.popen.poll()
                    if rc is not None:
                        with self.lock:
                            self.procs.pop(spec, None)
                        code = max(code, rc)
                if not self.procs:
                    if self.args.keep:
                        time.sleep(0.2)
                        continue
                    # grace period for a pending stage that adds workers
                    if idle_since is None:
                        idle_since = time.time()
                    elif time.time() - idle_since > 3.0:
                        break
                else:
                    idle_since = None
        finally:
            server.stop()
        return code

    def apply_stage(self, stage):
        version = int(stage["version"])
        cluster = stage["cluster"]
        workers = cluster["workers"]
        peers_csv = ",".join(workers)
        self.version = version
        want = set(self.local_specs(peers_csv))
        with self.lock:
            have = set(self.procs.keys())
        for spec in have - want:
            self.kill(spec)
        for spec in sorted(want - have):
            self.spawn(spec, peers_csv, version)

    def run(self):
        timeout = parse_duration(self.args.timeout)
        if self.args.delay:
            time.sleep(parse_duration(self.args.delay))
        need_config = self.args.watch or self.args.builtin_config_port
        if need_config and not self.args.config_server:
            self.start_builtin_config_server()
        if self.args.auto_recover:
            from kungfu_amd.launcher.monitor import monitored_run

            return monitored_run(self,
                                 parse_duration(self.args.auto_recover))
        if self.args.watch:
            return self.watch_run()
        if timeout:
            for spec in self.local_specs(self.peers):
                self.spawn(spec, self.peers, self.version)
            return self.wait_all(timeout)
        return self.simple_run()


def main(argv=None):
    args = parse_args(argv)
    runner = Runner(args)

    def on_sig(signum, frame):
        with runner.lock:
            specs = list(runner.procs.keys())
        for s in specs:
            runner.kill(s)
        sys.exit(128 + signum)

    signal.signal(signal.SIGINT, on_sig)
    signal.signal(signal.SIGTERM, on_sig)
    sys.exit(runner.run())


if __name__ == "__main__":
    main()
