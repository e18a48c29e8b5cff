"""kungfu-rrun: static remote job runner over ssh.

Reference parity: srcs/go/cmd/kungfu-rrun/rrun.go:19-43 +
utils/runner/remote RunStaticKungFuJob — unlike kungfu-distribute (which
starts one kungfu-run per host), rrun launches every WORKER process
directly over ssh with the full KUNGFU_* env protocol; workers mesh
peer-to-peer with no remote launcher in between.
"""
import argparse
import shlex
import subprocess
import sys
import threading


def gen_peers(hosts, np, port_base):
    """Round-robin-by-slot peer allocation (reference hostspec.go
    GenPeerList semantics, same as the C++ gen_peer_list)."""
    specs = []
    for h in hosts.split(","):
        parts = h.split(":")
        ip = parts[0]
        slots = int(parts[1]) if len(parts) > 1 and parts[1] else 1
        specs.append((ip, slots))
    peers = []
    used = {ip: 0 for ip, _ in specs}
    while len(peers) < np:
        progressed = False
        for ip, slots in specs:
            if len(peers) >= np:
                break
            if used[ip] < slots:
                peers.append("%s:%d" % (ip, port_base + used[ip]))
                used[ip] += 1
                progressed = True
        if not progressed:
            raise SystemExit("host capacity %d < np %d" %
                             (sum(s for _, s in specs), np))
    return peers


def build_commands(args, prog):
    peers = gen_peers(args.hosts, args.np, args.port_range)
    peer_csv = ",".join(peers)
    cmds = []
    local_idx = {}
    for rank, spec in enumerate(peers):
        ip = spec.rsplit(":", 1)[0]
        li = local_idx.get(ip, 0)
        local_idx[ip] = li + 1
        env = {
            "KUNGFU_SELF_SPEC": spec,
            "KUNGFU_INIT_PEERS": peer_csv,
            "KUNGFU_ALLREDUCE_STRATEGY": args.strategy,
            "KUNGFU_JOB_START_TIMESTAMP": "0",
            "HIP_VISIBLE_DEVICES": str(li),
            "CUDA_VISIBLE_DEVICES": str(li),
        }
        envs = " ".join("%s=%s" % (k, shlex.quote(v))
                        for k, v in env.items())
        inner = "%s %s" % (envs,
                           " ".join(shlex.quote(c) for c in prog))
        target = "%s@%s" % (args.user, ip) if args.user else ip
        cmds.append((rank, spec, ["ssh", "-o", "StrictHostKeyChecking=no",
                                  target, inner]))
    return cmds


def main(argv=None):
    p = argparse.ArgumentParser("kungfu-rrun")
    p.add_argument("-np", type=int, default=1)
    p.add_argument("-H", dest="hosts", required=True,
                   help="ip:slots,ip:slots")
    p.add_argument("-strategy", default="AUTO")
    p.add_argument("-port-range", dest="port_range", type=int,
                   default=30100, help="first worker port per host")
    p.add_argument("-u", dest="user", default="")
    p.add_argument("-timeout", type=float, default=0)
    p.add_argument("--dry-run", action="store_true")
    p.add_argument("prog", nargs=argparse.REMAINDER)
    args = p.parse_args(argv)
    prog = [a for a in args.prog if a != "--"]
    if not prog:
        p.error("no program given")
    cmds = build_commands(args, prog)
    if args.dry_run:
        for rank, spec, cmd in cmds:
            print("[%d %s] %s" % (rank, spec, " ".join(cmd)))
        return 0
    procs = []
    for rank, spec, cmd in cmds:
        procs.append((spec, subprocess.Popen(cmd)))

    rc = [0]

    def reap(spec, proc):
        code = proc.wait()
        if code:
            rc[0] = code

    threads = [threading.Thread(target=reap, args=sp) for sp in procs]
    for t in threads:
        t.start()
    for t in threads:
        t.join(args.timeout or None)
    return rc[0]


if __name__ == "__main__":
    sys.exit(main())
