"""kungfu-distribute: launch kungfu-run on every host of -H over ssh.

Reference parity: srcs/go/cmd/kungfu-distribute (+ utils/ssh/): each host
runs its own kungfu-run with the same flags and `-self <host-ip>`; workers
then mesh directly. `--dry-run` prints the per-host commands (used by the
unit test; this container has no multi-host fabric).
"""
import argparse
import shlex
import subprocess
import sys


def build_commands(args, rest):
    hosts = [h.split(":")[0] for h in args.hosts.split(",") if h]
    cmds = []
    for ip in hosts:
        inner = ["python3", "-m", "kungfu_amd.run", "-np", str(args.np),
                 "-H", args.hosts, "-self", ip, "-port", str(args.port),
                 "-strategy", args.strategy] + rest
        if args.user:
            target = "%s@%s" % (args.user, ip)
        else:
            target = ip
        cmds.append((ip, ["ssh", "-o", "StrictHostKeyChecking=no", target,
                          " ".join(shlex.quote(c) for c in inner)]))
    return cmds


def main(argv=None):
    p = argparse.ArgumentParser("kungfu-distribute")
    p.add_argument("-np", type=int, required=True)
    p.add_argument("-H", dest="hosts", required=True)
    p.add_argument("-port", type=int, default=38080)
    p.add_argument("-strategy", default="AUTO")
    p.add_argument("-u", dest="user", default=None, help="ssh user")
    p.add_argument("--dry-run", action="store_true")
    p.add_argument("prog", nargs=argparse.REMAINDER)
    args = p.parse_args(argv)
    rest = args.prog[1:] if args.prog and args.prog[0] == "--" else args.prog
    cmds = build_commands(args, rest)
    if args.dry_run:
        for ip, cmd in cmds:
            print("[%s] %s" % (ip, " ".join(cmd)))
        return 0
    procs = [(ip, subprocess.Popen(cmd)) for ip, cmd in cmds]
    code = 0
    for ip, proc in procs:
        rc = proc.wait()
        if rc != 0:
            print("[kungfu-distribute] host %s exited %d" % (ip, rc),
                  file=sys.stderr)
        code = max(code, rc)
    return code


if __name__ == "__main__":
    sys.exit(main())
