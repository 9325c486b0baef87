#!/usr/bin/env python3
"""Cluster cleanup (reference tools/kill-mxnet.py, redesigned around exact
PIDs instead of process-name pattern matching — pattern kills can take out
unrelated processes).

The launcher (tools/launch.py) records every worker pid under
`<rundir>/<worker_id>.pid` (default rundir: ~/.dtmx/run). This tool reads
those pidfiles and kills the exact recorded pids — locally, or over ssh for
remote hosts from the hostfile.

    python tools/kill_dtmx.py                 # kill everything in the rundir
    python tools/kill_dtmx.py -H hostfile     # also clean remote hosts
"""
import argparse
import os
import signal
import subprocess
import sys

DEFAULT_RUNDIR = os.path.expanduser(os.environ.get("DTMX_RUN_DIR",
                                                   "~/.dtmx/run"))


def kill_local(rundir: str) -> int:
    n = 0
    if not os.path.isdir(rundir):
        return 0
    for f in sorted(os.listdir(rundir)):
        if not f.endswith(".pid"):
            continue
        path = os.path.join(rundir, f)
        try:
            pid = int(open(path).read().strip())
            os.kill(pid, signal.SIGTERM)
            print(f"killed {f[:-4]} (pid {pid})")
            n += 1
        except (ValueError, ProcessLookupError, PermissionError) as e:
            print(f"skip {f}: {e}")
        finally:
            try:
                os.remove(path)
            except OSError:
                pass
    return n


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-H", "--hostfile", default=None)
    ap.add_argument("--rundir", default=DEFAULT_RUNDIR)
    args = ap.parse_args()
    n = kill_local(args.rundir)
    if args.hostfile:
        hosts = {line.strip() for line in open(args.hostfile)
                 if line.strip() and not line.startswith("#")}
        me = {"127.0.0.1", "localhost", os.uname().nodename}
        for h in sorted(hosts - me):
            subprocess.run(
                ["ssh", "-o", "StrictHostKeyChecking=no", h,
                 f"python3 {os.path.abspath(__file__)} --rundir "
                 f"{args.rundir}"], check=False)
    print(f"cleaned {n} local worker(s)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
