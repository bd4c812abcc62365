#!/usr/bin/env python3
"""Distributed launcher (parity role of tools/launch.py + dmlc-tracker).

MI355X-native: instead of the reference's ps-lite scheduler/server/worker
topology, training scales as one process per GPU with RCCL collectives;
this launcher spawns the local ranks (and, with --hosts, ssh-launches the
remote ranks) with the torch.distributed rendezvous env (MASTER_ADDR/PORT,
RANK, WORLD_SIZE, LOCAL_RANK).

  # single node, 8 GPUs:
  python tools/launch.py -n 8 python train.py --kv-store device
  # multi node (2 hosts x 8):
  python tools/launch.py -n 16 -H host1,host2 --master host1 python train.py
  # CPU CI of the multi-process path (gloo):
  python tools/launch.py -n 2 --backend gloo python tests/...
"""
import argparse
import os
import signal
import subprocess
import sys


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-n", "--num-workers", type=int, required=True)
    ap.add_argument("-H", "--hosts", default=None,
                    help="comma-separated host list (default: local only)")
    ap.add_argument("--master", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=29517)
    ap.add_argument("--backend", default="nccl", choices=["nccl", "gloo"])
    ap.add_argument("--launcher", default="local", choices=["local", "ssh"],
                    help="'local' spawns every rank on this host "
                         "(reference --launcher local CI mode)")
    ap.add_argument("command", nargs=argparse.REMAINDER)
    args = ap.parse_args()
    assert args.command, "no command given"

    hosts = args.hosts.split(",") if args.hosts else ["127.0.0.1"]
    per_host = args.num_workers // len(hosts)
    procs = []

    def env_for(rank, local_rank):
        env = dict(os.environ)
        env.update({
            "MASTER_ADDR": args.master,
            "MASTER_PORT": str(args.port),
            "WORLD_SIZE": str(args.num_workers),
            "RANK": str(rank),
            "LOCAL_RANK": str(local_rank),
            "MXA_DIST_BACKEND": args.backend,
        })
        return env

    rank = 0
    for hi, host in enumerate(hosts):
        for lr in range(per_host):
            if args.launcher == "local" or host in ("127.0.0.1",
                                                    "localhost"):
                p = subprocess.Popen(args.command, env=env_for(rank, lr))
            else:
                envs = " ".join(
                    f"{k}={v}" for k, v in env_for(rank, lr).items()
                    if k.startswith(("MASTER", "WORLD", "RANK", "LOCAL",
                                     "MXA_", "HSA_")))
                p = subprocess.Popen(
                    ["ssh", host, f"cd {os.getcwd()} && {envs} " +
                     " ".join(args.command)])
            procs.append(p)
            rank += 1

    def terminate(*_):
        for p in procs:
            if p.poll() is None:
                p.terminate()
    signal.signal(signal.SIGINT, terminate)
    signal.signal(signal.SIGTERM, terminate)

    rc = 0
    for p in procs:
        p.wait()
        rc = rc or p.returncode
    sys.exit(rc)


if __name__ == "__main__":
    main()
