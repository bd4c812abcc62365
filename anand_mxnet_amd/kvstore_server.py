"""KVStore server-role entry point (reference parity:
python/mxnet/kvstore_server.py — the process a `dist_*` job starts when
DMLC_ROLE is 'server' or 'scheduler').

MI355X-native design: there IS no parameter server — multi-node data
parallelism is RCCL all-reduce, so the only coordination process a job
needs is the torch.distributed TCP-store rendezvous host, which rank 0
embeds.  This module keeps the reference's launch contract working:
a process started with DMLC_ROLE=server/scheduler parks until the job's
workers finish (launchers written against the reference can keep
spawning the same role set; server slots simply cost nothing here).
"""
from __future__ import annotations

import os
import sys
import time


def _init_server_module():
    role = os.environ.get("DMLC_ROLE", "worker")
    if role not in ("server", "scheduler"):
        return False
    sys.stderr.write(
        f"[anand_mxnet_amd] role '{role}': parameter servers are not used "
        "(all-reduce architecture); idling until the job ends.\n")
    return True


class KVStoreServer:
    """Compatibility shell: run() blocks like the reference server loop."""

    def __init__(self, kvstore=None):
        self.kvstore = kvstore

    def run(self):
        # Workers tear this process down via the launcher when training
        # ends (reference: ps-lite Finalize barrier).
        while True:
            time.sleep(5)


if __name__ == "__main__":
    if _init_server_module():
        KVStoreServer().run()
