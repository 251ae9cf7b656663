"""2-process collective battery on one GPU (HIP IPC across processes).

Spawns tests/workers/collective_worker.py twice with gloo rendezvous on
127.0.0.1. Exercises the full multi-rank protocol: IPC heap exchange,
flag signal/wait, LL packets, parity scratch reuse, p2p ack credits.
"""

import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires GPU", allow_module_level=True)

REPO = Path(__file__).resolve().parent.parent
WORKER = REPO / "tests" / "workers" / "collective_worker.py"


import pytest


@pytest.mark.parametrize("world", [2, 4, 8])
def test_two_process_collectives(world):
    env_base = dict(os.environ)
    env_base.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(29471 + world),
        "WORLD_SIZE": str(world),
        "HSA_ENABLE_IPC_MODE_LEGACY": "0",
        "PYTHONPATH": str(REPO),
        "UCCL_TEST_LIGHT": "1" if world > 2 else "0",
        "UCCL_TEST_ALARM": "600",
    })
    procs = []
    for r in range(world):
        env = dict(env_base)
        env["RANK"] = str(r)
        procs.append(subprocess.Popen(
            [sys.executable, str(WORKER)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    ok = True
    for p in procs:
        try:
            out, _ = p.communicate(timeout=620)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    assert ok, "worker failure:\n" + "\n=====\n".join(outs)
    assert "ALL COLLECTIVE TESTS PASSED" in "".join(outs)
