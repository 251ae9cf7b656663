"""Native c10d "uccl" backend: torch.distributed + DDP end-to-end on 2
processes sharing one GPU."""

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
WORKER = REPO / "tests" / "workers" / "pg_worker.py"


def test_pg_backend_two_ranks():
    env_base = dict(os.environ)
    env_base.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29475",
        "WORLD_SIZE": "2", "PYTHONPATH": str(REPO),
    })
    ps = []
    for r in range(2):
        env = dict(env_base)
        env["RANK"] = str(r)
        ps.append(subprocess.Popen([sys.executable, str(WORKER)], env=env,
                                   stdout=subprocess.PIPE,
                                   stderr=subprocess.STDOUT))
    outs, ok = [], True
    for p in ps:
        try:
            out, _ = p.communicate(timeout=240)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    joined = "\n=====\n".join(outs)
    assert ok, joined
    assert "PG BACKEND ALL OK" in joined
