"""NCCL C-ABI drop-in (libuccl_nccl.so): 2-process bootstrap + collective
battery via ctypes on one GPU."""

import os
import subprocess
import sys
import tempfile
import uuid
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires GPU", allow_module_level=True)

REPO = Path(__file__).resolve().parent.parent
WORKER = REPO / "tests" / "workers" / "nccl_shim_worker.py"


def test_nccl_shim_two_ranks():
    id_path = os.path.join(tempfile.gettempdir(),
                           f"uccl_nccl_{uuid.uuid4().hex}.id")
    ps = []
    for r in range(2):
        env = dict(os.environ)
        env.update({"RANK": str(r), "WORLD_SIZE": "2",
                    "PYTHONPATH": str(REPO)})
        ps.append(subprocess.Popen(
            [sys.executable, str(WORKER), id_path], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs, ok = [], True
    for p in ps:
        try:
            out, _ = p.communicate(timeout=220)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    joined = "\n=====\n".join(outs)
    assert ok, joined
    assert joined.count("NCCL SHIM OK") == 2, joined
