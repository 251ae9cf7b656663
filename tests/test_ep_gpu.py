"""GPU-tier EP tests: world=1 directly, world=2 via subprocess pair on one
device (HIP IPC)."""

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
WORKER = REPO / "tests" / "workers" / "ep_worker.py"


def test_ep_world1():
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run([sys.executable, str(WORKER)], env=env,
                       capture_output=True, timeout=280)
    out = r.stdout.decode() + r.stderr.decode()
    assert r.returncode == 0, out
    assert "EP ALL OK" in out


def _run_world2(extra_env, port, world=2):
    env_base = dict(os.environ)
    env_base.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "WORLD_SIZE": str(world), "PYTHONPATH": str(REPO),
    })
    env_base.update(extra_env)
    ps = []
    for r in range(world):
        env = dict(env_base)
        env["RANK"] = str(r)
        ps.append(subprocess.Popen([sys.executable, str(WORKER)], env=env,
                                   stdout=subprocess.PIPE,
                                   stderr=subprocess.STDOUT))
    outs, ok = [], True
    for p in ps:
        try:
            out, _ = p.communicate(timeout=280)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    joined = "\n=====\n".join(outs)
    assert ok, joined
    assert joined.count("EP ALL OK") == world, joined


def test_ep_world2():
    _run_world2({}, 29473)


def test_ep_world2_forced_proxy():
    """Internode-emulation: no IPC mapping between the two ranks; all EP
    traffic flows GPU -> D2H command ring -> CPU proxy -> multipath
    reliable transport -> peer proxy -> peer GPU memory."""
    _run_world2({"UCCL_EP_FORCE_PROXY": "1"}, 29477)


def test_ep_world4_forced_proxy():
    _run_world2({"UCCL_EP_FORCE_PROXY": "1"}, 29479, world=4)
