"""torch.distributed (gloo, world 2, CPU) integration of the glue layers:
compat dispatch layout, elastic world detection, dist-coordinated p2p
object transfer. Mirrors how multi-rank bootstrap works on the GPU tier
without needing a GPU."""

import os
import socket
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
WORKER = REPO / "tests" / "workers" / "dist_cpu_worker.py"


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_dist_glue_world2():
    port = _free_port()
    env = dict(os.environ)
    env.update({"PYTHONPATH": str(REPO), "MASTER_ADDR": "127.0.0.1",
                "MASTER_PORT": str(port), "WORLD_SIZE": "2"})
    ps = []
    for rank in range(2):
        e = dict(env)
        e["RANK"] = str(rank)
        ps.append(subprocess.Popen([sys.executable, str(WORKER)], env=e,
                                   stdout=subprocess.PIPE,
                                   stderr=subprocess.STDOUT))
    outs, ok = [], True
    for p in ps:
        try:
            out, _ = p.communicate(timeout=200)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        outs.append(out.decode())
        ok = ok and p.returncode == 0
    joined = "\n====\n".join(outs)
    assert ok, joined
    assert "RANK0 OK" in joined and "RANK1 OK" in joined, joined
