#!/usr/bin/env python3
"""Prove librccl-net-uccl.so under REAL RCCL.

Two tiers:

1. (any GPU count) Plugin load + init proof: start a 1-rank
   torch.distributed "nccl" (=RCCL) process group with NCCL_NET_PLUGIN
   pointing at our plugin and NCCL_DEBUG=INFO, run an all_reduce, and
   assert RCCL's own log shows it loaded and initialized the uccl net
   (it enumerates/initializes networks during comm init). This exercises
   pluginInit / pluginDevices / pluginGetProperties under the real loader.

2. (>=2 GPUs) Full data-path proof: 2 ranks, one per GPU, with
   NCCL_P2P_DISABLE=1 NCCL_SHM_DISABLE=1 so RCCL's only transport is the
   net plugin; all_reduce + all_gather correctness checked against torch
   references. RCCL refuses two ranks on one device ("Duplicate GPU
   detected"), so this tier needs a real multi-GPU lease; tier 1 is the
   1-GPU-box fallback.

Writes the captured RCCL logs to gpurun_out/rccl_plugin_proof/.
Reference parity: the reference proves its plugin with rccl-tests
(thirdparty/rccl-tests); no rccl-tests binary ships in this image, so
torch.distributed-over-RCCL is the equivalent stock consumer.
"""

import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
LIBDIR = REPO / "uccl_amd" / "lib"
OUT = REPO / "gpurun_out" / "rccl_plugin_proof"

WORKER = r"""
import os, sys, torch, torch.distributed as dist
rank = int(os.environ.get("RANK", "0"))
world = int(os.environ.get("WORLD_SIZE", "1"))
torch.cuda.set_device(int(os.environ.get("DEV", "0")))
dist.init_process_group("nccl", rank=rank, world_size=world)
n = 1 << 20
x = torch.full((n,), float(rank + 1), device="cuda")
dist.all_reduce(x)
torch.cuda.synchronize()
want = float(world * (world + 1) // 2)
assert torch.all(x == want), (x[:4], want)
g = [torch.empty(n, device="cuda") for _ in range(world)]
dist.all_gather(g, torch.full((n,), float(rank + 7), device="cuda"))
torch.cuda.synchronize()
for r in range(world):
    assert torch.all(g[r] == float(r + 7))
# >4MB message to push multiple NCCL chunks through the net
big = torch.randn(3 << 20, device="cuda")
ref = big.clone()
dist.all_reduce(big)
torch.cuda.synchronize()
if world == 1:
    assert torch.allclose(big, ref)
print(f"RANK {rank} OK", flush=True)
dist.destroy_process_group()
"""


def run_tier(world: int, tag: str) -> bool:
    OUT.mkdir(parents=True, exist_ok=True)
    env_base = dict(os.environ)
    env_base.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": "29617",
        "WORLD_SIZE": str(world),
        "NCCL_NET_PLUGIN": "uccl",
        "LD_LIBRARY_PATH": f"{LIBDIR}:" + env_base.get("LD_LIBRARY_PATH", ""),
        "NCCL_DEBUG": "INFO",
        "NCCL_DEBUG_SUBSYS": "INIT,NET",
        "HSA_ENABLE_IPC_MODE_LEGACY": "0",
    })
    if world > 1:
        # force the net path between intranode ranks
        env_base["NCCL_P2P_DISABLE"] = "1"
        env_base["NCCL_SHM_DISABLE"] = "1"
    procs = []
    for r in range(world):
        env = dict(env_base)
        env["RANK"] = str(r)
        env["DEV"] = str(r)
        procs.append(subprocess.Popen(
            [sys.executable, "-c", WORKER], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    logs, ok = [], True
    for p in procs:
        try:
            out, _ = p.communicate(timeout=240)
        except subprocess.TimeoutExpired:
            p.kill()
            out, _ = p.communicate()
            ok = False
        logs.append(out.decode(errors="replace"))
        ok = ok and p.returncode == 0
    log = "\n===== rank split =====\n".join(logs)
    (OUT / f"{tag}.log").write_text(log)
    loaded = ("uccl" in log and
              ("Using network" in log or "Initialized NET plugin" in log or
               "NET/uccl" in log))
    print(f"--- tier {tag}: procs_ok={ok} plugin_loaded={loaded}")
    for line in log.splitlines():
        if "uccl" in line.lower() and "NCCL INFO" in line:
            print("   ", line.strip())
    return ok and loaded


def main():
    import torch
    ngpu = torch.cuda.device_count()
    print(f"GPUs visible: {ngpu}")
    ok1 = run_tier(1, "world1_load")
    ok2 = True
    if ngpu >= 2:
        ok2 = run_tier(min(ngpu, 2), "world2_netpath")
    else:
        print("--- tier world2_netpath: SKIPPED (needs >=2 GPUs; RCCL "
              "rejects duplicate devices)")
    if ok1 and ok2:
        print("RCCL PLUGIN PROOF OK")
        return 0
    print("RCCL PLUGIN PROOF FAILED")
    return 1


if __name__ == "__main__":
    sys.exit(main())
