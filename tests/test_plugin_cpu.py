"""RCCL net plugin (v6 ABI) tests: build + dlopen harness, no GPU/RCCL
needed. The harness drives listen/connect/accept/isend/irecv/test exactly
as RCCL's proxy would."""

import subprocess
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


import os

import pytest


@pytest.mark.parametrize("plane", ["multipath", "tcp"])
def test_plugin_harness(plane):
    from uccl_amd._build import build_plugin

    so = build_plugin()
    harness = so.parent / "plugin_test"
    env = dict(os.environ)
    env["UCCL_NET_TRANSPORT"] = plane
    r = subprocess.run([str(harness), str(so)], capture_output=True,
                       timeout=300, env=env)
    out = r.stdout.decode() + r.stderr.decode()
    assert r.returncode == 0, out
    assert "PLUGIN HARNESS OK" in out


def test_plugin_exports_symbol():
    from uccl_amd._build import build_plugin

    so = build_plugin()
    r = subprocess.run(["nm", "-D", str(so)], capture_output=True)
    assert b"ncclNetPlugin_v6" in r.stdout
