"""CLI smoke: python -m uccl_amd {info,env}."""

import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def _run(*args):
    return subprocess.run([sys.executable, "-m", "uccl_amd", *args],
                          cwd=REPO, capture_output=True, text=True,
                          timeout=180)


def test_cli_info():
    r = _run("info")
    assert r.returncode == 0, r.stderr
    assert "uccl_amd" in r.stdout
    assert "native ext" in r.stdout
    assert "librccl-net-uccl.so" in r.stdout


def test_cli_env():
    r = _run("env")
    assert r.returncode == 0, r.stderr
    for key in ("UCCL_TP_CC", "UCCL_NET_TRANSPORT", "UCCL_SYM_HEAP_MB"):
        assert key in r.stdout


def test_cli_unknown_command():
    r = _run("bogus")
    assert r.returncode == 2
