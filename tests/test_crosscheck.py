"""Numerical cross-validation against the reference implementation, run as
part of the CPU suite when the reference tree is present (it is mounted
read-only in the build environment; users without it skip)."""

import os
import subprocess
import sys

import pytest


@pytest.mark.skipif(not os.path.isdir("/root/reference/src"), reason="reference tree not mounted")
def test_gradients_match_reference():
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    result = subprocess.run(
        [sys.executable, os.path.join(repo, "scripts", "crosscheck_reference.py")],
        capture_output=True, text=True, timeout=600,
    )
    assert result.returncode == 0, result.stdout[-3000:] + result.stderr[-2000:]
    assert "all within 1e-6 of the reference" in result.stdout


@pytest.mark.skipif(not os.path.isdir("/root/reference/src"), reason="reference tree not mounted")
def test_symbol_method_ctor_parity():
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    result = subprocess.run(
        [sys.executable, os.path.join(repo, "scripts", "audit_parity.py")],
        capture_output=True, text=True, timeout=600,
    )
    assert result.returncode == 0, result.stdout[-3000:] + result.stderr[-1500:]
    assert "parity holds" in result.stdout
