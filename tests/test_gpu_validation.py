"""End-to-end runner-plane validation on real hardware — the
reference's gpucloud scenario suite (integration-test/gpucloud/
README.md:49-56) run against a live control plane + GPU runner via
scripts/gpu_validate.py. Uses the tiny presets so the whole pass stays
under ~2 minutes on an MI355X.
"""
import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(600)
def test_gpucloud_scenarios_small():
    proc = subprocess.run(
        [sys.executable, os.path.join(ROOT, "scripts", "gpu_validate.py"),
         "--small"],
        capture_output=True, text=True, timeout=560, cwd=ROOT)
    lines = [json.loads(l) for l in proc.stdout.splitlines()
             if l.startswith("{")]
    summary = next((l for l in lines if l.get("summary")), None)
    assert summary is not None, proc.stdout + proc.stderr
    failed = [l for l in lines if not l.get("summary") and not l["ok"]]
    assert summary["ok"], f"failed scenarios: {failed}\n{proc.stderr[-2000:]}"
    assert proc.returncode == 0
