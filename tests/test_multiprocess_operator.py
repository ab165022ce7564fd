"""Full multi-process deployment shape: the operator runs as its own OS process
(python -m grove_amd operator --serve), workloads are applied and observed over HTTP
with the CLI — the closest to the reference's operator-pod deployment this environment
can run."""
import json
import os
import subprocess
import sys
import time
import urllib.request

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
PORT = 18311


@pytest.mark.timeout(120)
def test_operator_process_end_to_end(tmp_path):
    cfg = tmp_path / "config.yaml"
    cfg.write_text(f"""
servers:
  api: {{enabled: true, host: 127.0.0.1, port: {PORT}}}
scheduler: {{default: amd-gang-scheduler}}
logLevel: warn
""")
    state = tmp_path / "state.json"
    proc = subprocess.Popen(
        [sys.executable, "-m", "grove_amd", "operator", "--config-file", str(cfg),
         "--virtual-nodes", "2", "--state-file", str(state)],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        base = f"http://127.0.0.1:{PORT}"
        for _ in range(100):
            try:
                urllib.request.urlopen(f"{base}/healthz", timeout=0.5)
                break
            except Exception:
                time.sleep(0.2)
        else:
            raise AssertionError(f"operator never became healthy:\n"
                                 f"{proc.stdout.read()[:2000]}")
        # apply via the CLI
        rc = subprocess.run(
            [sys.executable, "-m", "grove_amd", "apply", "-f", "samples/simple1.yaml",
             "--server", base], cwd=REPO, capture_output=True, text=True, timeout=30)
        assert rc.returncode == 0, rc.stdout + rc.stderr
        # wait for availability over HTTP
        deadline = time.monotonic() + 60
        while time.monotonic() < deadline:
            with urllib.request.urlopen(
                    f"{base}/apis/grove.io/v1alpha1/namespaces/default/"
                    f"podcliquesets/simple1", timeout=5) as r:
                pcs = json.loads(r.read())
            if (pcs.get("status") or {}).get("availableReplicas", 0) >= 1:
                break
            time.sleep(0.25)
        else:
            raise AssertionError(f"never available: {pcs.get('status')}")
        # kubectl-style get
        out = subprocess.run(
            [sys.executable, "-m", "grove_amd", "get", "pods", "--server", base,
             "-l", "app.kubernetes.io/part-of=simple1"],
            cwd=REPO, capture_output=True, text=True, timeout=30)
        assert out.returncode == 0 and out.stdout.count("simple1-0-") == 7
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=10)
        except subprocess.TimeoutExpired:
            proc.kill()
    # graceful shutdown wrote the state snapshot
    assert state.exists() and state.stat().st_size > 1000
