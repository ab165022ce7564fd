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


@pytest.mark.timeout(180)
def test_remote_node_agent(tmp_path):
    """Fully distributed shape: operator process + remote node-agent process (HTTP
    client kubelet) + pod processes, coordinated only over the wire."""
    port = 18327
    cfg = tmp_path / "config.yaml"
    cfg.write_text(f"servers:\n  api: {{enabled: true, host: 127.0.0.1, port: {port}}}\n"
                   f"logLevel: warn\n")
    base = f"http://127.0.0.1:{port}"
    env = {**os.environ, "GROVE_AGENT_TOKEN": "test-agent-token"}
    op = subprocess.Popen(
        [sys.executable, "-m", "grove_amd", "operator", "--config-file", str(cfg)],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
        env=env)
    agent = None
    try:
        for _ in range(100):
            try:
                urllib.request.urlopen(f"{base}/healthz", timeout=0.5)
                break
            except Exception:
                time.sleep(0.2)
        agent = subprocess.Popen(
            [sys.executable, "-m", "grove_amd", "agent", "--server", base,
             "--node-name", "remote-0", "--virtual-gpus", "0",
             "--poll-interval", "0.1"],
            cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True,
            env=env)
        pcs = {"apiVersion": "grove.io/v1alpha1", "kind": "PodCliqueSet",
               "metadata": {"name": "remote"},
               "spec": {"replicas": 1, "template": {"cliques": [{
                   "name": "w",
                   "annotations": {"grove.io/payload": "none"},
                   "spec": {"roleName": "w", "replicas": 2,
                            "podSpec": {"containers": [{
                                "name": "m", "image": "i",
                                "resources": {"requests": {"cpu": "1"}}}]}}}]}}}
        req = urllib.request.Request(
            f"{base}/apis/grove.io/v1alpha1/namespaces/default/podcliquesets",
            data=json.dumps(pcs).encode(), method="POST",
            headers={"Content-Type": "application/json"})
        urllib.request.urlopen(req, timeout=5)
        deadline = time.monotonic() + 90
        phases = []
        while time.monotonic() < deadline:
            with urllib.request.urlopen(
                    f"{base}/api/v1/namespaces/default/pods"
                    f"?labelSelector=app.kubernetes.io/part-of=remote",
                    timeout=5) as r:
                pods = json.loads(r.read())["items"]
            phases = [(p.get("status") or {}).get("phase") for p in pods]
            if len(pods) == 2 and all(ph == "Succeeded" for ph in phases):
                break
            time.sleep(0.3)
        else:
            raise AssertionError(
                f"pods never succeeded on the remote agent: {phases}\n"
                f"agent: {agent.stdout.read()[:1500] if agent.poll() is not None else '(running)'}")
        # pods ran on the agent's node
        assert all(p["spec"]["nodeName"] == "remote-0" for p in pods)
    finally:
        for proc in (agent, op):
            if proc is not None:
                proc.terminate()
                try:
                    proc.wait(timeout=10)
                except subprocess.TimeoutExpired:
                    proc.kill()
