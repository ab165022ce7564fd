"""Packaging coherence: the kustomization + Dockerfile +
manifests must deploy operator + topology-agent + CRDs as one consistent unit —
every referenced file exists, every image/Secret/SA/config reference resolves, and
the shipped operator config parses with the real loader."""
import os

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DEPLOY = os.path.join(REPO, "deploy")


def _docs(path):
    return [d for d in yaml.safe_load_all(open(path)) if d]


def _kustomize_resources():
    k = yaml.safe_load(open(os.path.join(DEPLOY, "kustomization.yaml")))
    return k["resources"], k


def test_kustomization_resources_exist_and_parse():
    resources, _k = _kustomize_resources()
    docs = []
    for rel in resources:
        path = os.path.normpath(os.path.join(DEPLOY, rel))
        assert os.path.exists(path), f"kustomization references missing {rel}"
        docs.extend(_docs(path))
    kinds = [d["kind"] for d in docs]
    # the unit ships all five CRDs + operator + agent
    assert kinds.count("CustomResourceDefinition") == 5
    for want in ("Namespace", "ServiceAccount", "ClusterRole",
                 "ClusterRoleBinding", "ConfigMap", "Secret", "Deployment",
                 "Service", "DaemonSet"):
        assert want in kinds, f"missing {want} in deploy unit"


def test_images_and_references_are_coherent():
    docs = []
    for f in ("operator.yaml", "topology-agent.yaml"):
        docs.extend(_docs(os.path.join(DEPLOY, f)))
    by_kind = {}
    for d in docs:
        by_kind.setdefault(d["kind"], []).append(d)

    # every container image is the one the Dockerfile builds (kustomize retags)
    images = set()
    for d in by_kind["Deployment"] + by_kind["DaemonSet"]:
        spec = d["spec"]["template"]["spec"]
        for ctr in spec.get("containers", []) + spec.get("initContainers", []):
            images.add(ctr["image"])
    assert images == {"grove-amd:latest"}

    # secret/config/SA references resolve within the unit
    secret_names = {d["metadata"]["name"] for d in by_kind.get("Secret", [])}
    cm_names = {d["metadata"]["name"] for d in by_kind.get("ConfigMap", [])}
    sa_names = {d["metadata"]["name"] for d in by_kind.get("ServiceAccount", [])}
    for d in by_kind["Deployment"] + by_kind["DaemonSet"]:
        spec = d["spec"]["template"]["spec"]
        assert spec["serviceAccountName"] in sa_names
        for vol in spec.get("volumes", []):
            if "configMap" in vol:
                assert vol["configMap"]["name"] in cm_names
        for ctr in spec.get("containers", []):
            for env in ctr.get("env", []):
                ref = (env.get("valueFrom") or {}).get("secretKeyRef")
                if ref:
                    assert ref["name"] in secret_names, \
                        f"{ctr['name']} references unknown secret {ref['name']}"

    # the agent points at the operator Service's DNS name + port
    svc = by_kind["Service"][0]
    svc_dns = (f"{svc['metadata']['name']}."
               f"{svc['metadata']['namespace']}.svc")
    agent_cmd = " ".join(
        by_kind["DaemonSet"][0]["spec"]["template"]["spec"]["containers"][0][
            "command"])
    assert svc_dns in agent_cmd
    assert str(svc["spec"]["ports"][0]["port"]) in agent_cmd


def test_shipped_operator_config_parses():
    import tempfile
    from grove_amd.config import load_configuration
    docs = _docs(os.path.join(DEPLOY, "operator.yaml"))
    cm = next(d for d in docs if d["kind"] == "ConfigMap")
    with tempfile.NamedTemporaryFile("w", suffix=".yaml", delete=False) as f:
        f.write(cm["data"]["config.yaml"])
        path = f.name
    cfg = load_configuration(path)
    assert cfg.api_server.enabled
    os.unlink(path)


def test_dockerfile_builds_the_referenced_entrypoints():
    df = open(os.path.join(REPO, "Dockerfile")).read()
    assert "COPY grove_amd/ grove_amd/" in df
    assert "COPY crds/ crds/" in df
    assert "build_all" in df  # native extensions compiled at image build
    assert "gfx950" in df
    assert 'ENTRYPOINT ["python", "-m", "grove_amd"]' in df
