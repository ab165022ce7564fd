"""API contract tests: namegen byte-compatibility, defaulting, validation."""
import pytest

from grove_amd.api import constants as c, namegen
from grove_amd.api.defaulting import default_podcliqueset, parse_duration_seconds
from grove_amd.api.validation import validate_podcliqueset
from grove_amd.kubecore.store import ApiError


# ---- namegen: values must match reference namegen.go byte-for-byte ----

def test_namegen_contract():
    assert namegen.headless_service_name("pcs", 2) == "pcs-2"
    assert namegen.headless_service_address("pcs", 0, "ns") == "pcs-0.ns.svc.cluster.local"
    assert namegen.pod_role_name("pcs") == "grove.io:pcs:pcs"
    assert namegen.pod_role_binding_name("pcs") == "grove.io:pcs:pcs"
    assert namegen.pod_service_account_name("pcs") == "pcs"
    assert namegen.initc_sa_token_secret_name("pcs") == "pcs-ic-sat"
    assert namegen.podclique_name("simple1", 0, "pca") == "simple1-0-pca"
    assert namegen.pcsg_name("simple1", 0, "sga") == "simple1-0-sga"
    assert namegen.base_podgang_name("simple1", 0) == "simple1-0"
    assert namegen.scaled_podgang_name("simple1-0-sga", 1) == "simple1-0-sga-1"
    assert namegen.pod_hostname("simple1-0-pca", 3) == "simple1-0-pca-3"
    assert namegen.extract_scaling_group_name("simple1-0-sga", "simple1", 0) == "sga"


def test_podgang_name_for_pcsg_member():
    # replicas below minAvailable -> base gang; above -> scaled gang 0-based
    assert namegen.podgang_name_for_pclq_in_pcsg("s", 0, "s-0-sg", 2, 0) == "s-0"
    assert namegen.podgang_name_for_pclq_in_pcsg("s", 0, "s-0-sg", 2, 1) == "s-0"
    assert namegen.podgang_name_for_pclq_in_pcsg("s", 0, "s-0-sg", 2, 2) == "s-0-sg-0"
    assert namegen.podgang_name_for_pclq_in_pcsg("s", 0, "s-0-sg", 2, 4) == "s-0-sg-2"


def test_label_constants():
    assert c.LABEL_PODGANG == "grove.io/podgang"
    assert c.LABEL_POD_TEMPLATE_HASH == "grove.io/pod-template-hash"
    assert c.POD_GANG_SCHEDULING_GATE == "grove.io/podgang-pending-creation"
    assert c.FINALIZER_PCS == "grove.io/podcliqueset.grove.io"
    assert c.ENV_PCSG_TEMPLATE_NUM_PODS == "GROVE_PCSG_TEMPLATE_NUM_PODS"


# ---- defaulting ----

def _pcs(cliques=None, sgs=None, **spec):
    return {
        "apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
        "metadata": {"name": "t"},
        "spec": {
            "replicas": 1,
            "template": {
                "cliques": cliques if cliques is not None else [
                    {"name": "a", "spec": {"roleName": "ra",
                                           "podSpec": {"containers": [{"name": "c"}]}}}],
                **({"podCliqueScalingGroups": sgs} if sgs else {}),
            },
            **spec,
        },
    }


def test_defaulting_fills_contract_defaults():
    pcs = _pcs()
    default_podcliqueset(pcs)
    t = pcs["spec"]["template"]
    assert t["cliqueStartupType"] == c.STARTUP_ANY_ORDER
    assert parse_duration_seconds(t["terminationDelay"]) == 4 * 3600
    assert t["headlessServiceConfig"]["publishNotReadyAddresses"] is True
    cl = t["cliques"][0]["spec"]
    assert cl["replicas"] == 1 and cl["minAvailable"] == 1
    assert cl["podSpec"]["restartPolicy"] == "Always"
    assert cl["podSpec"]["terminationGracePeriodSeconds"] == 30
    assert pcs["spec"]["updateStrategy"]["type"] == c.UPDATE_ROLLING_RECREATE


def test_defaulting_pcsg():
    pcs = _pcs(cliques=[{"name": "a", "spec": {"roleName": "r", "replicas": 2,
                                               "podSpec": {"containers": [{"name": "c"}]}}}],
               sgs=[{"name": "sg", "cliqueNames": ["a"],
                     "scaleConfig": {"maxReplicas": 5}}])
    default_podcliqueset(pcs)
    sg = pcs["spec"]["template"]["podCliqueScalingGroups"][0]
    assert sg["replicas"] == 1 and sg["minAvailable"] == 1
    assert sg["scaleConfig"]["minReplicas"] == 1


def test_duration_parse():
    assert parse_duration_seconds("4h") == 14400
    assert parse_duration_seconds("1h30m") == 5400
    assert parse_duration_seconds("250ms") == 0.25
    with pytest.raises(ValueError):
        parse_duration_seconds("5")


# ---- validation ----

def _valid_pcs():
    pcs = _pcs()
    default_podcliqueset(pcs)
    return pcs


def test_validation_accepts_valid():
    validate_podcliqueset(_valid_pcs())


def test_validation_rejects_no_cliques():
    pcs = _pcs(cliques=[])
    with pytest.raises(ApiError):
        validate_podcliqueset(pcs)


def test_validation_rejects_duplicate_clique():
    pcs = _pcs(cliques=[
        {"name": "a", "spec": {"roleName": "r", "podSpec": {"containers": [{"name": "c"}]}}},
        {"name": "a", "spec": {"roleName": "r", "podSpec": {"containers": [{"name": "c"}]}}},
    ])
    default_podcliqueset(pcs)
    with pytest.raises(ApiError):
        validate_podcliqueset(pcs)


def test_validation_minavailable_gt_replicas():
    pcs = _pcs(cliques=[{"name": "a", "spec": {
        "roleName": "r", "replicas": 2, "minAvailable": 3,
        "podSpec": {"containers": [{"name": "c"}]}}}])
    with pytest.raises(ApiError):
        validate_podcliqueset(pcs)


def test_validation_45_char_name_budget():
    pcs = _pcs()
    pcs["metadata"]["name"] = "x" * 30
    pcs["spec"]["template"]["cliques"][0]["name"] = "y" * 20
    default_podcliqueset(pcs)
    with pytest.raises(ApiError) as ei:
        validate_podcliqueset(pcs)
    assert "45-character" in str(ei.value)


def test_validation_startup_dag_cycle():
    cliques = [
        {"name": "a", "spec": {"roleName": "r", "startsAfter": ["b"],
                               "podSpec": {"containers": [{"name": "c"}]}}},
        {"name": "b", "spec": {"roleName": "r", "startsAfter": ["a"],
                               "podSpec": {"containers": [{"name": "c"}]}}},
    ]
    pcs = _pcs(cliques=cliques)
    pcs["spec"]["template"]["cliqueStartupType"] = c.STARTUP_EXPLICIT
    default_podcliqueset(pcs)
    with pytest.raises(ApiError) as ei:
        validate_podcliqueset(pcs)
    assert "cycle" in str(ei.value)


def test_validation_starts_after_requires_explicit():
    cliques = [
        {"name": "a", "spec": {"roleName": "r", "podSpec": {"containers": [{"name": "c"}]}}},
        {"name": "b", "spec": {"roleName": "r", "startsAfter": ["a"],
                               "podSpec": {"containers": [{"name": "c"}]}}},
    ]
    pcs = _pcs(cliques=cliques)
    default_podcliqueset(pcs)
    with pytest.raises(ApiError):
        validate_podcliqueset(pcs)


def test_validation_reserved_env_prefix():
    pcs = _pcs(cliques=[{"name": "a", "spec": {
        "roleName": "r",
        "podSpec": {"containers": [{"name": "c",
                                    "env": [{"name": "GROVE_PCS_NAME", "value": "x"}]}]}}}])
    default_podcliqueset(pcs)
    with pytest.raises(ApiError):
        validate_podcliqueset(pcs)


def test_validation_sg_member_autoscaling_conflict():
    pcs = _pcs(cliques=[{"name": "a", "spec": {
        "roleName": "r", "autoScalingConfig": {"maxReplicas": 4},
        "podSpec": {"containers": [{"name": "c"}]}}}],
        sgs=[{"name": "sg", "cliqueNames": ["a"]}])
    default_podcliqueset(pcs)
    with pytest.raises(ApiError):
        validate_podcliqueset(pcs)


def test_validation_immutable_updates():
    old = _valid_pcs()
    new = _valid_pcs()
    new["spec"]["template"]["cliques"] = [
        {"name": "z", "spec": {"roleName": "r",
                               "podSpec": {"containers": [{"name": "c"}]}}}]
    default_podcliqueset(new)
    with pytest.raises(ApiError):
        validate_podcliqueset(new, old)
