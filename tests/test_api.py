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


def test_validation_podspec_sanity():
    """validatePodSpec parity (validation/podcliqueset.go:591-642): operator-owned
    scheduling fields rejected on create; env names validated and deduped."""
    from grove_amd import Cluster
    from grove_amd.kubecore.store import ApiError

    def pcs(podspec_extra=None, env=None):
        ps = {"containers": [{"name": "m", "image": "i",
                              "env": env or []}]}
        ps.update(podspec_extra or {})
        return {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
                "metadata": {"name": "vps"},
                "spec": {"replicas": 1, "template": {"cliques": [{
                    "name": "w", "spec": {"roleName": "w", "replicas": 1,
                                          "podSpec": ps}}]}}}
    cl = Cluster()
    try:
        with pytest.raises(ApiError, match="nodeName"):
            cl.store.create(pcs({"nodeName": "n1"}))
        with pytest.raises(ApiError, match="topologySpreadConstraints"):
            cl.store.create(pcs({"topologySpreadConstraints": [
                {"maxSkew": 1, "topologyKey": "zone",
                 "whenUnsatisfiable": "DoNotSchedule"}]}))
        with pytest.raises(ApiError, match="invalid environment variable"):
            cl.store.create(pcs(env=[{"name": "1BAD", "value": "x"}]))
        with pytest.raises(ApiError, match="duplicate environment variable"):
            cl.store.create(pcs(env=[{"name": "A", "value": "1"},
                                     {"name": "A", "value": "2"}]))
        cl.store.create(pcs(env=[{"name": "GOOD_ONE", "value": "1"}]))
    finally:
        cl.stop()


def test_validation_resource_sharing_rules():
    """validateResourceSharingSpecs/filters parity (podcliqueset.go:139-233)."""
    from grove_amd import Cluster
    from grove_amd.kubecore.store import ApiError

    def pcs(sharing=None, sg_sharing=None, templates=None):
        tmpl = {"cliques": [
            {"name": "a", "spec": {"roleName": "a", "replicas": 1,
                                   "podSpec": {"containers": [
                                       {"name": "m", "image": "i"}]}}},
            {"name": "b", "spec": {"roleName": "b", "replicas": 1,
                                   "podSpec": {"containers": [
                                       {"name": "m", "image": "i"}]}}}],
            "podCliqueScalingGroups": [{
                "name": "sg", "cliqueNames": ["b"],
                **({"resourceSharing": sg_sharing} if sg_sharing else {})}]}
        if templates is not None:
            tmpl["resourceClaimTemplates"] = templates
        if sharing is not None:
            tmpl["resourceSharing"] = sharing
        return {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
                "metadata": {"name": "vrs"}, "spec": {"replicas": 1,
                                                      "template": tmpl}}
    tpl = [{"name": "t1", "templateSpec": {"spec": {"devices": {"requests": [
        {"name": "d", "deviceClassName": "x"}]}}}}]
    cl = Cluster()
    try:
        with pytest.raises(ApiError, match="scope"):
            cl.store.create(pcs(sharing=[{"name": "t1", "scope": "Bogus"}],
                                templates=tpl))
        with pytest.raises(ApiError, match="duplicate reference"):
            cl.store.create(pcs(sharing=[
                {"name": "t1", "scope": "AllReplicas"},
                {"name": "t1", "scope": "PerReplica"}], templates=tpl))
        with pytest.raises(ApiError, match="namespace must be empty"):
            cl.store.create(pcs(sharing=[{"name": "t1", "namespace": "other",
                                          "scope": "AllReplicas"}],
                                templates=tpl))
        with pytest.raises(ApiError, match="unknown clique"):
            cl.store.create(pcs(sharing=[{
                "name": "t1", "scope": "AllReplicas",
                "filter": {"childCliqueNames": ["nope"]}}], templates=tpl))
        # PCSG-scope filter may only reference member cliques
        with pytest.raises(ApiError, match="unknown clique"):
            cl.store.create(pcs(sg_sharing=[{
                "name": "t1", "scope": "PerReplica",
                "filter": {"childCliqueNames": ["a"]}}], templates=tpl))
        # claim template device requests required
        with pytest.raises(ApiError, match="device request"):
            cl.store.create(pcs(templates=[
                {"name": "t2", "templateSpec": {"spec": {}}}]))
        with pytest.raises(ApiError, match="duplicate template"):
            cl.store.create(pcs(templates=tpl + tpl))
        cl.store.create(pcs(sharing=[{
            "name": "t1", "scope": "AllReplicas",
            "filter": {"childCliqueNames": ["a"],
                       "childScalingGroupNames": ["sg"]}}], templates=tpl))
    finally:
        cl.stop()
