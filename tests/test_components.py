"""Component tests: HPA autoscaling loop, authorizer, resource sharing, lastErrors."""
import time

import pytest

from grove_amd.api import constants as c
from grove_amd.controllers.hpa import CPU_USAGE_ANNOTATION
from grove_amd.kubecore.identity import as_user
from grove_amd.kubecore.store import ApiError


def _pcs_with_hpa(name="h1"):
    return {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
            "metadata": {"name": name},
            "spec": {"replicas": 1, "template": {"cliques": [{
                "name": "web", "spec": {
                    "roleName": "web", "replicas": 2,
                    "minAvailable": 1,
                    "podSpec": {"containers": [{
                        "name": "m", "image": "img",
                        "resources": {"requests": {"cpu": "100m"}}}]},
                    "autoScalingConfig": {"minReplicas": 1, "maxReplicas": 6,
                                          "metrics": [{"type": "Resource", "resource": {
                                              "name": "cpu", "target": {
                                                  "type": "Utilization",
                                                  "averageUtilization": 50}}}]},
                }}]}}}


def _set_usage(cluster, selector, millicores):
    for p in cluster.store.list("Pod", "default", selector):
        cluster.store.patch(
            "Pod", "default", p["metadata"]["name"],
            lambda o: o["metadata"].setdefault("annotations", {}).update(
                {CPU_USAGE_ANNOTATION: f"{millicores}m"}))


class TestHPA:
    def test_scale_out_on_high_utilization(self, cluster):
        cluster.add_virtual_nodes(2)
        cluster.apply(_pcs_with_hpa())
        cluster.wait_pcs_available("h1", timeout=20)
        assert len(cluster.store.list("Pod", "default",
                                      {c.LABEL_PODCLIQUE: "h1-0-web"})) == 2
        # 150m usage vs 100m request = 150% util vs target 50% → scale out
        _set_usage(cluster, {c.LABEL_PODCLIQUE: "h1-0-web"}, 150)

        def scaled():
            q = cluster.store.get(c.KIND_PCLQ, "default", "h1-0-web")
            return int(q["spec"]["replicas"]) == 6  # ratio 3 → 2*3=6, capped at max
        cluster.wait_for(scaled, timeout=15, desc="HPA scale-out")
        cluster.wait_pods_ready({c.LABEL_PODCLIQUE: "h1-0-web"}, 6, timeout=20)

    def test_scale_in_on_low_utilization(self, cluster):
        cluster.add_virtual_nodes(2)
        cluster.apply(_pcs_with_hpa("h2"))
        cluster.wait_pcs_available("h2", timeout=20)
        _set_usage(cluster, {c.LABEL_PODCLIQUE: "h2-0-web"}, 10)  # 10% vs 50% target

        def scaled_in():
            q = cluster.store.get(c.KIND_PCLQ, "default", "h2-0-web")
            return int(q["spec"]["replicas"]) == 1
        cluster.wait_for(scaled_in, timeout=15, desc="HPA scale-in")

    def test_pcsg_scale_target(self, cluster):
        pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
               "metadata": {"name": "h3"},
               "spec": {"replicas": 1, "template": {
                   "cliques": [{"name": "w", "spec": {
                       "roleName": "w", "replicas": 1,
                       "podSpec": {"containers": [{
                           "name": "m", "image": "img",
                           "resources": {"requests": {"cpu": "100m"}}}]}}}],
                   "podCliqueScalingGroups": [{
                       "name": "sg", "cliqueNames": ["w"],
                       "scaleConfig": {"minReplicas": 1, "maxReplicas": 4,
                                       "metrics": [{"type": "Resource", "resource": {
                                           "name": "cpu", "target": {
                                               "type": "Utilization",
                                               "averageUtilization": 50}}}]}}]}}}
        cluster.add_virtual_nodes(2)
        cluster.apply(pcs)
        cluster.wait_pcs_available("h3", timeout=20)
        _set_usage(cluster, {c.LABEL_PCSG: "h3-0-sg"}, 200)

        def scaled():
            g = cluster.store.get(c.KIND_PCSG, "default", "h3-0-sg")
            return int(g["spec"]["replicas"]) == 4
        cluster.wait_for(scaled, timeout=15, desc="PCSG HPA scale-out")
        # scaled gangs follow
        cluster.wait_for(
            lambda: len(cluster.store.list(c.KIND_PODGANG, "default",
                                           {c.LABEL_PART_OF: "h3"})) == 4,
            timeout=15, desc="scaled PodGangs for HPA replicas")


class TestAuthorizer:
    def test_external_user_cannot_mutate_managed_resources(self, cluster, simple1_yaml):
        cluster.add_virtual_nodes(1)
        cluster.apply(simple1_yaml)
        cluster.wait_pcs_available("simple1", timeout=20)
        pod = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "simple1"})[0]
        with as_user("system:serviceaccount:default:intruder"):
            with pytest.raises(ApiError) as ei:
                cluster.store.delete("Pod", "default", pod["metadata"]["name"])
            assert ei.value.reason == "Forbidden"
            with pytest.raises(ApiError):
                cluster.store.patch(
                    c.KIND_PCLQ, "default", "simple1-0-pca",
                    lambda o: o["spec"].update(replicas=99))
        # operator identity still allowed (controllers keep functioning)
        cluster.store.patch("Pod", "default", pod["metadata"]["name"],
                            lambda o: o["metadata"]["labels"].update(x="y"))

    def test_escape_hatch_annotation(self, cluster, simple1_yaml):
        cluster.add_virtual_nodes(1)
        cluster.apply(simple1_yaml)
        cluster.wait_pcs_available("simple1", timeout=20)
        cluster.store.patch(
            c.KIND_PCS, "default", "simple1",
            lambda o: o["metadata"].setdefault("annotations", {}).update(
                {c.ANNOTATION_DISABLE_MANAGED_RESOURCE_PROTECTION: "true"}))
        pod = cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "simple1"})[0]
        with as_user("some-admin"):
            cluster.store.delete("Pod", "default", pod["metadata"]["name"])


class TestResourceSharing:
    def test_pcs_level_claims_injected(self, cluster):
        pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
               "metadata": {"name": "rc1"},
               "spec": {"replicas": 2, "template": {
                   "resourceClaimTemplates": [{
                       "name": "shared-mem",
                       "templateSpec": {"spec": {"devices": {"requests": [{
                           "name": "m",
                           "deviceClassName": "mem.example.com"}]}}}}],
                   "resourceSharing": [
                       {"name": "shared-mem", "scope": "PerReplica"}],
                   "cliques": [{"name": "w", "spec": {
                       "roleName": "w", "replicas": 1,
                       "podSpec": {"containers": [{
                           "name": "m", "image": "img",
                           "resources": {"requests": {"cpu": "1"}}}]}}}]}}}
        cluster.add_virtual_nodes(1)
        cluster.apply(pcs)
        cluster.wait_pcs_available("rc1", timeout=20)
        claims = sorted(x["metadata"]["name"] for x in
                        cluster.store.list("ResourceClaim", "default"))
        assert claims == ["rc1-0-shared-mem", "rc1-1-shared-mem"]
        pod = cluster.store.list("Pod", "default",
                                 {c.LABEL_PODCLIQUE: "rc1-0-w"})[0]
        assert pod["spec"]["resourceClaims"] == [
            {"name": "shared-mem", "resourceClaimName": "rc1-0-shared-mem"}]
        assert pod["spec"]["containers"][0]["resources"]["claims"] == [
            {"name": "shared-mem"}]

    def test_auto_xgmi_domain(self):
        from grove_amd import Cluster
        cl = Cluster(auto_xgmi_domain=True, use_native_scheduler=None).start()
        try:
            cl.add_virtual_nodes(1, gpus=8)
            pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
                   "metadata": {"name": "xg"},
                   "spec": {"replicas": 1, "template": {"cliques": [{
                       "name": "inf", "spec": {
                           "roleName": "inf", "replicas": 2,
                           "podSpec": {"containers": [{
                               "name": "m", "image": "img",
                               "resources": {"requests": {
                                   c.AMD_GPU_RESOURCE: "1"}}}]}}}]}}}
            cl.store.create(pcs)
            cl.wait_pcs_available("xg", timeout=20)
            claim = cl.store.try_get("ResourceClaim", "default", "xg-0-xgmi-default")
            assert claim is not None
            assert claim["spec"]["devices"]["requests"][0]["deviceClassName"] == \
                "xgmi.amd.com"
            pod = cl.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "xg-0-inf"})[0]
            assert pod["spec"]["resourceClaims"][0]["resourceClaimName"] == \
                "xg-0-xgmi-default"
        finally:
            cl.stop()


class TestLastErrors:
    def test_reconcile_error_recorded(self, cluster):
        cluster.add_virtual_nodes(1)
        # inject a failing mutator on Service creation to break the PCS sync
        def bomb(obj, old):
            raise RuntimeError("boom: service quota exceeded")
        cluster.store.register_validator("Service", bomb)
        pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
               "metadata": {"name": "err1"},
               "spec": {"replicas": 1, "template": {"cliques": [{
                   "name": "w", "spec": {"roleName": "w",
                                         "podSpec": {"containers": [{
                                             "name": "m", "image": "img"}]}}}]}}}
        cluster.store.create(pcs)

        def recorded():
            p = cluster.store.get(c.KIND_PCS, "default", "err1")
            errs = (p.get("status") or {}).get("lastErrors") or []
            return any("boom" in e.get("description", "") for e in errs)
        cluster.wait_for(recorded, timeout=15, desc="lastErrors recorded")
        p = cluster.store.get(c.KIND_PCS, "default", "err1")
        assert p["status"]["lastErrors"][0]["code"] == "ERR_RECONCILE"


class TestXGMIGroups:
    def _pcs(self, pcs_ann=None, clique_anns=(None, None), gpu=(True, True)):
        cliques = []
        for i, (ann, has_gpu) in enumerate(zip(clique_anns, gpu)):
            res = {"cpu": "1"}
            if has_gpu:
                res[c.AMD_GPU_RESOURCE] = "1"
            cl = {"name": f"c{i}", "spec": {
                "roleName": f"c{i}", "replicas": 1,
                "podSpec": {"containers": [{"name": "m", "image": "i",
                                            "resources": {"requests": res}}]}}}
            if ann is not None:
                cl["annotations"] = {c.ANNOTATION_XGMI_GROUP: ann}
            cliques.append(cl)
        pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
               "metadata": {"name": "xgrp"},
               "spec": {"replicas": 1, "template": {"cliques": cliques}}}
        if pcs_ann is not None:
            pcs["metadata"]["annotations"] = {c.ANNOTATION_XGMI_GROUP: pcs_ann}
        return pcs

    def test_group_hierarchy_and_optout(self, cluster):
        cluster.add_virtual_nodes(1, gpus=8)
        # PCS-level group "fabric"; clique c1 opts out with "none"
        cluster.store.create(self._pcs(pcs_ann="fabric", clique_anns=(None, "none")))
        cluster.wait_pcs_available("xgrp", timeout=20)
        claims = [x["metadata"]["name"]
                  for x in cluster.store.list("ResourceClaim", "default")]
        assert claims == ["xgrp-0-xgmi-fabric"]
        p0 = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "xgrp-0-c0"})[0]
        assert p0["spec"]["resourceClaims"][0]["resourceClaimName"] == \
            "xgrp-0-xgmi-fabric"
        p1 = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "xgrp-0-c1"})[0]
        assert "resourceClaims" not in p1["spec"]

    def test_non_gpu_clique_silently_skips_inherited(self, cluster):
        cluster.add_virtual_nodes(1, gpus=8)
        cluster.store.create(self._pcs(pcs_ann="fabric", gpu=(True, False)))
        cluster.wait_pcs_available("xgrp", timeout=20)
        p1 = cluster.store.list("Pod", "default", {c.LABEL_PODCLIQUE: "xgrp-0-c1"})[0]
        assert "resourceClaims" not in p1["spec"]

    def test_explicit_group_on_non_gpu_rejected(self, cluster):
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError):
            cluster.store.create(self._pcs(clique_anns=("g1", None),
                                           gpu=(False, True)))

    def test_invalid_group_name_rejected(self, cluster):
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError):
            cluster.store.create(self._pcs(pcs_ann="Bad_Name!"))

    def test_group_annotation_immutable(self, cluster):
        cluster.add_virtual_nodes(1, gpus=8)
        cluster.store.create(self._pcs(pcs_ann="fabric"))
        cluster.wait_pcs_available("xgrp", timeout=20)
        from grove_amd.kubecore.store import ApiError
        with pytest.raises(ApiError):
            cluster.store.patch(
                c.KIND_PCS, "default", "xgrp",
                lambda o: o["metadata"]["annotations"].update(
                    {c.ANNOTATION_XGMI_GROUP: "other"}))

    def test_pcsg_members_share_group_claim(self, cluster):
        pcs = {"apiVersion": c.API_VERSION, "kind": c.KIND_PCS,
               "metadata": {"name": "xsg",
                            "annotations": {c.ANNOTATION_XGMI_GROUP: "fab"}},
               "spec": {"replicas": 1, "template": {
                   "cliques": [{"name": "w", "spec": {
                       "roleName": "w", "replicas": 1,
                       "podSpec": {"containers": [{
                           "name": "m", "image": "i",
                           "resources": {"requests": {
                               c.AMD_GPU_RESOURCE: "1"}}}]}}}],
                   "podCliqueScalingGroups": [{"name": "sg", "cliqueNames": ["w"],
                                               "replicas": 2, "minAvailable": 1}]}}}
        cluster.add_virtual_nodes(1, gpus=8)
        cluster.store.create(pcs)
        cluster.wait_pcs_available("xsg", timeout=20)
        cluster.wait_pods_ready({c.LABEL_PART_OF: "xsg"}, 2, timeout=20)
        for p in cluster.store.list("Pod", "default", {c.LABEL_PART_OF: "xsg"}):
            refs = p["spec"].get("resourceClaims") or []
            assert any(r["resourceClaimName"] == "xsg-0-xgmi-fab" for r in refs), \
                p["metadata"]["name"]
