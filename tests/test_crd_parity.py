"""CRD surface parity: every field in the reference CRD schemas must be implemented
(declared in grove_amd/api/schema.py), passed through (embedded upstream types), or
listed in KNOWN_GAPS with a reason. Runs only where /root/reference is mounted."""
import functools
import glob

import pytest
import yaml

from grove_amd.api.schema import SCHEMAS, KNOWN_GAPS, declared_paths

pytestmark = pytest.mark.reference

CRD_GLOBS = [
    "/root/reference/operator/api/core/v1alpha1/crds/*.yaml",
    "/root/reference/scheduler/api/core/v1alpha1/crds/*.yaml",
]


def reference_paths(crd) -> set:
    out = set()

    def walk(schema, prefix, depth=0):
        if depth > 14:
            return
        for k, v in (schema.get("properties") or {}).items():
            path = f"{prefix}.{k}" if prefix else k
            out.add(path)
            if v.get("type") == "array" and "items" in v:
                walk(v["items"], path, depth + 1)
            else:
                walk(v, path, depth + 1)

    ver = crd["spec"]["versions"][0]
    walk(ver["schema"]["openAPIV3Schema"], "")
    return out


@functools.lru_cache(maxsize=1)
def _crds():
    files = []
    for g in CRD_GLOBS:
        files.extend(glob.glob(g))
    out = {}
    for f in files:
        crd = yaml.safe_load(open(f))
        out[crd["metadata"]["name"]] = crd
    return out


def test_all_reference_crds_covered():
    crds = _crds()
    assert set(crds) == set(SCHEMAS), "CRD inventory mismatch"


@pytest.mark.parametrize("crd_name", sorted(SCHEMAS))
def test_crd_fields_implemented(crd_name):
    crds = _crds()
    ref = reference_paths(crds[crd_name])
    ours = declared_paths(crd_name)
    passthrough_roots = [p[:-2] for p in ours if p.endswith(".*")]
    gaps = set(KNOWN_GAPS.get(crd_name, []))
    missing = []
    for path in sorted(ref):
        if path in ours or path in gaps:
            continue
        if any(path == root or path.startswith(root + ".")
               for root in passthrough_roots):
            continue
        missing.append(path)
    assert not missing, (
        f"{crd_name}: {len(missing)} reference fields not implemented/declared:\n  "
        + "\n  ".join(missing[:40]))


@pytest.mark.parametrize("crd_name", sorted(SCHEMAS))
def test_crd_versions_and_scope(crd_name):
    crds = _crds()
    crd = crds[crd_name]
    ver = crd["spec"]["versions"][0]
    assert ver["name"] == "v1alpha1"
    scope = crd["spec"]["scope"]
    if crd_name == "clustertopologybindings.grove.io":
        assert scope == "Cluster"
    else:
        assert scope == "Namespaced"


def test_committed_crds_match_renderer():
    """crds/*.yaml are generated artifacts — they must match api/crds.py exactly."""
    import os
    from grove_amd.api.crds import render_all
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for crd in render_all():
        path = os.path.join(repo, "crds", f"{crd['metadata']['name']}.yaml")
        assert os.path.exists(path), f"missing {path} (run install-crds --output-dir crds)"
        on_disk = yaml.safe_load(open(path))
        assert on_disk == crd, f"{path} drifted — regenerate with install-crds"


# ---------------------------------------------------------------- full-schema parity
def _strip_descriptions(node, is_prop_map=False):
    """Remove schema-metadata description strings without touching fields literally
    named "description" (keys of a `properties` map are field names)."""
    if isinstance(node, dict):
        if is_prop_map:
            return {k: _strip_descriptions(v) for k, v in node.items()}
        out = {}
        for k, v in node.items():
            if k == "description" and isinstance(v, str):
                continue
            out[k] = _strip_descriptions(v, is_prop_map=(k == "properties"))
        return out
    if isinstance(node, list):
        return [_strip_descriptions(x) for x in node]
    return node


def _deep_diff(a, b, path=""):
    out = []
    if type(a) != type(b):
        return [f"{path}: type {type(a).__name__} != {type(b).__name__}"]
    if isinstance(a, dict):
        for k in sorted(set(a) | set(b)):
            if k not in a:
                out.append(f"{path}.{k}: only in ours")
            elif k not in b:
                out.append(f"{path}.{k}: missing in ours")
            else:
                out.extend(_deep_diff(a[k], b[k], f"{path}.{k}"))
    elif isinstance(a, list):
        if len(a) != len(b):
            return [f"{path}: list len {len(a)} != {len(b)}"]
        for i, (x, y) in enumerate(zip(a, b)):
            out.extend(_deep_diff(x, y, f"{path}[{i}]"))
    elif a != b:
        out.append(f"{path}: {a!r} != {b!r}")
    return out


@pytest.mark.parametrize("crd_name", sorted(SCHEMAS))
def test_crd_schema_byte_compatible(crd_name):
    """The rendered CRD is semantically identical to the reference's generated YAML:
    full structural openAPIV3Schema (types/enums/bounds/CEL/list-type markers),
    printer columns, subresources, names, scope — only descriptions and the
    controller-gen version annotation are ignored."""
    from grove_amd.api.crds import render_crd
    ref = _crds()[crd_name]
    ours = render_crd(crd_name)
    ref_v = ref["spec"]["versions"][0]
    our_v = ours["spec"]["versions"][0]
    diffs = []
    diffs += _deep_diff(ours["spec"]["names"], ref["spec"]["names"], "names")
    if ours["spec"]["scope"] != ref["spec"]["scope"]:
        diffs.append("scope differs")
    if ours["spec"]["group"] != ref["spec"]["group"]:
        diffs.append("group differs")
    for key in ("name", "served", "storage"):
        if our_v.get(key) != ref_v.get(key):
            diffs.append(f"versions[0].{key} differs")
    diffs += _deep_diff(our_v.get("additionalPrinterColumns"),
                        ref_v.get("additionalPrinterColumns"), "printerColumns")
    diffs += _deep_diff(our_v.get("subresources"), ref_v.get("subresources"),
                        "subresources")
    diffs += _deep_diff(
        _strip_descriptions(our_v["schema"]["openAPIV3Schema"]),
        _strip_descriptions(ref_v["schema"]["openAPIV3Schema"]), "schema")
    assert not diffs, f"{crd_name}: {len(diffs)} diffs:\n  " + "\n  ".join(diffs[:40])


# ---------------------------------------------------------------- schema admission
def test_schema_admission_rejects_type_violations():
    """The store enforces the SAME structural schema the CRDs publish: wrong types,
    bad enums, pattern violations and missing required fields are rejected
    server-side (wire the schema into admission)."""
    from grove_amd import Cluster
    from grove_amd.kubecore.store import ApiError

    def pcs(**spec_over):
        spec = {"replicas": 1, "template": {"cliques": [{
            "name": "a", "spec": {"roleName": "a", "replicas": 1,
                                  "podSpec": {"containers": [
                                      {"name": "m", "image": "i"}]}}}]}}
        spec.update(spec_over)
        return {"apiVersion": "grove.io/v1alpha1", "kind": "PodCliqueSet",
                "metadata": {"name": "sv"}, "spec": spec}

    cl = Cluster()
    try:
        # wrong scalar type
        bad = pcs(replicas="three")
        with pytest.raises(ApiError) as e:
            cl.store.create(bad)
        assert "spec.replicas" in str(e.value.message)
        # bad enum
        bad = pcs()
        bad["spec"]["template"]["cliqueStartupType"] = "Sideways"
        with pytest.raises(ApiError) as e:
            cl.store.create(bad)
        assert "cliqueStartupType" in str(e.value.message)
        # missing required field inside clique spec
        bad = pcs()
        del bad["spec"]["template"]["cliques"][0]["spec"]["roleName"]
        with pytest.raises(ApiError) as e:
            cl.store.create(bad)
        assert "roleName" in str(e.value.message)
        # pattern violation on topology domain
        bad = pcs()
        bad["spec"]["template"]["topologyConstraint"] = {
            "pack": {"required": "NOT-A-DOMAIN!"}}
        with pytest.raises(ApiError) as e:
            cl.store.create(bad)
        assert "pack.required" in str(e.value.message)
        # int32 overflow
        bad = pcs(replicas=2**40)
        with pytest.raises(ApiError) as e:
            cl.store.create(bad)
        assert "range" in str(e.value.message)
        # array-of-strings type violation
        bad = pcs()
        bad["spec"]["template"]["cliques"][0]["spec"]["startsAfter"] = [3]
        with pytest.raises(ApiError):
            cl.store.create(bad)
        # valid object passes and gets schema defaults applied
        good = cl.store.create(pcs())
        assert good["spec"]["template"]["cliqueStartupType"] == \
            "CliqueStartupTypeAnyOrder"
        assert good["spec"]["template"]["headlessServiceConfig"][
            "publishNotReadyAddresses"] is True
    finally:
        cl.stop()


def test_schema_admission_podgang_and_ctb():
    from grove_amd import Cluster
    from grove_amd.kubecore.store import ApiError
    cl = Cluster()
    try:
        with pytest.raises(ApiError) as e:
            cl.store.create({"apiVersion": "scheduler.grove.io/v1alpha1",
                             "kind": "PodGang", "metadata": {"name": "g"},
                             "spec": {"podgroups": [{"name": "x"}]}})
        msg = str(e.value.message)
        assert "minReplicas" in msg and "podReferences" in msg
        with pytest.raises(ApiError) as e:
            cl.store.create({"apiVersion": "grove.io/v1alpha1",
                             "kind": "ClusterTopologyBinding",
                             "metadata": {"name": "t"},
                             "spec": {"levels": []}})
        assert "at least 1 items" in str(e.value.message)
    finally:
        cl.stop()
