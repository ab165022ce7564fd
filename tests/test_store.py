"""kubecore store semantics: rv/generation, finalizers, GC, watches, selectors."""
import threading

import pytest

from grove_amd.kubecore.store import ApiError, Store, owner_reference


def mk(kind, name, ns="default", labels=None, spec=None):
    return {"apiVersion": "v1", "kind": kind,
            "metadata": {"name": name, "namespace": ns, "labels": labels or {}},
            "spec": spec or {}}


def test_create_get_update_generation():
    s = Store()
    o = s.create(mk("Pod", "p1", spec={"a": 1}))
    assert o["metadata"]["uid"] and o["metadata"]["generation"] == 1
    rv = o["metadata"]["resourceVersion"]
    o["spec"]["a"] = 2
    o2 = s.update(o)
    assert o2["metadata"]["generation"] == 2
    assert o2["metadata"]["resourceVersion"] != rv
    # status update does not bump generation
    o2["status"] = {"ok": True}
    o3 = s.update_status(o2)
    assert o3["metadata"]["generation"] == 2
    assert o3["status"] == {"ok": True}


def test_conflict_on_stale_rv():
    s = Store()
    o = s.create(mk("Pod", "p1"))
    stale = dict(o, spec={"x": 1})
    s.update(o)  # bumps rv
    with pytest.raises(ApiError) as ei:
        s.update(stale)
    assert ei.value.reason == "Conflict"


def test_generate_name():
    s = Store()
    o = s.create({"kind": "Pod", "metadata": {"generateName": "web-"}, "spec": {}})
    assert o["metadata"]["name"].startswith("web-")
    assert len(o["metadata"]["name"]) > 4


def test_label_selector_list():
    s = Store()
    s.create(mk("Pod", "a", labels={"app": "x", "tier": "1"}))
    s.create(mk("Pod", "b", labels={"app": "x"}))
    s.create(mk("Pod", "c", labels={"app": "y"}))
    assert len(s.list("Pod", "default", {"app": "x"})) == 2
    assert len(s.list("Pod", "default", {"app": "x", "tier": "1"})) == 1
    assert len(s.list("Pod")) == 3


def test_finalizer_two_phase_delete():
    s = Store()
    o = mk("Pod", "p1")
    o["metadata"]["finalizers"] = ["keep.io/me"]
    s.create(o)
    s.delete("Pod", "default", "p1")
    cur = s.get("Pod", "default", "p1")
    assert cur["metadata"]["deletionTimestamp"]
    # removing the finalizer completes the delete
    cur["metadata"]["finalizers"] = []
    s.update(cur)
    assert s.try_get("Pod", "default", "p1") is None


def test_owner_cascade_gc():
    s = Store()
    parent = s.create(mk("PodClique", "par"))
    child = mk("Pod", "ch")
    child["metadata"]["ownerReferences"] = [owner_reference(parent)]
    s.create(child)
    s.delete("PodClique", "default", "par")
    assert s.try_get("Pod", "default", "ch") is None


def test_watch_events():
    s = Store()
    w = s.watch("Pod")
    s.create(mk("Pod", "p1"))
    ev, obj = w.queue.get(timeout=1)
    assert ev == "ADDED" and obj["metadata"]["name"] == "p1"
    s.patch("Pod", "default", "p1", lambda o: o["spec"].update(x=1))
    ev, obj = w.queue.get(timeout=1)
    assert ev == "MODIFIED"
    s.delete("Pod", "default", "p1")
    ev, obj = w.queue.get(timeout=1)
    assert ev == "DELETED"
    w.stop()


def test_patch_retry_loop():
    s = Store()
    s.create(mk("Pod", "p1", spec={"n": 0}))

    def bump(o):
        o["spec"]["n"] += 1

    threads = [threading.Thread(target=lambda: s.patch("Pod", "default", "p1", bump))
               for _ in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert s.get("Pod", "default", "p1")["spec"]["n"] == 8


def test_delete_collection_with_selector():
    s = Store()
    for i in range(4):
        s.create(mk("PodClique", f"q{i}", labels={"grp": "a" if i < 2 else "b"}))
    n = s.delete_collection("PodClique", "default", {"grp": "a"})
    assert n == 2
    assert len(s.list("PodClique")) == 2


def test_cluster_scoped_kind():
    s = Store()
    s.create({"kind": "Node", "metadata": {"name": "n0"}, "spec": {}})
    assert s.get("Node", None, "n0")["metadata"].get("namespace") in (None, "")


def test_clientset_typed_surface():
    """Generated-client analog: the typed Clientset drives both the
    in-process Store and the HTTP client with identical ergonomics."""
    from grove_amd.kubecore.clientset import for_store, for_server
    from grove_amd.kubecore.store import Store
    from grove_amd.kubecore.apiserver import ApiServer
    st = Store()
    cs = for_store(st)
    cs.podcliquesets("default").create({
        "apiVersion": "grove.io/v1alpha1",
        "metadata": {"name": "cs1"},
        "spec": {"replicas": 1, "template": {"cliques": [{
            "name": "w", "spec": {"roleName": "w", "replicas": 1,
                                  "podSpec": {"containers": [
                                      {"name": "m", "image": "i"}]}}}]}}})
    assert cs.podcliquesets("default").get("cs1")["spec"]["replicas"] == 1
    cs.podcliquesets("default").patch("cs1",
                                      lambda o: o["spec"].update(replicas=2))
    assert cs.podcliquesets().get("cs1")["spec"]["replicas"] == 2
    cs.nodes().create({"apiVersion": "v1", "metadata": {"name": "n0"},
                       "spec": {}, "status": {}})
    assert len(cs.nodes().list()) == 1
    # watch resume through the typed surface
    rv = cs.podcliquesets().get("cs1")["metadata"]["resourceVersion"]
    cs.podcliquesets().patch("cs1", lambda o: o["metadata"].setdefault(
        "labels", {}).update(x="1"))
    events = []
    for ev, obj in cs.podcliquesets().watch(resource_version=rv):
        events.append((ev, obj["metadata"]["name"]))
        break
    assert events == [("MODIFIED", "cs1")]
    # same surface over HTTP
    api = ApiServer(st, port=18633).start()
    try:
        hcs = for_server(api.url)
        assert hcs.podcliquesets("default").get("cs1")["spec"]["replicas"] == 2
        items, cont, rv2 = hcs.podcliquesets("default").list_page(limit=10)
        assert len(items) == 1 and rv2 is not None
    finally:
        api.stop()
    cs.podcliquesets().delete("cs1")
    assert cs.podcliquesets().try_get("cs1") is None


def test_store_race_stress_switchinterval():
    """-race analog: hammer the store from many threads with the GIL
    switch interval forced to 1 µs (maximal preemption) and assert invariants —
    monotonic resourceVersions, no lost updates on optimistic patches, label index
    consistency."""
    import sys
    import threading
    from grove_amd.kubecore.store import Store
    st = Store()
    st.create({"kind": "Pod", "metadata": {"name": "rc", "labels": {"k": "0"}},
               "spec": {"counter": 0}})
    old_interval = sys.getswitchinterval()
    sys.setswitchinterval(1e-6)
    errs = []

    def bump(n):
        try:
            for _ in range(n):
                st.patch("Pod", "default", "rc",
                         lambda o: o["spec"].update(
                             counter=o["spec"]["counter"] + 1),
                         retries=10000)
        except Exception as e:  # pragma: no cover
            errs.append(e)

    def churn_labels(n):
        try:
            for i in range(n):
                st.patch("Pod", "default", "rc",
                         lambda o: o["metadata"]["labels"].update(k=str(i)),
                         retries=10000)
        except Exception as e:  # pragma: no cover
            errs.append(e)
    try:
        ts = [threading.Thread(target=bump, args=(200,)) for _ in range(4)] + \
             [threading.Thread(target=churn_labels, args=(100,))]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
    finally:
        sys.setswitchinterval(old_interval)
    assert not errs, errs[:3]
    cur = st.get("Pod", "default", "rc")
    assert cur["spec"]["counter"] == 800, "lost update under contention"
    # label index still consistent with the final object
    v = cur["metadata"]["labels"]["k"]
    assert any(p["metadata"]["name"] == "rc"
               for p in st.list("Pod", "default", {"k": v}))
