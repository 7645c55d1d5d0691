import os
import time


def test_session_lifecycle(store):
    sid = store.create_session(name="svc", project="proj", tags=["a"])
    assert store.get_session(sid)["name"] == "svc"
    assert store.resolve_session(None) == sid
    assert store.revision(sid) == 0
    store.set_config_object(sid, "endpoints", {"a": 1})
    assert store.revision(sid) == 1
    assert store.get_config_object(sid, "endpoints") == {"a": 1}
    assert store.get_config_object(sid, "missing", {"d": 1}) == {"d": 1}


def test_params(store):
    sid = store.create_session()
    store.set_params(sid, {"serving_base_url": "http://x", "freq": 0.5})
    assert store.get_params(sid) == {"serving_base_url": "http://x", "freq": 0.5}


def test_model_registry_query(store):
    m1 = store.register_model(name="alpha", project="p1", tags=["t1"])
    time.sleep(0.01)
    m2 = store.register_model(name="alpha-v2", project="p1", tags=["t1", "t2"],
                              published=True)
    time.sleep(0.01)
    m3 = store.register_model(name="beta", project="p2")

    # newest first
    ids = [m.model_id for m in store.query_models()]
    assert ids == [m3.model_id, m2.model_id, m1.model_id]
    assert [m.model_id for m in store.query_models(project="p1")] == \
        [m2.model_id, m1.model_id]
    # name is a regex
    assert len(store.query_models(name="^alpha")) == 2
    assert len(store.query_models(name="^alpha$")) == 1
    assert [m.model_id for m in store.query_models(tags=["t2"])] == [m2.model_id]
    assert [m.model_id for m in store.query_models(only_published=True)] == \
        [m2.model_id]


def test_model_file_copy(store, tmp_path):
    f = tmp_path / "model.bin"
    f.write_bytes(b"weights")
    rec = store.register_model(name="m", path=str(f))
    local = store.get_model_local_path(rec.model_id)
    assert os.path.isfile(local)
    assert open(local, "rb").read() == b"weights"


def test_artifact_store(store, tmp_path):
    sid = store.create_session()
    f = tmp_path / "preprocess.py"
    f.write_text("class Preprocess:\n    pass\n")
    digest = store.upload_artifact(sid, "py_code_x", str(f))
    art = store.get_artifact(sid, "py_code_x")
    assert art["sha256"] == digest
    assert os.path.isfile(art["path"])


def test_cross_instance_visibility(tmp_path):
    from clearml_serving_amd.store import ServingStore

    s1 = ServingStore(str(tmp_path / "shared"))
    s2 = ServingStore(str(tmp_path / "shared"))
    sid = s1.create_session(name="visible")
    assert s2.get_session(sid)["name"] == "visible"
    s1.set_config_object(sid, "endpoints", {"x": 1})
    assert s2.revision(sid) == 1
    assert s2.get_config_object(sid, "endpoints") == {"x": 1}


def test_concurrent_writers(tmp_path):
    """Two processes' worth of writers against one store (WAL): no lost
    updates, monotonic revision."""
    import threading

    from clearml_serving_amd.store import ServingStore

    s1 = ServingStore(str(tmp_path / "c"))
    s2 = ServingStore(str(tmp_path / "c"))
    sid = s1.create_session(name="c")

    def writer(store, key):
        for i in range(20):
            store.set_config_object(sid, key, {"i": i})

    t1 = threading.Thread(target=writer, args=(s1, "a"))
    t2 = threading.Thread(target=writer, args=(s2, "b"))
    t1.start(); t2.start(); t1.join(); t2.join()
    assert s1.get_config_object(sid, "a") == {"i": 19}
    assert s1.get_config_object(sid, "b") == {"i": 19}
    assert s1.revision(sid) == 40


def test_instance_keepalive_no_revision_bump(tmp_path):
    """Serving-container pings register liveness WITHOUT bumping the
    session revision (a ping must not look like a config change)."""
    import time as _t

    from clearml_serving_amd.store import ServingStore

    s = ServingStore(str(tmp_path / "k"))
    sid = s.create_session(name="k")
    rev = s.revision(sid)
    s.ping_instance(sid, "hostA:1", {"pid": 1})
    s.ping_instance(sid, "hostB:2", {"pid": 2})
    assert s.revision(sid) == rev
    inst = s.list_instances(sid)
    assert {i["instance_id"] for i in inst} == {"hostA:1", "hostB:2"}
    # re-ping updates, not duplicates
    s.ping_instance(sid, "hostA:1", {"pid": 1})
    assert len(s.list_instances(sid)) == 2
    # stale filtering
    old = _t.time() - 10_000
    s._conn.execute("UPDATE instances SET last_ping=? WHERE instance_id=?",
                    (old, "hostB:2"))
    s._conn.commit()
    assert [i["instance_id"] for i in s.list_instances(sid, max_age_sec=600)] \
        == ["hostA:1"]
