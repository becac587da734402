"""MemKube thread-safety: the embedded platform drives controllers, the
HTTP apiserver and the local runtime against one MemKube from separate
threads — CRUD + watches must stay consistent under contention."""

import threading

import pytest

from substratus_amd.api import types as apiv1
from substratus_amd.kube.memapi import Conflict, MemKube, NotFound


@pytest.mark.timeout(120)
def test_concurrent_crud_consistent():
    kube = MemKube()
    events = []
    kube.watch(None, lambda ev, obj: events.append(ev))
    errs = []

    def worker(tid: int) -> None:
        try:
            for i in range(50):
                name = f"m-{tid}-{i}"
                kube.create(apiv1.new_object("Model", name,
                                             {"image": "x"}))
                kube.patch("Model", name,
                           {"status": {"ready": True}})
                obj = kube.get("Model", name)
                assert obj["status"]["ready"] is True
                kube.delete("Model", name)
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errs, errs[:3]
    assert kube.list("Model") == []
    # every create/patch/delete produced exactly one event
    assert len(events) == 8 * 50 * 3


@pytest.mark.timeout(60)
def test_create_conflict_and_rv_monotonic():
    kube = MemKube()
    kube.create(apiv1.new_object("Model", "dup", {"image": "x"}))
    with pytest.raises(Conflict):
        kube.create(apiv1.new_object("Model", "dup", {"image": "y"}))
    rv1 = int(kube.get("Model", "dup")["metadata"]["resourceVersion"])
    kube.patch("Model", "dup", {"spec": {"image": "z"}})
    rv2 = int(kube.get("Model", "dup")["metadata"]["resourceVersion"])
    assert rv2 > rv1
    gen = kube.get("Model", "dup")["metadata"]["generation"]
    assert gen == 2  # spec change bumps generation
    # status-only change does NOT bump generation
    kube.patch("Model", "dup", {"status": {"ready": True}})
    assert kube.get("Model", "dup")["metadata"]["generation"] == 2
    with pytest.raises(NotFound):
        kube.get("Model", "missing")
