"""Thread-safety stress tests for the in-memory apiserver (the Python
equivalent of the reference's -race coverage — SURVEY.md §5 notes the repo's
documented custom-metrics race history)."""
import threading

from active_monitor_amd.kube import (
    AlreadyExistsError,
    ConflictError,
    MemoryApiServer,
    NotFoundError,
)


def test_concurrent_create_update_delete_consistent():
    s = MemoryApiServer()
    errs = []
    N_THREADS, N_OPS = 8, 200

    def worker(tid):
        try:
            for i in range(N_OPS):
                name = f"obj-{tid}-{i % 20}"
                obj = {
                    "apiVersion": "activemonitor.keikoproj.io/v1alpha1",
                    "kind": "HealthCheck",
                    "metadata": {"name": name, "namespace": "health"},
                    "spec": {"repeatAfterSec": i, "workflow": {}},
                }
                try:
                    s.create(obj)
                except AlreadyExistsError:
                    pass
                try:
                    got = s.get("activemonitor.keikoproj.io/v1alpha1",
                                "HealthCheck", "health", name)
                    got["status"] = {"successCount": i}
                    s.update_status(got)
                except (NotFoundError, ConflictError):
                    pass
                if i % 7 == 0:
                    try:
                        s.delete("activemonitor.keikoproj.io/v1alpha1",
                                 "HealthCheck", "health", name)
                    except NotFoundError:
                        pass
        except Exception as e:  # pragma: no cover
            errs.append(e)

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(N_THREADS)]
    [t.start() for t in threads]
    [t.join() for t in threads]
    assert errs == []
    # store internally consistent: every remaining object has a unique rv and
    # the owner index only references live keys
    objs = s.list("activemonitor.keikoproj.io/v1alpha1", "HealthCheck", "health")
    rvs = [o["metadata"]["resourceVersion"] for o in objs]
    assert len(rvs) == len(set(rvs))
    for uid, keys in s._by_owner.items():
        for key in keys:
            assert key in s._objects


def test_concurrent_generate_name_unique():
    s = MemoryApiServer()
    names, lock = [], threading.Lock()

    def worker():
        for _ in range(100):
            created = s.create({
                "apiVersion": "argoproj.io/v1alpha1", "kind": "Workflow",
                "metadata": {"generateName": "wf-", "namespace": "health"},
                "spec": {},
            })
            with lock:
                names.append(created["metadata"]["name"])

    threads = [threading.Thread(target=worker) for _ in range(6)]
    [t.start() for t in threads]
    [t.join() for t in threads]
    assert len(names) == 600
    assert len(set(names)) == 600
