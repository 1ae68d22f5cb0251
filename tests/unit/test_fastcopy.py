"""fastcopy tests: Python and native implementations must agree exactly."""
import pytest

from active_monitor_amd.utils import fastcopy

IMPLS = [("python", fastcopy._py_deep_copy, fastcopy._py_snapshot)]
if fastcopy.NATIVE:
    from active_monitor_amd import _amcore

    IMPLS.append(("native", _amcore.deep_copy, _amcore.snapshot))

OBJ = {
    "apiVersion": "activemonitor.keikoproj.io/v1alpha1",
    "kind": "HealthCheck",
    "metadata": {
        "name": "x", "namespace": "health", "labels": {"a": "b"},
        "ownerReferences": [{"uid": "u1", "controller": True}],
    },
    "spec": {
        "repeatAfterSec": 60,
        "workflow": {"resource": {"source": {"inline": "spec: {}\n"}}},
        "nested": [1, 2.5, True, None, "s", [{"deep": ("t", 1)}]],
    },
    "status": {"successCount": 3, "startedAt": "2026-01-01T00:00:00Z"},
}


@pytest.mark.parametrize("name,dc,snap", IMPLS, ids=[i[0] for i in IMPLS])
def test_deep_copy_private_and_equal(name, dc, snap):
    c = dc(OBJ)
    assert c == OBJ
    assert c is not OBJ
    assert c["spec"] is not OBJ["spec"]
    assert c["spec"]["workflow"] is not OBJ["spec"]["workflow"]
    assert c["metadata"]["ownerReferences"] is not OBJ["metadata"]["ownerReferences"]
    c["spec"]["workflow"]["x"] = 1
    assert "x" not in OBJ["spec"]["workflow"]


@pytest.mark.parametrize("name,dc,snap", IMPLS, ids=[i[0] for i in IMPLS])
def test_snapshot_shares_spec_copies_meta_status(name, dc, snap):
    s = snap(OBJ)
    assert s == OBJ
    assert s["spec"] is OBJ["spec"]          # shared
    assert s["metadata"] is not OBJ["metadata"]  # private
    assert s["status"] is not OBJ["status"]      # private
    s["metadata"]["name"] = "changed"
    assert OBJ["metadata"]["name"] == "x"


@pytest.mark.parametrize("name,dc,snap", IMPLS, ids=[i[0] for i in IMPLS])
def test_scalars_and_leaves(name, dc, snap):
    for v in ("s", 1, 2.5, True, None, b"b"):
        assert dc(v) == v
    assert snap([1, 2]) == [1, 2]
    assert snap("x") == "x"


@pytest.mark.parametrize("name,dc,snap", IMPLS, ids=[i[0] for i in IMPLS])
def test_non_json_payload_falls_back_to_deepcopy(name, dc, snap):
    class Custom:
        def __init__(self, v):
            self.v = v

    obj = {"spec": Custom(7)}
    c = dc(obj)
    assert c["spec"] is not obj["spec"]
    assert c["spec"].v == 7


def test_native_is_built_in_tree():
    """The repo policy: the native extension builds and loads (GPU boxes set
    AM_REQUIRE_NATIVE=1 to make a silent fallback impossible)."""
    assert fastcopy.NATIVE, "run `python setup.py build_ext --inplace`"
