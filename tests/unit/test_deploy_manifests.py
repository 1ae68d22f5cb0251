"""Deploy manifest sanity (VERDICT r1 missing #4: vendored Argo install).

The Argo install (deploy/deploy-argo.yaml) must provide the pieces
active-monitor depends on — all 8 Argo CRDs, the workflow-controller with
the ``instanceID: activemonitor-workflows`` configmap and the 1800s TTL
(reference deploy/deploy-argo.yaml:1162-1238) — and the e2e script must
reference objects that actually exist in the manifests it applies.
"""
import os
import re

import yaml

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

ARGO_CRDS = {
    "workflows.argoproj.io",
    "workflowtemplates.argoproj.io",
    "clusterworkflowtemplates.argoproj.io",
    "cronworkflows.argoproj.io",
    "workfloweventbindings.argoproj.io",
    "workflowtasksets.argoproj.io",
    "workflowtaskresults.argoproj.io",
    "workflowartifactgctasks.argoproj.io",
}


def _load(path):
    with open(os.path.join(REPO, path)) as f:
        return [d for d in yaml.safe_load_all(f) if d]


def test_argo_install_provides_all_crds_and_instance_id():
    docs = _load("deploy/deploy-argo.yaml")
    crds = {
        d["metadata"]["name"]
        for d in docs
        if d["kind"] == "CustomResourceDefinition"
    }
    assert crds == ARGO_CRDS

    cm = next(
        d for d in docs
        if d["kind"] == "ConfigMap"
        and d["metadata"]["name"] == "workflow-controller-configmap"
    )
    cfg = yaml.safe_load(cm["data"]["config"])
    # the label the controller stamps on every submitted workflow
    # (engine/parse.py WF_INSTANCE_ID) must match the controller scoping
    assert cfg["instanceID"] == "activemonitor-workflows"
    assert cfg["ttlStrategy"]["secondsAfterCompletion"] == 1800

    deploy = next(d for d in docs if d["kind"] == "Deployment")
    assert deploy["metadata"]["name"] == "workflow-controller"
    assert deploy["metadata"]["namespace"] == "health"
    container = deploy["spec"]["template"]["spec"]["containers"][0]
    assert "--configmap" in container["args"]
    ns = next(d for d in docs if d["kind"] == "Namespace")
    assert ns["metadata"]["name"] == "health"


def test_instance_id_matches_controller_constant():
    from active_monitor_amd.engine.parse import WF_INSTANCE_ID

    docs = _load("deploy/deploy-argo.yaml")
    cm = next(
        d for d in docs
        if d["kind"] == "ConfigMap"
        and d["metadata"]["name"] == "workflow-controller-configmap"
    )
    cfg = yaml.safe_load(cm["data"]["config"])
    assert cfg["instanceID"] == WF_INSTANCE_ID


def test_e2e_script_references_real_objects():
    with open(os.path.join(REPO, "hack", "e2e-kind.sh")) as f:
        script = f.read()
    assert os.access(os.path.join(REPO, "hack", "e2e-kind.sh"), os.X_OK)

    # every file the script applies exists
    for m in re.finditer(r"(?:apply|create) (?:cluster )?(?:--\S+ )*-f (\S+)", script):
        path = m.group(1)
        assert os.path.exists(os.path.join(REPO, path)), path

    # deployment rollouts the script waits on exist in the applied manifests
    deploy_names = set()
    for path in ("deploy/deploy-argo.yaml", "deploy/deploy-active-monitor.yaml"):
        for d in _load(path):
            if d.get("kind") == "Deployment":
                deploy_names.add(d["metadata"]["name"])
    for m in re.finditer(r"rollout status deploy/(\S+)", script):
        assert m.group(1) in deploy_names, m.group(1)

    # the CR it waits on is the example it applied
    example = _load("examples/inline-hello.yaml")[0]
    assert example["metadata"]["name"] == "inline-hello"
    assert "get hc inline-hello" in script


def test_dockerfile_ships_native_extension():
    """the image must build and require the native
    extension, not silently ship the pure-Python fallback."""
    with open(os.path.join(REPO, "Dockerfile")) as f:
        df = f.read()
    assert "COPY native ./native" in df
    assert "setup.py" in df
    assert "AM_REQUIRE_NATIVE=1" in df
    assert "import active_monitor_amd._amcore" in df
