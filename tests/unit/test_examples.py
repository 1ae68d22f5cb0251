"""Every shipped example must parse into a valid HealthCheck with a loadable
workflow definition (the reference's examples/ are its canonical CRs)."""
import pathlib

import pytest
import yaml

from active_monitor_amd.api import HealthCheck
from active_monitor_amd.engine.parse import (
    parse_remedy_workflow_from_healthcheck,
    parse_workflow_from_healthcheck,
)

EXAMPLES = sorted(
    p for p in (pathlib.Path(__file__).parents[2] / "examples").rglob("*.yaml")
)


def test_examples_exist():
    assert len(EXAMPLES) >= 15


@pytest.mark.parametrize("path", EXAMPLES, ids=lambda p: str(p.relative_to(p.parents[1])))
def test_example_parses(path):
    doc = yaml.safe_load(path.read_text())
    assert doc["kind"] == "HealthCheck"
    assert doc["apiVersion"] == "activemonitor.keikoproj.io/v1alpha1"
    hc = HealthCheck.from_dict(doc)
    assert hc.name
    spec = hc.spec
    assert spec.workflow.resource is not None
    assert spec.workflow.generate_name
    # schedulable: repeatAfterSec, cron, or deliberately paused fixture
    assert spec.repeat_after_sec > 0 or spec.schedule.cron or "paused" in hc.name

    # inline sources must produce a submittable workflow
    if spec.workflow.resource.source.inline is not None:
        wf_spec, labels = parse_workflow_from_healthcheck(hc)
        assert "entrypoint" in wf_spec
        assert wf_spec["podGC"] == {"strategy": "OnPodCompletion"}
        assert labels["workflows.argoproj.io/controller-instanceid"] == "activemonitor-workflows"
    if (
        not spec.remedy_workflow.is_empty()
        and spec.remedy_workflow.resource is not None
        and spec.remedy_workflow.resource.source.inline is not None
    ):
        r_spec, _ = parse_remedy_workflow_from_healthcheck(hc)
        assert "entrypoint" in r_spec

    # spec round-trips through the wire format losslessly
    assert HealthCheck.from_dict(hc.to_dict()).to_dict() == hc.to_dict()


def test_crd_manifest_in_sync():
    """config/crd/bases must match the generator output (make manifests)."""
    from active_monitor_amd.api.crd import healthcheck_crd_yaml

    path = (
        pathlib.Path(__file__).parents[2]
        / "config/crd/bases/activemonitor.keikoproj.io_healthchecks.yaml"
    )
    assert yaml.safe_load(path.read_text()) == yaml.safe_load(healthcheck_crd_yaml())
