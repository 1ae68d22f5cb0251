"""Workflow parse/injection unit tests
(reference: healthcheck_controller.go:876-1125;
healthcheck_controller_unit_test.go:102-256)."""
import pytest

from active_monitor_amd.api import (
    ArtifactLocation,
    HealthCheck,
    HealthCheckSpec,
    ObjectMeta,
    RemedyWorkflow,
    ResourceObject,
    Workflow,
)
from active_monitor_amd.engine.parse import (
    WorkflowParseError,
    parse_remedy_workflow_from_healthcheck,
    parse_workflow_from_healthcheck,
)


def hc_with_inline(inline, remedy_inline=None, repeat=30, timeout=0):
    spec = HealthCheckSpec(
        repeat_after_sec=repeat,
        workflow=Workflow(
            generate_name="t-",
            timeout=timeout,
            resource=ResourceObject(
                namespace="health", service_account="sa",
                source=ArtifactLocation(inline=inline),
            ),
        ),
    )
    if remedy_inline is not None:
        spec.remedy_workflow = RemedyWorkflow(
            generate_name="t-remedy-",
            resource=ResourceObject(
                namespace="health", service_account="rsa",
                source=ArtifactLocation(inline=remedy_inline),
            ),
        )
    return HealthCheck(metadata=ObjectMeta(name="t", namespace="health"), spec=spec)


BASIC = "apiVersion: argoproj.io/v1alpha1\nkind: Workflow\nspec:\n  entrypoint: e\n"


def test_injections_basic():
    hc = hc_with_inline(BASIC, timeout=45)
    spec, labels = parse_workflow_from_healthcheck(hc)
    assert spec["podGC"] == {"strategy": "OnPodCompletion"}
    assert spec["serviceAccountName"] == "sa"
    assert spec["activeDeadlineSeconds"] == 45
    assert labels == {"workflows.argoproj.io/controller-instanceid": "activemonitor-workflows"}


def test_timeout_defaults_from_repeat_after_sec():
    """Workflow.Timeout==0 → mutated to RepeatAfterSec and used as deadline
    (reference :980-986)."""
    hc = hc_with_inline(BASIC, repeat=77, timeout=0)
    spec, _ = parse_workflow_from_healthcheck(hc)
    assert hc.spec.workflow.timeout == 77  # spec mutation, feeds backoff calc
    assert spec["activeDeadlineSeconds"] == 77


def test_existing_deadline_kept():
    y = BASIC + "  activeDeadlineSeconds: 9\n"
    hc = hc_with_inline(y, timeout=45)
    spec, _ = parse_workflow_from_healthcheck(hc)
    assert spec["activeDeadlineSeconds"] == 9
    assert hc.spec.workflow.timeout == 45  # NOT round-tripped (hc-wf path)


def test_existing_podgc_kept():
    y = BASIC + "  podGC:\n    strategy: OnWorkflowSuccess\n"
    spec, _ = parse_workflow_from_healthcheck(hc_with_inline(y))
    assert spec["podGC"] == {"strategy": "OnWorkflowSuccess"}


def test_labels_extracted_and_stringified():
    y = (
        "metadata:\n  labels:\n    team: sre\n    num: 7\n    flag: true\n"
        + BASIC
    )
    _, labels = parse_workflow_from_healthcheck(hc_with_inline(y))
    assert labels["team"] == "sre"
    assert labels["num"] == "7"  # values stringified like fmt.Sprintf("%v")
    assert labels["flag"] == "True"
    assert labels["workflows.argoproj.io/controller-instanceid"] == "activemonitor-workflows"


def test_workflow_can_override_instance_id_label():
    y = (
        'metadata:\n  labels:\n    "workflows.argoproj.io/controller-instanceid": custom\n'
        + BASIC
    )
    _, labels = parse_workflow_from_healthcheck(hc_with_inline(y))
    assert labels["workflows.argoproj.io/controller-instanceid"] == "custom"


def test_non_map_metadata_treated_as_unset():
    # reference :930-932 "metadata is not a map, treating as unset"
    y = "metadata: just-a-string\n" + BASIC
    _, labels = parse_workflow_from_healthcheck(hc_with_inline(y))
    assert labels == {"workflows.argoproj.io/controller-instanceid": "activemonitor-workflows"}


def test_non_map_labels_use_default():
    y = "metadata:\n  labels: not-a-map\n" + BASIC
    _, labels = parse_workflow_from_healthcheck(hc_with_inline(y))
    assert labels == {"workflows.argoproj.io/controller-instanceid": "activemonitor-workflows"}


def test_missing_spec_errors():
    with pytest.raises(WorkflowParseError, match="invalid workflow, missing spec"):
        parse_workflow_from_healthcheck(
            hc_with_inline("apiVersion: argoproj.io/v1alpha1\nkind: Workflow\n")
        )


def test_non_map_spec_errors():
    with pytest.raises(WorkflowParseError, match="invalid workflow, spec is not a map"):
        parse_workflow_from_healthcheck(
            hc_with_inline("spec: not-a-map\n")
        )


def test_invalid_yaml_errors():
    with pytest.raises(WorkflowParseError, match="Invalid spec file passed"):
        parse_workflow_from_healthcheck(hc_with_inline("a: [unclosed\n"))


def test_non_mapping_document_errors():
    with pytest.raises(WorkflowParseError, match="not a mapping"):
        parse_workflow_from_healthcheck(hc_with_inline("- a\n- b\n"))


# -- remedy variants (reference :1002-1125) --------------------------------


def test_remedy_missing_spec_message():
    hc = hc_with_inline(BASIC, remedy_inline="kind: Workflow\n")
    with pytest.raises(WorkflowParseError, match="Invalid remedy workflow, missing spec"):
        parse_remedy_workflow_from_healthcheck(hc)


def test_remedy_deadline_defaults_from_repeat_and_roundtrips():
    hc = hc_with_inline(BASIC, remedy_inline=BASIC, repeat=55)
    spec, _ = parse_remedy_workflow_from_healthcheck(hc)
    assert spec["activeDeadlineSeconds"] == 55
    assert hc.spec.remedy_workflow.timeout == 55  # round-trip (reference :1110-1112)


def test_remedy_existing_numeric_deadline_roundtrips():
    hc = hc_with_inline(BASIC, remedy_inline=BASIC + "  activeDeadlineSeconds: 33\n", repeat=55)
    spec, _ = parse_remedy_workflow_from_healthcheck(hc)
    assert spec["activeDeadlineSeconds"] == 33
    assert hc.spec.remedy_workflow.timeout == 33


def test_remedy_non_numeric_deadline_uses_default():
    hc = hc_with_inline(
        BASIC, remedy_inline=BASIC + "  activeDeadlineSeconds: soon\n", repeat=55
    )
    spec, _ = parse_remedy_workflow_from_healthcheck(hc)
    assert hc.spec.remedy_workflow.timeout == 55  # reference :1114-1119
