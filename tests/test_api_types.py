"""Policy API type tests (reference api/upgrade/v1alpha1/upgrade_spec.go)."""

import pytest
from pydantic import ValidationError

from k8s_operator_libs_amd.api.upgrade.v1alpha1 import (
    DrainSpec,
    DriverUpgradePolicySpec,
    IntOrString,
    PodDeletionSpec,
    WaitForCompletionSpec,
)


def test_policy_defaults():
    p = DriverUpgradePolicySpec()
    assert p.auto_upgrade is False
    assert p.max_parallel_upgrades == 1
    assert p.max_unavailable == "25%"
    assert p.wait_for_completion is None
    assert p.pod_deletion is None
    assert p.drain_spec is None


def test_policy_reference_wire_format():
    """A policy document authored against the reference kubebuilder schema
    (json tags of upgrade_spec.go:27-110) parses and round-trips byte-
    compatibly: ``drain`` (spec.go:48) and ``timeoutSeconds``
    (spec.go:63,77,104) are the wire names."""
    ref_doc = {
        "autoUpgrade": True,
        "maxParallelUpgrades": 4,
        "maxUnavailable": 3,
        "drain": {"enable": True, "timeoutSeconds": 60},
        "podDeletion": {"force": True, "timeoutSeconds": 120, "deleteEmptyDir": True},
        "waitForCompletion": {"podSelector": "app=job", "timeoutSeconds": 30},
    }
    p = DriverUpgradePolicySpec.model_validate(ref_doc)
    assert p.max_parallel_upgrades == 4
    assert p.max_unavailable == 3
    assert p.drain_spec.enable and p.drain_spec.timeout_seconds == 60
    assert p.pod_deletion.force and p.pod_deletion.delete_emptydir_data
    assert p.pod_deletion.timeout_seconds == 120
    assert p.wait_for_completion.pod_selector == "app=job"
    assert p.wait_for_completion.timeout_seconds == 30
    # round-trip: emitted wire shape uses reference names and re-parses equal
    wire = p.model_dump(by_alias=True, exclude_none=True)
    assert "drain" in wire and "drainSpec" not in wire
    assert wire["podDeletion"]["timeoutSeconds"] == 120
    assert "timeoutSecond" not in wire["waitForCompletion"]
    assert DriverUpgradePolicySpec.model_validate(wire) == p


def test_policy_legacy_round1_aliases_still_accepted():
    """Round-1 wire names (``drainSpec``, ``timeoutSecond``) remain valid
    on input so existing in-repo documents keep parsing."""
    p = DriverUpgradePolicySpec.model_validate(
        {
            "drainSpec": {"enable": True, "timeoutSeconds": 60},
            "waitForCompletion": {"podSelector": "app=job", "timeoutSecond": 30},
        }
    )
    assert p.drain_spec.enable and p.drain_spec.timeout_seconds == 60
    assert p.wait_for_completion.timeout_seconds == 30
    # but output is normalized to the reference format
    wire = p.model_dump(by_alias=True, exclude_none=True)
    assert "drain" in wire and "drainSpec" not in wire


def test_nested_defaults_match_reference():
    assert PodDeletionSpec().timeout_seconds == 300  # spec.go:72-77
    assert DrainSpec().timeout_seconds == 300  # spec.go:100-104
    assert DrainSpec().enable is False
    assert WaitForCompletionSpec().timeout_seconds == 0


def test_validation_rejects_negatives():
    with pytest.raises(ValidationError):
        DriverUpgradePolicySpec(maxParallelUpgrades=-1)
    with pytest.raises(ValidationError):
        DrainSpec(timeoutSeconds=-5)


def test_max_unavailable_percent_validation():
    assert DriverUpgradePolicySpec(maxUnavailable="50%").max_unavailable == "50%"
    with pytest.raises(ValidationError):
        DriverUpgradePolicySpec(maxUnavailable="nonsense")


@pytest.mark.parametrize(
    "value,total,round_up,expected",
    [
        ("25%", 8, True, 2),
        ("25%", 10, True, 3),  # 2.5 rounds up
        ("25%", 10, False, 2),
        ("100%", 7, True, 7),
        ("0%", 5, True, 0),
        (3, 100, True, 3),
        (None, 100, True, 0),
    ],
)
def test_int_or_percent_scaling(value, total, round_up, expected):
    assert IntOrString.scaled_value(value, total, round_up) == expected


def test_deep_copy_is_independent():
    p = DriverUpgradePolicySpec(drainSpec={"enable": True})
    q = p.deep_copy()
    q.drain_spec.enable = False
    assert p.drain_spec.enable is True


def test_openapi_v3_schema_for_crd_embedding():
    from k8s_operator_libs_amd.api.upgrade.v1alpha1 import openapi_v3_schema

    schema = openapi_v3_schema()
    props = schema["properties"]
    assert props["autoUpgrade"]["default"] is False
    assert props["maxParallelUpgrades"]["default"] == 1
    assert props["maxParallelUpgrades"]["minimum"] == 0
    assert props["maxUnavailable"]["x-kubernetes-int-or-string"] is True
    assert props["maxUnavailable"]["default"] == "25%"
    assert props["drain"]["properties"]["timeoutSeconds"]["default"] == 300
    assert props["podDeletion"]["properties"]["timeoutSeconds"]["default"] == 300
    # structural: no $refs/anyOf remain anywhere
    import json
    text = json.dumps(schema)
    assert "$ref" not in text and "anyOf" not in text
