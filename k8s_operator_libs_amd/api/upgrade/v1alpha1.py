"""Driver-upgrade policy API types (group ``upgrade.amd.com``, v1alpha1).

Capability parity with the reference's
``api/upgrade/v1alpha1/upgrade_spec.go:27-110``: the four spec fragments a
consuming operator embeds in its own CRD to configure the rolling
driver-upgrade state machine.  Field names, defaults and validation semantics
match the reference's kubebuilder markers:

- ``DriverUpgradePolicySpec.auto_upgrade`` default False (spec.go:32)
- ``max_parallel_upgrades`` default 1, minimum 0, 0 = unlimited (spec.go:33-38)
- ``max_unavailable`` int-or-percent, default "25%" (spec.go:39-45)
- ``WaitForCompletionSpec`` pod_selector/timeout_seconds (spec.go:52-64)
- ``PodDeletionSpec`` force/timeout 300 s/delete_emptydir_data (spec.go:67-83)
- ``DrainSpec`` enable/force/pod_selector/timeout 300 s/delete_emptydir_data
  (spec.go:86-110)

Pydantic replaces Go's kubebuilder validation + generated deepcopy: models are
validated on construction, ``model_copy(deep=True)`` is the DeepCopy
equivalent, and the JSON aliases are the exact wire names used inside a CRD
(``autoUpgrade``, ``maxParallelUpgrades``, ...).
"""

from __future__ import annotations

import math
import re
from typing import Optional, Union

from pydantic import AliasChoices, BaseModel, ConfigDict, Field, field_validator

_PERCENT_RE = re.compile(r"^(\d+)%$")


class IntOrString:
    """Kubernetes ``intstr.IntOrString`` semantics for int-or-percent fields."""

    @staticmethod
    def scaled_value(value: Union[int, str, None], total: int, round_up: bool) -> int:
        """Mirror of apimachinery ``intstr.GetScaledValueFromIntOrPercent``.

        An int (or int-like string) is returned as-is; ``"25%"`` scales against
        ``total`` and rounds up or down per ``round_up``.  The in-place mode
        rounds **up** when computing maxUnavailable (upgrade_inplace.go:54-60).
        """
        if value is None:
            return 0
        if isinstance(value, int):
            return value
        value = value.strip()
        m = _PERCENT_RE.match(value)
        if m:
            pct = int(m.group(1))
            exact = pct * total / 100.0
            return math.ceil(exact) if round_up else math.floor(exact)
        return int(value)

    @staticmethod
    def validate(value: Union[int, str, None]) -> Union[int, str, None]:
        if value is None or isinstance(value, int):
            return value
        if _PERCENT_RE.match(value.strip()):
            return value.strip()
        return int(value)  # raises ValueError on junk


class _SpecBase(BaseModel):
    model_config = ConfigDict(populate_by_name=True, extra="forbid")

    def deep_copy(self):
        """DeepCopy equivalent of zz_generated.deepcopy.go."""
        return self.model_copy(deep=True)


class WaitForCompletionSpec(_SpecBase):
    """Wait for selected workload pods to complete before deletion
    (upgrade_spec.go:52-64)."""

    # Label selector (string form, e.g. "app=training-job") of pods to wait on.
    pod_selector: str = Field(default="", alias="podSelector")
    # 0 means wait forever.  Wire name matches the reference json tag
    # ``timeoutSeconds`` (upgrade_spec.go:63); the round-1 name
    # ``timeoutSecond`` is still accepted on input for compatibility.
    timeout_seconds: int = Field(
        default=0,
        ge=0,
        serialization_alias="timeoutSeconds",
        validation_alias=AliasChoices("timeoutSeconds", "timeoutSecond", "timeout_seconds"),
    )


class PodDeletionSpec(_SpecBase):
    """Controlled deletion of selected pods before driver restart
    (upgrade_spec.go:67-83)."""

    force: bool = Field(default=False)
    # Wire name ``timeoutSeconds`` per the reference json tag
    # (upgrade_spec.go:77); round-1's ``timeoutSecond`` accepted on input.
    timeout_seconds: int = Field(
        default=300,
        ge=0,
        serialization_alias="timeoutSeconds",
        validation_alias=AliasChoices("timeoutSeconds", "timeoutSecond", "timeout_seconds"),
    )
    delete_emptydir_data: bool = Field(default=False, alias="deleteEmptyDir")


class DrainSpec(_SpecBase):
    """Full node drain configuration (upgrade_spec.go:86-110)."""

    enable: bool = Field(default=False)
    force: bool = Field(default=False)
    pod_selector: str = Field(default="", alias="podSelector")
    timeout_seconds: int = Field(default=300, ge=0, alias="timeoutSeconds")
    delete_emptydir_data: bool = Field(default=False, alias="deleteEmptyDir")


class DriverUpgradePolicySpec(_SpecBase):
    """Top-level driver upgrade policy (upgrade_spec.go:27-50)."""

    auto_upgrade: bool = Field(default=False, alias="autoUpgrade")
    # 0 = no limit on concurrently upgrading nodes.
    max_parallel_upgrades: int = Field(default=1, ge=0, alias="maxParallelUpgrades")
    # int or percent string; percent is relative to total managed nodes.
    max_unavailable: Optional[Union[int, str]] = Field(
        default="25%", alias="maxUnavailable"
    )
    wait_for_completion: Optional[WaitForCompletionSpec] = Field(
        default=None, alias="waitForCompletion"
    )
    pod_deletion: Optional[PodDeletionSpec] = Field(default=None, alias="podDeletion")
    # Wire name ``drain`` per the reference json tag (upgrade_spec.go:48);
    # round-1's ``drainSpec`` accepted on input for compatibility.
    drain_spec: Optional[DrainSpec] = Field(
        default=None,
        serialization_alias="drain",
        validation_alias=AliasChoices("drain", "drainSpec", "drain_spec"),
    )

    @field_validator("max_unavailable")
    @classmethod
    def _validate_max_unavailable(cls, v):
        return IntOrString.validate(v)


def openapi_v3_schema() -> dict:
    """Structural OpenAPI v3 schema of DriverUpgradePolicySpec for embedding
    in a consumer CRD (the kubebuilder controller-gen analogue: the
    reference's consumers get this from the marker comments on
    upgrade_spec.go).  Defaults and minimums match the field definitions."""

    def prune(schema: dict) -> dict:
        # pydantic emits $defs/anyOf forms CRDs don't accept; inline and
        # simplify to the structural-schema subset
        defs = schema.pop("$defs", {})

        def walk(node):
            if isinstance(node, dict):
                if "$ref" in node:
                    ref = node.pop("$ref").rsplit("/", 1)[-1]
                    node.update(walk(dict(defs[ref])))
                if "anyOf" in node:
                    # int-or-string fields -> x-kubernetes-int-or-string
                    options = node.pop("anyOf")
                    types = {o.get("type") for o in options}
                    if types >= {"integer", "string"}:
                        node["x-kubernetes-int-or-string"] = True
                    elif len(options) == 1:
                        node.update(walk(options[0]))
                    else:
                        # optional nested spec (Type | null): take the object arm
                        arm = next((o for o in options if o.get("type") != "null"),
                                   options[0])
                        node.update(walk(dict(arm)))
                node.pop("title", None)
                for v in list(node.values()):
                    walk(v)
            elif isinstance(node, list):
                for v in node:
                    walk(v)
            return node

        return walk(schema)

    return prune(DriverUpgradePolicySpec.model_json_schema(by_alias=True))
