"""crdutil tests (reference pkg/crdutil/crdutil_test.go:61-262)."""

import os
import textwrap

import pytest

from k8s_operator_libs_amd.core.errors import NotFoundError
from k8s_operator_libs_amd.crdutil import (
    CRD_OPERATION_APPLY,
    CRD_OPERATION_DELETE,
    CrdUtilError,
    parse_crds_from_paths,
    process_crds,
    walk_crd_paths,
)

CRD_TMPL = """\
apiVersion: apiextensions.k8s.io/v1
kind: CustomResourceDefinition
metadata:
  name: {plural}.{group}
spec:
  group: {group}
  scope: Namespaced
  names:
    kind: {kind}
    plural: {plural}
  versions:
    - name: v1
      served: true
      storage: true
"""


def write(path, content):
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "w") as fh:
        fh.write(content)
    return path


@pytest.fixture
def crd_dir(tmp_path):
    d = tmp_path / "crds"
    write(str(d / "widgets.yaml"), CRD_TMPL.format(group="amd.com", kind="Widget", plural="widgets"))
    write(
        str(d / "nested" / "gadgets.yml"),
        CRD_TMPL.format(group="amd.com", kind="Gadget", plural="gadgets"),
    )
    # multi-doc file with a non-CRD document mixed in (must be skipped)
    write(
        str(d / "multi.yaml"),
        CRD_TMPL.format(group="amd.com", kind="Gizmo", plural="gizmos")
        + "---\n"
        + textwrap.dedent(
            """\
            apiVersion: v1
            kind: ConfigMap
            metadata:
              name: not-a-crd
            """
        )
        + "---\n"
        + CRD_TMPL.format(group="amd.com", kind="Doodad", plural="doodads"),
    )
    write(str(d / "notes.txt"), "ignored, not yaml")
    return str(d)


def test_walk_recursive_and_sorted(crd_dir):
    paths = walk_crd_paths([crd_dir])
    names = [os.path.basename(p) for p in paths]
    assert names == ["multi.yaml", "gadgets.yml", "widgets.yaml"]


def test_walk_missing_path_raises():
    with pytest.raises(CrdUtilError):
        walk_crd_paths(["/does/not/exist"])


def test_parse_skips_non_crds(crd_dir):
    crds = parse_crds_from_paths([crd_dir])
    kinds = sorted(c["spec"]["names"]["kind"] for c in crds)
    assert kinds == ["Doodad", "Gadget", "Gizmo", "Widget"]


def test_parse_single_file(crd_dir):
    crds = parse_crds_from_paths([os.path.join(crd_dir, "widgets.yaml")])
    assert len(crds) == 1


def test_parse_bad_yaml_raises(tmp_path):
    p = write(str(tmp_path / "bad.yaml"), "a: [unclosed")
    with pytest.raises(CrdUtilError):
        parse_crds_from_paths([p])


def test_apply_creates_and_serves(client, crd_dir):
    n = process_crds(client, [crd_dir], CRD_OPERATION_APPLY)
    assert n == 4
    got = client.get("apiextensions.k8s.io/v1", "CustomResourceDefinition", "widgets.amd.com")
    assert got["spec"]["names"]["kind"] == "Widget"
    # servability: the CR kind is usable immediately after apply returns
    client.create({"apiVersion": "amd.com/v1", "kind": "Widget",
                   "metadata": {"name": "w", "namespace": "default"}})


def test_apply_is_update_on_existing(client, crd_dir, tmp_path):
    process_crds(client, [crd_dir], CRD_OPERATION_APPLY)
    rv1 = client.get("apiextensions.k8s.io/v1", "CustomResourceDefinition",
                     "widgets.amd.com")["metadata"]["resourceVersion"]
    # re-apply an updated widget CRD (adds a label)
    updated = CRD_TMPL.format(group="amd.com", kind="Widget", plural="widgets").replace(
        "metadata:\n  name: widgets.amd.com",
        "metadata:\n  name: widgets.amd.com\n  labels:\n    rev: '2'",
    )
    p = write(str(tmp_path / "w2.yaml"), updated)
    process_crds(client, [p], CRD_OPERATION_APPLY)
    got = client.get("apiextensions.k8s.io/v1", "CustomResourceDefinition", "widgets.amd.com")
    assert got["metadata"]["labels"]["rev"] == "2"
    assert got["metadata"]["resourceVersion"] != rv1


def test_delete_idempotent(client, crd_dir):
    process_crds(client, [crd_dir], CRD_OPERATION_APPLY)
    assert process_crds(client, [crd_dir], CRD_OPERATION_DELETE) == 4
    with pytest.raises(NotFoundError):
        client.get("apiextensions.k8s.io/v1", "CustomResourceDefinition", "widgets.amd.com")
    # second delete: all already absent, still succeeds
    assert process_crds(client, [crd_dir], CRD_OPERATION_DELETE) == 4


def test_variadic_multi_dir(client, crd_dir, tmp_path):
    other = write(
        str(tmp_path / "other" / "x.yaml"),
        CRD_TMPL.format(group="net.amd.com", kind="Link", plural="links"),
    )
    n = process_crds(client, [crd_dir, other], CRD_OPERATION_APPLY)
    assert n == 5


def test_no_crds_found_raises(client, tmp_path):
    empty = tmp_path / "empty"
    empty.mkdir()
    with pytest.raises(CrdUtilError):
        process_crds(client, [str(empty)], CRD_OPERATION_APPLY)


def test_unknown_operation_raises(client, crd_dir):
    with pytest.raises(CrdUtilError):
        process_crds(client, [crd_dir], "upsert")


def test_nodemaintenance_fixture_applies(client):
    fixture = os.path.join(os.path.dirname(__file__), "..", "hack", "crds")
    n = process_crds(client, [fixture], CRD_OPERATION_APPLY)
    assert n == 1
    client.create({
        "apiVersion": "maintenance.amd.com/v1alpha1", "kind": "NodeMaintenance",
        "metadata": {"name": "nm", "namespace": "default"},
        "spec": {"nodeName": "n1", "requestorID": "amd.gpu.operator"},
    })


def test_apply_crds_cli_example(tmp_path, crd_dir, capsys):
    import examples.apply_crds as cli

    rc = cli.main(["--crds-path", crd_dir, "--operation", "apply", "--fake"])
    assert rc == 0


def test_generated_consumer_crd_applies(client, tmp_path):
    """examples/generate_crd.py output is a valid, servable CRD whose policy
    schema round-trips through crdutil apply."""
    import examples.generate_crd as gen
    import yaml as _yaml

    crd_path = tmp_path / "amdgpudrivers.yaml"
    crd_path.write_text(_yaml.safe_dump(gen.build_crd(), sort_keys=False))
    assert process_crds(client, [str(crd_path)], CRD_OPERATION_APPLY) == 1
    client.create({
        "apiVersion": "driver.amd.com/v1alpha1", "kind": "AMDGPUDriver",
        "metadata": {"name": "default"},
        "spec": {"driverVersion": "6.4",
                 "driverUpgradePolicy": {"autoUpgrade": True,
                                          "maxParallelUpgrades": 2}},
    })
    got = client.get("driver.amd.com/v1alpha1", "AMDGPUDriver", "default")
    assert got["spec"]["driverUpgradePolicy"]["maxParallelUpgrades"] == 2


class TestApplyRaces:
    def test_update_conflict_retries_until_success(self, client, tmp_path):
        """A concurrent writer bumping the CRD between our GET and UPDATE
        forces RetryOnConflict behaviour (crdutil.go:214-249)."""

        p = write(str(tmp_path / "w.yaml"),
                  CRD_TMPL.format(group="amd.com", kind="Widget", plural="widgets"))
        process_crds(client, [p], CRD_OPERATION_APPLY)

        real_update = client.update
        bumps = {"n": 0}

        def racing_update(obj):
            if bumps["n"] < 2:
                bumps["n"] += 1
                # concurrent writer invalidates our resourceVersion
                live = client.get("apiextensions.k8s.io/v1",
                                  "CustomResourceDefinition", "widgets.amd.com")
                live["metadata"]["labels"] = {"race": str(bumps["n"])}
                real_update(live)
            return real_update(obj)

        client.update = racing_update
        try:
            updated = CRD_TMPL.format(group="amd.com", kind="Widget",
                                      plural="widgets").replace(
                "metadata:\n  name: widgets.amd.com",
                "metadata:\n  name: widgets.amd.com\n  labels:\n    final: 'yes'")
            p2 = write(str(tmp_path / "w2.yaml"), updated)
            process_crds(client, [p2], CRD_OPERATION_APPLY)
        finally:
            client.update = real_update
        got = client.get("apiextensions.k8s.io/v1", "CustomResourceDefinition",
                         "widgets.amd.com")
        assert got["metadata"]["labels"]["final"] == "yes"
        assert bumps["n"] == 2  # retried through both injected conflicts

    def test_create_race_falls_back_to_update(self, client, tmp_path):
        """Another applier creating the CRD between our GET (404) and CREATE
        must turn into an update, not an error (crdutil.go:224-231)."""
        p = write(str(tmp_path / "w.yaml"),
                  CRD_TMPL.format(group="amd.com", kind="Widget", plural="widgets"))
        real_create = client.create
        raced = {"done": False}

        def racing_create(obj):
            if obj.get("kind") == "CustomResourceDefinition" and not raced["done"]:
                raced["done"] = True
                real_create(obj)  # the other applier wins the create
            return real_create(obj)  # ours now raises AlreadyExists

        client.create = racing_create
        try:
            process_crds(client, [p], CRD_OPERATION_APPLY)
        finally:
            client.create = real_create
        assert client.get("apiextensions.k8s.io/v1", "CustomResourceDefinition",
                          "widgets.amd.com")

    def test_wait_for_crds_times_out_on_never_served(self, client):
        from k8s_operator_libs_amd.crdutil import CrdUtilError, wait_for_crds

        ghost = {"apiVersion": "apiextensions.k8s.io/v1",
                 "kind": "CustomResourceDefinition",
                 "metadata": {"name": "ghosts.amd.com"},
                 "spec": {"group": "amd.com", "scope": "Namespaced",
                          "names": {"kind": "Ghost", "plural": "ghosts"},
                          "versions": [{"name": "v1", "served": True}]}}
        # never created on the cluster: discovery can't serve it
        with pytest.raises(CrdUtilError):
            wait_for_crds(client, [ghost], timeout=0.3)
