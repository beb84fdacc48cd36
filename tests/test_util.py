"""Tests for driver-name registry, key getters, StringSet, KeyedMutex
(reference pkg/upgrade/util.go)."""

import threading

import pytest

from k8s_operator_libs_amd.upgrade import util
from k8s_operator_libs_amd.upgrade.util import InvalidDriverNameError, KeyedMutex, StringSet


def test_default_driver_name_keys():
    assert util.get_upgrade_state_label_key() == "amd.com/amdgpu-driver-upgrade-state"
    assert util.get_upgrade_skip_node_label_key() == "amd.com/amdgpu-driver-upgrade.skip"
    assert (
        util.get_upgrade_wait_for_safe_driver_load_annotation_key()
        == "amd.com/amdgpu-driver-upgrade.driver-wait-for-safe-load"
    )
    assert util.get_event_reason() == "AMDGPUDriverUpgrade"


def test_set_driver_name_changes_keys():
    util.set_driver_name("anic")
    assert util.get_upgrade_state_label_key() == "amd.com/anic-driver-upgrade-state"
    assert util.get_upgrade_requested_annotation_key() == "amd.com/anic-driver-upgrade-requested"
    assert util.get_upgrade_requestor_mode_annotation_key() == "amd.com/anic-driver-upgrade-requestor-mode"


def test_invalid_driver_name_rejected():
    for bad in ("", "UPPER", "has space", "-lead", "trail-", "a_b"):
        with pytest.raises(InvalidDriverNameError):
            util.set_driver_name(bad)


def test_skip_drain_selector_excludes_marked_pods():
    assert util.get_upgrade_skip_drain_pod_selector() == (
        "amd.com/amdgpu-driver-upgrade-drain.skip!=true"
    )


def test_string_set_basicity():
    s = StringSet()
    assert s.add_if_absent("a")
    assert not s.add_if_absent("a")
    assert s.has("a")
    assert len(s) == 1
    s.remove("a")
    assert not s.has("a")
    s.remove("a")  # idempotent


def test_string_set_concurrent_guard():
    s = StringSet()
    won = []

    def worker():
        if s.add_if_absent("node-1"):
            won.append(1)

    threads = [threading.Thread(target=worker) for _ in range(32)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert len(won) == 1


def test_keyed_mutex_serializes_per_key():
    km = KeyedMutex()
    order = []

    def worker(i):
        with km.lock("node-A"):
            order.append(("enter", i))
            order.append(("exit", i))

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    # strictly alternating enter/exit means no two holders overlapped
    for j in range(0, len(order), 2):
        assert order[j][0] == "enter" and order[j + 1][0] == "exit"
        assert order[j][1] == order[j + 1][1]


def test_keyed_mutex_distinct_keys_do_not_block():
    km = KeyedMutex()
    km.acquire("a")
    done = threading.Event()

    def other():
        with km.lock("b"):
            done.set()

    t = threading.Thread(target=other)
    t.start()
    assert done.wait(2.0), "lock on a different key must not block"
    km.release("a")
    t.join()
