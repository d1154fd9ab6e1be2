"""Node-annotation mutex semantics (reference nodelock.go:13-107)."""
from datetime import datetime, timedelta, timezone

import pytest

from k8s_device_plugin_amd.utils import nodelock
from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
from k8s_device_plugin_amd.utils.types import NODE_LOCK_ANNO, NodeInfo


@pytest.fixture
def client():
    c = FakeKubeClient()
    c.add_node(NodeInfo(name="n1"))
    return c


def test_acquire_and_release(client):
    nodelock.lock_node(client, "n1")
    assert NODE_LOCK_ANNO in client.get_node("n1").annotations
    nodelock.release_node_lock(client, "n1")
    assert NODE_LOCK_ANNO not in client.get_node("n1").annotations


def test_contended_lock_fails_after_retries(client, monkeypatch):
    monkeypatch.setattr(nodelock, "RETRY_DELAY_S", 0.01)
    client.patch_node_annotations(
        "n1", {NODE_LOCK_ANNO: nodelock._now_str()})
    with pytest.raises(nodelock.NodeLockError):
        nodelock.lock_node(client, "n1")
    # the fresh lock is untouched
    assert NODE_LOCK_ANNO in client.get_node("n1").annotations


def test_expired_lock_is_broken(client, monkeypatch):
    monkeypatch.setattr(nodelock, "RETRY_DELAY_S", 0.01)
    stale = (datetime.now(timezone.utc) - timedelta(minutes=6)).strftime(
        "%Y-%m-%dT%H:%M:%SZ")
    client.patch_node_annotations("n1", {NODE_LOCK_ANNO: stale})
    nodelock.lock_node(client, "n1")  # breaks + re-acquires
    held = client.get_node("n1").annotations[NODE_LOCK_ANNO]
    assert held != stale


def test_garbage_timestamp_is_broken(client, monkeypatch):
    monkeypatch.setattr(nodelock, "RETRY_DELAY_S", 0.01)
    client.patch_node_annotations("n1", {NODE_LOCK_ANNO: "not-a-time"})
    nodelock.lock_node(client, "n1")
    assert nodelock._parse(
        client.get_node("n1").annotations[NODE_LOCK_ANNO]) is not None


def test_release_is_idempotent(client):
    nodelock.release_node_lock(client, "n1")
    nodelock.release_node_lock(client, "n1")
