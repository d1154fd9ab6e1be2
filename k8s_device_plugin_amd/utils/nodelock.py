"""Node mutex implemented as a node annotation.

Reference behavior: /root/reference/pkg/util/nodelock/nodelock.go:13-107 —
``4pd.io/mutex.lock=<RFC3339 time>``; acquiring retries 5 times with backoff;
a lock older than 5 minutes is considered stale and is broken.
"""
from __future__ import annotations

import logging
import time
from datetime import datetime, timezone

from .kubeclient import KubeClient
from .types import NODE_LOCK_ANNO, NODE_LOCK_EXPIRE_SECONDS

log = logging.getLogger(__name__)

MAX_RETRIES = 5
RETRY_DELAY_S = 1.0


class NodeLockError(RuntimeError):
    pass


def _now_str() -> str:
    return datetime.now(timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


def _parse(ts: str):
    try:
        return datetime.strptime(ts, "%Y-%m-%dT%H:%M:%SZ").replace(tzinfo=timezone.utc)
    except ValueError:
        return None


def set_node_lock(client: KubeClient, node_name: str) -> None:
    node = client.get_node(node_name)
    if NODE_LOCK_ANNO in node.annotations:
        raise NodeLockError(f"node {node_name} is locked")
    client.patch_node_annotations(node_name, {NODE_LOCK_ANNO: _now_str()})


def lock_node(client: KubeClient, node_name: str) -> None:
    """Acquire with retries; break expired locks (reference nodelock.go:81-107)."""
    for attempt in range(MAX_RETRIES):
        node = client.get_node(node_name)
        holder = node.annotations.get(NODE_LOCK_ANNO)
        if holder is None:
            try:
                set_node_lock(client, node_name)
                return
            except NodeLockError:
                pass  # raced; retry
        else:
            t = _parse(holder)
            if t is None or (
                datetime.now(timezone.utc) - t
            ).total_seconds() > NODE_LOCK_EXPIRE_SECONDS:
                log.warning("node %s lock expired (%s); breaking", node_name, holder)
                release_node_lock(client, node_name)
                continue
        time.sleep(RETRY_DELAY_S * (attempt + 1))
    raise NodeLockError(f"could not lock node {node_name} after {MAX_RETRIES} tries")


def release_node_lock(client: KubeClient, node_name: str) -> None:
    node = client.get_node(node_name)
    if NODE_LOCK_ANNO in node.annotations:
        client.patch_node_annotations(node_name, {NODE_LOCK_ANNO: None})
