"""Daemon entry points: plugin restart loop + scheduler CLI wiring.

Reference behavior: cmd/device-plugin/nvidia/main.go:154-238 (restart on
kubelet-socket re-creation / SIGHUP) and cmd/scheduler/main.go:48-94.
"""
import os
import threading
import time

import pytest

from k8s_device_plugin_amd.plugin.config import PluginConfig
from k8s_device_plugin_amd.plugin.main import PluginDaemon, _SocketWatch
from k8s_device_plugin_amd.scheduler.main import parse_args as sched_parse_args
from k8s_device_plugin_amd.utils.kubeclient import FakeKubeClient
from k8s_device_plugin_amd.utils.types import NodeInfo


def test_socket_watch_triggers_only_on_recreation(tmp_path):
    sock = tmp_path / "kubelet.sock"
    sock.write_bytes(b"")
    w = _SocketWatch(str(sock))
    assert not w.changed()
    # socket vanishing (kubelet down) is not a restart trigger
    sock.unlink()
    assert not w.changed()
    # re-creation is
    time.sleep(0.01)
    sock.write_bytes(b"")
    assert w.changed()
    assert not w.changed()


def _cfg(tmp_path):
    return PluginConfig(
        node_name="n1",
        plugin_socket_dir=str(tmp_path),
        kubelet_socket=str(tmp_path / "kubelet.sock"),
        config_file=str(tmp_path / "absent.json"),
        register_interval_s=0.1,
        health_interval_s=0.1,
    )


def test_plugin_session_stop_and_reload(tmp_path):
    client = FakeKubeClient()
    client.add_node(NodeInfo(name="n1"))
    daemon = PluginDaemon(_cfg(tmp_path), client)

    # reload: set HUP while the session is running
    result = {}

    def run():
        result["reason"] = daemon.run_session()

    t = threading.Thread(target=run)
    t.start()
    time.sleep(0.3)
    daemon.request_reload()
    t.join(timeout=10)
    assert not t.is_alive()
    assert result["reason"] == "reload"
    # session registered node annotations through the RegisterLoop
    node = client.get_node("n1")
    assert any("node-handshake" in k for k in node.annotations)

    # stop: pre-set stop -> session winds down immediately
    daemon.request_stop()
    assert daemon.run_session() == "stop"


def test_plugin_session_kubelet_restart(tmp_path):
    client = FakeKubeClient()
    client.add_node(NodeInfo(name="n1"))
    daemon = PluginDaemon(_cfg(tmp_path), client)
    result = {}

    def run():
        result["reason"] = daemon.run_session()

    t = threading.Thread(target=run)
    t.start()
    time.sleep(0.3)
    (tmp_path / "kubelet.sock").write_bytes(b"")  # kubelet came (back) up
    t.join(timeout=10)
    assert not t.is_alive()
    assert result["reason"] == "kubelet-restart"


def test_scheduler_args_defaults_match_reference():
    a = sched_parse_args([])
    assert a.http_bind == "0.0.0.0:443"
    assert a.scheduler_name == "vgpu-scheduler"
    assert a.metrics_bind_address == ":9395"
    a = sched_parse_args(["--default-mem", "2048", "--default-cores", "10"])
    assert (a.default_mem, a.default_cores) == (2048, 10)


def test_zoo_cli_smoke(capsys, monkeypatch):
    """The in-cluster benchmark entry point (benchmarks/ai-benchmark Jobs)
    runs a case end-to-end on CPU."""
    import torch.nn as nn

    from k8s_device_plugin_amd.models import zoo

    tiny = zoo.BenchCase("tiny", lambda: nn.Linear(8, 4), "inference", 2, (8,))
    monkeypatch.setitem(zoo.CASES, "tiny", tiny)
    zoo.main(["--cases", "tiny", "--steps", "3", "--warmup", "1",
              "--device", "cpu"])
    out = capsys.readouterr().out
    assert "tiny" in out and "samples_per_s" in out
