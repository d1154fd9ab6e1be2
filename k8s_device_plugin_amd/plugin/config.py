"""Device-plugin configuration.

Reference: cmd/device-plugin/nvidia/vgpucfg.go:15-107 — CLI flags
(--device-split-count default 10, --device-memory-scaling,
--device-cores-scaling, --disable-core-limit, --resource-name) overridable
per node by a JSON ConfigMap mounted at /config/config.json.  Precedence:
node JSON > CLI > defaults.
"""
from __future__ import annotations

import argparse
import json
import logging
import os
from dataclasses import dataclass, field, replace
from typing import List, Optional

log = logging.getLogger(__name__)

DEFAULT_HOOK_PATH = "/usr/local/vgpu"


@dataclass
class PluginConfig:
    node_name: str = ""
    resource_name: str = "amd.com/gpu"
    device_split_count: int = 10
    device_memory_scaling: float = 1.0
    device_cores_scaling: float = 1.0
    disable_core_limit: bool = False
    hook_path: str = DEFAULT_HOOK_PATH       # host dir with libvgpu-hip.so
    plugin_socket_dir: str = "/var/lib/kubelet/device-plugins"
    kubelet_socket: str = "/var/lib/kubelet/device-plugins/kubelet.sock"
    config_file: str = "/config/config.json"
    register_interval_s: float = 30.0
    health_interval_s: float = 5.0
    # per-process runtime reservation charged against the quota inside the
    # container (VGPU_CONTEXT_OVERHEAD); 0 = off
    context_overhead_mb: int = 0
    # "envvar" (ROCR_VISIBLE_DEVICES + DeviceSpecs) or "cdi-annotations"
    # (additionally name CDI devices in the Allocate response annotations;
    # reference --device-list-strategy, main.go:61-70)
    device_list_strategy: str = "envvar"
    cdi_spec_dir: str = "/var/run/cdi"
    # desired compute-partition mode applied at startup: "keep" (default),
    # or SPX/DPX/QPX/CPX — CPX advertises each XCD as a 32-CU hard-isolated
    # device (the MIG-strategy analog; plugin/partition.py)
    compute_partition: str = "keep"
    # time-slicing/replica config (reference rm/device_map.go:37-317):
    # per-device replica counts overriding device_split_count, plus an
    # optional resource rename.  Populated from the node JSON's
    # "timeslicing" block; device key "*" matches all.
    replica_overrides: dict = field(default_factory=dict)


def parse_args(argv: Optional[List[str]] = None) -> PluginConfig:
    p = argparse.ArgumentParser("amd-vgpu-device-plugin")
    c = PluginConfig()
    p.add_argument("--node-name", default=os.environ.get("NodeName", os.environ.get("NODE_NAME", "")))
    p.add_argument("--resource-name", default=c.resource_name)
    p.add_argument("--device-split-count", type=int, default=c.device_split_count)
    p.add_argument("--device-memory-scaling", type=float, default=c.device_memory_scaling)
    p.add_argument("--device-cores-scaling", type=float, default=c.device_cores_scaling)
    p.add_argument("--disable-core-limit", action="store_true", default=False)
    p.add_argument("--hook-path", default=os.environ.get("HOOK_PATH", c.hook_path))
    p.add_argument("--plugin-socket-dir", default=c.plugin_socket_dir)
    p.add_argument("--kubelet-socket", default=c.kubelet_socket)
    p.add_argument("--config-file", default=c.config_file)
    p.add_argument("--device-list-strategy", default=c.device_list_strategy,
                   choices=["envvar", "cdi-annotations"])
    p.add_argument("--context-overhead-mb", type=int, default=c.context_overhead_mb)
    p.add_argument("--cdi-spec-dir", default=c.cdi_spec_dir)
    p.add_argument("--compute-partition", default=c.compute_partition,
                   choices=["keep", "SPX", "DPX", "QPX", "CPX"])
    a = p.parse_args(argv)
    cfg = PluginConfig(
        node_name=a.node_name,
        resource_name=a.resource_name,
        device_split_count=a.device_split_count,
        device_memory_scaling=a.device_memory_scaling,
        device_cores_scaling=a.device_cores_scaling,
        disable_core_limit=a.disable_core_limit,
        hook_path=a.hook_path,
        plugin_socket_dir=a.plugin_socket_dir,
        kubelet_socket=a.kubelet_socket,
        config_file=a.config_file,
        device_list_strategy=a.device_list_strategy,
        cdi_spec_dir=a.cdi_spec_dir,
        context_overhead_mb=a.context_overhead_mb,
        compute_partition=a.compute_partition,
    )
    return apply_node_config(cfg)


def apply_node_config(cfg: PluginConfig) -> PluginConfig:
    """Per-node JSON override (vgpucfg.go:81-107 readFromConfigFile)."""
    path = cfg.config_file
    if not path or not os.path.exists(path):
        return cfg
    try:
        with open(path) as f:
            data = json.load(f)
    except (OSError, json.JSONDecodeError) as e:
        log.error("bad node config %s: %s", path, e)
        return cfg
    for entry in data.get("nodeconfig", []):
        if entry.get("name") == cfg.node_name:
            log.info("applying node config override for %s", cfg.node_name)
            cfg = replace(
                cfg,
                device_split_count=int(entry.get("devicesplitcount", cfg.device_split_count)),
                device_memory_scaling=float(entry.get("devicememoryscaling", cfg.device_memory_scaling)),
                device_cores_scaling=float(entry.get("devicecorescaling", cfg.device_cores_scaling)),
                compute_partition=str(entry.get("computepartition", cfg.compute_partition)),
            )
            break
    return apply_time_slicing(cfg, data.get("timeslicing") or {})


def apply_time_slicing(cfg: PluginConfig, ts: dict) -> PluginConfig:
    """Reference rm/device_map.go:37-317 replica machinery, MI355X-sized:

    "timeslicing": {"resources": [
        {"name": "amd.com/gpu", "rename": "amd.com/gpu.shared",
         "replicas": 20, "devices": ["GPU-abc", ...]}   # devices optional
    ]}

    A matching entry overrides the fan-out (replicas) for the listed
    devices ("*" / omitted = all) and may rename the advertised resource.
    """
    overrides = dict(cfg.replica_overrides)
    resource = cfg.resource_name
    for entry in ts.get("resources", []):
        if entry.get("name") and entry["name"] != cfg.resource_name:
            continue
        replicas = int(entry.get("replicas", 0))
        if replicas <= 0:
            log.warning("timeslicing entry without positive replicas: %r",
                        entry)
            continue
        for dev in entry.get("devices") or ["*"]:
            overrides[str(dev)] = replicas
        if entry.get("rename"):
            resource = str(entry["rename"])
            log.info("timeslicing renames resource %s -> %s",
                     cfg.resource_name, resource)
    if overrides == cfg.replica_overrides and resource == cfg.resource_name:
        return cfg
    return replace(cfg, replica_overrides=overrides, resource_name=resource)
