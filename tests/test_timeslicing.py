"""Time-slicing/replica config block (reference rm/device_map.go:37-317).

The node JSON ConfigMap may carry a "timeslicing" block: per-device
replica counts overriding --device-split-count and an optional resource
rename — the MI355X rendering of the reference's config-file replica
machinery (which it partially disables upstream, main.go:316-352; we keep
the useful subset: fan-out control per device + rename)."""
import json

from k8s_device_plugin_amd.plugin.config import (
    PluginConfig,
    apply_node_config,
    apply_time_slicing,
)
from k8s_device_plugin_amd.plugin.rm import ResourceManager
from test_plugin import make_kfd_tree

from k8s_device_plugin_amd.plugin.kfd import enumerate_gpus


def _gpus(tmp_path, n=2):
    topo, pci = make_kfd_tree(tmp_path, n_gpus=n)
    return enumerate_gpus(str(topo), str(pci))


class TestReplicaOverrides:
    def test_star_overrides_all(self, tmp_path):
        gpus = _gpus(tmp_path)
        rm = ResourceManager(gpus, split_count=10,
                             replica_overrides={"*": 20})
        assert len(rm.fake_devices()) == 40
        assert all(i.count == 20 for i in rm.api_devices())

    def test_per_device_override(self, tmp_path):
        gpus = _gpus(tmp_path)
        rm = ResourceManager(
            gpus, split_count=10,
            replica_overrides={gpus[0].uuid: 3})
        counts = {i.id: i.count for i in rm.api_devices()}
        assert counts[gpus[0].uuid] == 3
        assert counts[gpus[1].uuid] == 10
        assert len(rm.fake_devices()) == 13

    def test_no_override_uses_split_count(self, tmp_path):
        gpus = _gpus(tmp_path)
        rm = ResourceManager(gpus, split_count=4)
        assert len(rm.fake_devices()) == 8


class TestConfigParsing:
    def test_timeslicing_block(self, tmp_path):
        cfgfile = tmp_path / "config.json"
        cfgfile.write_text(json.dumps({
            "timeslicing": {"resources": [
                {"name": "amd.com/gpu", "replicas": 20,
                 "devices": ["GPU-a", "GPU-b"]},
            ]},
        }))
        cfg = PluginConfig(node_name="n1", config_file=str(cfgfile))
        out = apply_node_config(cfg)
        assert out.replica_overrides == {"GPU-a": 20, "GPU-b": 20}
        assert out.resource_name == "amd.com/gpu"

    def test_rename(self):
        cfg = PluginConfig(node_name="n1")
        out = apply_time_slicing(cfg, {"resources": [
            {"name": "amd.com/gpu", "rename": "amd.com/gpu.shared",
             "replicas": 16},
        ]})
        assert out.resource_name == "amd.com/gpu.shared"
        assert out.replica_overrides == {"*": 16}

    def test_other_resource_ignored(self):
        cfg = PluginConfig(node_name="n1")
        out = apply_time_slicing(cfg, {"resources": [
            {"name": "other.com/gpu", "replicas": 16},
        ]})
        assert out.replica_overrides == {}
        assert out.resource_name == "amd.com/gpu"

    def test_nonpositive_replicas_ignored(self):
        cfg = PluginConfig(node_name="n1")
        out = apply_time_slicing(cfg, {"resources": [
            {"name": "amd.com/gpu", "replicas": 0},
        ]})
        assert out.replica_overrides == {}
