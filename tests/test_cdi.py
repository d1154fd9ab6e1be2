"""CDI spec generation + annotation codec (reference pkg/.../cdi/cdi.go,
TestCDIAllocateResponse in plugin/server_test.go:30-173)."""
import json

from k8s_device_plugin_amd.plugin import cdi
from k8s_device_plugin_amd.plugin.kfd import PhysicalGPU


def mkgpu(i):
    return PhysicalGPU(
        index=i, node_id=i + 1, gpu_id=1000 + i, uuid=f"GPU-{i:016x}",
        cu_count=256, mem_bytes=288 << 30, numa_node=i % 2,
        pci_bdf=f"0000:0{i}:00.0", drm_render_minor=128 + i,
        gfx_target="gfx950",
    )


def test_spec_shape():
    spec = cdi.generate_spec([mkgpu(0), mkgpu(1)], hook_path="/usr/local")
    assert spec["kind"] == "amd.com/gpu"
    assert spec["cdiVersion"] == "0.5.0"
    assert len(spec["devices"]) == 3  # 2 GPUs + the composite "all"
    d0 = spec["devices"][0]
    assert d0["name"] == "GPU-0000000000000000"
    paths = [n["path"] for n in d0["containerEdits"]["deviceNodes"]]
    assert paths == ["/dev/kfd", "/dev/dri/card0", "/dev/dri/renderD128"]
    mounts = d0["containerEdits"]["mounts"]
    assert mounts[0]["hostPath"] == "/usr/local/vgpu/libvgpu-hip.so"
    assert mounts[1]["containerPath"] == "/etc/ld.so.preload"


def test_spec_without_hook_path_has_no_mounts():
    spec = cdi.generate_spec([mkgpu(0)])
    assert "mounts" not in spec["devices"][0]["containerEdits"]


def test_write_spec_atomic(tmp_path):
    path = cdi.write_spec([mkgpu(0)], spec_dir=str(tmp_path))
    spec = json.loads(open(path).read())
    assert spec["kind"] == "amd.com/gpu"
    assert path.endswith("amd.com-gpu.json")
    # rewrite over existing
    cdi.write_spec([mkgpu(0), mkgpu(1)], spec_dir=str(tmp_path))
    spec = json.loads(open(path).read())
    assert len(spec["devices"]) == 3  # 2 GPUs + the composite "all"


def test_annotation_roundtrip():
    ann = cdi.annotations(["GPU-a", "GPU-b"])
    assert ann == {"cdi.k8s.io/vgpu-amd": "amd.com/gpu=GPU-a,amd.com/gpu=GPU-b"}
    assert cdi.parse_annotation(ann["cdi.k8s.io/vgpu-amd"]) == ["GPU-a", "GPU-b"]
    assert cdi.annotations([]) == {}
    assert cdi.parse_annotation("nvidia.com/gpu=X") == []


def test_all_composite_device(tmp_path):
    """nvcdi-style composite: one CDI device named "all" holding every
    GPU's nodes exactly once (/dev/kfd deduplicated)."""
    from test_plugin import make_kfd_tree

    from k8s_device_plugin_amd.plugin import cdi
    from k8s_device_plugin_amd.plugin.kfd import enumerate_gpus

    topo, pci = make_kfd_tree(tmp_path, n_gpus=2)
    gpus = enumerate_gpus(str(topo), str(pci))
    spec = cdi.generate_spec(gpus)
    byname = {d["name"]: d for d in spec["devices"]}
    assert "all" in byname
    paths = [n["path"] for n in byname["all"]["containerEdits"]["deviceNodes"]]
    assert paths.count("/dev/kfd") == 1
    assert "/dev/dri/renderD128" in paths and "/dev/dri/renderD129" in paths
