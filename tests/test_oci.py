"""OCI spec shim tests (reference pkg/oci runtime_exec_test.go:27-100 and
spec modify semantics)."""
import json
import os

import pytest

from k8s_device_plugin_amd.utils.oci import (
    FileSpec,
    SyscallExecRuntime,
    inject_devices,
    inject_env,
    inject_mounts,
)


def write_spec(tmp_path):
    p = tmp_path / "config.json"
    p.write_text(json.dumps({
        "ociVersion": "1.0.2",
        "process": {"env": ["PATH=/usr/bin"]},
        "mounts": [{"destination": "/proc", "type": "proc"}],
    }))
    return p


def test_load_modify_flush_roundtrip(tmp_path):
    p = write_spec(tmp_path)
    fs = FileSpec(str(p))
    fs.load()
    fs.modify(
        inject_env({"VGPU_DEVICE_MEMORY_LIMIT": "73728m", "PATH": "/override-ignored"}),
        inject_mounts([{
            "destination": "/usr/local/vgpu/libvgpu-hip.so",
            "source": "/usr/local/vgpu/libvgpu-hip.so",
            "type": "bind", "options": ["ro", "bind"],
        }]),
    )
    fs.flush()
    out = json.loads(p.read_text())
    env = out["process"]["env"]
    assert "VGPU_DEVICE_MEMORY_LIMIT=73728m" in env
    assert env.count("PATH=/usr/bin") == 1  # existing key not duplicated
    assert any(m["destination"].endswith("libvgpu-hip.so") for m in out["mounts"])
    # idempotent re-apply
    fs2 = FileSpec(str(p))
    fs2.load()
    fs2.modify(inject_env({"VGPU_DEVICE_MEMORY_LIMIT": "73728m"}))
    fs2.flush()
    out2 = json.loads(p.read_text())
    assert out2["process"]["env"].count("VGPU_DEVICE_MEMORY_LIMIT=73728m") == 1


def test_inject_devices(tmp_path):
    p = write_spec(tmp_path)
    fs = FileSpec(str(p))
    fs.load()
    fs.modify(inject_devices(["/dev/null"]))  # exists everywhere
    fs.flush()
    out = json.loads(p.read_text())
    devs = out["linux"]["devices"]
    assert devs[0]["path"] == "/dev/null"
    assert devs[0]["major"] == 1 and devs[0]["minor"] == 3
    assert out["linux"]["resources"]["devices"][0]["allow"] is True


def test_flush_requires_load(tmp_path):
    fs = FileSpec(str(tmp_path / "missing.json"))
    with pytest.raises(RuntimeError):
        fs.flush()


def test_exec_runtime_validates_target(tmp_path):
    with pytest.raises(FileNotFoundError):
        SyscallExecRuntime(str(tmp_path / "no-such-runc"))
    rt = SyscallExecRuntime("/bin/true")
    assert rt.path == "/bin/true"
