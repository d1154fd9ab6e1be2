"""Codec round-trip tests (reference analog: pkg/util/util_test.go:33-64)."""
import pytest

from k8s_device_plugin_amd.utils import codec
from k8s_device_plugin_amd.utils.types import ContainerDevice, DeviceInfo


def mk_dev(i, health=True):
    return DeviceInfo(
        id=f"MI355X-GPU-{i:02d}",
        count=10,
        devmem=294912,
        devcore=100,
        type="AMD-Instinct-MI355X",
        numa=i // 4,
        health=health,
    )


class TestNodeDevices:
    def test_roundtrip(self):
        devs = [mk_dev(i) for i in range(8)]
        s = codec.encode_node_devices(devs)
        back = codec.decode_node_devices(s)
        assert len(back) == 8
        for a, b in zip(devs, back):
            assert (a.id, a.count, a.devmem, a.devcore, a.type, a.numa, a.health) == (
                b.id, b.count, b.devmem, b.devcore, b.type, b.numa, b.health)

    def test_wire_format(self):
        s = codec.encode_node_devices([mk_dev(0)])
        assert s == "MI355X-GPU-00,10,294912,100,AMD-Instinct-MI355X,0,true:"

    def test_unhealthy(self):
        s = codec.encode_node_devices([mk_dev(0, health=False)])
        assert ",false:" in s
        assert codec.decode_node_devices(s)[0].health is False

    def test_decode_no_colon_raises(self):
        with pytest.raises(codec.CodecError):
            codec.decode_node_devices("garbage")

    def test_decode_wrong_fields_raises(self):
        with pytest.raises(codec.CodecError):
            codec.decode_node_devices("a,b,c:")

    def test_empty_entries_skipped(self):
        assert codec.decode_node_devices(":") == []


class TestContainerDevices:
    def test_roundtrip(self):
        cd = [
            ContainerDevice(uuid="MI355X-GPU-00", type="AMD", usedmem=73728, usedcores=25),
            ContainerDevice(uuid="MI355X-GPU-01", type="AMD", usedmem=147456, usedcores=50),
        ]
        s = codec.encode_container_devices(cd)
        assert s == ("MI355X-GPU-00,AMD,73728,25:MI355X-GPU-01,AMD,147456,50:")
        back = codec.decode_container_devices(s)
        assert [(d.uuid, d.type, d.usedmem, d.usedcores) for d in back] == [
            ("MI355X-GPU-00", "AMD", 73728, 25),
            ("MI355X-GPU-01", "AMD", 147456, 50),
        ]

    def test_empty(self):
        assert codec.decode_container_devices("") == []
        assert codec.encode_container_devices([]) == ""


class TestPodDevices:
    def test_roundtrip(self):
        checklist = {"AMD": "vgpu.amd.com/devices-to-allocate"}
        pd = {
            "AMD": [
                [ContainerDevice(uuid="u0", type="AMD", usedmem=1000, usedcores=10)],
                [],  # a container with no GPUs keeps its slot
                [ContainerDevice(uuid="u1", type="AMD", usedmem=2000, usedcores=20)],
            ]
        }
        annos = codec.encode_pod_devices(checklist, pd)
        assert annos["vgpu.amd.com/devices-to-allocate"] == (
            "u0,AMD,1000,10:;;u1,AMD,2000,20:;"
        )
        back = codec.decode_pod_devices(checklist, annos)
        assert len(back["AMD"]) == 3
        assert [(d.uuid, d.usedmem) for d in back["AMD"][0]] == [("u0", 1000)]
        assert back["AMD"][1] == []
        assert [(d.uuid, d.usedmem) for d in back["AMD"][2]] == [("u1", 2000)]

    def test_decode_missing_anno(self):
        assert codec.decode_pod_devices({"AMD": "key"}, {}) == {}
        assert codec.decode_pod_devices({"AMD": "key"}, {"other": "x"}) == {}
