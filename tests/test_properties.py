"""Property-based invariants (hypothesis): wire codec round-trips, CU-mask
allocator disjointness, scheduler fit never over-commits."""
from hypothesis import given, settings, strategies as st

from k8s_device_plugin_amd.ops.cumask import CoreMaskAllocator, cus_for_percent
from k8s_device_plugin_amd.scheduler.score import NodeUsage, fit_in_certain_device
from k8s_device_plugin_amd.utils.codec import (
    decode_container_devices,
    decode_node_devices,
    encode_container_devices,
    encode_node_devices,
    decode_node_xgmi,
    encode_node_xgmi,
)
from k8s_device_plugin_amd.utils.types import (
    ContainerDevice,
    ContainerDeviceRequest,
    DeviceInfo,
    DeviceUsage,
)

uuid_st = st.from_regex(r"GPU-[0-9a-f]{4,16}", fullmatch=True)


# min_size=1: the reference's decoder errors on an empty annotation by
# design (util.go:78-109), mirrored here
@given(st.lists(st.tuples(uuid_st, st.integers(1, 100), st.integers(0, 1 << 40),
                          st.integers(0, 100), st.integers(0, 7),
                          st.booleans()), min_size=1, max_size=16))
@settings(max_examples=50, deadline=None)
def test_node_devices_roundtrip(items):
    devs = [DeviceInfo(id=u, count=c, devmem=m, devcore=cc,
                       type="AMD-Instinct-MI355X", numa=n, health=h, index=i)
            for i, (u, c, m, cc, n, h) in enumerate(items)]
    decoded = decode_node_devices(encode_node_devices(devs))
    assert len(decoded) == len(devs)
    for a, b in zip(decoded, devs):
        assert (a.id, a.count, a.devmem, a.devcore, a.numa, a.health) == \
            (b.id, b.count, b.devmem, b.devcore, b.numa, b.health)


@given(st.lists(st.tuples(uuid_st, st.integers(0, 1 << 30), st.integers(0, 100)),
                max_size=8))
@settings(max_examples=50, deadline=None)
def test_container_devices_roundtrip(items):
    devs = [ContainerDevice(uuid=u, type="AMD", usedmem=m, usedcores=c)
            for (u, m, c) in items]
    decoded = decode_container_devices(encode_container_devices(devs))
    assert len(decoded) == len(devs)
    for a, b in zip(decoded, devs):
        assert (a.uuid, a.usedmem, a.usedcores) == (b.uuid, b.usedmem, b.usedcores)


@given(st.dictionaries(uuid_st, st.lists(uuid_st, max_size=7), max_size=8))
@settings(max_examples=50, deadline=None)
def test_xgmi_roundtrip(adj):
    adj = {u: sorted(set(p)) for u, p in adj.items()}
    assert decode_node_xgmi(encode_node_xgmi(adj)) == adj


@given(st.lists(st.integers(1, 100), min_size=1, max_size=12))
@settings(max_examples=100, deadline=None)
def test_cumask_allocations_disjoint(percents):
    """However requests arrive, granted masks are pairwise disjoint, sized
    to cus_for_percent, and free() returns the CUs."""
    alloc = CoreMaskAllocator()
    granted = []
    for p in percents:
        m = alloc.alloc("GPU-x", p)
        if m is not None:
            granted.append((p, m))
    union = 0
    for p, m in granted:
        assert bin(m).count("1") == cus_for_percent(p)
        assert union & m == 0
        union |= m
    assert alloc.used_count("GPU-x") == bin(union).count("1")
    for _, m in granted:
        alloc.free("GPU-x", m)
    assert alloc.used_count("GPU-x") == 0


@given(
    st.lists(st.tuples(st.integers(0, 10), st.integers(0, 294912),
                       st.integers(0, 100)), min_size=1, max_size=8),
    st.integers(1, 4),
    st.integers(0, 294912),
    st.integers(0, 100),
)
@settings(max_examples=100, deadline=None)
def test_fit_never_overcommits(devstates, nums, memreq, cores):
    devices = [
        DeviceUsage(id=f"GPU-{i}", index=i, used=u, count=10, usedmem=um,
                    totalmem=294912, totalcore=100, usedcores=uc, numa=0,
                    type="AMD-Instinct-MI355X", health=True)
        for i, (u, um, uc) in enumerate(devstates)
    ]
    node = NodeUsage(devices=devices)
    req = ContainerDeviceRequest(nums=nums, type="AMD", memreq=memreq,
                                 mem_percentage_req=101, coresreq=cores)
    fit, devs = fit_in_certain_device(node, req, {})
    if fit:
        picks = devs["AMD"]
        assert len(picks) == nums
        assert len({p.uuid for p in picks}) == nums  # distinct devices
        for p in picks:
            d = node.devices[p.idx]
            assert d.id == p.uuid
            assert d.usedmem + p.usedmem <= d.totalmem
            assert d.usedcores + p.usedcores <= d.totalcore
            assert d.used < d.count


@given(st.text(alphabet=st.characters(blacklist_categories=("Cs",)),
               max_size=200))
@settings(max_examples=200, deadline=None)
def test_decoders_never_crash_on_garbage(s):
    """Malformed node/pod annotations must raise CodecError (or parse),
    never crash with an arbitrary exception — the scheduler ingests these
    from any node object in the cluster."""
    from k8s_device_plugin_amd.utils.codec import (
        CodecError,
        decode_container_devices,
        decode_node_devices,
        decode_pod_devices,
    )
    from k8s_device_plugin_amd.utils.types import IN_REQUEST_DEVICES

    for fn in (decode_node_devices, decode_container_devices):
        try:
            fn(s)
        except CodecError:
            pass
    try:
        decode_pod_devices(IN_REQUEST_DEVICES,
                           {IN_REQUEST_DEVICES["AMD"]: s})
    except CodecError:
        pass
    decode_node_xgmi(s)  # xgmi decode is total: any text -> dict


def test_codec_roundtrip_with_cpx_partition_uuids():
    """CPX partition UUIDs carry a '.' suffix (GPU-<id>.<k>) — the wire
    codec's separators are ',' ':' ';' so dotted IDs must round-trip
    through both the node and container encodings."""
    from k8s_device_plugin_amd.utils.codec import (
        decode_container_devices,
        decode_node_devices,
        encode_container_devices,
        encode_node_devices,
    )
    from k8s_device_plugin_amd.utils.types import ContainerDevice, DeviceInfo

    devs = [DeviceInfo(id=f"GPU-00abc.{k}", count=2, devmem=36864,
                       devcore=100, type="AMD-Instinct-MI355X-CPX",
                       numa=0, health=True, index=k) for k in range(8)]
    back = decode_node_devices(encode_node_devices(devs))
    assert [d.id for d in back] == [d.id for d in devs]

    cds = [ContainerDevice(uuid=f"GPU-00abc.{k}", type="AMD",
                           usedmem=36864, usedcores=100) for k in range(3)]
    back2 = decode_container_devices(encode_container_devices(cds))
    assert [c.uuid for c in back2] == [c.uuid for c in cds]
