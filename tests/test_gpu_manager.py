"""AMD GPU manager tests with a fake KFD sysfs tree (reference analog:
pkg/worker/nvidia_test.go fakes GPUInfoClient)."""

import os

import pytest

from beta9_amd.worker.gpu import (
    AllocationError,
    GPUManager,
    discover_devices,
    make_fake_topology,
)


@pytest.fixture
def topo(tmp_path):
    return make_fake_topology(str(tmp_path), n_gpus=8)


def test_discovery(topo):
    devs = discover_devices(topo)
    assert len(devs) == 8
    assert devs[0].gfx_target.startswith("gfx95")
    assert devs[0].render_path == "/dev/dri/renderD128"
    assert devs[3].vram_bytes == 288 * (1 << 30)
    # full xGMI mesh
    for d in devs:
        assert len(d.xgmi_peers) == 7


def test_discovery_no_kfd(tmp_path):
    assert discover_devices(str(tmp_path / "missing")) == []


def test_assign_unassign(topo):
    m = GPUManager(topology_path=topo)
    assert m.gpu_type == "MI355X"
    assert m.free_count == 8
    a = m.assign("c1", 2)
    assert len(a.ordinals) == 2
    assert m.free_count == 6
    assert a.env["HIP_VISIBLE_DEVICES"] == ",".join(map(str, a.ordinals))
    assert a.env["ROCR_VISIBLE_DEVICES"] == a.env["HIP_VISIBLE_DEVICES"]
    assert "/dev/kfd" in a.devices and len(a.devices) == 3
    m.unassign("c1")
    assert m.free_count == 8


def test_exhaustion(topo):
    m = GPUManager(topology_path=topo)
    m.assign("c1", 8)
    with pytest.raises(AllocationError):
        m.assign("c2", 1)


def test_xgmi_aware_choice(tmp_path):
    """Two 4-GPU xGMI islands: a 4-GPU request must stay on one island."""
    nodes = os.path.join(str(tmp_path), "nodes")
    import beta9_amd.worker.gpu as g

    # build two islands manually
    os.makedirs(os.path.join(nodes, "0"), exist_ok=True)
    with open(os.path.join(nodes, "0", "properties"), "w") as f:
        f.write("simd_count 0\n")
    for i in range(8):
        nd = os.path.join(nodes, str(i + 1))
        os.makedirs(os.path.join(nd, "io_links"), exist_ok=True)
        with open(os.path.join(nd, "properties"), "w") as f:
            f.write(f"simd_count 1024\ngfx_target_version 90500\ndrm_render_minor {128+i}\n")
        island = i // 4
        li = 0
        for j in range(island * 4, island * 4 + 4):
            if j == i:
                continue
            ld = os.path.join(nd, "io_links", str(li))
            os.makedirs(ld, exist_ok=True)
            with open(os.path.join(ld, "properties"), "w") as f:
                f.write(f"type {g.IOLINK_TYPE_XGMI}\nnode_from {i+1}\nnode_to {j+1}\n")
            li += 1
    m = GPUManager(topology_path=nodes)
    # occupy GPU 0 so naive first-fit would pick {1,2,3,4} crossing islands
    m.assign("blocker", 1)
    assert m._assigned["blocker"] == [0]
    a = m.assign("c1", 4)
    assert a.ordinals == [4, 5, 6, 7]  # the intact island


def test_rendezvous_provisioning(topo):
    m = GPUManager(topology_path=topo)
    a = m.assign("c1", 8)
    env = m.provision_rendezvous(a)
    assert env["WORLD_SIZE"] == "8"
    assert env["MASTER_ADDR"] == "127.0.0.1"
    assert int(env["MASTER_PORT"]) > 0
    assert env["NCCL_SOCKET_IFNAME"] == "lo"
    assert env["HSA_ENABLE_IPC_MODE_LEGACY"] == "0"
    # single GPU: no rendezvous
    b = m.unassign("c1") or m.assign("c2", 1)
    assert m.provision_rendezvous(b) == {}


def test_device_spec(topo):
    m = GPUManager(topology_path=topo)
    spec = m.device_spec()
    assert spec["kind"] == "amd.com/gpu"
    assert len(spec["devices"]) == 8
    assert spec["devices"][0]["nodes"][0] == "/dev/kfd"


def test_fractional_packing(topo):
    """Thunder-analog GPU sharing: fractional containers best-fit pack onto
    one shared device, keeping whole GPUs free for exclusive requests
    (reference: pkg/worker/thunder.go:31-138 selection worker.go:128-137)."""
    m = GPUManager(topology_path=topo)
    a = m.assign("f1", 1, fraction=0.5)
    b = m.assign("f2", 1, fraction=0.5)
    # both halves land on the SAME device; 7 whole GPUs untouched
    assert a.ordinals == b.ordinals
    assert m.free_count == 7
    assert a.env["B9_GPU_MEMORY_FRACTION"] == "0.5"
    assert len(a.devices) == 2 and "/dev/kfd" in a.devices
    # the shared device is exhausted: a third half must take a NEW device
    c = m.assign("f3", 1, fraction=0.5)
    assert c.ordinals != a.ordinals
    assert m.free_count == 6
    # release one half of the first device -> next half packs back onto it
    m.unassign("f1")
    d = m.assign("f4", 1, fraction=0.5)
    assert d.ordinals == a.ordinals
    # releasing everything restores whole-device capacity
    for cid in ("f2", "f3", "f4"):
        m.unassign(cid)
    assert m.free_count == 8
    assert m.free_units == 8.0


def test_fractional_vs_exclusive(topo):
    m = GPUManager(topology_path=topo)
    m.assign("big", 7)  # one whole device left
    m.assign("f1", 1, fraction=0.25)
    # the only device is now shared: exclusive requests must fail...
    assert m.free_count == 0
    with pytest.raises(AllocationError):
        m.assign("excl", 1)
    # ...but fractions that still fit succeed, and oversubscription fails
    m.assign("f2", 1, fraction=0.5)
    assert not m.can_fit(1, 0.5)
    with pytest.raises(AllocationError):
        m.assign("f3", 1, fraction=0.5)
    m.assign("f3", 1, fraction=0.25)
    assert m.free_units == pytest.approx(0.0)


def test_fractional_invalid(topo):
    m = GPUManager(topology_path=topo)
    with pytest.raises(AllocationError):
        m.assign("bad", 1, fraction=-0.5)
