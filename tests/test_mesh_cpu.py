"""DeviceMesh axis-group collectives on a 2x2 mesh over gloo ws=4
(reference: collective plumbing tests without GPUs)."""
import torch

from dist_utils import run_distributed

import alpa_amd as aa


def _mesh22_worker(rank, world_size):
    mesh = aa.full_mesh((2, 2))
    # axis 1 (rows): ranks {0,1} and {2,3}; axis 0 (cols): {0,2}, {1,3}
    t = torch.tensor([float(rank)])
    row = t.clone()
    mesh.all_reduce(row, axis=1)
    col = t.clone()
    mesh.all_reduce(col, axis=0)
    full = t.clone()
    mesh.all_reduce(full)
    # all-gather along axis 1
    out = torch.zeros(2)
    mesh.all_gather(out, t, axis=1)
    return {"row": row, "col": col, "full": full, "gather": out,
            "coord": torch.tensor(mesh.coord)}


def test_mesh_2x2_axis_collectives():
    res = run_distributed(_mesh22_worker, world_size=4)
    # row sums: {0+1, 2+3}; col sums: {0+2, 1+3}
    expect_row = [1.0, 1.0, 5.0, 5.0]
    expect_col = [2.0, 4.0, 2.0, 4.0]
    for r in range(4):
        assert float(res[r]["row"]) == expect_row[r]
        assert float(res[r]["col"]) == expect_col[r]
        assert float(res[r]["full"]) == 6.0
    assert res[0]["gather"].tolist() == [0.0, 1.0]
    assert res[3]["gather"].tolist() == [2.0, 3.0]
    assert res[2]["coord"].tolist() == [1, 0]


def _bcast_worker(rank, world_size):
    mesh = aa.full_mesh((1, world_size))
    t = torch.tensor([float(rank) + 1])
    mesh.broadcast(t, src_coord=1, axis=1)
    return t


def test_mesh_broadcast():
    res = run_distributed(_bcast_worker, world_size=2)
    assert all(float(r) == 2.0 for r in res)


def test_virtual_mesh_slicing():
    """VirtualMesh (reference VirtualPhysicalMesh slice_2d:1888): slice
    an unallocated mesh into submeshes and reshape without creating
    process groups."""
    from alpa_amd.mesh import VirtualMesh, full_virtual_mesh
    vm = VirtualMesh(tuple(range(8)), (2, 4))
    sub = vm.slice_ranks(0, 4)
    assert sub.ranks == (0, 1, 2, 3) and sub.num_devices == 4
    r = sub.reshape((2, 2))
    assert r.shape == (2, 2) and r.ranks == (0, 1, 2, 3)
    assert r.rank_grid().tolist() == [[0, 1], [2, 3]]
    assert full_virtual_mesh(8).num_devices == 8


def _alive_worker(rank, world_size):
    import alpa_amd as aa
    mesh = aa.DeviceMesh(list(range(world_size)), (world_size, 1))
    return mesh.check_alive(timeout_s=30.0)


def test_check_alive_probe():
    """Liveness probe (reference MeshHostWorker.check_alive +
    pipeline_check_alive polling): healthy mesh answers True on every
    rank within the timeout."""
    from dist_utils import run_distributed
    results = run_distributed(_alive_worker, world_size=2, timeout=120)
    assert all(results), results


def test_check_alive_single_process():
    import alpa_amd as aa
    from alpa_amd.mesh import VirtualMesh  # noqa: F401
    mesh = aa.DeviceMesh([0], (1, 1))
    assert mesh.check_alive()
