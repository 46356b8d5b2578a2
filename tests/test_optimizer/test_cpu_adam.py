

def test_disk_offload_adam(tmp_path):
    """Disk-backed states (nvme_optimizer equivalent): math matches CPUAdam,
    states live in memory-mapped files under the offload dir."""
    import os

    import torch

    from colossalai_amd.nn.optimizer import CPUAdam, DiskOffloadAdam

    torch.manual_seed(0)
    p1 = torch.randn(1000, requires_grad=True)
    p2 = p1.detach().clone().requires_grad_(True)
    g = torch.randn(1000)
    ref = CPUAdam([p1], lr=1e-2)
    disk = DiskOffloadAdam([p2], lr=1e-2, offload_dir=str(tmp_path))
    for _ in range(3):
        p1.grad = g.clone()
        p2.grad = g.clone()
        ref.step()
        disk.step()
    torch.testing.assert_close(p1, p2)
    files = [f for f in os.listdir(tmp_path) if f.startswith("state_")]
    assert len(files) == 2 and os.path.getsize(tmp_path / files[0]) == 4000
