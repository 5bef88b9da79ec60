"""Torch + ceph_amd runtime coexistence (the 8-GPU SCALE-run shape).

torch bundles its own HIP runtime; if ceph_amd loads /opt/rocm's first,
torch.cuda sees zero devices (two runtimes in one process). bench.py
therefore initialises torch before ceph_amd for distributed runs — this
test pins the working order and the parity of our kernels when running on
torch's already-initialised runtime."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_torch_first_then_ceph_amd_parity():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("torch sees no GPU")
    torch.cuda.set_device(0)
    _ = torch.zeros(8, device="cuda")  # force torch HIP runtime init

    import ceph_amd
    import oracle
    assert ceph_amd.device_count() >= 1
    ctx = ceph_amd.EcContext(4, 2, "reed_sol_van", device=0)
    try:
        rng = np.random.default_rng(1)
        data = [rng.integers(0, 256, 4096, dtype=np.uint8)
                for _ in range(4)]
        got = ctx.encode_chunks(data)
        want = oracle.encode("reed_sol_van", 4, 2, data)
        for g, w in zip(got, want):
            assert np.array_equal(g, w)
    finally:
        ctx.close()
