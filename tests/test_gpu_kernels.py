"""GPU kernel parity tier: each HIP kernel vs its torch/CPU
reference twin.  Marked gpu; runs on an MI355X via gpurun."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)


@pytest.fixture(scope="module")
def ops():
    from distributedes_amd.ops.hip import HipOps, hip_available, load_error
    assert hip_available(), f"HIP extension must be built on a GPU box: {load_error()}"
    return HipOps()


@pytest.fixture(scope="module")
def ref():
    from distributedes_amd.ops.reference import TorchOps
    return TorchOps()


def test_noise_fill_matches_cpu_philox(ops):
    from distributedes_amd.core.noise import fill_noise_table_cpu
    n = 1 << 16
    gpu = ops.fill_table(n + 3, 987654321, "cuda")   # non-multiple-of-4 tail
    cpu = fill_noise_table_cpu(n + 3, 987654321)
    # same Philox bits; Box-Muller transcendentals differ by libm ulps
    torch.testing.assert_close(gpu.cpu(), cpu, rtol=1e-5, atol=1e-5)


def test_noise_fill_deterministic(ops):
    a = ops.fill_table(1 << 12, 7, "cuda")
    b = ops.fill_table(1 << 12, 7, "cuda")
    assert torch.equal(a, b)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("D", [4096, 5003])   # even/odd row alignment paths
def test_perturb_matches_ref(ops, ref, dtype, D):
    torch.manual_seed(0)
    P = 33
    table = torch.randn(1 << 18, device="cuda")
    theta = torch.randn(D, device="cuda")
    offsets = (torch.randint(0, (table.numel() - D) // 16, (P,),
                             device="cuda") * 16).to(torch.int64)
    signs = torch.where(torch.rand(P, device="cuda") < 0.5, -1.0, 1.0)
    got = ops.perturb(theta, table, offsets, signs, 0.07, out_dtype=dtype)
    want = ref.perturb(theta, table, offsets, signs, 0.07, out_dtype=dtype)
    if dtype == torch.float32:
        torch.testing.assert_close(got, want, rtol=1e-6, atol=1e-6)
    else:
        torch.testing.assert_close(got.float(), want.float(), rtol=1e-2, atol=1e-2)
        # bf16 rounding must match torch's cast exactly
        assert (got.view(torch.uint16) == want.view(torch.uint16)).float().mean() > 0.999


def test_perturb_sigma_vec(ops, ref):
    torch.manual_seed(1)
    P, D = 9, 1024
    table = torch.randn(1 << 16, device="cuda")
    theta = torch.randn(D, device="cuda")
    sigma_vec = torch.rand(D, device="cuda") * 0.2
    offsets = (torch.randint(0, (table.numel() - D) // 16, (P,),
                             device="cuda") * 16).to(torch.int64)
    signs = torch.ones(P, device="cuda")
    got = ops.perturb(theta, table, offsets, signs, 0.0, sigma_vec=sigma_vec)
    want = ref.perturb(theta, table, offsets, signs, 0.0, sigma_vec=sigma_vec)
    torch.testing.assert_close(got, want, rtol=1e-6, atol=1e-6)


@pytest.mark.parametrize("want_sq", [False, True])
def test_weighted_noise_sum_matches_ref(ops, ref, want_sq):
    torch.manual_seed(2)
    P, D = 64, 5003
    table = torch.randn(1 << 18, device="cuda")
    offsets = (torch.randint(0, (table.numel() - D) // 16, (P,),
                             device="cuda") * 16).to(torch.int64)
    signs = torch.where(torch.rand(P, device="cuda") < 0.5, -1.0, 1.0)
    weights = torch.randn(P, device="cuda")
    weights[::5] = 0.0  # exercise the skip branch
    got = ops.weighted_noise_sum(table, offsets, signs, weights, D, want_sq)
    want = ref.weighted_noise_sum(table, offsets, signs, weights, D, want_sq)
    if want_sq:
        torch.testing.assert_close(got[0], want[0], rtol=1e-4, atol=1e-4)
        torch.testing.assert_close(got[1], want[1], rtol=1e-4, atol=1e-4)
    else:
        torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)


def test_native_extension_is_loaded_on_gpu():
    """Guard against silent eager fallback: on a GPU box the engine must pick
    the HIP backend."""
    from distributedes_amd.core.engine import _make_ops
    ops = _make_ops(torch.device("cuda"))
    assert ops.name == "hip"


@pytest.mark.parametrize("table_dtype", [torch.bfloat16])
def test_bf16_table_fill_and_perturb(ops, ref, table_dtype):
    n = 1 << 14
    gpu = ops.fill_table(n, 4242, "cuda", dtype=table_dtype)
    cpu = ref.fill_table(n, 4242, "cpu", dtype=table_dtype)
    # same Philox bits + same bf16 rounding -> near-exact
    torch.testing.assert_close(gpu.cpu().float(), cpu.float(),
                               rtol=1e-2, atol=1e-2)
    D = 1024
    theta = torch.randn(D, device="cuda")
    offsets = (torch.randint(0, (n - D) // 16, (8,), device="cuda") * 16).to(torch.int64)
    signs = torch.ones(8, device="cuda")
    got = ops.perturb(theta, gpu, offsets, signs, 0.1)
    # isolate the perturb op from fill-side bf16-ulp boundaries: feed the ref
    # the GPU-filled table itself
    want = ref.perturb(theta.cpu(), gpu.cpu(), offsets.cpu(), signs.cpu(), 0.1)
    torch.testing.assert_close(got.cpu(), want, rtol=1e-5, atol=1e-6)
    acc = ops.weighted_noise_sum(gpu, offsets, signs,
                                 torch.rand(8, device="cuda"), D)
    assert torch.isfinite(acc).all()


def test_centered_rank_kernel_bitexact(ops):
    from distributedes_amd.core.rank import centered_ranks
    torch.manual_seed(0)
    for n in (2, 17, 1000, 8192):
        f = torch.randn(n, device="cuda")
        f[:: max(1, n // 7)] = 1.25   # force ties
        u = torch.empty_like(f)
        ops.ext.centered_rank(f, u)
        ref = centered_ranks(f.cpu())
        assert torch.equal(u.cpu(), ref)   # integer counting: bit-exact
