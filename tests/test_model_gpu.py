"""GPU model-level tests: full FNO on cuda vs the same weights on CPU."""

import pytest
import torch

import dfno_amd as dfno
from oracle import oracle_fno

pytestmark = pytest.mark.gpu


def _copy_params(dst, src):
    with torch.no_grad():
        sd = {k: v.to(next(iter(dst.parameters())).device if any(True for _ in dst.parameters()) else "cpu")
              for k, v in src.state_dict().items()}
        dst.load_state_dict({k: v for k, v in sd.items()})


@pytest.mark.parametrize("dtype,tt", [(torch.float32, 3e-4), (torch.float64, 1e-10)])
def test_fno_gpu_matches_cpu(dtype, tt):
    torch.manual_seed(0)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1))
    in_shape = [2, 3, 16, 12, 4]
    cpu_model = dfno.DistributedFNONd(P_x, in_shape, 6, 12, (4, 3, 2),
                                      num_blocks=2, dtype=dtype)
    gpu_model = dfno.DistributedFNONd(P_x, in_shape, 6, 12, (4, 3, 2),
                                      num_blocks=2, device=torch.device("cuda"),
                                      dtype=dtype)
    gpu_model.load_state_dict({k: v.cuda() for k, v in cpu_model.state_dict().items()})

    x = torch.rand(*in_shape, dtype=dtype)
    y_cpu = cpu_model(x)
    y_gpu = gpu_model(x.cuda())
    assert torch.allclose(y_gpu.cpu(), y_cpu, rtol=tt, atol=tt), \
        f"max {(y_gpu.cpu() - y_cpu).abs().max()}"

    # gradients
    gy = torch.rand_like(y_cpu)
    y_cpu.backward(gy)
    y_gpu.backward(gy.cuda())
    for (n, pc), (_, pg) in zip(cpu_model.named_parameters(), gpu_model.named_parameters()):
        if pc.grad is None:
            assert pg.grad is None or pg.grad.numel() == 0
            continue
        assert torch.allclose(pg.grad.cpu(), pc.grad, rtol=tt * 10, atol=tt * 10), \
            f"{n}: max {(pg.grad.cpu() - pc.grad).abs().max()}"


def test_fno_gpu_matches_reference_oracle_fp64():
    torch.manual_seed(1)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1, 1))
    in_shape = [1, 2, 10, 10, 8, 1]
    model = dfno.DistributedFNONd(P_x, in_shape, 6, 8, (3, 3, 3, 2),
                                  num_blocks=2, device=torch.device("cuda"),
                                  dtype=torch.float64)
    x = torch.rand(*in_shape, dtype=torch.float64, device="cuda")
    y = model(x)

    cpu_model = dfno.DistributedFNONd(P_x, in_shape, 6, 8, (3, 3, 3, 2),
                                      num_blocks=2, dtype=torch.float64)
    cpu_model.load_state_dict({k: v.cpu() for k, v in model.state_dict().items()})
    y_ref = oracle_fno(cpu_model, x.cpu())
    assert torch.allclose(y.cpu(), y_ref, rtol=1e-10, atol=1e-10), \
        f"max {(y.cpu() - y_ref).abs().max()}"


def test_native_extension_required_on_gpu():
    # the HIP extension must be importable on a GPU box — ops refuse to run
    # on CUDA tensors without it (no silent eager fallback)
    from dfno_amd import _ext
    assert _ext.get(required=True) is not None
    assert _ext._find_prebuilt() is not None, "extension .so must be in-tree"
