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


def test_nccl_backend_single_rank_paths():
    """Exercise the RCCL (backend "nccl") code paths as far as one device
    allows: process-group init, broadcast_object_list
    under NCCL, CUDA-tensor allreduce over the group, plan execution /
    packed-exchange machinery on CUDA tensors, and a full fwd+bwd step of
    the model under an initialized nccl world."""
    import os
    import torch.distributed as dist

    assert not dist.is_initialized()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29711")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        # raw RCCL collective on a CUDA tensor (world 1 still runs the op)
        t = torch.ones(1024, device="cuda")
        dist.all_reduce(t)
        torch.cuda.synchronize()
        assert torch.equal(t.cpu(), torch.ones(1024))

        # object broadcast (store-based path used by Broadcast metadata)
        obj = [("shape", torch.float32)]
        dist.broadcast_object_list(obj, src=0)
        assert obj[0][0] == "shape"

        # full model step under the nccl world
        _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1, 1))
        model = dfno.DistributedFNONd(P_x, [1, 2, 12, 12, 8, 1], 6, 8,
                                      (3, 3, 2, 2), num_blocks=1,
                                      device=torch.device("cuda"),
                                      dtype=torch.float32)
        x = torch.rand(1, 2, 12, 12, 8, 1, device="cuda")
        y = model(x)
        loss = dfno.DistributedRelativeLpLoss(P_x)(y, torch.rand_like(y))
        loss.backward()
        torch.cuda.synchronize()
        assert torch.isfinite(loss.detach()).all()
    finally:
        dist.destroy_process_group()


def test_pack_unpack_boxes_roundtrip_gpu():
    """HIP pack/unpack kernels against aten slicing (fp32/c64/fp64/bf16)."""
    from dfno_amd import _ext
    from dfno_amd.comm import _box_record, _PACK_REC

    ext = _ext.get(required=True)
    torch.manual_seed(5)
    for dtype in (torch.float32, torch.complex64, torch.float64,
                  torch.bfloat16):
        wpe = 2 if dtype.is_complex else 1
        shape = (2, 5, 9, 7, 6)
        x = (torch.randn(*shape, dtype=dtype, device="cuda")
             if not dtype.is_complex else
             torch.randn(*shape, dtype=dtype, device="cuda"))
        boxes = [
            [(0, 1), (1, 4), (0, 9), (2, 5), (0, 6)],
            [(1, 2), (0, 5), (3, 8), (0, 7), (1, 4)],
            [(0, 2), (2, 3), (0, 4), (6, 7), (5, 6)],
        ]
        recs, off, mx = [], 0, 1
        for b in boxes:
            rec, n = _box_record(shape, b, wpe, off)
            recs.append(rec)
            off += n
            mx = max(mx, n)
        desc = torch.tensor(recs, dtype=torch.int64, device="cuda")
        assert desc.shape[1] == _PACK_REC
        word = (torch.float64 if dtype == torch.float64
                else torch.bfloat16 if dtype == torch.bfloat16
                else torch.float32)
        flat = torch.empty(off, dtype=word, device="cuda")
        xw = (torch.view_as_real(x) if dtype.is_complex else x).reshape(-1)
        ext.pack_boxes(xw.contiguous(), flat, desc, mx)
        # reference pack via aten slicing
        ref_parts = []
        for b in boxes:
            sl = tuple(slice(a, c) for a, c in b)
            piece = x[sl].contiguous()
            pw = (torch.view_as_real(piece) if dtype.is_complex else piece).reshape(-1)
            ref_parts.append(pw)
        ref = torch.cat(ref_parts)
        assert torch.equal(flat, ref)

        # unpack back into a zeroed tensor and compare against scatter
        y = torch.zeros_like(x)
        yw = (torch.view_as_real(y) if dtype.is_complex else y).reshape(-1)
        ext.unpack_boxes(flat, yw, desc, mx)
        y_ref = torch.zeros_like(x)
        for b in boxes:
            sl = tuple(slice(a, c) for a, c in b)
            y_ref[sl] = x[sl]
        assert torch.equal(y, y_ref)


@pytest.mark.parametrize("in_shape,t_out,modes", [
    ([1, 2, 8, 8, 64, 1], 30, (3, 3, 12, 8)),   # flagship (z,t): pinned kernels
    ([1, 2, 10, 10, 32, 1], 24, (3, 3, 8, 6)),  # generic-shape kernels
])
def test_fno_zt_fused_path_matches_default(monkeypatch, in_shape, t_out, modes):
    """DFNO_ZT=1 routes the trailing (z, t) pair through the fused 2-D
    kernels (csrc/dft2d.hip, incl. the stash); output and grads must match
    the default 1-D chain."""
    torch.manual_seed(3)
    _, P_x, _ = dfno.create_standard_partitions((1,) * len(in_shape))
    dev = torch.device("cuda")

    def run(zt):
        monkeypatch.setenv("DFNO_ZT", "1" if zt else "0")
        torch.manual_seed(7)
        model = dfno.DistributedFNONd(P_x, in_shape, t_out, 8, modes,
                                      num_blocks=2, device=dev)
        x = torch.rand(*in_shape, device=dev, requires_grad=True)
        y = model(x)
        loss = y.square().mean()
        loss.backward()
        return (y.detach().cpu(), x.grad.cpu(),
                {n: p.grad.cpu() for n, p in model.named_parameters()
                 if p.grad is not None})

    y0, gx0, gp0 = run(False)
    y1, gx1, gp1 = run(True)
    assert torch.allclose(y1, y0, rtol=1e-4, atol=1e-4), \
        f"y {(y1-y0).abs().max()}"
    assert torch.allclose(gx1, gx0, rtol=1e-4, atol=1e-4), \
        f"gx {(gx1-gx0).abs().max()}"
    for n in gp0:
        assert torch.allclose(gp1[n], gp0[n], rtol=1e-4, atol=1e-3), \
            f"{n}: {(gp1[n]-gp0[n]).abs().max()}"


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_training_converges_gpu(dtype):
    """The full fused stack (heads, mix backwards, stash, fused Adam) must
    optimize end-to-end: 150 steps toward a teacher model's output halve
    the relative-Lp loss."""
    from dfno_amd.optim import Adam

    torch.manual_seed(7)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1, 1))
    kw = dict(num_blocks=2, device=torch.device("cuda"), dtype=dtype)
    model = dfno.DistributedFNONd(P_x, [1, 2, 8, 8, 8, 1], 6, 8,
                                  (2, 2, 2, 2), **kw)
    teacher = dfno.DistributedFNONd(P_x, [1, 2, 8, 8, 8, 1], 6, 8,
                                    (2, 2, 2, 2), **kw)
    crit = dfno.DistributedRelativeLpLoss(P_x)
    opt = Adam(model.parameters(), lr=1e-3)
    x = torch.rand(1, 2, 8, 8, 8, 1, device="cuda", dtype=dtype)
    with torch.no_grad():
        y = teacher(x)
    first = None
    for _ in range(150):
        opt.zero_grad(set_to_none=True)
        loss = crit(model(x), y)
        loss.backward()
        opt.step()
        if first is None:
            first = float(loss.detach())
    last = float(loss.detach())
    assert last < 0.5 * first, f"{dtype}: loss {first} -> {last}"
