"""hipGraph-captured inference (dfno_amd/serve.py) on an MI355X."""

import time

import pytest
import torch

import dfno_amd as dfno
from dfno_amd.serve import GraphedEval

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("in_shape,t_out,modes", [
    ([1, 1, 64, 64, 10], 20, (4, 4, 4)),       # NS-like 2D+time (launch-bound)
    ([1, 2, 12, 12, 8, 1], 6, (3, 3, 2, 2)),   # small 3D+time
])
def test_graphed_eval_matches_eager(in_shape, t_out, modes):
    torch.manual_seed(4)
    _, P_x, _ = dfno.create_standard_partitions((1,) * len(in_shape))
    model = dfno.DistributedFNONd(P_x, in_shape, t_out, 12, modes,
                                  num_blocks=4, device=torch.device("cuda"))
    model.eval()
    x = torch.rand(*in_shape, device="cuda")
    with torch.no_grad():
        y_ref = model(x)
    ge = GraphedEval(model, x)
    y = ge(x)
    assert torch.allclose(y, y_ref, rtol=1e-5, atol=1e-5), \
        f"max {(y - y_ref).abs().max()}"

    # a second input must flow through the replay correctly
    x2 = torch.rand_like(x)
    with torch.no_grad():
        y2_ref = model(x2)
    y2 = ge(x2)
    assert torch.allclose(y2, y2_ref, rtol=1e-5, atol=1e-5), \
        f"max {(y2 - y2_ref).abs().max()}"

    # shape guard
    with pytest.raises(ValueError):
        ge(torch.rand(2, *in_shape[1:], device="cuda"))


def test_graphed_eval_faster_when_launch_bound():
    """On the small NS grid the replayed graph must not be slower than the
    eager eval (it removes ~all per-kernel launch overhead)."""
    torch.manual_seed(4)
    _, P_x, _ = dfno.create_standard_partitions((1, 1, 1, 1, 1))
    model = dfno.DistributedFNONd(P_x, [1, 1, 64, 64, 10], 20, 20, (4, 4, 4),
                                  num_blocks=4, device=torch.device("cuda"))
    model.eval()
    x = torch.rand(1, 1, 64, 64, 10, device="cuda")
    ge = GraphedEval(model, x)

    def clock(fn, iters=30):
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    with torch.no_grad():
        dt_eager = clock(lambda: model(x))
    dt_graph = clock(lambda: ge(x))
    print(f"eager {dt_eager*1e3:.3f} ms vs graph {dt_graph*1e3:.3f} ms")
    assert dt_graph < dt_eager * 1.10   # at least not slower; usually much faster
