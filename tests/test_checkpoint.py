"""Offline checkpoint resharding (dfno_amd.checkpoint): P -> Q -> P
roundtrips must be exact, and resharded shards must load into real models.
The reference has no resharding (same-partition load only).
"""

import torch

from dist_utils import run_dist

from dfno_amd.checkpoint import reshard_checkpoint, _build_model

CFG = dict(in_shape=[1, 2, 8, 8, 6, 1], out_t=6, width=6,
           modes=(2, 2, 2, 2), num_blocks=2)
PSERIAL = (1, 1, 1, 1, 1, 1)
P2 = (1, 1, 2, 1, 1, 1)
P4 = (1, 1, 2, 2, 1, 1)


def _save_serial(tmp_path):
    torch.manual_seed(7)
    m = _build_model(PSERIAL, 0, **CFG)
    torch.save(m.state_dict(), tmp_path / "model_0000.pt")
    return m


def test_reshard_roundtrip_1_2_1(tmp_path):
    m0 = _save_serial(tmp_path)
    d2 = tmp_path / "p2"
    reshard_checkpoint(tmp_path, PSERIAL, P2, out_dir=d2, **CFG)
    d1 = tmp_path / "back"
    reshard_checkpoint(d2, P2, PSERIAL, out_dir=d1, **CFG)
    s0 = m0.state_dict()
    s1 = torch.load(d1 / "model_0000.pt", weights_only=False)
    assert set(s0) == set(s1)
    for k in s0:
        assert torch.equal(s0[k], s1[k]), k


def test_reshard_shards_load_strict(tmp_path):
    _save_serial(tmp_path)
    d4 = tmp_path / "p4"
    paths = reshard_checkpoint(tmp_path, PSERIAL, P4, out_dir=d4, **CFG)
    assert len(paths) == 4
    for q in range(4):
        m = _build_model(P4, q, **CFG)
        state = torch.load(d4 / f"model_{q:04d}.pt", weights_only=False)
        m.load_state_dict(state, strict=True)


def test_reshard_4_to_2(tmp_path):
    _save_serial(tmp_path)
    d4 = tmp_path / "p4"
    reshard_checkpoint(tmp_path, PSERIAL, P4, out_dir=d4, **CFG)
    d2 = tmp_path / "p2"
    reshard_checkpoint(d4, P4, P2, out_dir=d2, **CFG)
    d1 = tmp_path / "final"
    reshard_checkpoint(d2, P2, PSERIAL, out_dir=d1, **CFG)
    s0 = _build_model(PSERIAL, 0, **CFG).state_dict()  # fresh init differs
    s1 = torch.load(d1 / "model_0000.pt", weights_only=False)
    src = torch.load(tmp_path / "model_0000.pt", weights_only=False)
    for k in src:
        assert torch.equal(src[k], s1[k]), k


def _load_dist_body(rank, world, d2, cfg, pshape):
    import torch as t
    from dfno_amd.partition import Partition
    from dfno_amd.nn.fno import DistributedFNONd

    P_x = Partition(tuple(range(world)), pshape)
    m = DistributedFNONd(P_x, cfg["in_shape"], cfg["out_t"], cfg["width"],
                         cfg["modes"], num_blocks=cfg["num_blocks"],
                         dtype=t.float32)
    state = t.load(f"{d2}/model_{rank:04d}.pt", weights_only=False)
    m.load_state_dict(state, strict=True)
    # and the loaded model must run
    from dfno_amd.partition import compute_distribution_info
    info = compute_distribution_info(P_x, cfg["in_shape"])
    x = t.rand(*cfg["in_shape"])
    y = m(x[info["slice"]].clone())
    assert t.isfinite(y).all()


def test_resharded_loads_into_real_2rank_model(tmp_path):
    _save_serial(tmp_path)
    d2 = tmp_path / "p2"
    reshard_checkpoint(tmp_path, PSERIAL, P2, out_dir=d2, **CFG)
    run_dist(_load_dist_body, 2, str(d2), CFG, P2)


# 2D+time NS-style config on a (1,1,2,2,1) world: the pencil P_y folds onto
# a PREFIX of ranks (odd transform count), so some ranks hold no spectral
# shards at all -- the harvest/emit passes must handle them.
CFG2D = dict(in_shape=[1, 1, 8, 8, 4], out_t=6, width=4,
             modes=(2, 2, 2), num_blocks=2)
P2D_SERIAL = (1, 1, 1, 1, 1)
P2D_4 = (1, 1, 2, 2, 1)


def test_reshard_2d_inactive_py_ranks(tmp_path):
    torch.manual_seed(3)
    m = _build_model(P2D_SERIAL, 0, **CFG2D)
    torch.save(m.state_dict(), tmp_path / "model_0000.pt")
    d4 = tmp_path / "p4"
    paths = reshard_checkpoint(tmp_path, P2D_SERIAL, P2D_4, out_dir=d4, **CFG2D)
    assert len(paths) == 4
    back = tmp_path / "back"
    reshard_checkpoint(d4, P2D_4, P2D_SERIAL, out_dir=back, **CFG2D)
    src = m.state_dict()
    out = torch.load(back / "model_0000.pt", weights_only=False)
    assert set(src) == set(out)
    for k in src:
        assert torch.equal(src[k], out[k]), k
