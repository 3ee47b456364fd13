"""UMA/eSCN path on GPU: HIP-backend engine vs the fp64 CPU oracle, and
the bf16-autocast bench mode sanity."""
import numpy as np
import pytest
import torch

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")
pytestmark = pytest.mark.gpu


@requires_gpu
def test_uma_engine_gpu_vs_oracle():
    from distmlip_amd.structures import diamond_si
    from distmlip_amd.uma_model import UMAConfig, UMACore
    from distmlip_amd.uma_runtime import UmaSpmdEngine
    from oracle.graph_ref import brute_force_neighbors
    from oracle.uma_ref import uma_oracle_forward

    s = diamond_si((8, 2, 2), jitter=0.2, seed=2)
    s.species = np.asarray(s.species) % 3
    cfg = UMAConfig(n_elements=3, sphere_channels=128, num_layers=4,
                    avg_degree=40.0)
    core = UMACore.seeded(cfg, seed=0)

    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 0.0)
    ref = uma_oracle_forward(core.double(), s, g["src"], g["dst"],
                             g["offsets"], dtype=torch.float64)

    eng = UmaSpmdEngine(core.float(), world=1, threads=4)
    out = eng.step(s)
    assert abs(out["energy"].item() - ref["energy"].item()) < 5e-3 * max(
        1.0, abs(ref["energy"].item()))
    F = np.zeros((s.num_atoms, 3))
    F[out["global_ids_owned"]] = out["forces_owned"].double().cpu().numpy()
    dF = np.abs(F - ref["forces"].numpy()).max()
    fscale = max(1.0, np.abs(ref["forces"].numpy()).max())
    assert dF < 1e-3 * fscale, f"UMA GPU force error {dF} (scale {fscale})"


@requires_gpu
def test_uma_engine_gpu_chunked_matches(monkeypatch):
    """The node-range-chunked rotation-kernel message pass (forced small
    chunks) must match the single-chunk path."""
    from distmlip_amd.structures import diamond_si
    from distmlip_amd.uma_model import UMAConfig, UMACore
    from distmlip_amd.uma_runtime import UmaSpmdEngine

    s = diamond_si((6, 2, 2), jitter=0.2, seed=3)
    s.species = np.asarray(s.species) % 3
    cfg = UMAConfig(n_elements=3, sphere_channels=64, num_layers=2)
    core = UMACore.seeded(cfg, seed=1).float()
    ref = UmaSpmdEngine(core, world=1, threads=4).step(s)
    monkeypatch.setenv("DM_UMA_CHUNK", "3000")
    monkeypatch.setenv("DM_UMA_DEG_CHUNK", "5000")
    got = UmaSpmdEngine(core, world=1, threads=4).step(s)
    assert abs(ref["energy"].item() - got["energy"].item()) < 1e-4 * max(
        1.0, abs(ref["energy"].item()))
    scale = ref["forces_owned"].abs().max().item()
    dF = (ref["forces_owned"] - got["forces_owned"]).abs().max().item()
    assert dF < 1e-3 * max(1.0, scale), (dF, scale)


@requires_gpu
def test_uma_engine_bf16_autocast_close():
    """The bf16-autocast bench mode stays within bf16-resolution of the
    fp32 engine on the same inputs (no silent divergence)."""
    from distmlip_amd.structures import diamond_si
    from distmlip_amd.uma_model import UMAConfig, UMACore
    from distmlip_amd.uma_runtime import UmaSpmdEngine

    s = diamond_si((6, 2, 2), jitter=0.2, seed=3)
    s.species = np.asarray(s.species) % 3
    cfg = UMAConfig(n_elements=3, sphere_channels=64, num_layers=2)
    core = UMACore.seeded(cfg, seed=1).float()
    outs = {}
    for ac in (False, True):
        eng = UmaSpmdEngine(core, world=1, threads=4, autocast_bf16=ac)
        outs[ac] = eng.step(s)
    e0, e1 = outs[False]["energy"].item(), outs[True]["energy"].item()
    assert abs(e0 - e1) < 2e-2 * max(1.0, abs(e0)), (e0, e1)
    f0 = outs[False]["forces_owned"]
    f1 = outs[True]["forces_owned"]
    scale = f0.abs().max().item()
    assert (f0 - f1).abs().max().item() < 5e-2 * max(1.0, scale)


@requires_gpu
def test_rot_kernels_vs_torch():
    """dm_rot_{gather,scatter,dD}_f32 against the einsum composition:
    values + gradients through both Functions."""
    from distmlip_amd import so3, uma_ops

    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    N, E, C = 300, 2000, 128
    h = torch.randn(N, 9, C, device=dev)
    vec = torch.randn(E, 3, device=dev)
    R = so3.edge_align_rotation(vec.double())
    D = so3.wigner_D_batch(R, 2).float()
    idx = torch.randint(0, N, (E,), device=dev, dtype=torch.int32)

    hr = h.clone().requires_grad_(True)
    Dr = D.clone().requires_grad_(True)
    ref = torch.einsum("est,etc->esc", Dr, hr[idx.long()])
    lw = torch.randn_like(ref)
    (ref * lw).sum().backward()

    hk = h.clone().requires_grad_(True)
    Dk = D.clone().requires_grad_(True)
    got = uma_ops._RotGather.apply(hk, Dk, idx)
    assert torch.allclose(got, ref, atol=1e-5)
    (got * lw).sum().backward()
    assert torch.allclose(hk.grad, hr.grad, atol=1e-4)
    assert torch.allclose(Dk.grad, Dr.grad, atol=1e-3), \
        (Dk.grad - Dr.grad).abs().max().item()

    # scatter: dst-sorted edges over N nodes
    dst = torch.sort(torch.randint(0, N, (E,), device=dev)).values.int()
    rp = torch.zeros(N + 1, dtype=torch.int64, device=dev)
    rp[1:] = torch.bincount(dst.long(), minlength=N)
    rp = torch.cumsum(rp, 0).int()
    mt = torch.randn(E, 9, C, device=dev)
    mtr = mt.clone().requires_grad_(True)
    Dr2 = D.clone().requires_grad_(True)
    rot = torch.einsum("ets,etc->esc", Dr2, mtr)     # D^T apply
    ref2 = torch.zeros(N, 9, C, device=dev).index_add(0, dst.long(), rot)
    lw2 = torch.randn_like(ref2)
    (ref2 * lw2).sum().backward()

    mtk = mt.clone().requires_grad_(True)
    Dk2 = D.clone().requires_grad_(True)
    got2 = uma_ops._RotScatter.apply(mtk, Dk2, rp, dst)
    assert torch.allclose(got2, ref2, atol=1e-4), \
        (got2 - ref2).abs().max().item()
    (got2 * lw2).sum().backward()
    assert torch.allclose(mtk.grad, mtr.grad, atol=1e-4)
    assert torch.allclose(Dk2.grad, Dr2.grad, atol=1e-3)


@requires_gpu
def test_so2_split_matches_cat():
    """so2_conv_split (the kernel path's GEMM split) == so2_conv over the
    concatenated tensor."""
    from distmlip_amd import uma_ops
    from distmlip_amd.uma_model import UMAConfig, UMACore

    torch.manual_seed(1)
    dev = torch.device("cuda:0")
    cfg = UMAConfig(n_elements=3, sphere_channels=128)
    core = UMACore.seeded(cfg, seed=0).float().to(dev)
    core.requires_grad_(False)
    blk = core.blocks[0]
    E, C = 3000, cfg.sphere_channels
    xs = torch.randn(E, 9, C, device=dev)
    xd = torch.randn(E, 9, C, device=dev)
    gate = torch.rand(E, 3, device=dev)
    ref = uma_ops.so2_conv(blk.msg, torch.cat([xs, xd], dim=2), gate, C)
    got = uma_ops.so2_conv_split(blk.msg, xs, xd, gate, C)
    assert torch.allclose(got, ref, atol=1e-4), \
        (got - ref).abs().max().item()
