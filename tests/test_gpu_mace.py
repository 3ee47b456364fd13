"""MACE path on GPU: HIP-backend engine vs the fp64 CPU oracle."""
import numpy as np
import pytest
import torch

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")
pytestmark = pytest.mark.gpu


@requires_gpu
def test_mace_engine_gpu_vs_oracle():
    from distmlip_amd.mace_model import MACEConfig, MACECore
    from distmlip_amd.mace_runtime import MaceSpmdEngine
    from distmlip_amd.structures import diamond_si
    from oracle.graph_ref import brute_force_neighbors
    from oracle.mace_ref import mace_oracle_forward

    s = diamond_si((8, 2, 2), jitter=0.1, seed=2)
    s.species = np.asarray(s.species) % 3
    cfg = MACEConfig(n_elements=3, channels=128, avg_num_neighbors=40.0,
                     atomic_inter_scale=0.7, atomic_inter_shift=0.1)
    core = MACECore.seeded(cfg, seed=0)

    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 0.0)
    ref = mace_oracle_forward(core.double(), s, g["src"], g["dst"],
                              g["offsets"], dtype=torch.float64,
                              compute_stress=True)

    eng = MaceSpmdEngine(core.float(), world=1, threads=4)
    out = eng.step(s, calc_stresses=True)

    assert abs(out["energy"].item() - ref["energy"].item()) < 5e-3 * max(
        1.0, abs(ref["energy"].item()))
    F = np.zeros((s.num_atoms, 3))
    F[out["global_ids_owned"]] = out["forces_owned"].double().cpu().numpy()
    dF = np.abs(F - ref["forces"].numpy()).max()
    assert dF < 1e-4, f"MACE GPU force error {dF} exceeds 1e-4 eV/A"
    dS = np.abs(out["stress"].cpu().numpy() - ref["stress"].numpy()).max()
    assert dS < 1e-3, dS


@requires_gpu
def test_mace_engine_gpu_chunked_matches(monkeypatch):
    """The node-range-chunked fused message pass (the >6M-edge memory
    path, forced small here) must match the unchunked fused path."""
    from distmlip_amd.mace_model import MACEConfig, MACECore
    from distmlip_amd.mace_runtime import MaceSpmdEngine
    from distmlip_amd.structures import diamond_si

    s = diamond_si((6, 2, 2), jitter=0.1, seed=3)
    s.species = np.asarray(s.species) % 3
    cfg = MACEConfig(n_elements=3, channels=64)
    core = MACECore.seeded(cfg, seed=1).float()
    ref = MaceSpmdEngine(core, world=1, threads=4).step(s)
    monkeypatch.setenv("DM_MACE_CHUNK", "3000")
    got = MaceSpmdEngine(core, world=1, threads=4).step(s)
    assert abs(ref["energy"].item() - got["energy"].item()) < 1e-5
    dF = (ref["forces_owned"] - got["forces_owned"]).abs().max().item()
    assert dF < 1e-4, dF


@requires_gpu
def test_mace_engine_gpu_checkpointed_matches():
    """checkpoint='on' (the big-workload path) must match checkpoint-off
    bit-for-bit on the same inputs (same kernels, same order)."""
    from distmlip_amd.mace_model import MACEConfig, MACECore
    from distmlip_amd.mace_runtime import MaceSpmdEngine
    from distmlip_amd.structures import diamond_si

    s = diamond_si((6, 2, 2), jitter=0.1, seed=3)
    s.species = np.asarray(s.species) % 3
    cfg = MACEConfig(n_elements=3, channels=64)
    core = MACECore.seeded(cfg, seed=1).float()
    outs = {}
    for ck in ("off", "on"):
        eng = MaceSpmdEngine(core, world=1, threads=4, checkpoint=ck)
        outs[ck] = eng.step(s)
    assert outs["off"]["energy"].item() == outs["on"]["energy"].item()
    assert torch.equal(outs["off"]["forces_owned"],
                       outs["on"]["forces_owned"])


@requires_gpu
def test_fused_tp_kernel_vs_torch():
    """dm_mace_tp_{fwd,bwd}_f32 against the torch broadcast composition,
    values + all four gradients, for both interaction shapes (l=0-only
    first layer; l=0+1 second layer)."""
    import os

    from distmlip_amd import mace_ops, so3
    from distmlip_amd.mace_model import MACEConfig, MACECore

    torch.manual_seed(0)
    dev = torch.device("cuda:0")
    cfg = MACEConfig(n_elements=3, channels=128)
    core = MACECore.seeded(cfg, seed=0).float().to(dev)
    E = 4000
    for li in (0, 1):
        inter = core.interactions[li]
        C = cfg.channels
        P = len(inter.paths)
        Y = torch.randn(E, 16, device=dev)
        w = torch.randn(E, P, C, device=dev)
        g0 = torch.randn(E, C, 1, device=dev)
        g1 = torch.randn(E, C, 3, device=dev) if li == 1 else None

        args = [t.clone().requires_grad_(True)
                for t in (g0, Y, w) if t is not None]
        x0r, Yr, wr = args
        x1r = g1.clone().requires_grad_(True) if g1 is not None else None
        gathered = {0: x0r}
        if x1r is not None:
            gathered[1] = x1r
        os.environ["DM_MACE_TP"] = "bcast"
        try:
            ref = mace_ops.conv_tp_messages(inter, gathered, Yr, wr)
        finally:
            os.environ.pop("DM_MACE_TP", None)
        loss_w = {l3: torch.randn_like(ref[l3]) for l3 in ref}
        sum(
            (ref[l3] * loss_w[l3]).sum() for l3 in ref).backward()

        x0k = g0[:, :, 0].clone().requires_grad_(True)
        x1k = g1.clone().requires_grad_(True) if g1 is not None else None
        Yk = Y.clone().requires_grad_(True)
        wk = w.clone().requires_grad_(True)
        outs = mace_ops.conv_tp_hip(inter, x0k, x1k, Yk, wk)
        loss = 0
        for l3 in ref:
            # kernel layout [E, d3, C] vs torch [E, C, d3]
            got = outs[l3].permute(0, 2, 1)
            assert torch.allclose(got, ref[l3], atol=2e-5), (
                li, l3, (got - ref[l3]).abs().max().item())
            loss = loss + (got * loss_w[l3]).sum()
        # absent l3 blocks must come out zero
        for l3 in range(4):
            if l3 not in ref:
                assert outs[l3].abs().max().item() == 0.0, (li, l3)
        loss.backward()
        assert torch.allclose(x0k.grad, x0r.grad[:, :, 0], atol=2e-4), li
        if x1k is not None:
            assert torch.allclose(x1k.grad, x1r.grad, atol=2e-4), li
        assert torch.allclose(wk.grad, wr.grad, atol=2e-4), li
        dY_err = (Yk.grad - Yr.grad).abs().max().item()
        assert dY_err < 2e-3, (li, dY_err)   # dY sums over C=128 lanes


@requires_gpu
def test_fused_symc_kernel_vs_torch():
    """dm_mace_symc_{fwd,bwd}_f32 against the torch combo-GEMM
    contraction: values + dx gradients, both product shapes (hidden
    0e+1o; scalar-only last layer)."""
    import os

    from distmlip_amd import mace_ops
    from distmlip_amd.mace_model import MACEConfig, MACECore

    torch.manual_seed(1)
    dev = torch.device("cuda:0")
    cfg = MACEConfig(n_elements=3, channels=128)
    core = MACECore.seeded(cfg, seed=0).float().to(dev)
    core.requires_grad_(False)
    N = 3000
    species = torch.randint(0, 3, (N,), device=dev)
    for li in (0, 1):
        prod = core.products[li]
        x = {l: torch.randn(N, cfg.channels, 2 * l + 1, device=dev)
             for l in range(4)}
        xr = {l: t.clone().requires_grad_(True) for l, t in x.items()}
        os.environ["DM_MACE_SYMC"] = "torch"
        try:
            ref = mace_ops.symmetric_contract(prod, xr, species,
                                              cfg.correlation)
        finally:
            os.environ.pop("DM_MACE_SYMC", None)
        lw = {lo: torch.randn_like(ref[lo]) for lo in ref}
        sum((ref[lo] * lw[lo]).sum() for lo in ref).backward()

        xk = {l: t.clone().requires_grad_(True) for l, t in x.items()}
        got = mace_ops.symmetric_contract_hip(prod, xk, species,
                                              cfg.correlation)
        loss = 0
        for lo in ref:
            err = (got[lo] - ref[lo]).abs().max().item()
            assert err < 5e-4, (li, lo, err)
            loss = loss + (got[lo] * lw[lo]).sum()
        loss.backward()
        for l in range(4):
            derr = (xk[l].grad - xr[l].grad).abs().max().item()
            scale = xr[l].grad.abs().max().item()
            assert derr < 1e-3 * max(1.0, scale), (li, l, derr, scale)
