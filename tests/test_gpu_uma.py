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
