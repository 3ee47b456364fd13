"""Per-edge frame-gauge covariance of the UMA message path.

The z-aligned edge frame (so3.edge_align_rotation) is only defined up
to a rotation about z (the reference-axis pick, escn_md.py's
perpendicular construction).  The message math must be EXACTLY gauge
invariant -- this is what makes the per-edge argmin branch harmless
(DESIGN.md §13 addendum 2 relies on it).  Checked here in fp64:
  * block_message: invariant under D -> D(Rz(theta)) D per edge;
  * edge_degree_embed: uses only m=0 rows, which Rz fixes -> equal.
"""
import numpy as np
import torch

from distmlip_amd import so3, uma_ops
from distmlip_amd.uma_model import UMAConfig, UMACore


def _gauge_pair(E, seed):
    """(D, D_gauged): same edges, reference frames differing by a random
    per-edge rotation about z."""
    torch.manual_seed(seed)
    vec = torch.randn(E, 3, dtype=torch.float64)
    R = so3.edge_align_rotation(vec)
    th = torch.rand(E, dtype=torch.float64) * (2 * np.pi)
    c, s = torch.cos(th), torch.sin(th)
    Rz = torch.zeros(E, 3, 3, dtype=torch.float64)
    Rz[:, 0, 0] = c
    Rz[:, 0, 1] = -s
    Rz[:, 1, 0] = s
    Rz[:, 1, 1] = c
    Rz[:, 2, 2] = 1.0
    Rg = Rz @ R
    # both are valid z-aligning frames for the same vectors
    vhat = vec / vec.norm(dim=1, keepdim=True)
    for M in (R, Rg):
        z = torch.einsum("est,et->es", M, vhat)
        assert torch.allclose(z, torch.tensor([0.0, 0.0, 1.0],
                                              dtype=torch.float64)
                              .expand(E, 3), atol=1e-12)
    return so3.wigner_D_batch(R, 2), so3.wigner_D_batch(Rg, 2)


def test_block_message_gauge_invariant():
    E, C = 64, 32
    cfg = UMAConfig(n_elements=3, sphere_channels=C, num_layers=2)
    core = UMACore.seeded(cfg, seed=0).double()
    blk = core.blocks[0]
    D, Dg = _gauge_pair(E, seed=1)
    torch.manual_seed(2)
    x_src = torch.randn(E, cfg.S, C, dtype=torch.float64)
    x_dst = torch.randn(E, cfg.S, C, dtype=torch.float64)
    x_edge = torch.randn(E, cfg.num_gauss + 2 * cfg.spec_emb,
                         dtype=torch.float64)
    m0 = uma_ops.block_message(blk, cfg, x_src, x_dst, x_edge, D,
                               D.transpose(-1, -2))
    m1 = uma_ops.block_message(blk, cfg, x_src, x_dst, x_edge, Dg,
                               Dg.transpose(-1, -2))
    assert (m0 - m1).abs().max().item() < 1e-12, \
        (m0 - m1).abs().max().item()


def test_edge_degree_embed_gauge_invariant():
    E = 64
    cfg = UMAConfig(n_elements=3, sphere_channels=32, num_layers=2)
    core = UMACore.seeded(cfg, seed=0).double()
    D, Dg = _gauge_pair(E, seed=3)
    torch.manual_seed(4)
    x_edge = torch.randn(E, cfg.num_gauss + 2 * cfg.spec_emb,
                         dtype=torch.float64)
    m0 = uma_ops.edge_degree_embed(core, x_edge, D.transpose(-1, -2))
    m1 = uma_ops.edge_degree_embed(core, x_edge, Dg.transpose(-1, -2))
    assert (m0 - m1).abs().max().item() < 1e-12
