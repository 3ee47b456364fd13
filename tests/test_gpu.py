"""GPU parity tests (MI355X): HIP kernels vs plain-torch, and the full
distributed E+F forward vs the fp64 CPU oracle (forces within the
north-star 1e-4 eV/A)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs HIP GPU")


@requires_gpu
class TestKernels:
    def setup_method(self, _):
        torch.manual_seed(0)
        self.dev = torch.device("cuda:0")

    def _csr(self, idx, n_rows):
        """host-side reference CSR build for random indices"""
        idx = idx.cpu().numpy()
        order = np.argsort(idx, kind="stable")
        rp = np.zeros(n_rows + 1, dtype=np.int64)
        np.add.at(rp, idx + 1, 1)
        rp = np.cumsum(rp)
        return (torch.tensor(order, dtype=torch.int32, device=self.dev),
                torch.tensor(rp, dtype=torch.int32, device=self.dev))

    def test_gather_fwd_bwd(self):
        from distmlip_amd.ops import _Gather
        N, E, D = 500, 3000, 64
        x = torch.randn(N, D, device=self.dev, requires_grad=True)
        idx64 = torch.randint(0, N, (E,))
        idx = idx64.to(torch.int32).to(self.dev)
        perm, rp = self._csr(idx64, N)
        out = _Gather.apply(x, idx, perm, rp)
        ref = x.detach()[idx64.to(self.dev)]
        assert torch.allclose(out, ref)
        g = torch.randn_like(out)
        out.backward(g)
        xr = x.detach().clone().requires_grad_(True)
        xr[idx64.to(self.dev)].backward(g)
        assert torch.allclose(x.grad, xr.grad, atol=1e-4, rtol=1e-4)

    def test_seg_sum_vs_index_add(self):
        from distmlip_amd.ops import raw_seg_sum
        for D in (64, 128, 3, 9):
            N, E = 400, 5000
            idx64 = torch.sort(torch.randint(0, N, (E,))).values
            msg = torch.randn(E, D, device=self.dev)
            base = torch.randn(N, D, device=self.dev)
            rp = torch.zeros(N + 1, dtype=torch.int64)
            rp.index_add_(0, idx64 + 1, torch.ones(E, dtype=torch.int64))
            rp = torch.cumsum(rp, 0).to(torch.int32).to(self.dev)
            out = raw_seg_sum(msg, rp, N, base)
            ref = base.clone().index_add_(0, idx64.to(self.dev), msg)
            assert torch.allclose(out, ref, atol=1e-4), D

    def test_gather_add3(self):
        from distmlip_amd.ops import _GatherAdd3

        class PD:
            pass
        N, E, D = 300, 4000, 64
        zs = torch.randn(N, D, device=self.dev, requires_grad=True)
        zd = torch.randn(N, D, device=self.dev, requires_grad=True)
        src64 = torch.randint(0, N, (E,))
        dst64 = torch.sort(torch.randint(0, N, (E,))).values
        ze = torch.randn(E, D, device=self.dev, requires_grad=True)
        sperm, srp = self._csr(src64, N)
        _, drp = self._csr(dst64, N)
        src = src64.to(torch.int32).to(self.dev)
        dst = dst64.to(torch.int32).to(self.dev)
        z, h = _GatherAdd3.apply(zs, zd, ze, src, dst, sperm, srp, drp)
        ref = zs.detach()[src64] + zd.detach()[dst64] + ze.detach()
        assert torch.allclose(z, ref, atol=1e-5)
        assert torch.allclose(h, torch.nn.functional.silu(ref), atol=1e-5)
        g = torch.randn_like(h)
        h.backward(g)
        zs2 = zs.detach().clone().requires_grad_(True)
        zd2 = zd.detach().clone().requires_grad_(True)
        ze2 = ze.detach().clone().requires_grad_(True)
        torch.nn.functional.silu(zs2[src64] + zd2[dst64] + ze2).backward(g)
        assert torch.allclose(zs.grad, zs2.grad, atol=1e-3, rtol=1e-4)
        assert torch.allclose(zd.grad, zd2.grad, atol=1e-3, rtol=1e-4)
        assert torch.allclose(ze.grad, ze2.grad, atol=1e-5)


@requires_gpu
def test_edge_geom_rbf_fwd_bwd():
    """Fused geometry+RBF kernel vs the torch composition (fwd + grads)."""
    import math

    from distmlip_amd.model import bond_expansion_from_dist
    from distmlip_amd.ops import _EdgeGeomRbf
    torch.manual_seed(3)
    dev = torch.device("cuda:0")
    N, E = 200, 3000

    class PD:
        pass
    pd = PD()
    src64 = torch.randint(0, N, (E,))
    dst64 = torch.sort(torch.randint(0, N, (E,))).values
    pd.src = src64.to(torch.int32).to(dev)
    pd.dst = dst64.to(torch.int32).to(dev)

    def csr(idx):
        import numpy as np
        idx = idx.numpy()
        order = np.argsort(idx, kind="stable")
        rp = np.zeros(N + 1, dtype=np.int64)
        np.add.at(rp, idx + 1, 1)
        return (torch.tensor(order, dtype=torch.int32, device=dev),
                torch.tensor(np.cumsum(rp), dtype=torch.int32, device=dev))
    pd.src_perm, pd.src_row_ptr = csr(src64)
    _, pd.row_ptr = csr(dst64)

    pos = (torch.rand(N, 3, device=dev) * 20).requires_grad_(True)
    off = (torch.randn(E, 3, device=dev) * 2).requires_grad_(True)
    freqs = (torch.arange(1, 10, device=dev, dtype=torch.float32) * math.pi)
    bv, bd, exp = _EdgeGeomRbf.apply(pos, off, freqs, 6.0, 5, pd)

    pos2 = pos.detach().clone().requires_grad_(True)
    off2 = off.detach().clone().requires_grad_(True)
    bv2 = pos2[dst64.to(dev)] + off2 - pos2[src64.to(dev)]
    bd2 = torch.linalg.norm(bv2, dim=1)
    exp2 = bond_expansion_from_dist(bd2, freqs, 6.0, 5)
    assert torch.allclose(bv, bv2, atol=1e-5)
    assert torch.allclose(bd, bd2, atol=1e-5)
    assert torch.allclose(exp, exp2, atol=1e-5), (exp - exp2).abs().max()

    g_bv = torch.randn_like(bv)
    g_bd = torch.randn_like(bd)
    g_exp = torch.randn_like(exp)
    torch.autograd.backward([bv, bd, exp], [g_bv, g_bd, g_exp])
    torch.autograd.backward([bv2, bd2, exp2], [g_bv, g_bd, g_exp])
    assert torch.allclose(pos.grad, pos2.grad, atol=2e-4, rtol=1e-3), \
        (pos.grad - pos2.grad).abs().max()
    assert torch.allclose(off.grad, off2.grad, atol=2e-4, rtol=1e-3)


@requires_gpu
def test_gated_combine_fwd_bwd():
    from distmlip_amd.ops import _GatedCombine
    torch.manual_seed(1)
    dev = torch.device("cuda:0")
    E, D = 4000, 64
    for has_w, has_base in [(True, True), (True, False), (False, True),
                            (False, False)]:
        c = torch.randn(E, D, device=dev, requires_grad=True)
        g = torch.randn(E, D, device=dev, requires_grad=True)
        w = torch.randn(E, D, device=dev, requires_grad=True) if has_w else None
        base = torch.randn(E, D, device=dev, requires_grad=True) if has_base else None
        out = _GatedCombine.apply(c, g, w, base)
        ref = torch.nn.functional.silu(c.detach().clone().requires_grad_(True))
        c2 = c.detach().clone().requires_grad_(True)
        g2 = g.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True) if has_w else None
        b2 = base.detach().clone().requires_grad_(True) if has_base else None
        r = torch.nn.functional.silu(c2) * torch.sigmoid(g2)
        if has_w:
            r = r * w2
        if has_base:
            r = b2 + r
        assert torch.allclose(out, r, atol=2e-6), (has_w, has_base)
        go = torch.randn_like(out)
        out.backward(go)
        r.backward(go)
        assert torch.allclose(c.grad, c2.grad, atol=2e-5)
        assert torch.allclose(g.grad, g2.grad, atol=2e-5)
        if has_w:
            assert torch.allclose(w.grad, w2.grad, atol=2e-5)
        if has_base:
            assert torch.allclose(base.grad, b2.grad)


@requires_gpu
def test_gated_combine_packed_fwd_bwd():
    """Packed cg [2,E,D] variant matches the separate-tensor op + grads."""
    from distmlip_amd.ops import _GatedCombinePacked
    torch.manual_seed(2)
    dev = torch.device("cuda:0")
    E, D = 4000, 64
    for has_w, has_base in [(True, True), (False, False)]:
        cg = torch.randn(2, E, D, device=dev, requires_grad=True)
        w = torch.randn(E, D, device=dev, requires_grad=True) if has_w else None
        base = torch.randn(E, D, device=dev, requires_grad=True) if has_base else None
        out = _GatedCombinePacked.apply(cg, w, base)
        cg2 = cg.detach().clone().requires_grad_(True)
        r = torch.nn.functional.silu(cg2[0]) * torch.sigmoid(cg2[1])
        if has_w:
            w2 = w.detach().clone().requires_grad_(True)
            r = r * w2
        if has_base:
            b2 = base.detach().clone().requires_grad_(True)
            r = b2 + r
        assert torch.allclose(out, r, atol=2e-6), (has_w, has_base)
        go = torch.randn_like(out)
        out.backward(go)
        r.backward(go)
        assert torch.allclose(cg.grad, cg2.grad, atol=2e-5)
        if has_w:
            assert torch.allclose(w.grad, w2.grad, atol=2e-5)
        if has_base:
            assert torch.allclose(base.grad, b2.grad)


def test_edge_mlp_fused_vs_unfused():
    """The fused first-layer edge-MLP kernel (GEMM-in-kernel, LDS weights)
    matches GEMM + gather_add3/4 within fp32 noise, incl. input grads."""
    from distmlip_amd.ops import _EdgeMlp3, _EdgeMlp4
    import numpy as np
    torch.manual_seed(4)
    dev = torch.device("cuda:0")
    N, E, Din, Dout = 500, 30000, 64, 128

    def csr_of(idx64, n):
        order = torch.argsort(idx64, stable=True)
        rp = torch.from_numpy(np.searchsorted(
            idx64[order].numpy(), np.arange(n + 1)).astype(np.int32)).to(dev)
        return order.to(torch.int32).to(dev), rp

    src64 = torch.randint(0, N, (E,))
    dst64 = torch.sort(torch.randint(0, N, (E,))).values
    _, drp = csr_of(dst64, N)
    sperm, srp = csr_of(src64, N)
    erow = torch.randn(E, Din, device=dev, requires_grad=True)
    wt = torch.randn(Din, Dout, device=dev)
    bias = torch.randn(Dout, device=dev)
    zs = torch.randn(N, Dout, device=dev, requires_grad=True)
    zd = torch.randn(N, Dout, device=dev, requires_grad=True)
    src = src64.to(torch.int32).to(dev)
    dst = dst64.to(torch.int32).to(dev)
    z, h = _EdgeMlp3.apply(erow, wt, bias, zs, zd, src, dst, sperm, srp, drp)
    e2 = erow.detach().clone().requires_grad_(True)
    zs2 = zs.detach().clone().requires_grad_(True)
    zd2 = zd.detach().clone().requires_grad_(True)
    zref = e2 @ wt + bias + zs2[src64] + zd2[dst64]
    href = torch.nn.functional.silu(zref)
    assert torch.allclose(z, zref, atol=2e-4), (z - zref).abs().max()
    assert torch.allclose(h, href, atol=2e-4)
    go = torch.randn_like(h)
    h.backward(go)
    href.backward(go)
    assert torch.allclose(erow.grad, e2.grad, atol=2e-3), \
        (erow.grad - e2.grad).abs().max()
    assert torch.allclose(zs.grad, zs2.grad, atol=2e-3)
    assert torch.allclose(zd.grad, zd2.grad, atol=2e-3)

    # 4-input (line-graph) form through a minimal pd bundle
    class PD:
        pass
    B = 800
    l_src64 = torch.randint(0, B, (E,))
    l_dst64 = torch.sort(torch.randint(0, B, (E,))).values
    c64 = torch.randint(0, N, (E,))
    pd = PD()
    pd.l_src = l_src64.to(torch.int32).to(dev)
    pd.l_dst = l_dst64.to(torch.int32).to(dev)
    pd.center = c64.to(torch.int32).to(dev)
    pd.line_src_perm, pd.line_src_row_ptr = csr_of(l_src64, B)
    _, pd.line_row_ptr = csr_of(l_dst64, B)
    pd.center_perm, pd.center_row_ptr = csr_of(c64, N)
    a = torch.randn(E, Din, device=dev, requires_grad=True)
    z1 = torch.randn(B, Dout, device=dev, requires_grad=True)
    z2 = torch.randn(B, Dout, device=dev)
    zv = torch.randn(N, Dout, device=dev)
    _z4, h4 = _EdgeMlp4.apply(a, wt, bias, z1, z2, zv, pd)
    a2 = a.detach().clone().requires_grad_(True)
    z1b = z1.detach().clone().requires_grad_(True)
    href4 = torch.nn.functional.silu(
        a2 @ wt + bias + z1b[l_src64] + z2[l_dst64] + zv[c64])
    assert torch.allclose(h4, href4, atol=2e-4)
    go4 = torch.randn_like(h4)
    h4.backward(go4)
    href4.backward(go4)
    assert torch.allclose(a.grad, a2.grad, atol=2e-3)
    assert torch.allclose(z1.grad, z1b.grad, atol=2e-3)


def test_gather_add3_inplace_only_without_grad():
    """Regression: grad mode is off inside Function.forward, so the in-place
    no-grad silu must key on input.requires_grad — with grad inputs, z must
    stay the PRE-activation sum (run 15 caught silu written over z)."""
    from distmlip_amd.ops import _GatherAdd3
    torch.manual_seed(3)
    dev = torch.device("cuda:0")
    N, E, D = 300, 2000, 64
    src64 = torch.randint(0, N, (E,))
    dst64 = torch.sort(torch.randint(0, N, (E,))).values
    import numpy as np
    rp = torch.from_numpy(np.searchsorted(
        dst64.numpy(), np.arange(N + 1)).astype(np.int32)).to(dev)
    sp_order = torch.argsort(src64, stable=True)
    sp = sp_order.to(torch.int32).to(dev)
    srp = torch.from_numpy(np.searchsorted(
        src64[sp_order].numpy(), np.arange(N + 1)).astype(np.int32)).to(dev)
    zs = torch.randn(N, D, device=dev, requires_grad=True)
    zd = torch.randn(N, D, device=dev)
    ze = torch.randn(E, D, device=dev)
    src = src64.to(torch.int32).to(dev)
    dst = dst64.to(torch.int32).to(dev)
    z, h = _GatherAdd3.apply(zs, zd, ze, src, dst, sp, srp, rp)
    ref = zs.detach()[src64] + zd[dst64] + ze
    assert torch.allclose(z, ref, atol=1e-5)          # pre-activation kept
    assert torch.allclose(h, torch.nn.functional.silu(ref), atol=1e-5)
    assert z.data_ptr() != h.data_ptr()
    # and with NO grad inputs the two alias (the memory optimisation)
    z2, h2 = _GatherAdd3.apply(zs.detach(), zd, ze, src, dst, sp, srp, rp)
    assert z2.data_ptr() == h2.data_ptr()
    assert torch.allclose(h2, torch.nn.functional.silu(ref), atol=1e-5)


def _gpu_model(core, P, devices=None):
    from distmlip_amd.chgnet import CHGNet_Dist
    model = CHGNet_Dist.from_existing(core, dtype=torch.float32)
    model.enable_distributed_mode(devices if devices is not None else [0] * P)
    return model


@requires_gpu
@pytest.mark.parametrize("P", [1, 2])
def test_e2e_energy_forces_vs_oracle(P):
    """Full distributed GPU forward (HIP kernels) vs fp64 CPU oracle."""
    from distmlip_amd.model import CHGNetCore
    from distmlip_amd.pes import Potential_Dist
    from distmlip_amd.structures import diamond_si
    from oracle.chgnet_ref import oracle_forward
    from oracle.graph_ref import brute_force_neighbors

    s = diamond_si((12, 2, 2), jitter=0.12, seed=2)
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    core = CHGNetCore.seeded(seed=0)
    ref = oracle_forward(core.double(), s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)
    model = _gpu_model(core.float(), P)
    pot = Potential_Dist(model, calc_forces=True)
    E, F, _, _ = pot.forward(s)
    assert abs(E.item() - ref["energy"].item()) < 5e-3 * max(1.0, abs(ref["energy"].item()))
    dF = (F.double().cpu() - ref["forces"]).abs().max().item()
    assert dF < 1e-4, f"force error {dF} exceeds 1e-4 eV/A"


@requires_gpu
def test_fused_policy_equivalence():
    """Both first-layer paths (fused edge-MLP kernel vs rocBLAS +
    gather_add) must agree end-to-end through the engine — guards the
    grad-mode/size policy branch in chgnet._use_fused."""
    import os

    from distmlip_amd.model import CHGNetCore
    from distmlip_amd.runtime import SpmdEngine
    from distmlip_amd.structures import diamond_si

    s = diamond_si((10, 2, 2), jitter=0.1, seed=5)
    core = CHGNetCore.seeded(seed=0)
    outs = {}
    for mode in ("DM_FUSED_MLP", "DM_NO_FUSED_MLP"):
        os.environ[mode] = "1"
        try:
            eng = SpmdEngine(core.float(), world=1, threads=4)
            outs[mode] = eng.step(s)
        finally:
            os.environ.pop(mode, None)
    dE = abs(outs["DM_FUSED_MLP"]["energy"].item()
             - outs["DM_NO_FUSED_MLP"]["energy"].item())
    dF = (outs["DM_FUSED_MLP"]["forces_owned"]
          - outs["DM_NO_FUSED_MLP"]["forces_owned"]).abs().max().item()
    assert dE < 1e-3, dE
    assert dF < 2e-4, dF


@requires_gpu
def test_spmd_engine_world1_vs_oracle():
    """The bench/production path (SpmdEngine at world=1, per-rank geometry,
    HIP kernels) against the fp64 oracle."""
    from distmlip_amd.model import CHGNetCore
    from distmlip_amd.runtime import SpmdEngine
    from distmlip_amd.structures import diamond_si
    from oracle.chgnet_ref import oracle_forward
    from oracle.graph_ref import brute_force_neighbors

    s = diamond_si((12, 2, 2), jitter=0.12, seed=2)
    g = brute_force_neighbors(s.frac_coords, s.lattice, s.pbc, 6.0, 3.0)
    core = CHGNetCore.seeded(seed=0)
    ref = oracle_forward(core.double(), s, g["src"], g["dst"], g["offsets"],
                         g["within_bond_r"], dtype=torch.float64)
    eng = SpmdEngine(core.float(), world=1, threads=4)
    out = eng.step(s)
    assert abs(out["energy"].item() - ref["energy"].item()) < 5e-3 * max(
        1.0, abs(ref["energy"].item()))
    F = np.zeros((s.num_atoms, 3))
    F[out["global_ids_owned"]] = out["forces_owned"].double().cpu().numpy()
    dF = np.abs(F - ref["forces"].numpy()).max()
    assert dF < 1e-4, f"engine force error {dF}"


@requires_gpu
def test_gpu_graph_matches_cpu_builder():
    """GPU cell-list graph build == native CPU builder: identical edge,
    bond and line-edge SETS (fp64 decisions replicated exactly)."""
    from distmlip_amd import capi, gpu_graph
    from distmlip_amd.structures import bcc_li
    s = bcc_li(8, jitter=0.12, seed=4)      # 1024 atoms, 28 A cubic
    assert gpu_graph.supported(s, 6.0)
    pd = gpu_graph.build(s, 6.0, 3.0, 1e-8, True, torch.device("cuda:0"))
    cpu = capi.get_subgraphs_fast(s.cart_coords, 6.0, s.pbc, s.lattice, 1,
                                  3.0, 1e-8, 4, True, s.frac_coords,
                                  return_csr=True)
    cpu_t, cpu_csr = cpu

    def keys(src, dst, off):
        src = np.asarray(src, dtype=np.int64)
        dst = np.asarray(dst, dtype=np.int64)
        off = np.asarray(off, dtype=np.int64)
        return set(zip(src.tolist(), dst.tolist(), off[:, 0].tolist(),
                       off[:, 1].tolist(), off[:, 2].tolist()))

    g_src = pd.src.cpu().numpy()
    g_dst = pd.dst.cpu().numpy()
    g_off = pd.off_i8.cpu().numpy()
    k_gpu = keys(g_src, g_dst, g_off)
    k_cpu = keys(cpu_t[0][0], cpu_t[1][0], cpu_csr[0]["offsets_i8"])
    assert k_gpu == k_cpu, (len(k_gpu), len(k_cpu),
                            len(k_gpu ^ k_cpu))

    # bond sets via the underlying edge keys
    mde = pd.map_de.cpu().numpy()
    b_gpu = keys(g_src[mde], g_dst[mde], g_off[mde])
    cmde = np.asarray(cpu_t[14][0])
    b_cpu = keys(np.asarray(cpu_t[0][0])[cmde], np.asarray(cpu_t[1][0])[cmde],
                 cpu_csr[0]["offsets_i8"][cmde])
    assert b_gpu == b_cpu

    # line-edge sets as (edge-key(src bde), edge-key(dst bde))
    def ek(src, dst, off, i):
        return (int(src[i]), int(dst[i]), int(off[i, 0]), int(off[i, 1]),
                int(off[i, 2]))
    ls = pd.l_src.cpu().numpy()
    ld = pd.l_dst.cpu().numpy()
    lines_gpu = set((ek(g_src, g_dst, g_off, mde[a]),
                     ek(g_src, g_dst, g_off, mde[b]))
                    for a, b in zip(ls.tolist(), ld.tolist()))
    cls = np.asarray(cpu_t[9][0])
    cld = np.asarray(cpu_t[10][0])
    cude = np.asarray(cpu_t[15][0])
    inv_ude = np.empty_like(cude)
    inv_ude[cude] = np.arange(len(cude))
    csrc = np.asarray(cpu_t[0][0])
    cdst = np.asarray(cpu_t[1][0])
    coff = cpu_csr[0]["offsets_i8"]
    lines_cpu = set((ek(csrc, cdst, coff, cmde[inv_ude[a]]),
                     ek(csrc, cdst, coff, cmde[inv_ude[b]]))
                    for a, b in zip(cls.tolist(), cld.tolist()))
    assert lines_gpu == lines_cpu


@requires_gpu
def test_engine_gpu_build_matches_cpu_build():
    """E+F through the GPU-built graph == through the CPU-built graph."""
    from distmlip_amd.model import CHGNetCore
    from distmlip_amd.runtime import SpmdEngine
    from distmlip_amd.structures import bcc_li
    s = bcc_li(8, jitter=0.12, seed=4)
    core = CHGNetCore.seeded(seed=0).float()
    e_gpu = SpmdEngine(core, world=1, threads=4, gpu_build="auto")
    e_cpu = SpmdEngine(core, world=1, threads=4, gpu_build="off")
    o1 = e_gpu.step(s)
    o2 = e_cpu.step(s)
    assert abs(o1["energy"].item() - o2["energy"].item()) < 1e-4 * max(
        1.0, abs(o2["energy"].item()))
    F1 = np.zeros((s.num_atoms, 3))
    F1[o1["global_ids_owned"]] = o1["forces_owned"].cpu().numpy()
    F2 = np.zeros((s.num_atoms, 3))
    F2[o2["global_ids_owned"]] = o2["forces_owned"].cpu().numpy()
    assert np.abs(F1 - F2).max() < 5e-4, np.abs(F1 - F2).max()


@requires_gpu
def test_native_so_loaded():
    """Guard against silent eager fallback: the HIP extension must be the
    library the process actually loaded."""
    from distmlip_amd.ops import hip_lib
    lib = hip_lib()
    assert "libdistmlip_hip" in (lib._name or "")
    import distmlip_amd.capi as capi
    assert "libdistmlip_graph" in capi.graph_lib()._name
