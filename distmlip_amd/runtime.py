"""SPMD runtime — one process per GPU over torch.distributed (RCCL/xGMI).

The MI355X-native replacement for the reference's single-process
multi-GPU loop (chgnet.py:296-368 + the synchronous p2p slice copies of
dist.py:356): each rank owns one slab partition, computes its partition's
kernels, and exchanges border-node features with its slab neighbors via
point-to-point isend/irecv (RCCL over xGMI; slab topology means <= 2 peers
per rank).  The ONLY collective on the data path is the final scalar
energy all-reduce (SURVEY.md §2 "MI355X-native equivalent").

`HaloExchange` is an autograd.Function: forward replaces each ghost
("from") slice with the owner's fresh "to" slice; backward routes the
ghost-slice gradients back to the owners and ADDS them into the "to"
slices' gradients (the distributed transpose of the copy).  This is what
makes `torch.autograd.backward(E_local)` on every rank produce exactly the
reference's forces (pes.py:121-124) without a central GPU.

Divergences from the reference flow (documented in DESIGN.md):
  * positions and bond geometry are computed PER-RANK from the replicated
    structure (removes the GPU0-centralized bond_vec serialization point,
    chgnet.py:96-100); ghost BOND geometry still arrives by one
    bond-halo at forward start, exactly like chgnet.py:157-164.
  * forces are assembled by one reverse halo-add of position gradients;
    each rank ends with exact forces for its owned atoms.
"""
from __future__ import annotations

from copy import deepcopy
from typing import List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist

from distmlip_amd.conv import (_AtomConvFn, _BondConvFn,
                               conv_fn_available)
from distmlip_amd.chgnet import (
    PartitionData,
    _packed_weights,
    gated_mlp_split3,
    gated_mlp_split4,
)
from distmlip_amd.dist import Distributed
from distmlip_amd.model import (
    CHGNetCore,
    compute_theta,
    fourier_expansion,
)
from distmlip_amd.ops_base import default_ops_factory


def halo_plan(markers, rank: int, P: int) -> List[Tuple[int, int, int, int, int]]:
    """[(peer, send_start, send_end, recv_start, recv_end)] from the marker
    contract (dist.py:44-51): send my 'to peer' slice, receive into my
    'from peer' slice."""
    plan = []
    for q in range(P):
        if q == rank:
            continue
        ss, se = int(markers[1 + q]), int(markers[1 + q + 1])
        rs, re = int(markers[1 + P + q]), int(markers[1 + P + q + 1])
        if ss != se or rs != re:
            plan.append((q, ss, se, rs, re))
    return plan


import os as _os

# emergency fallback: DM_HALO_GLOO=1 stages halo slices through host
# memory over a gloo group
_GLOO_HALO_GROUP = None

# selection tolerance of the graph builder (dist.py tol=1e-8 mirror)
_SEL_TOL = 1e-8


def _exchange(feat: torch.Tensor, plan, reverse: bool = False) -> dict:
    """Grouped p2p exchange (batch_isend_irecv: one RCCL group, deadlock-
    free by construction); returns {peer: recv_buffer}.  reverse=True swaps
    roles (send my recv-slices, receive for my send-slices) — the backward
    direction."""
    if _GLOO_HALO_GROUP is not None and feat.device.type == "cuda":
        # explicit device sync: .cpu() orders only the CALLING thread's
        # current stream; inside autograd-engine callbacks that is not
        # guaranteed to be the stream that produced `feat` (observed as a
        # DETERMINISTIC stale-send on the UMA P=2 backward, where adding
        # any instrumentation sync made the corruption vanish)
        torch.cuda.synchronize(feat.device)
        recvs_cpu = _exchange_via(feat.cpu(), plan, reverse, _GLOO_HALO_GROUP)
        return {q: b.to(feat.device) for q, b in recvs_cpu.items()}
    return _exchange_via(feat, plan, reverse, None)


def _exchange_via(feat, plan, reverse, group):
    recvs, p2p, keep = {}, [], []
    for (q, ss, se, rs, re) in plan:
        c, d = (ss, se) if reverse else (rs, re)
        if d > c:
            buf = torch.empty((d - c,) + tuple(feat.shape[1:]),
                              dtype=feat.dtype, device=feat.device)
            recvs[q] = buf
            p2p.append(dist.P2POp(dist.irecv, buf, q, group=group))
    for (q, ss, se, rs, re) in plan:
        a, b = (rs, re) if reverse else (ss, se)
        if b > a:
            sbuf = feat[a:b].contiguous()
            keep.append(sbuf)                        # alive until wait
            p2p.append(dist.P2POp(dist.isend, sbuf, q, group=group))
    if p2p:
        for r in dist.batch_isend_irecv(p2p):
            r.wait()
    return recvs


class _HaloSeq:
    """RCCL p2p messages match by POSTING ORDER (no tags): every rank must
    run its halo exchanges in the same order.  Forward order is program
    order; backward order is enforced to be exactly reversed — a violation
    raises instead of silently crossing buffers.  One instance per engine
    step (created in SpmdEngine.step), so engines in the same process
    cannot cross-contaminate sequence numbers."""

    def __init__(self):
        self.fwd = 0
        self.expect_bwd = None


class HaloExchange(torch.autograd.Function):
    @staticmethod
    def forward(ctx, feat, plan, seq_state):
        ctx.plan = plan
        ctx.seq_state = seq_state
        ctx.seq = seq_state.fwd
        seq_state.fwd += 1
        recvs = _exchange(feat.detach(), plan, reverse=False)
        out = feat.detach().clone()
        for (q, ss, se, rs, re) in plan:
            if re > rs:
                out[rs:re] = recvs[q]
        return out

    @staticmethod
    def backward(ctx, grad):
        # the autograd engine drains its ready queue in descending
        # sequence-number order, so live exchanges run in strictly
        # DECREASING forward order on every rank (dead branches are skipped
        # identically on all ranks — graph structure is rank-invariant).
        if (dd := _os.environ.get("DM_HALO_DUMP_DIR")):
            import numpy as _np
            _rk = dist.get_rank() if dist.is_initialized() else 0
            _np.save(f"{dd}/halo_bwd_in_seq{ctx.seq}_{_rk}.npy",
                     grad.detach().double().cpu().numpy())
        st = ctx.seq_state
        if st.expect_bwd is None:
            st.expect_bwd = st.fwd
        if ctx.seq >= st.expect_bwd:
            raise RuntimeError(
                f"halo backward out of order: got seq {ctx.seq} after "
                f"{st.expect_bwd} — rank-divergent exchange order")
        st.expect_bwd = ctx.seq
        plan = ctx.plan
        g = grad.contiguous()
        recvs = _exchange(g, plan, reverse=True)     # send ghost grads home
        g = g.clone()
        for (q, ss, se, rs, re) in plan:
            if re > rs:
                g[rs:re] = 0                         # ghosts aren't my vars
        for (q, ss, se, rs, re) in plan:
            if se > ss:
                g[ss:se] += recvs[q]                 # peers' ghost grads
        return g, None, None


class SpmdEngine:
    """Per-rank CHGNet E+F engine (the bench/production path)."""

    def __init__(self, core: CHGNetCore, world: int, threads: int = 8,
                 use_bond_graph: bool = True, device: Optional[str] = None,
                 ops=None, graph_backend=None, checkpoint: str = "auto",
                 gpu_build: str = "auto"):
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        assert world == 1 or dist.is_initialized(), \
            "world > 1 needs an initialized torch.distributed process group"
        global _GLOO_HALO_GROUP
        if world > 1 and _os.environ.get("DM_HALO_GLOO") == "1" \
                and _GLOO_HALO_GROUP is None:
            _GLOO_HALO_GROUP = dist.new_group(backend="gloo")
        self.world = world
        self.config = core.config
        self.use_bond_graph = use_bond_graph and core.config.use_bond_graph
        if device is None:
            device = f"cuda:{torch.cuda.current_device()}"
        self.device = torch.device(device)
        self.core = deepcopy(core).to(self.device).eval()
        # inference engine: frozen parameters mean no GEMM saves its
        # [E,*] input for dW and backward drops every dW GEMM — large
        # memory and time win on the force backward
        self.core.requires_grad_(False)
        self.ops = ops if ops is not None else default_ops_factory(self.device)
        if (self.device.type == "cuda"
                and _os.environ.get("DM_NO_TUNABLEOP") != "1"
                and hasattr(torch.cuda, "tunable")):
            # per-shape GEMM algorithm tuning (rocBLAS vs hipBLASLt):
            # measured -14% step time at li100k (run 29, 177 -> 151 ms).
            # A pre-tuned result file for the bench shapes ships with the
            # package (validators pin GPU + library versions, so it only
            # loads on a matching box); missing shapes still tune at
            # first use, amortized into warmup.  DM_NO_TUNABLEOP=1
            # disables.
            torch.cuda.tunable.enable(True)
            pretuned = _os.path.join(_os.path.dirname(__file__),
                                     "tunableop_gfx950.csv")
            if _os.path.exists(pretuned) and not getattr(
                    SpmdEngine, "_tunableop_loaded", False):
                try:
                    torch.cuda.tunable.read_file(pretuned)
                except Exception:
                    pass
                SpmdEngine._tunableop_loaded = True
        self.threads = threads
        self.graph_backend = graph_backend
        self.float_th = next(self.core.parameters()).dtype
        # activation checkpointing of the conv segments: at ~42M edges the
        # saved [E,64] activations of 4 blocks would exceed the 288 GB HBM;
        # recompute trades ~1/3 more kernel work for ~6x less live memory.
        # Halo exchanges stay OUTSIDE checkpointed segments (a re-run of a
        # comm op during backward would desynchronize the ranks).
        self.checkpoint = checkpoint
        # GPU-resident graph build for the single-partition path
        # (distmlip_amd/gpu_graph.py); "auto" engages on diagonal-lattice
        # full-PBC structures when world == 1 and no custom backend is set
        self.gpu_build = gpu_build

    # -- graph ------------------------------------------------------------

    def build_graph(self, structure, skin: float = 0.0) -> Distributed:
        # focused slab build: each rank builds only its slab + margin —
        # per-rank graph cost stays O(atoms per slab), the scaling-critical
        # property (a full build per rank would grow with TOTAL atoms)
        focus = self.rank if (self.world > 1 and self.graph_backend is None) \
            else -1
        return Distributed.create_distributed(
            cart_coords=structure.cart_coords,
            frac_coords=structure.frac_coords,
            lattice_matrix=structure.lattice,
            num_partitions=self.world, pbc=structure.pbc,
            cutoff=self.config.cutoff + skin,
            three_body_cutoff=self.config.three_body_cutoff +
            (skin if self.use_bond_graph else 0.0),
            use_bond_graph=self.use_bond_graph,
            num_threads=self.threads, backend=self.graph_backend,
            focus_partition=focus)

    # -- Verlet-skin graph reuse across MD steps ----------
    #
    # Build once with cutoffs enlarged by `skin`; while no atom has moved
    # more than skin/2 since the build, the enlarged edge list is a
    # superset of the true one, and per-step distance MASKS on the shared
    # message weights make the result EXACTLY what a fresh build would
    # give: masked terms are exact fp zeros, so every reduction is
    # bit-identical to the fresh-graph sum.  This removes the per-step
    # CPU rebuild the reference pays (pes.py:75-85; 31 s at 1M atoms).

    def step_verlet(self, structure, skin: float = 1.0, **kw):
        rebuild = True
        if getattr(self, "_vcache", None) is not None:
            c = self._vcache
            # a lattice change (NPT / cell rescale) invalidates the stored
            # edge superset even with unchanged frac coords — compare the
            # cached lattice, not just displacement
            if (c["skin"] == skin and len(c["frac"]) == structure.num_atoms
                    and np.array_equal(c["lattice"],
                                       np.asarray(structure.lattice))):
                d = structure.frac_coords - c["frac"]
                d = (d + 0.5) % 1.0 - 0.5                 # minimal image
                disp2 = ((d @ structure.lattice) ** 2).sum(1)
                rebuild = float(disp2.max()) > (skin / 2) ** 2
        if rebuild:
            self._vcache = {
                "skin": skin,
                "lattice": np.asarray(structure.lattice).copy(),
                "frac": structure.frac_coords.copy(),
                "dist_info": self.build_graph(structure, skin=skin),
                "rebuilds": getattr(self, "_vcache", None) and
                self._vcache["rebuilds"] + 1 or 1,
            }
            frac_eff = None
        else:
            # atoms that re-wrapped across a periodic boundary since the
            # build would invalidate the stored edge images; evaluate at
            # build-frame coordinates + minimal-image displacement instead
            d = structure.frac_coords - self._vcache["frac"]
            frac_eff = self._vcache["frac"] + ((d + 0.5) % 1.0 - 0.5)
        return self.step(structure, dist_info=self._vcache["dist_info"],
                         mask_cutoffs=True, frac_override=frac_eff, **kw)

    # -- one E+F step ------------------------------------------------------

    def step(self, structure, dist_info: Optional[Distributed] = None,
             calc_stresses: bool = False, mask_cutoffs: bool = False,
             frac_override: Optional[np.ndarray] = None):
        """Returns (total_energy_scalar, owned_forces[n_owned,3]).

        total_energy includes scale/shift + element refs (pes.py:109-113);
        owned forces are fully assembled (reverse halo-add applied)."""
        r, P = self.rank, self.world
        cfg, core, ops, dev = self.config, self.core, self.ops, self.device
        ft = self.float_th

        gpu_pd = None
        if dist_info is None:
            from distmlip_amd import gpu_graph
            if (self.gpu_build == "auto"
                    and self.graph_backend is None
                    and dev.type == "cuda"
                    and _os.environ.get("DM_NO_GPU_BUILD") != "1"
                    and gpu_graph.supported(structure, cfg.cutoff)):
                if P == 1:
                    gpu_pd = gpu_graph.build(
                        structure, cfg.cutoff, cfg.three_body_cutoff, 1e-8,
                        self.use_bond_graph, dev,
                        frac_override=frac_override)
                else:
                    # GPU SPMD slab build: full-box GPU NL + on-device
                    # partition assembly (replaces the 0.4 s/rank CPU
                    # focused build — the N=8 scaling bottleneck)
                    gpu_pd = gpu_graph.build_partition(
                        structure, P, r, cfg.cutoff,
                        cfg.three_body_cutoff, 1e-8, self.use_bond_graph,
                        dev, frac_override=frac_override)
            else:
                dist_info = self.build_graph(structure)
        halo_seq = _HaloSeq()                 # fresh guard per step
        if gpu_pd is not None and P == 1:
            pd = gpu_pd
            plan, line_plan = [], []
            gids = None                       # identity: local ids == global
            n_owned = pd.n_atoms
        elif gpu_pd is not None:
            pd = gpu_pd
            plan = halo_plan(pd.markers, r, P)
            line_plan = halo_plan(pd.line_markers, r, P) \
                if self.use_bond_graph else None
            gids = pd.global_ids
            n_owned = pd.n_owned
        else:
            pd = PartitionData(dist_info, r, dev, self.use_bond_graph)
            plan = halo_plan(dist_info.markers[r], r, P)
            line_plan = halo_plan(dist_info.line_markers[r], r, P) \
                if self.use_bond_graph else None
            gids = np.asarray(dist_info.global_ids[r])
            n_owned = dist_info.num_owned_atoms(r)

        def _halo(feat, pl):
            return feat if not pl else HaloExchange.apply(feat, pl, halo_seq)

        # ---- local geometry (per-rank; no GPU0 serialization point)
        lat0 = torch.tensor(np.asarray(structure.lattice), dtype=ft, device=dev)
        strain = lat0.new_zeros(3, 3)
        if calc_stresses:
            strain.requires_grad_(True)
        lattice = lat0 @ (torch.eye(3, device=dev, dtype=ft) + strain)

        frac_src = np.asarray(frac_override if frac_override is not None
                              else structure.frac_coords)
        frac_local = torch.tensor(frac_src if gids is None else frac_src[gids],
                                  dtype=ft, device=dev)
        pos = frac_local @ lattice
        if not pos.requires_grad:
            pos.requires_grad_(True)
        pos.retain_grad()

        spec = np.asarray(structure.species)
        species_local = torch.tensor(spec if gids is None else spec[gids],
                                     dtype=torch.long, device=dev)

        if gpu_pd is not None:
            off_local = pd.off_i8.to(ft)
            csr = None
        elif (csr := dist_info.csr_parts[r]
              if getattr(dist_info, "csr_parts", None) else None) is not None:
            # int8 integer images straight from the builder: 15 MB H2D at
            # 5M edges instead of a 120 MB host-side f64 gather+convert
            off_local = torch.from_numpy(csr["offsets_i8"]).to(dev).to(ft)
        else:
            egids = np.asarray(dist_info.L2G_DE_mapping_list[r])
            off_local = torch.tensor(np.asarray(dist_info.py_offsets)[egids],
                                     dtype=ft, device=dev)
        offshift = off_local @ lattice

        bond_vec, bond_dist, bond_expansion = ops.edge_geom_rbf(
            pos, offshift, core.rbf_freq_atom, cfg.cutoff,
            cfg.cutoff_exponent, pd)

        # Verlet-skin mode: the edge list is a superset built at
        # cutoff+skin; mask the shared message weights by the TRUE cutoffs
        # so every out-of-range contribution is an exact zero.
        # The mask distances are recomputed in FP64 from the same
        # frac/lattice/offset inputs the builder uses: a fresh build
        # selects edges/bonds by fp64 `d < cutoff`, and an fp32 distance
        # flips SELECTION-boundary members (weight layers carry a bias,
        # so a flipped 3 A bond shifts forces O(1) — 54 flipped atoms at
        # li100k, run 34 forensics).
        edge_mask = None
        d64sq = None
        if mask_cutoffs:
            with torch.no_grad():
                # from the ORIGINAL fp64 frac/lattice (the fp32 model
                # tensors carry ~1e-5 position rounding at a 130 A box —
                # wider than the selection-flip window)
                lat64 = torch.tensor(np.asarray(structure.lattice),
                                     dtype=torch.float64, device=dev)
                frac64 = torch.tensor(
                    frac_src if gids is None else frac_src[gids],
                    dtype=torch.float64, device=dev)
                pos64 = frac64 @ lat64
                off64 = off_local.double() @ lat64
                bv64 = (pos64[pd.dst.long()] + off64 - pos64[pd.src.long()])
                d64sq = (bv64 * bv64).sum(1)
            # the builder selects by d^2 < r^2 + tol (fpis.c:760-764,
            # tol=1e-8): replicate the COMPARATOR, not just the radius —
            # strict `<` would drop exactly-on-boundary members a fresh
            # build keeps (tests/test_verlet_skin.py boundary test)
            edge_mask = (d64sq < cfg.cutoff ** 2 + _SEL_TOL).to(
                ft).unsqueeze(1)

        v = core.atom_embedding(species_local)
        e = core.bond_embedding(bond_expansion)

        use_bg = self.use_bond_graph
        if use_bg:
            # owned bond geometry from local edges; ghosts by one halo each
            # (reference chgnet.py:129-164).  map_ude is a permutation of
            # the owned range and the halo fills every ghost slot, so the
            # buffers start uninitialized (no zero-fill pass).
            nd_dist = torch.empty(pd.n_bonds, dtype=ft, device=dev).index_copy(
                0, pd.map_ude, bond_dist[pd.map_de])
            nd_vec = torch.empty(pd.n_bonds, 3, dtype=ft, device=dev).index_copy(
                0, pd.map_ude, bond_vec[pd.map_de])
            nd_dist = _halo(nd_dist.unsqueeze(1), line_plan).squeeze(1)
            nd_vec = _halo(nd_vec, line_plan)

            bond_mask = None
            if mask_cutoffs:
                with torch.no_grad():
                    nd_d64sq = torch.empty(pd.n_bonds, dtype=torch.float64,
                                           device=dev).index_copy(
                        0, pd.map_ude, d64sq[pd.map_de])
                    nd_d64sq = _halo(nd_d64sq.unsqueeze(1),
                                     line_plan).squeeze(1)
                bond_mask = (nd_d64sq < cfg.three_body_cutoff ** 2
                             + _SEL_TOL).to(ft).unsqueeze(1)
            exp3 = ops.rbf_env(nd_dist, core.rbf_freq_bond,
                               cfg.three_body_cutoff, cfg.cutoff_exponent)
            theta = compute_theta(
                ops.gather(nd_vec, pd.l_src, csr=pd.line_src_csr),
                ops.gather(nd_vec, pd.l_dst, csr=pd.line_dst_csr))
            a = core.angle_embedding(fourier_expansion(theta, core.angle_freq))
            n = torch.empty(pd.n_bonds, cfg.dim, dtype=ft, device=dev).index_copy(
                0, pd.map_ude, e[pd.map_de])
            n = _halo(n, line_plan)

        d = cfg.dim
        ckpt = self.checkpoint == "on" or (
            self.checkpoint == "auto" and len(pd.src) > 10_000_000)

        def _ck(fn, *args):
            if ckpt:
                return torch.utils.checkpoint.checkpoint(
                    fn, *args, use_reentrant=False)
            return fn(*args)

        # shared message weights (chgnet.py:272-294).  Under checkpointing
        # the [E,64]/[B,64] weight tensors are recomputed inside each
        # segment from the small [*,9] expansions instead of being held
        # for the whole forward (2x [E,64] = 22 GB at 1M atoms).
        def _m(w, m):
            return w if m is None else w * m

        if ckpt:
            w_ab = w_bb = None
            w_3b = None
        else:
            w_ab = _m(core.atom_bond_weights(bond_expansion), edge_mask)
            w_bb = _m(core.bond_bond_weights(bond_expansion), edge_mask)
            w_3b = _m(core.threebody_bond_weights(exp3), bond_mask) \
                if use_bg else None

        def atom_conv_body(layer_idx_t, v, e, bexp):
            blk = core.atom_convs[int(layer_idx_t)]
            wbb = w_bb if w_bb is not None else \
                _m(core.bond_bond_weights(bexp), edge_mask)
            wab = w_ab if w_ab is not None else \
                _m(core.atom_bond_weights(bexp), edge_mask)
            pe = _packed_weights(blk.edge_mlp)
            if d % 2 == 0 and conv_fn_available(ops, pe):
                # recording path: one Function per block, hand-sequenced
                # reverse pass (distmlip_amd/conv.py)
                pn = _packed_weights(blk.node_mlp)
                return _AtomConvFn.apply(v, e, wbb.contiguous(),
                                         wab.contiguous(), pd, ops,
                                         pe[:4], pn[:4], d)
            e = gated_mlp_split3(blk.edge_mlp, v, e, pd, ops, d,
                                 w=wbb, base=e)
            msg = gated_mlp_split3(blk.node_mlp, v, e, pd, ops, d, w=wab)
            v = ops.scatter_edges(msg, pd, base=v)
            return v, e

        def atom_conv(layer_i, v, e):
            return _ck(atom_conv_body,
                       torch.tensor(layer_i % cfg.n_blocks), v, e,
                       bond_expansion)

        # -- whole-graph hand-sequenced reverse (single partition, no
        # verlet masks): every block + the final conv in ONE Function
        # whose backward is sequenced by hand (conv.py)
        from distmlip_amd.conv import _WholeGraphFn, whole_graph_available
        # measured A/B (r2, same box): keep-mode wins at li100k (137.1 vs
        # 140.5 ms/step); the streamed-recompute mode LOSES at si1m
        # (2486 vs 2081 ms — the lean re-runs outweigh the saved adds), so
        # checkpointed steps stay on the op-by-op path unless forced
        wg_ckpt_ok = _os.environ.get("DM_WHOLE_GRAPH_CKPT") == "1"
        if (P == 1 and not mask_cutoffs and d % 2 == 0
                and (not ckpt or wg_ckpt_ok)
                and whole_graph_available(
                    ops, _packed_weights(core.atom_convs[0].edge_mlp))):
            wbb_m = w_bb if w_bb is not None else \
                core.bond_bond_weights(bond_expansion)
            wab_m = w_ab if w_ab is not None else \
                core.atom_bond_weights(bond_expansion)
            w3_m = None
            if use_bg:
                w3_m = w_3b if w_3b is not None else \
                    core.threebody_bond_weights(exp3)
            packs = {"atom": [(_packed_weights(b.edge_mlp),
                               _packed_weights(b.node_mlp))
                              for b in core.atom_convs]}
            if use_bg:
                packs["bond"] = [_packed_weights(b.bond_mlp)
                                 for b in core.bond_convs]
                packs["angle"] = [_packed_weights(b.angle_mlp)
                                  for b in core.bond_convs]
            v_final, v_mid = _WholeGraphFn.apply(
                e, a if use_bg else None, wbb_m.contiguous(),
                wab_m.contiguous(),
                w3_m.contiguous() if w3_m is not None else None, v, pd,
                ops, packs, d, cfg.n_blocks, use_bg, ckpt)
            site_props = core.sitewise_readout(v_mid)
            v = v_final
            return self._finish(structure, core, pos, strain, plan,
                                species_local, n_owned, gids, v,
                                site_props, calc_stresses, P)

        for layer_i in range(cfg.n_blocks - 1):           # chgnet.py:296-368
            v, e = atom_conv(layer_i, v, e)
            if use_bg:
                n = n.index_copy(0, pd.map_ude, e[pd.map_de])   # edge_to_bond
                n = _halo(n, line_plan)
                v = _halo(v, plan)

                blk = core.bond_convs[layer_i]

                def bond_body(n, a, v, e3, _blk=blk):
                    w3 = w_3b if w_3b is not None else \
                        _m(core.threebody_bond_weights(e3), bond_mask)
                    pb = _packed_weights(_blk.bond_mlp)
                    if (d % 2 == 0 and conv_fn_available(ops, pb)
                            and hasattr(ops, "r_gather_add4")):
                        mask_l = None
                        if bond_mask is not None:
                            mask_l = ops.r_gather_ldst(
                                bond_mask.contiguous(), pd)
                        return _BondConvFn.apply(
                            n, a.contiguous(), v, w3.contiguous(), pd, ops,
                            pb[:4], d, mask_l)
                    wl = ops.gather(w3, pd.l_src, csr=pd.line_src_csr)
                    if bond_mask is not None:
                        # Verlet mode: a message into a MASKED (superset)
                        # bond would later leak back into a live atom edge
                        # through bond_to_edge — kill by destination too
                        wl = wl * ops.gather(bond_mask, pd.l_dst,
                                             csr=pd.line_dst_csr)
                    msg = gated_mlp_split4(_blk.bond_mlp, n, a, v, pd, ops, d,
                                           w=wl)
                    return ops.scatter_lines(msg, pd, base=n)

                n = _ck(bond_body, n, a, v, exp3)
                e = e.index_copy(0, pd.map_de, n[pd.map_ude])    # bond_to_edge

                if layer_i < cfg.n_blocks - 2:
                    # the LAST bond block's angle update is dead compute
                    # (a never feeds the energy after it; the reference
                    # still computes it, chgnet.py:353-368) — skip it and
                    # the halo that feeds it
                    n = _halo(n, line_plan)

                    def angle_body(n, a, v, _blk=blk):
                        return gated_mlp_split4(_blk.angle_mlp, n, a, v, pd,
                                                ops, d, base=a)

                    a = _ck(angle_body, n, a, v)
            else:
                v = _halo(v, plan)

        site_props = core.sitewise_readout(v)              # chgnet.py:391-398

        v, e = atom_conv(-1, v, e)                         # final atom block
        v = _halo(v, plan)

        return self._finish(structure, core, pos, strain, plan,
                            species_local, n_owned, gids, v, site_props,
                            calc_stresses, P)

    def _finish(self, structure, core, pos, strain, plan, species_local,
                n_owned, gids, v, site_props, calc_stresses, P):
        """Readout + force assembly shared by the op-by-op and
        whole-graph paths (reference pes.py:101-145 semantics)."""
        dev = self.device
        atom_e = core.final_layer(v)
        e_local_raw = atom_e[:n_owned].sum()

        refs_local = core.element_refs[species_local[:n_owned]].sum()

        # forces: backward through std * E_local (mean/refs are pos-free)
        loss = core.data_std * e_local_raw
        grads = [pos, strain] if calc_stresses else [pos]
        gv = torch.autograd.grad(loss, grads)
        pos_grad = gv[0]

        # reverse halo-add of position gradients -> exact owned forces
        recvs = _exchange(pos_grad, plan, reverse=True)
        pos_grad = pos_grad.clone()
        for (q, ss, se, rs, re) in plan:
            if se > ss:
                pos_grad[ss:se] += recvs[q]
        forces_owned = -pos_grad[:n_owned]

        # scalar reductions for reporting
        scal = torch.stack([e_local_raw.detach(), refs_local.detach()])
        if P > 1:
            if not getattr(self, "_validated", False):
                # one-time partition sanity: owned sets must tile the
                # structure exactly across ranks
                t = torch.tensor([float(n_owned)], device=dev)
                dist.all_reduce(t)
                assert int(round(t.item())) == structure.num_atoms, (
                    f"owned atoms across ranks {int(t.item())} != "
                    f"{structure.num_atoms} — rank-divergent partitioning")
                self._validated = True
            dist.all_reduce(scal)
        total_e = core.data_std.detach() * scal[0] + core.data_mean.detach() \
            + scal[1]

        out = {"energy": total_e, "forces_owned": forces_owned,
               "site_props_owned": site_props[:n_owned].detach(),
               "n_owned": n_owned,
               "global_ids_owned": (np.arange(n_owned) if gids is None
                                    else gids[:n_owned])}
        if calc_stresses:
            sg = gv[1].detach().clone()
            if P > 1:
                dist.all_reduce(sg)
            volume = float(np.abs(np.linalg.det(np.asarray(structure.lattice))))
            out["stress"] = -sg / volume * -160.21766208
        return out
