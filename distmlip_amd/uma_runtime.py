"""UMA/eSCN SPMD runtime — one process per GPU over torch.distributed.

MI355X-native replacement for the reference's thread-pool-per-partition
loop (implementations/uma/escn_md.py:442-500: per-layer block forward +
atom_transfer halo, ThreadPoolExecutor + CUDA events): each rank owns one
slab partition; the halo is the same HaloExchange autograd Function the
CHGNet/MACE engines use, issued once per layer at the reference's
transfer point (escn_md.py:496 — and after the edge-degree embedding,
escn_md.py:416).  bf16 runs under autocast (GEMM-heavy ops in bf16,
reductions in fp32) — the BASELINE config #5 dtype.
"""
from __future__ import annotations

import os as _os
from copy import deepcopy
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

from distmlip_amd import so3, uma_ops
from distmlip_amd.chgnet import PartitionData
from distmlip_amd.dist import Distributed
from distmlip_amd.ops_base import default_ops_factory
from distmlip_amd.runtime import HaloExchange, _HaloSeq, _exchange, halo_plan
from distmlip_amd.uma_model import UMACore


def _flat(x):
    return x.reshape(x.shape[0], -1)


class UmaSpmdEngine:
    """Per-rank UMA E+F engine (config #5 path)."""

    def __init__(self, core: UMACore, world: int, threads: int = 8,
                 device: Optional[str] = None, ops=None,
                 checkpoint: str = "auto", autocast_bf16: bool = False):
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        assert world == 1 or dist.is_initialized()
        self.world = world
        self.config = core.config
        if device is None:
            device = f"cuda:{torch.cuda.current_device()}"
        self.device = torch.device(device)
        self.core = deepcopy(core).to(self.device).eval()
        self.core.requires_grad_(False)
        self.ops = ops if ops is not None else default_ops_factory(self.device)
        self.threads = threads
        self.checkpoint = checkpoint
        self.autocast_bf16 = autocast_bf16
        self.float_th = self.core.sphere_embedding.dtype

    def build_graph(self, structure) -> Distributed:
        focus = self.rank if self.world > 1 else -1
        return Distributed.create_distributed(
            cart_coords=structure.cart_coords,
            frac_coords=structure.frac_coords,
            lattice_matrix=structure.lattice,
            num_partitions=self.world, pbc=structure.pbc,
            cutoff=self.config.cutoff, three_body_cutoff=0.0,
            use_bond_graph=False, num_threads=self.threads,
            focus_partition=focus)

    def step(self, structure, dist_info: Optional[Distributed] = None,
             calc_stresses: bool = False):
        r, P = self.rank, self.world
        cfg, core, ops, dev = self.config, self.core, self.ops, self.device
        ft = self.float_th
        C, S = cfg.sphere_channels, cfg.S

        gpu_pd = None
        if dist_info is None:
            from distmlip_amd import gpu_graph
            if (dev.type == "cuda"
                    and not getattr(self.ops, "is_reference", False)
                    and _os.environ.get("DM_NO_GPU_BUILD") != "1"
                    and gpu_graph.supported(structure, cfg.cutoff)):
                if P == 1:
                    gpu_pd = gpu_graph.build(structure, cfg.cutoff, 0.0,
                                             1e-8, False, dev)
                else:
                    gpu_pd = gpu_graph.build_partition(
                        structure, P, r, cfg.cutoff, 0.0, 1e-8, False, dev)
            else:
                dist_info = self.build_graph(structure)
        if gpu_pd is not None and P == 1:
            pd = gpu_pd
            plan = []
            gids = np.arange(pd.n_atoms)
            n_owned = pd.n_atoms
        elif gpu_pd is not None:
            pd = gpu_pd
            plan = halo_plan(pd.markers, r, P)
            gids = pd.global_ids
            n_owned = pd.n_owned
        else:
            pd = PartitionData(dist_info, r, dev, use_bond_graph=False)
            plan = halo_plan(dist_info.markers[r], r, P)
            gids = np.asarray(dist_info.global_ids[r])
            n_owned = dist_info.num_owned_atoms(r)
        halo_seq = _HaloSeq()

        def _halo(x):
            if not plan:
                if _os.environ.get("DM_FORCE_HALO_NODE") == "1":
                    return HaloExchange.apply(_flat(x), [],
                                              halo_seq).view(-1, S, C)
                return x
            out = HaloExchange.apply(_flat(x), plan, halo_seq)
            if _os.environ.get("DM_UMA_HALO_HOOK") == "1":
                out.register_hook(lambda g: None)
            return out.view(-1, S, C)

        lat0 = torch.tensor(np.asarray(structure.lattice), dtype=ft,
                            device=dev)
        strain = lat0.new_zeros(3, 3)
        if calc_stresses:
            strain.requires_grad_(True)
        lattice = lat0 @ (torch.eye(3, device=dev, dtype=ft) + strain)
        frac_local = torch.tensor(
            np.asarray(structure.frac_coords)[gids], dtype=ft, device=dev)
        pos = frac_local @ lattice
        if not pos.requires_grad:
            pos.requires_grad_(True)
        pos.retain_grad()
        spec = np.asarray(structure.species)
        species = torch.tensor(spec[gids], dtype=torch.long, device=dev)

        if gpu_pd is not None:
            off_local = pd.off_i8.to(ft)
        elif (csr := dist_info.csr_parts[r]
              if getattr(dist_info, "csr_parts", None) else None) is not None:
            off_local = torch.from_numpy(csr["offsets_i8"]).to(dev).to(ft)
        else:
            egids = np.asarray(dist_info.L2G_DE_mapping_list[r])
            off_local = torch.tensor(np.asarray(dist_info.py_offsets)[egids],
                                     dtype=ft, device=dev)

        src_l, dst_l = pd.src.long(), pd.dst.long()
        vectors = pos[dst_l] + off_local @ lattice - pos[src_l]
        lengths = torch.linalg.norm(vectors, dim=1)
        def _want(name):
            st = _os.environ.get("DM_UMA_DUMP_STAGES", "")
            return _os.environ.get("DM_UMA_DUMP_DIR") and (
                st == "1" or name in st.split(","))

        def _dbg(t, name):
            if _want(name):
                _dd = _os.environ["DM_UMA_DUMP_DIR"]
                t.register_hook(lambda g, _n=name: np.save(
                    f"{_dd}/grad_{_n}_{r}.npy",
                    g.detach().double().cpu().numpy()))
            return t

        vectors = _dbg(vectors, "vectors")
        lengths = _dbg(lengths, "lengths")
        if _os.environ.get("DM_UMA_DUMP_ARG"):
            _dd = _os.environ["DM_UMA_DUMP_DIR"]
            np.save(f"{_dd}/vec_{r}.npy",
                    vectors.detach().double().cpu().numpy())
            np.save(f"{_dd}/argmin_{r}.npy",
                    vectors.detach().abs().argmin(1).cpu().numpy())
            np.save(f"{_dd}/srcdst_{r}.npy",
                    torch.stack([pd.src.long(),
                                 pd.dst.long()]).cpu().numpy())

        with torch.autocast("cuda", dtype=torch.bfloat16,
                            enabled=self.autocast_bf16
                            and dev.type == "cuda"):
            R = so3.edge_align_rotation(vectors)
            D = so3.wigner_D_batch(R, cfg.lmax)
            Dinv = D.transpose(-1, -2)
            x_edge = uma_ops.edge_scalars(core, lengths, species[src_l],
                                          species[dst_l])
            x_edge = _dbg(x_edge, "xedge")
            D = _dbg(D, "D")

            x = pos.new_zeros(len(species), S, C)
            x = x.index_put(
                (torch.arange(len(species), device=dev),
                 torch.zeros(len(species), dtype=torch.long, device=dev)),
                core.sphere_embedding[species])

            # edge-degree embedding + scatter (escn_md.py:416 transfer)
            ac = self.autocast_bf16 and dev.type == "cuda"

            def _f32(t):
                # the HIP scatter kernel is f32; under bf16 autocast the
                # messages arrive bf16 — cast only then (never downcast
                # the fp64 CPU test path)
                return t.float() if ac else t

            E_all = len(pd.src)
            deg_chunk = int(_os.environ.get("DM_UMA_DEG_CHUNK",
                                            2_000_000))
            if E_all > deg_chunk:
                accd = torch.zeros_like(x)
                step_e = deg_chunk
                for e0 in range(0, E_all, step_e):
                    e1 = min(e0 + step_e, E_all)
                    md = uma_ops.edge_degree_embed(core, x_edge[e0:e1],
                                                   Dinv[e0:e1])
                    accd = accd.index_add(0, dst_l[e0:e1], _f32(md))
                x = x + accd / cfg.avg_degree
            else:
                med = uma_ops.edge_degree_embed(core, x_edge, Dinv)
                x = x + ops.scatter_edges(
                    _f32(_flat(med)).contiguous(), pd
                ).view(-1, S, C) / cfg.avg_degree
            x = _halo(x)

            x = _dbg(x, "postedge")

            chunk_edges = int(_os.environ.get("DM_UMA_CHUNK",
                                               1_500_000))
            # when the chunked message pass is active its per-chunk
            # checkpoints already bound the [E,S,2C] rotation
            # transients; an outer per-layer checkpoint on top would
            # recompute every chunk forward a second time (mirrors the
            # MACE engine; measured there: 3.03 -> 2.30 s/step at 512k)
            ckpt = self.checkpoint == "on" or (
                self.checkpoint == "auto"
                and 4_000_000 < len(pd.src) <= chunk_edges)
            src_csr = (pd.src_perm, pd.src_row_ptr) \
                if hasattr(pd, "src_perm") else None

            # per-edge rotated tensors ([E, S, 2C]) reach ~100 GB at 11M
            # edges — the message pass runs over contiguous node-range
            # chunks (dst-sorted edges), bounding transients
            E_tot = len(pd.src)
            N_loc = x.shape[0]
            if E_tot > chunk_edges:
                rp = pd.row_ptr.long().cpu().numpy()
                ranges = []            # (n0, n1, e0, e1), node-aligned
                n0 = 0
                for n in range(1, len(rp)):
                    if rp[n] - rp[n0] >= chunk_edges or n == len(rp) - 1:
                        ranges.append((n0, n, int(rp[n0]), int(rp[n])))
                        n0 = n
            else:
                ranges = [(0, N_loc, 0, E_tot)]
            use_rot = uma_ops.rot_kernels_available(x, cfg.lmax) \
                and not getattr(self.ops, "is_reference", False)

            for li, blk in enumerate(core.blocks):
                def body(x, _blk=blk):
                    h = uma_ops.rms_norm(x, _blk.norm1, cfg.lmax)
                    if use_rot:
                        # fused rotation kernels (include/distmlip_hip.h):
                        # rotate-gather per edge, split SO(2) GEMMs,
                        # rotate-scatter per node over the dst CSR
                        hf = h.float().contiguous()
                        gatef = torch.sigmoid(
                            _blk.edge_mlp(x_edge)[:, :cfg.lmax + 1])
                        outs = []
                        for (n0, n1, e0, e1) in ranges:
                            def rchunk(hf_, gate_, _n0=n0, _n1=n1,
                                       _e0=e0, _e1=e1, _b=_blk):
                                Dc = D[_e0:_e1].float()
                                xs = uma_ops._RotGather.apply(
                                    hf_, Dc, pd.src[_e0:_e1])
                                xd = uma_ops._RotGather.apply(
                                    hf_, Dc, pd.dst[_e0:_e1])
                                mt = uma_ops.so2_conv_split(
                                    _b.msg, xs, xd, gate_[_e0:_e1], C)
                                rp_loc = (pd.row_ptr[_n0:_n1 + 1]
                                          - pd.row_ptr[_n0]).contiguous()
                                dst_rel = (pd.dst[_e0:_e1].int()
                                           - _n0).contiguous()
                                return uma_ops._RotScatter.apply(
                                    mt.float(), Dc, rp_loc, dst_rel)
                            if len(ranges) > 1 and torch.is_grad_enabled():
                                outs.append(torch.utils.checkpoint.checkpoint(
                                    rchunk, hf, gatef, use_reentrant=False))
                            else:
                                outs.append(rchunk(hf, gatef))
                        msg = torch.cat(outs, dim=0) if len(outs) > 1 \
                            else outs[0]
                        x2 = x + msg / cfg.avg_degree
                        return x2 + uma_ops.node_ffn(_blk, cfg, x2)
                    if len(ranges) == 1:
                        hf = _flat(h).contiguous()
                        x_src = ops.gather(hf, pd.src,
                                           csr=src_csr).view(-1, S, C)
                        x_dst = ops.gather(hf, pd.dst,
                                           csr=(None, pd.row_ptr)
                                           ).view(-1, S, C)
                        msg = uma_ops.block_message(_blk, cfg, x_src,
                                                    x_dst, x_edge, D, Dinv)
                        x2 = x + ops.scatter_edges(
                            _f32(_flat(msg)).contiguous(), pd
                        ).view(-1, S, C) / cfg.avg_degree
                    else:
                        acc = torch.zeros_like(x)
                        for (_n0, _n1, e0, e1) in ranges:
                            # nested checkpoint: without it the recompute
                            # backward holds EVERY chunk's [E_c, S, 2C]
                            # rotation graph at once (~20 GB x n_chunks)
                            def _chunk(hs, hd, xe, Dc, Dic, _b=_blk):
                                return uma_ops.block_message(
                                    _b, cfg, hs, hd, xe, Dc, Dic)
                            if torch.is_grad_enabled():
                                m = torch.utils.checkpoint.checkpoint(
                                    _chunk, h[src_l[e0:e1]],
                                    h[dst_l[e0:e1]], x_edge[e0:e1],
                                    D[e0:e1], Dinv[e0:e1],
                                    use_reentrant=False)
                            else:
                                m = _chunk(h[src_l[e0:e1]],
                                           h[dst_l[e0:e1]], x_edge[e0:e1],
                                           D[e0:e1], Dinv[e0:e1])
                            acc = acc.index_add(0, dst_l[e0:e1], _f32(m))
                        x2 = x + acc / cfg.avg_degree
                    return x2 + uma_ops.node_ffn(_blk, cfg, x2)

                if ckpt:
                    x = torch.utils.checkpoint.checkpoint(
                        body, x, use_reentrant=False)
                else:
                    x = body(x)
                x = _dbg(x, f"postlayer{li}")
                if li < cfg.num_layers - 1:
                    x = _halo(x)          # escn_md.py:496 transfer point
                    x = _dbg(x, f"posthalo{li}")

            es = uma_ops.energy_head(core, _f32(x))

        loss = es[:n_owned].sum()
        grads = [pos, strain] if calc_stresses else [pos]
        gv = torch.autograd.grad(loss, grads)
        pos_grad = gv[0]
        if _os.environ.get("DM_UMA_DUMP_DIR"):
            np.save(f"{_os.environ['DM_UMA_DUMP_DIR']}/prehalo_{r}.npy",
                    pos_grad.detach().double().cpu().numpy())
            np.save(f"{_os.environ['DM_UMA_DUMP_DIR']}/gids_{r}.npy",
                    np.asarray(gids))
        recvs = _exchange(pos_grad, plan, reverse=True)
        pos_grad = pos_grad.clone()
        for (q, ss, se, rs, re) in plan:
            if se > ss:
                pos_grad[ss:se] += recvs[q]
        forces_owned = -pos_grad[:n_owned]

        scal = loss.detach().clone()
        if P > 1:
            dist.all_reduce(scal)
        out = {"energy": scal, "forces_owned": forces_owned,
               "n_owned": n_owned, "global_ids_owned": gids[:n_owned]}
        if calc_stresses:
            sg = gv[1].detach().clone()
            if P > 1:
                dist.all_reduce(sg)
            volume = float(np.abs(np.linalg.det(
                np.asarray(structure.lattice))))
            out["stress"] = -sg / volume * -160.21766208
        return out
