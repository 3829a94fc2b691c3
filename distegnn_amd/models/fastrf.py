"""FastRF — radial-field network with virtual nodes (coords only).

Parity with reference models/FastRF.py (GCL_RF_vel :47-173, FastRF
:176-193): same parameterization (phi, phi_v, edge_mlp, edge_mlp_rv,
edge_mlp_vr, coord_mlp_vel; module names/shapes match for checkpoint
compatibility; the layer's default activation is LeakyReLU(0.2) —
reference :51). MI355X-first conventions follow models/fastegnn.py:
channels-major [B, C, *] virtual state, CSR segment reductions, fused
flat-buffer weighted-average all-reduce, no host syncs.
"""

from __future__ import annotations

import torch
from torch import nn

from .. import ops
from ..ops.linear import SplitKLinear as Linear
from ..parallel import comm


class GCLRFVel(nn.Module):
    """One FastRF layer (reference GCL_RF_vel)."""

    def __init__(self, edge_attr_nf, hidden_nf, virtual_channels, world_size,
                 act_fn=None):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.LeakyReLU(0.2)
        self.hidden_nf = hidden_nf
        self.world_size = world_size
        self.virtual_channels = virtual_channels
        self.epsilon = 1e-8

        def tanh_net(in_nf):
            last = Linear(hidden_nf, hidden_nf, bias=False)
            nn.init.xavier_uniform_(last.weight, gain=0.001)
            return nn.Sequential(Linear(in_nf, hidden_nf), act_fn, last,
                                 nn.Tanh())

        self.phi = tanh_net(1 + edge_attr_nf)
        self.phi_v = tanh_net(1 + virtual_channels)
        self.edge_mlp = nn.Sequential(
            Linear(hidden_nf, hidden_nf), act_fn, Linear(hidden_nf, 1))
        self.edge_mlp_rv = nn.Sequential(
            Linear(hidden_nf, hidden_nf), act_fn, Linear(hidden_nf, 1))
        self.edge_mlp_vr = nn.Sequential(
            Linear(hidden_nf, hidden_nf), act_fn, Linear(hidden_nf, 1))
        self.coord_mlp_vel = nn.Sequential(
            Linear(1, hidden_nf), act_fn, Linear(hidden_nf, 1))

    def forward(self, edge_index, coord, vel, virtual_coord, batch,
                edge_attr=None, *, rowptr=None, ptr=None, counts=None,
                counts_global=None, num_graphs=None, pool_chunks=None,
                colptr=None, col_perm=None):
        """virtual_coord: [B, C, 3] channels-major."""
        n = coord.size(0)
        b = num_graphs if num_graphs is not None else virtual_coord.size(0)
        c = self.virtual_channels
        row = edge_index[0]
        dist_active = self.world_size > 1 and comm.is_distributed()

        coord_diff = (ops.gather_rows(coord, row, rowptr)
                      - ops.gather_rows(coord, edge_index[1], colptr,
                                        col_perm))
        radial = coord_diff.pow(2).sum(1, keepdim=True)
        vdiff = (ops.gather_rows(virtual_coord, batch, ptr,
                                 chunks=pool_chunks) - coord.unsqueeze(1))
        vradial = vdiff.norm(p=2, dim=-1, keepdim=True)      # [N, C, 1]

        # edge model: phi([r^2, a]) -> [M, H]
        edge_feat = self.phi(torch.cat([radial, edge_attr], dim=1))

        coord_mean = ops.graph_mean_pool(coord, batch, b, ptr=ptr,
                                         counts=counts, chunks=pool_chunks)
        m_x = virtual_coord - coord_mean.unsqueeze(1)
        gram = torch.matmul(m_x, m_x.transpose(1, 2))        # [B, C, C]
        v_in = torch.cat([vradial, ops.gather_rows(gram, batch, ptr,
                                                   chunks=pool_chunks)],
                         dim=-1)
        v_feat = self.phi_v(v_in)                            # [N, C, H]

        # real-node coordinate update
        trans = coord_diff * self.edge_mlp(edge_feat)
        agg = ops.segment_mean(trans, row, n, rowptr=rowptr)
        coord = coord + agg
        trans_v = (-vdiff * self.edge_mlp_rv(v_feat)).mean(dim=1)
        coord = coord + trans_v
        coord = coord + vel * self.coord_mlp_vel(
            vel.norm(dim=-1, keepdim=True))

        # virtual-node coordinate update (+ weighted all-reduce)
        trans_x = vdiff * self.edge_mlp_vr(v_feat)           # [N, C, 3]
        agg_vc = ops.graph_mean_pool(
            trans_x.reshape(n, -1), batch, b, ptr=ptr, counts=counts,
            chunks=pool_chunks).reshape(b, c, 3)
        if dist_active:
            agg_vc = comm.fused_weighted_average_reduce(
                [agg_vc], counts, counts_global)
        virtual_coord = virtual_coord + agg_vc
        return coord, virtual_coord


class FastRF(nn.Module):
    """Stack of GCLRFVel layers (reference FastRF :176-193)."""

    def __init__(self, edge_attr_nf, hidden_nf, virtual_channels, world_size,
                 act_fn=None, n_layers=4):
        super().__init__()
        assert virtual_channels > 0, (
            f"Channels of virtual node must be greater than 0 "
            f"(got {virtual_channels})")
        self.hidden_nf = hidden_nf
        self.n_layers = n_layers
        self.virtual_channels = virtual_channels
        for i in range(n_layers):
            self.add_module(f"gcl_{i}",
                            GCLRFVel(edge_attr_nf, hidden_nf,
                                     virtual_channels, world_size))

    def forward(self, node_loc, node_vel, loc_mean, edge_index, data_batch,
                edge_attr=None, *, rowptr=None, ptr=None, counts=None,
                counts_global=None, pool_chunks=None, colptr=None,
                col_perm=None):
        b = loc_mean.size(0)
        if counts is None:
            counts = torch.bincount(data_batch, minlength=b).to(
                node_loc.dtype)
        if counts_global is None:
            counts_global = comm.global_counts(counts)
        virtual_loc = loc_mean.unsqueeze(1).expand(
            b, self.virtual_channels, 3).contiguous()
        loc = node_loc
        for i in range(self.n_layers):
            loc, virtual_loc = self._modules[f"gcl_{i}"](
                edge_index, loc, node_vel, virtual_loc, data_batch,
                edge_attr, rowptr=rowptr, ptr=ptr, counts=counts,
                counts_global=counts_global, num_graphs=b,
                pool_chunks=pool_chunks, colptr=colptr, col_perm=col_perm)
        return loc, virtual_loc.transpose(1, 2)  # [B, 3, C] API parity
