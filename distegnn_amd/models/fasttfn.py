"""FastTFN — virtual-node machinery with a 1-layer TFN real-node update.

Parity with reference models/FastTFN.py (TFN_GCL_vel :9-230, FastTFN
:233-264): the FastEGNN virtual pathway with the real-real coordinate
update delegated to a single OursTFN layer over the edge graph (reference
builds a DGL graph per call, :129-150; here an EdgeGraph). Not
distributed (no all-reduce in the reference). Channels-major [B, C, *]
virtual layout; reference parameter shapes preserved.
"""

from __future__ import annotations

import torch
from torch import nn

from .. import ops
from ..ops.linear import SplitKLinear as Linear
from .se3.graph import EdgeGraph
from .tfn import OursTFN, _SH_TO_XYZ, _XYZ_TO_SH


class TFNGCLVel(nn.Module):
    """One FastTFN layer (reference TFN_GCL_vel)."""

    def __init__(self, node_feat_nf, node_feat_out_nf, node_attr_nf,
                 edge_attr_nf, hidden_nf, virtual_channels, num_degrees=2,
                 act_fn=None, residual=True, attention=False,
                 normalize=False, tanh=False, gravity=None):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.SiLU()
        self.residual = residual
        self.attention = attention
        self.normalize = normalize
        self.hidden_nf = hidden_nf
        self.tanh = tanh
        self.virtual_channels = virtual_channels
        self.epsilon = 1e-8

        # real-real coordinate update: 1-layer TFN (reference :36-37)
        self.tfn_layer = OursTFN(num_layers=1, num_channels=1, edge_dim=0,
                                 div=1, act_fn=act_fn,
                                 num_degrees=num_degrees)
        self.edge_mlp = nn.Sequential(
            Linear(2 * node_feat_nf + 1 + edge_attr_nf, hidden_nf), act_fn,
            Linear(hidden_nf, hidden_nf), act_fn)
        self.edge_mlp_virtual = nn.Sequential(
            Linear(2 * node_feat_nf + 1 + virtual_channels, hidden_nf),
            act_fn, Linear(hidden_nf, hidden_nf), act_fn)
        if attention:
            self.att_mlp = nn.Sequential(Linear(hidden_nf, 1), nn.Sigmoid())
            self.att_mlp_virtual = nn.Sequential(Linear(hidden_nf, 1),
                                                 nn.Sigmoid())

        def coord_head():
            last = Linear(hidden_nf, 1, bias=False)
            nn.init.xavier_uniform_(last.weight, gain=0.001)
            mods = [Linear(hidden_nf, hidden_nf), act_fn, last]
            if tanh:
                mods.append(nn.Tanh())
            return nn.Sequential(*mods)

        self.coord_mlp_r_virtual = coord_head()
        self.coord_mlp_v_virtual = coord_head()
        self.gravity = gravity
        if gravity is not None:
            self.gravity_mlp = nn.Sequential(
                Linear(node_feat_nf, hidden_nf), act_fn,
                Linear(hidden_nf, 1))
        self.node_mlp = nn.Sequential(
            Linear(3 * hidden_nf + node_attr_nf, hidden_nf), act_fn,
            Linear(hidden_nf, node_feat_out_nf))
        self.node_mlp_virtual = nn.Sequential(
            Linear(2 * hidden_nf, hidden_nf), act_fn,
            Linear(hidden_nf, node_feat_out_nf))

    def forward(self, h, edge_index, coord, vel, virtual_coord, virtual_feat,
                batch, charges, edge_attr=None, node_attr=None, *,
                rowptr=None, ptr=None, counts=None, counts_global=None,
                num_graphs=None, pool_chunks=None, colptr=None,
                col_perm=None):
        n = coord.size(0)
        b = num_graphs if num_graphs is not None else virtual_coord.size(0)
        c = self.virtual_channels
        row = edge_index[0]

        coord_diff = (ops.gather_rows(coord, row, rowptr)
                      - ops.gather_rows(coord, edge_index[1], colptr,
                                        col_perm))
        radial = coord_diff.pow(2).sum(1, keepdim=True)
        if self.normalize:
            coord_diff = coord_diff / (radial.sqrt().detach() + self.epsilon)
        vdiff = (ops.gather_rows(virtual_coord, batch, ptr,
                                 chunks=pool_chunks) - coord.unsqueeze(1))
        vradial = vdiff.norm(p=2, dim=-1, keepdim=True)

        h_row = ops.gather_rows(h, row, rowptr)
        h_col = ops.gather_rows(h, edge_index[1], colptr, col_perm)
        edge_feat = self.edge_mlp(
            torch.cat([h_row, h_col, radial.to(h.dtype),
                       edge_attr.to(h.dtype)], dim=1))
        if self.attention:
            edge_feat = edge_feat * self.att_mlp(edge_feat)

        coord_mean = ops.graph_mean_pool(coord, batch, b, ptr=ptr,
                                         counts=counts, chunks=pool_chunks)
        m_x = virtual_coord - coord_mean.unsqueeze(1)
        gram = torch.matmul(m_x, m_x.transpose(1, 2))
        v_in = torch.cat([
            h.unsqueeze(1).expand(n, c, h.size(1)),
            ops.gather_rows(virtual_feat, batch, ptr, chunks=pool_chunks),
            vradial.to(h.dtype),
            ops.gather_rows(gram, batch, ptr, chunks=pool_chunks).to(h.dtype),
        ], dim=-1)
        v_msg = self.edge_mlp_virtual(v_in)
        if self.attention:
            v_msg = v_msg * self.att_mlp_virtual(v_msg)

        # real coordinate update via the TFN layer (reference :129-150)
        G = EdgeGraph(edge_index, n)
        G.ndata["f"] = charges.unsqueeze(2)
        G.ndata["f1"] = vel[:, _XYZ_TO_SH].unsqueeze(1)
        G.edata["d"] = (coord.index_select(0, edge_index[1])
                        - coord.index_select(0, edge_index[0]))
        trans_r = self.tfn_layer(G)["1"].view(coord.size())[:, _SH_TO_XYZ]
        coord = coord + trans_r
        trans_v = (-vdiff * self.coord_mlp_r_virtual(v_msg)).mean(dim=1)
        coord = coord + trans_v
        if self.gravity is not None:
            coord = coord + self.gravity_mlp(h) * self.gravity.to(h.device)

        trans_x = vdiff * self.coord_mlp_v_virtual(v_msg)
        agg_vc = ops.graph_mean_pool(
            trans_x.reshape(n, -1), batch, b, ptr=ptr, counts=counts,
            chunks=pool_chunks).reshape(b, c, 3)
        virtual_coord = virtual_coord + agg_vc

        agg_e = ops.segment_mean(edge_feat, row, n, rowptr=rowptr)
        agg_v = v_msg.mean(dim=1)
        cat = [h, agg_e, agg_v] + ([node_attr] if node_attr is not None
                                   else [])
        h_out = self.node_mlp(torch.cat(cat, dim=1))
        if self.residual:
            h_out = h + h_out

        agg_vf = ops.graph_mean_pool(
            v_msg.reshape(n, -1), batch, b, ptr=ptr, counts=counts,
            chunks=pool_chunks).reshape(b, c, self.hidden_nf)
        vf_out = self.node_mlp_virtual(
            torch.cat([virtual_feat, agg_vf], dim=-1))
        if self.residual:
            vf_out = virtual_feat + vf_out
        return h_out, coord, vf_out, virtual_coord


class FastTFN(nn.Module):
    """Stack of TFNGCLVel layers (reference FastTFN :233-264)."""

    def __init__(self, node_feat_nf, node_attr_nf, edge_attr_nf, hidden_nf,
                 virtual_channels, device="cpu", act_fn=None, n_layers=4,
                 residual=True, attention=False, normalize=False, tanh=False,
                 gravity=None):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.SiLU()
        assert virtual_channels > 0
        self.hidden_nf = hidden_nf
        self.n_layers = n_layers
        self.node_attr_nf = node_attr_nf
        self.virtual_channels = virtual_channels
        self.virtual_node_feat = nn.Parameter(
            torch.randn(1, hidden_nf, virtual_channels))
        self.W = nn.Parameter(torch.randn(1, virtual_channels, 3))
        self.embedding_in = nn.Linear(node_feat_nf, hidden_nf)
        if gravity is not None:
            gravity = torch.tensor(gravity)
        for i in range(n_layers):
            self.add_module(f"gcl_{i}", TFNGCLVel(
                hidden_nf, hidden_nf, node_attr_nf, edge_attr_nf, hidden_nf,
                virtual_channels=virtual_channels, act_fn=act_fn,
                residual=residual, attention=attention, normalize=normalize,
                tanh=tanh, gravity=gravity))

    def forward(self, node_feat, node_loc, node_vel, loc_mean, edge_index,
                data_batch, charges, edge_attr=None, node_attr=None, *,
                rowptr=None, ptr=None, counts=None, counts_global=None,
                pool_chunks=None, colptr=None, col_perm=None):
        b = loc_mean.size(0)
        if counts is None:
            counts = torch.bincount(data_batch, minlength=b).to(
                node_loc.dtype)
        virtual_feat = self.virtual_node_feat.transpose(1, 2).expand(
            b, self.virtual_channels, self.hidden_nf).contiguous()
        virtual_loc = loc_mean.unsqueeze(1).expand(
            b, self.virtual_channels, 3).contiguous()
        h = self.embedding_in(node_feat)
        loc = node_loc
        for i in range(self.n_layers):
            h, loc, virtual_feat, virtual_loc = self._modules[f"gcl_{i}"](
                h, edge_index, loc, node_vel, virtual_loc, virtual_feat,
                data_batch, charges, edge_attr=edge_attr,
                node_attr=node_attr, rowptr=rowptr, ptr=ptr, counts=counts,
                counts_global=counts_global, num_graphs=b,
                pool_chunks=pool_chunks, colptr=colptr, col_perm=col_perm)
        return loc, virtual_loc.transpose(1, 2)
