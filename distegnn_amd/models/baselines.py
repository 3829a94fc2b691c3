"""Non-virtual-node baseline models (reference models/basic.py).

EGNN (scalarization-based layer, :280-337), GNN (:359-399), RF_vel
(:413-464), Linear_dynamics (:402-410), plus the shared building blocks
BaseMLP (:167-191), EquivariantScalarNet (:194-237), InvariantScalarNet
(:240-277) and EGMN (:339-356). Module naming follows the reference so
checkpoints stay loadable; aggregations go through the ops dispatch
(CSR segment kernels on GPU when rowptr metadata is supplied; these
baselines accept plain edge_index too and then use index_select paths).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from .. import ops


def aggregate(message, row_index, n_node, aggr="sum", mask=None):
    """Edge->node aggregation (reference basic.py:22-47)."""
    if mask is not None:
        message = message * mask.unsqueeze(-1)
    if aggr == "sum":
        return ops.reference.segment_sum(message, row_index, n_node)
    if aggr == "mean":
        if mask is not None:
            s = ops.reference.segment_sum(message, row_index, n_node)
            c = ops.reference.segment_sum(mask.unsqueeze(-1).expand_as(message),
                                          row_index, n_node)
            return s / c.clamp(min=1)
        return ops.reference.segment_mean(message, row_index, n_node)
    raise NotImplementedError(f"Unknown aggregation method: {aggr}")


class BaseMLP(nn.Module):
    def __init__(self, input_dim, hidden_dim, output_dim, activation,
                 residual=False, last_act=False, flat=False):
        super().__init__()
        self.residual = residual
        if flat:
            activation = nn.Tanh()
            hidden_dim = 4 * hidden_dim
        if residual:
            assert output_dim == input_dim
        mods = [nn.Linear(input_dim, hidden_dim), activation,
                nn.Linear(hidden_dim, output_dim)]
        if last_act:
            mods.append(activation)
        self.mlp = nn.Sequential(*mods)

    def forward(self, x):
        return self.mlp(x) + x if self.residual else self.mlp(x)


def _gram_scalars(vectors, scalars, norm):
    z = torch.stack(vectors, dim=-1) if isinstance(vectors, list) else vectors
    k = z.shape[-1]
    s = torch.einsum("bij,bjk->bik", z.transpose(-1, -2), z).reshape(-1, k * k)
    if norm:
        s = F.normalize(s, p=2, dim=-1)
    if scalars is not None:
        s = torch.cat((s, scalars), dim=-1)
    return z, s


class EquivariantScalarNet(nn.Module):
    """Universal O(n)-equivariant net via gram scalarization (:194-237)."""

    def __init__(self, n_vector_input, hidden_dim, activation,
                 n_scalar_input=0, norm=True, flat=True):
        super().__init__()
        self.input_dim = n_vector_input * n_vector_input + n_scalar_input
        self.hidden_dim = hidden_dim
        self.output_dim = hidden_dim
        self.norm = norm
        self.in_scalar_net = BaseMLP(self.input_dim, hidden_dim, hidden_dim,
                                     activation, last_act=True, flat=flat)
        self.out_vector_net = BaseMLP(hidden_dim, hidden_dim, n_vector_input,
                                      activation, flat=flat)
        self.out_scalar_net = BaseMLP(hidden_dim, hidden_dim,
                                      self.output_dim, activation, flat=flat)

    def forward(self, vectors, scalars=None):
        z, s = _gram_scalars(vectors, scalars, self.norm)
        s = self.in_scalar_net(s)
        vec_scalar = self.out_vector_net(s)
        vector = torch.einsum("bij,bj->bi", z, vec_scalar)
        return vector, self.out_scalar_net(s)


class InvariantScalarNet(nn.Module):
    """Universal O(n)-invariant net via gram scalarization (:240-277)."""

    def __init__(self, n_vector_input, hidden_dim, output_dim, activation,
                 n_scalar_input=0, norm=True, last_act=False, flat=False):
        super().__init__()
        self.input_dim = n_vector_input * n_vector_input + n_scalar_input
        self.norm = norm
        self.scalar_net = BaseMLP(self.input_dim, hidden_dim, output_dim,
                                  activation, last_act=last_act, flat=flat)

    def forward(self, vectors, scalars=None):
        _, s = _gram_scalars(vectors, scalars, self.norm)
        return self.scalar_net(s)


class E_GCL(nn.Module):
    """Classic EGNN conv layer (reference basic.py:69-164)."""

    def __init__(self, input_nf, output_nf, hidden_nf, edges_in_d=0,
                 nodes_att_dim=0, act_fn=None, recurrent=True,
                 coords_weight=1.0, attention=False, clamp=False,
                 norm_diff=False, tanh=False):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.ReLU()
        self.coords_weight = coords_weight
        self.recurrent = recurrent
        self.attention = attention
        self.tanh = tanh
        self.clamp = clamp
        self.edge_mlp = nn.Sequential(
            nn.Linear(input_nf * 2 + 1 + edges_in_d, hidden_nf), act_fn,
            nn.Linear(hidden_nf, hidden_nf), act_fn)
        self.node_mlp = nn.Sequential(
            nn.Linear(hidden_nf + input_nf + nodes_att_dim, hidden_nf),
            act_fn, nn.Linear(hidden_nf, output_nf))
        last = nn.Linear(hidden_nf, 1, bias=False)
        nn.init.xavier_uniform_(last.weight, gain=0.001)
        mods = [nn.Linear(hidden_nf, hidden_nf), act_fn, last]
        if tanh:
            mods.append(nn.Tanh())
            self.coords_range = nn.Parameter(torch.ones(1)) * 3
        self.coord_mlp = nn.Sequential(*mods)
        if attention:
            self.att_mlp = nn.Sequential(nn.Linear(hidden_nf, 1),
                                         nn.Sigmoid())

    def coord2radial(self, edge_index, coord):
        row, col = edge_index[0], edge_index[1]
        coord_diff = coord[row] - coord[col]
        radial = coord_diff.pow(2).sum(1, keepdim=True)
        norm = torch.sqrt(radial + 1e-8)
        return radial, coord_diff / (norm + 1)

    def forward(self, h, edge_index, coord, edge_attr=None, node_attr=None):
        row = edge_index[0]
        radial, coord_diff = self.coord2radial(edge_index, coord)
        parts = [h[row], h[edge_index[1]], radial]
        if edge_attr is not None:
            parts.append(edge_attr)
        edge_feat = self.edge_mlp(torch.cat(parts, dim=1))
        if self.attention:
            edge_feat = edge_feat * self.att_mlp(edge_feat)
        trans = coord_diff * self.coord_mlp(edge_feat)
        coord = coord + aggregate(trans, row, coord.size(0),
                                  "mean") * self.coords_weight
        agg = aggregate(edge_feat, row, h.size(0), "sum")
        cat = [h, agg] + ([node_attr] if node_attr is not None else [])
        out = self.node_mlp(torch.cat(cat, dim=1))
        if self.recurrent:
            out = h + out
        return out, coord, edge_attr


class EGNN_Layer(nn.Module):
    """Scalarization-based EGNN layer (reference basic.py:280-316)."""

    def __init__(self, in_edge_nf, hidden_nf, activation=None, with_v=False,
                 flat=False, norm=False):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.with_v = with_v
        self.edge_message_net = InvariantScalarNet(
            n_vector_input=1, hidden_dim=hidden_nf, output_dim=hidden_nf,
            activation=activation, n_scalar_input=2 * hidden_nf + in_edge_nf,
            norm=norm, last_act=True, flat=flat)
        self.coord_net = BaseMLP(hidden_nf, hidden_nf, 1, activation,
                                 flat=flat)
        self.node_net = BaseMLP(2 * hidden_nf, hidden_nf, hidden_nf,
                                activation, flat=flat)
        self.node_v_net = (BaseMLP(hidden_nf, hidden_nf, 1, activation,
                                   flat=flat) if with_v else None)

    def forward(self, x, h, edge_index, edge_fea, v=None):
        row, col = edge_index[0], edge_index[1]
        rij = x[row] - x[col]
        hij = torch.cat((h[row], h[col], edge_fea), dim=-1)
        message = self.edge_message_net(vectors=[rij], scalars=hij)
        coord_message = self.coord_net(message)
        f = rij * coord_message
        tot_f = aggregate(f, row, x.shape[0], "mean").clamp(min=-100, max=100)
        if v is not None:
            x = x + self.node_v_net(h) * v + tot_f
        else:
            x = x + tot_f
        tot_message = aggregate(message, row, x.shape[0], "mean")
        h = self.node_net(torch.cat((h, tot_message), dim=-1))
        return x, v, h


class EGNN(nn.Module):
    """EGNN baseline (reference basic.py:319-337)."""

    def __init__(self, n_layers, in_node_nf, in_edge_nf, hidden_nf,
                 activation=None, device="cpu", with_v=False, flat=False,
                 norm=False):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.layers = nn.ModuleList()
        self.n_layers = n_layers
        self.with_v = with_v
        self.embedding = nn.Linear(in_node_nf, hidden_nf)
        for _ in range(n_layers):
            self.layers.append(EGNN_Layer(in_edge_nf, hidden_nf,
                                          activation=activation,
                                          with_v=with_v, flat=flat,
                                          norm=norm))

    def forward(self, x, h, edge_index, edge_fea, v=None):
        h = self.embedding(h)
        for layer in self.layers:
            x, v, h = layer(x, h, edge_index, edge_fea, v=v)
        return (x, v, h) if v is not None else (x, h)


class EGMN(nn.Module):
    """Stacked EquivariantScalarNet (reference basic.py:339-356)."""

    def __init__(self, n_layers, n_vector_input, hidden_dim, n_scalar_input,
                 activation=None, device="cpu", norm=False, flat=False):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.layers = nn.ModuleList()
        self.n_layers = n_layers
        for i in range(n_layers):
            self.layers.append(EquivariantScalarNet(
                n_vector_input=n_vector_input + i, hidden_dim=hidden_dim,
                activation=activation,
                n_scalar_input=n_scalar_input if i == 0 else hidden_dim,
                norm=norm, flat=flat))

    def forward(self, vectors, scalars):
        cur = vectors
        for layer in self.layers:
            vector, scalars = layer(cur, scalars)
            cur.append(vector)
        return cur[-1], scalars


class GNN_Layer(nn.Module):
    def __init__(self, in_edge_nf, hidden_nf, activation=None, with_v=False,
                 flat=False):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.with_v = with_v
        self.edge_message_net = BaseMLP(in_edge_nf + 2 * hidden_nf,
                                        hidden_nf, hidden_nf, activation,
                                        flat=flat)
        self.node_net = BaseMLP(2 * hidden_nf, hidden_nf, hidden_nf,
                                activation, flat=flat)

    def forward(self, h, edge_index, edge_fea):
        row = edge_index[0]
        hij = torch.cat((h[row], h[edge_index[1]], edge_fea), dim=-1)
        message = self.edge_message_net(hij)
        agg = aggregate(message, row, h.shape[0], "mean")
        return h + self.node_net(torch.cat((agg, h), dim=-1))


class GNN(nn.Module):
    """Plain message-passing GNN (reference basic.py:359-399)."""

    def __init__(self, n_layers, in_node_nf, in_edge_nf, hidden_nf,
                 activation=None, device="cpu", flat=False):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.layers = nn.ModuleList()
        self.n_layers = n_layers
        self.embedding = nn.Linear(in_node_nf, hidden_nf)
        for _ in range(n_layers):
            self.layers.append(GNN_Layer(in_edge_nf, hidden_nf,
                                         activation=activation, flat=flat))
        self.decoder = nn.Sequential(nn.Linear(hidden_nf, hidden_nf),
                                     activation, nn.Linear(hidden_nf, 3))

    def forward(self, h, edge_index, edge_fea):
        h = self.embedding(h)
        for layer in self.layers:
            h = layer(h, edge_index, edge_fea)
        return self.decoder(h)


class Linear_dynamics(nn.Module):
    """x + v * t with a learnable scalar t (reference basic.py:402-410)."""

    def __init__(self, device="cpu"):
        super().__init__()
        self.time = nn.Parameter(torch.ones(1))

    def forward(self, x, v):
        return x + v * self.time


class GCL_rf_vel(nn.Module):
    """Radial-field layer with velocity (reference basic.py:436-464)."""

    def __init__(self, nf=64, edge_attr_nf=0, act_fn=None,
                 coords_weight=1.0):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.LeakyReLU(0.2)
        self.coords_weight = coords_weight
        self.coord_mlp_vel = nn.Sequential(nn.Linear(1, nf), act_fn,
                                           nn.Linear(nf, 1))
        last = nn.Linear(nf, 1, bias=False)
        nn.init.xavier_uniform_(last.weight, gain=0.001)
        self.phi = nn.Sequential(nn.Linear(1 + edge_attr_nf, nf), act_fn,
                                 last, nn.Tanh())

    def forward(self, x, vel_norm, vel, edge_index, edge_attr=None):
        row, col = edge_index[0], edge_index[1]
        x_diff = x[row] - x[col]
        radial = x_diff.pow(2).sum(1, keepdim=True).sqrt()
        m_ij = x_diff * self.phi(torch.cat([radial, edge_attr], dim=1))
        agg = aggregate(m_ij, row, x.size(0), "mean")
        x = x + agg * self.coords_weight
        x = x + vel * self.coord_mlp_vel(vel_norm)
        return x, edge_attr


class RF_vel(nn.Module):
    """Radial-field baseline (reference basic.py:413-433)."""

    def __init__(self, hidden_nf, edge_attr_nf=0, device="cpu", act_fn=None,
                 n_layers=4):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.SiLU()
        self.hidden_nf = hidden_nf
        self.n_layers = n_layers
        for i in range(n_layers):
            self.add_module(f"gcl_{i}",
                            GCL_rf_vel(nf=hidden_nf,
                                       edge_attr_nf=edge_attr_nf,
                                       act_fn=act_fn))

    def forward(self, vel_norm, x, edges, vel, edge_attr):
        for i in range(self.n_layers):
            x, _ = self._modules[f"gcl_{i}"](x, vel_norm, vel, edges,
                                             edge_attr)
        return x
