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
from .. import ops as ops_mod


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


class EquivariantEdgeScalarNet(nn.Module):
    """Edge-pair O(n)-equivariant net (reference basic.py:467-506)."""

    def __init__(self, n_vector_input, hidden_dim, activation,
                 n_scalar_input=0, norm=True, flat=False):
        super().__init__()
        self.input_dim = n_vector_input * n_vector_input + n_scalar_input
        self.hidden_dim = hidden_dim
        self.output_dim = hidden_dim
        self.norm = norm
        self.in_scalar_net = BaseMLP(self.input_dim, hidden_dim, hidden_dim,
                                     activation, last_act=True, flat=flat)
        self.out_vector_net = BaseMLP(hidden_dim, hidden_dim,
                                      n_vector_input * n_vector_input,
                                      activation, flat=flat)

    def forward(self, vectors_i, vectors_j, scalars=None):
        z_i, z_j = vectors_i, vectors_j                       # [M, 3, K]
        k = z_i.shape[-1]
        s = torch.einsum("bij,bjk->bik", z_j.transpose(-1, -2),
                         z_i).reshape(-1, k * k)
        if self.norm:
            s = F.normalize(s, p=2, dim=-1)
        if scalars is not None:
            s = torch.cat((s, scalars), dim=-1)
        s = self.in_scalar_net(s)
        vec_scalar = self.out_vector_net(s).reshape(-1, k, k)
        vector = torch.einsum("bij,bjk->bik", z_j, vec_scalar)
        return vector, s


class PoolingLayer(nn.Module):
    """EGHN pooling layer (reference basic.py:509-539)."""

    def __init__(self, in_edge_nf, hidden_nf, n_vector_input,
                 activation=None, flat=False):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.edge_message_net = EquivariantEdgeScalarNet(
            n_vector_input=n_vector_input, hidden_dim=hidden_nf,
            activation=activation,
            n_scalar_input=2 * hidden_nf + in_edge_nf, norm=True, flat=flat)
        self.node_net = BaseMLP(2 * hidden_nf, hidden_nf, hidden_nf,
                                activation, flat=flat)

    def forward(self, vectors, h, edge_index, edge_fea):
        row = edge_index[0]
        hij = torch.cat((h[row], h[edge_index[1]], edge_fea), dim=-1)
        vec_out, message = self.edge_message_net(
            vectors_i=vectors[row], vectors_j=vectors[edge_index[1]],
            scalars=hij)
        dim, v = vec_out.shape[-2], vec_out.shape[-1]
        agg_vec = aggregate(vec_out.reshape(-1, dim * v), row, h.shape[0],
                            "mean").reshape(-1, dim, v)
        vectors = vectors + agg_vec
        tot = aggregate(message, row, h.shape[0], "sum")
        h = self.node_net(torch.cat((h, tot), dim=-1)) + h
        return vectors, h


class PoolingNet(nn.Module):
    """EGHN pooling network (reference basic.py:542-566)."""

    def __init__(self, n_layers, in_edge_nf, n_vector_input, hidden_nf,
                 output_nf, activation=None, device="cpu", flat=False):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.layers = nn.ModuleList([
            PoolingLayer(in_edge_nf, hidden_nf,
                         n_vector_input=n_vector_input,
                         activation=activation, flat=flat)
            for _ in range(n_layers)])
        self.n_layers = n_layers
        self.pooling = nn.Sequential(nn.Linear(hidden_nf, 8 * hidden_nf),
                                     nn.Tanh(),
                                     nn.Linear(8 * hidden_nf, output_nf))

    def forward(self, vectors, h, edge_index, edge_fea):
        if isinstance(vectors, list):
            vectors = torch.stack(vectors, dim=-1)
        for layer in self.layers:
            vectors, h = layer(vectors, h, edge_index, edge_fea)
        return self.pooling(h)


def _adj_matmul(edge_index, n, dense):
    """(unweighted adjacency) @ dense — replaces torch_sparse.spmm
    (reference basic.py:663,668) with the deterministic CSR SpMM
    composition (ops.spmm_adj: sort -> rowptr -> gather -> segment sum)."""
    return ops_mod.spmm_adj(edge_index, n, dense)


class EGHN(nn.Module):
    """Equivariant Graph Hierarchy Network (reference basic.py:569-731)."""

    def __init__(self, in_node_nf, in_edge_nf, hidden_nf, n_cluster,
                 layer_per_block=3, layer_pooling=3, layer_decoder=1,
                 flat=False, activation=None, device="cpu", norm=False,
                 with_v=True):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.embedding = nn.Linear(in_node_nf, hidden_nf)
        self.current_pooling_plan = None
        self.n_cluster = n_cluster
        self.with_v = with_v
        self.low_force_net = EGNN(n_layers=layer_per_block,
                                  in_node_nf=hidden_nf,
                                  in_edge_nf=in_edge_nf, hidden_nf=hidden_nf,
                                  activation=activation, with_v=with_v,
                                  flat=flat, norm=norm)
        self.low_pooling = PoolingNet(n_vector_input=3, hidden_nf=hidden_nf,
                                      output_nf=n_cluster,
                                      activation=activation,
                                      in_edge_nf=in_edge_nf,
                                      n_layers=layer_pooling, flat=flat)
        self.high_force_net = EGNN(n_layers=layer_per_block,
                                   in_node_nf=hidden_nf, in_edge_nf=1,
                                   hidden_nf=hidden_nf,
                                   activation=activation, with_v=with_v,
                                   flat=flat)
        nvi = 4 if with_v else 3
        if layer_decoder == 1:
            self.kinematics_net = EquivariantScalarNet(
                n_vector_input=nvi, hidden_dim=hidden_nf,
                activation=activation, n_scalar_input=2 * hidden_nf,
                norm=True, flat=flat)
        else:
            self.kinematics_net = EGMN(
                n_vector_input=nvi, hidden_dim=hidden_nf,
                activation=activation, n_scalar_input=2 * hidden_nf,
                norm=True, flat=flat, n_layers=layer_decoder)

    def get_cut_loss(self, a):
        a = F.normalize(a, p=2, dim=2)
        eye = torch.eye(a.shape[-1], device=a.device)
        return torch.norm(a - eye, p="fro", dim=[1, 2]).mean()

    @staticmethod
    def construct_edges(a, n_node):
        h_edge_fea = a.reshape(-1)
        p = a.shape[1]
        h_row = torch.arange(p, device=a.device).unsqueeze(-1) \
            .expand(-1, p).reshape(-1)
        h_col = torch.arange(p, device=a.device).unsqueeze(0) \
            .expand(p, -1).reshape(-1)
        h_row = h_row.unsqueeze(0).expand(a.shape[0], -1)
        h_col = h_col.unsqueeze(0).expand(a.shape[0], -1)
        offset = (torch.arange(a.shape[0], device=a.device)
                  * n_node).unsqueeze(-1)
        h_row = (h_row + offset).reshape(-1)
        h_col = (h_col + offset).reshape(-1)
        h_edge_mask = torch.ones_like(h_row)
        h_edge_mask[torch.arange(p, device=a.device) * (p + 1)] = 0
        return h_row, h_col, h_edge_fea, h_edge_mask

    def forward(self, x, h, edge_index, edge_fea, local_edge_index,
                local_edge_fea, n_node, v=None, node_mask=None,
                node_nums=None):
        h = self.embedding(h)
        row, col = edge_index[0], edge_index[1]

        new_x, new_v, h = self.low_force_net(x, h, edge_index, edge_fea, v=v)
        nf = new_x - x

        if node_nums is None:
            x_mean = x.reshape(-1, n_node, x.shape[-1]).mean(
                1, keepdim=True).expand(-1, n_node, -1).reshape(
                -1, x.shape[-1])
        else:
            pooled = (x.reshape(-1, n_node, x.shape[-1]).sum(1).T
                      / node_nums).T.unsqueeze(1)
            x_mean = pooled.expand(-1, n_node, -1).reshape(-1, x.shape[-1])
        pooling_fea = self.low_pooling(vectors=[x - x_mean, nf, v], h=h,
                                       edge_index=local_edge_index,
                                       edge_fea=local_edge_fea)
        hard = F.one_hot(pooling_fea.argmax(-1), self.n_cluster).float()
        pooling = F.softmax(pooling_fea, dim=1)
        self.current_pooling_plan = hard

        s = pooling.reshape(-1, n_node, pooling.shape[-1])     # [B, N, P]
        s_t = s.transpose(-2, -1)
        p_index = torch.ones_like(nf)[..., 0]
        if node_mask is not None:
            p_index = p_index * node_mask
        p_index = p_index.reshape(-1, n_node, 1)
        count = torch.einsum("bij,bjk->bik", s_t, p_index).clamp_min(1e-5)
        _x = x.reshape(-1, n_node, x.shape[-1])
        _h = h.reshape(-1, n_node, h.shape[-1])
        _nf = nf.reshape(-1, n_node, nf.shape[-1])
        big_x = torch.einsum("bij,bjk->bik", s_t, _x) / count
        big_h = torch.einsum("bij,bjk->bik", s_t, _h) / count
        big_nf = torch.einsum("bij,bjk->bik", s_t, _nf) / count
        if v is not None:
            big_v = torch.einsum("bij,bjk->bik", s_t,
                                 v.reshape(-1, n_node, v.shape[-1])) / count
            big_v = big_v.reshape(-1, big_v.shape[-1])
        else:
            big_v = None
        big_x = big_x.reshape(-1, big_x.shape[-1])
        big_h = big_h.reshape(-1, big_h.shape[-1])

        a = _adj_matmul(local_edge_index, x.shape[0], pooling)
        a = a.reshape(-1, n_node, a.shape[-1])
        big_a = torch.einsum("bij,bjk->bik", s_t, a)
        self.cut_loss = self.get_cut_loss(big_a)
        aa = _adj_matmul(edge_index, x.shape[0], pooling)
        aa = aa.reshape(-1, n_node, aa.shape[-1])
        big_aa = torch.einsum("bij,bjk->bik", s_t, aa)

        h_row, h_col, h_edge_fea, _ = self.construct_edges(
            big_aa, big_aa.shape[-1])
        h_new_x, h_new_v, h_new_h = self.high_force_net(
            big_x, big_h, (h_row, h_col), h_edge_fea.unsqueeze(-1), v=big_v)
        h_nf = h_new_x - big_x

        p = big_aa.shape[1]
        l_nf = torch.einsum("bij,bjk->bik", s,
                            h_nf.reshape(-1, p, 3)).reshape(-1, 3)
        l_x = torch.einsum("bij,bjk->bik", s,
                           big_x.reshape(-1, p, 3)).reshape(-1, 3)
        if v is not None:
            l_v = torch.einsum("bij,bjk->bik", s,
                               big_v.reshape(-1, p, 3)).reshape(-1, 3)
            vectors = [l_nf, x - l_x, v - l_v, nf]
        else:
            vectors = [l_nf, x - l_x, nf]
        l_h = torch.einsum("bij,bjk->bik", s,
                           h_new_h.reshape(-1, p, h_new_h.shape[-1]))
        l_h = l_h.reshape(-1, l_h.shape[-1])
        l_kin, h_out = self.kinematics_net(
            vectors=vectors, scalars=torch.cat((h, l_h), dim=-1))
        _l_x = torch.einsum("bij,bjk->bik", s,
                            (big_x + h_nf).reshape(-1, p, 3)).reshape(-1, 3)
        x_out = _l_x + l_kin
        return (x_out, v, h_out) if v is not None else (x_out, h_out)


class FullMLP(nn.ModuleList):
    """Non-equivariant MLP baseline (reference basic.py:734-749)."""

    def __init__(self, in_node_nf, hidden_nf, n_layers, activation=None,
                 flat=False, device="cpu"):
        super().__init__()
        activation = activation if activation is not None else nn.SiLU()
        self.layers = nn.ModuleList([
            BaseMLP(hidden_nf, hidden_nf, hidden_nf, activation,
                    residual=True, last_act=True, flat=flat)
            for _ in range(n_layers)])
        self.embedding = nn.Linear(in_node_nf, hidden_nf)
        self.output = nn.Linear(hidden_nf, 3)

    def forward(self, x):
        x = self.embedding(x)
        for layer in self.layers:
            x = layer(x)
        return self.output(x)
