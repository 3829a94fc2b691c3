"""Model zoo + factory (reference main.py:58-92 ``get_model`` parity)."""

from __future__ import annotations

from .fastegnn import FastEGNN, EGCLVel

_SCHNET_CUTOFFS = {"nbody_100": 1.0, "protein": 10.0, "Water-3D": 0.035}


def get_model(model_config, world_size: int, dataset_name: str):
    name = model_config.model_name
    if name == "FastEGNN":
        return FastEGNN(
            node_feat_nf=model_config.node_feat_nf,
            node_attr_nf=model_config.node_attr_nf,
            edge_attr_nf=model_config.edge_attr_nf,
            normalize=model_config.normalize,
            hidden_nf=model_config.hidden_nf,
            n_layers=model_config.n_layers,
            virtual_channels=model_config.virtual_channels,
            gravity=None,
            world_size=world_size,
        )
    if name == "FastRF":
        from .fastrf import FastRF

        return FastRF(edge_attr_nf=model_config.edge_attr_nf,
                      hidden_nf=model_config.hidden_nf,
                      n_layers=model_config.n_layers,
                      virtual_channels=model_config.virtual_channels,
                      world_size=world_size)
    if name in ("FastSchNet", "SchNet"):
        cutoff = _SCHNET_CUTOFFS.get(dataset_name)
        if cutoff is None:
            raise ValueError(f"no SchNet cutoff for dataset {dataset_name}")
        if name == "FastSchNet":
            from .fastschnet import FastSchNet

            return FastSchNet(node_feat_nf=model_config.node_feat_nf,
                              node_attr_nf=model_config.node_attr_nf,
                              edge_attr_nf=model_config.edge_attr_nf,
                              hidden_nf=model_config.hidden_nf,
                              virtual_channels=model_config.virtual_channels,
                              n_layers=model_config.n_layers,
                              normalize=model_config.normalize,
                              gravity=None, cutoff=cutoff)
        from .schnet import SchNet

        return SchNet(hidden_channels=model_config.hidden_nf,
                      max_num_neighbors=200000, cutoff=cutoff)
    if name == "EGNN":
        from .baselines import EGNN

        return EGNN(n_layers=model_config.n_layers,
                    in_node_nf=model_config.node_feat_nf,
                    in_edge_nf=model_config.edge_attr_nf,
                    hidden_nf=model_config.hidden_nf, with_v=True)
    if name == "RF":
        from .baselines import RF_vel

        return RF_vel(hidden_nf=model_config.hidden_nf,
                      edge_attr_nf=model_config.edge_attr_nf,
                      n_layers=model_config.n_layers)
    if name == "TFN":
        from .tfn import OurDynamics

        return OurDynamics(nf=model_config.hidden_nf // 2,
                           n_layers=model_config.n_layers, model="tfn",
                           num_degrees=2, div=1)
    if name == "FastTFN":
        from .fasttfn import FastTFN

        return FastTFN(node_feat_nf=model_config.node_feat_nf,
                       node_attr_nf=model_config.node_attr_nf,
                       edge_attr_nf=model_config.edge_attr_nf,
                       hidden_nf=model_config.hidden_nf,
                       virtual_channels=model_config.virtual_channels,
                       n_layers=model_config.n_layers,
                       normalize=model_config.normalize, gravity=None)
    if name == "Linear":
        from .baselines import Linear_dynamics

        return Linear_dynamics()
    raise NotImplementedError(f"Model {name} Not Implemented")


__all__ = ["get_model", "FastEGNN", "EGCLVel"]
