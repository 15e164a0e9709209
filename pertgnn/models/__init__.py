from .pert_gnn import SAGEDeterministic, TransformerConv

__all__ = ["SAGEDeterministic", "TransformerConv"]
