"""Pure-PyTorch (eager) reference implementations of every op in the framework.

These are the numerical oracle: each HIP kernel is tested against the function
here of the same name (fp32, deterministic).  The math mirrors the reference
repo's dependency-internal op surface (kernels K1–K14):

  * edge attention  = PyG 2.4.0 TransformerConv semantics with heads=1
    (reference model.py:25-52):
      out_i = W_skip x_i + b_skip
            + sum_e softmax_i(<q_i, k_src(e)+e_e>/sqrt(H)) * (v_src(e)+e_e)
  * segment softmax = torch_geometric.utils.softmax (max-subtracted, per dst)
  * pattern pool    = x * p / n  ->  segment-sum by graph (model.py:106-107)
  * quantile loss   = mean(max(tau*e, (tau-1)*e)) (pert_gnn.py:191-193)
"""
from __future__ import annotations

import math

import torch


# ---------------------------------------------------------------------------
# segment primitives
# ---------------------------------------------------------------------------

def segment_softmax(logits: torch.Tensor, dst: torch.Tensor, num_nodes: int) -> torch.Tensor:
    """Numerically-stable softmax over edge logits grouped by destination node.

    Mirrors ``torch_geometric.utils.softmax`` (reference dependency surface,
    SURVEY.md K5).  ``logits``: [E], ``dst``: [E] int64, returns alpha [E].
    """
    if logits.numel() == 0:
        return logits
    seg_max = torch.full((num_nodes,), float("-inf"), dtype=logits.dtype, device=logits.device)
    seg_max = seg_max.scatter_reduce(0, dst, logits, reduce="amax", include_self=True)
    shifted = logits - seg_max.index_select(0, dst)
    expv = shifted.exp()
    seg_sum = torch.zeros(num_nodes, dtype=logits.dtype, device=logits.device)
    seg_sum = seg_sum.index_add(0, dst, expv)
    # rows with no incoming edges never appear in dst; no div-by-zero possible
    return expv / seg_sum.index_select(0, dst)


def scatter_sum(src: torch.Tensor, index: torch.Tensor, dim_size: int) -> torch.Tensor:
    """Segment sum of row vectors: out[index[e]] += src[e].  src: [E,H]."""
    out = torch.zeros(dim_size, *src.shape[1:], dtype=src.dtype, device=src.device)
    return out.index_add(0, index, src)


# ---------------------------------------------------------------------------
# K3-K6: fused edge attention (TransformerConv heads=1 semantics)
# ---------------------------------------------------------------------------

def edge_attention(
    q: torch.Tensor,        # [N,H]  W_q x (dst side)
    k: torch.Tensor,        # [N,H]  W_k x (src side)
    v: torch.Tensor,        # [N,H]  W_v x (src side)
    e: torch.Tensor,        # [E,H]  W_e edge_embed — added to both key and value
    edge_index: torch.Tensor,  # [2,E] (src, dst)
    num_nodes: int,
    skip: torch.Tensor | None = None,  # [N,H] W_skip x + b — added to output
    return_alpha: bool = False,
):
    """PyG-2.4.0 TransformerConv message+aggregate with heads=1.

    Reference call site: model.py:100,104 via SURVEY.md §2.3 semantics.
    """
    src, dst = edge_index[0], edge_index[1]
    h = q.shape[1]
    ke = k.index_select(0, src) + e
    ve = v.index_select(0, src) + e
    logits = (q.index_select(0, dst) * ke).sum(-1) / math.sqrt(h)
    alpha = segment_softmax(logits, dst, num_nodes)
    out = scatter_sum(alpha.unsqueeze(-1) * ve, dst, num_nodes)
    if skip is not None:
        out = out + skip
    if return_alpha:
        return out, alpha
    return out


# ---------------------------------------------------------------------------
# K1 + K11: categorical embedding sum + feature concat
# ---------------------------------------------------------------------------

def embed_concat_node(x_raw: torch.Tensor, cat_X: torch.Tensor, tables: list[torch.Tensor]) -> torch.Tensor:
    """x = [x_raw ‖ sum_i table_i[cat_X[:,i]]]  (model.py:87-90)."""
    acc = tables[0].index_select(0, cat_X[:, 0])
    for i in range(1, len(tables)):
        acc = acc + tables[i].index_select(0, cat_X[:, i])
    return torch.cat([x_raw, acc], dim=1)


def embed_concat_edge(edge_attr: torch.Tensor, interface_table: torch.Tensor, rpctype_table: torch.Tensor) -> torch.Tensor:
    """edge_embeds = [ifc[attr[:,0]] ‖ rpc[attr[:,1]]]  (model.py:91-97)."""
    return torch.cat(
        [interface_table.index_select(0, edge_attr[:, 0]),
         rpctype_table.index_select(0, edge_attr[:, 1])],
        dim=1,
    )


# ---------------------------------------------------------------------------
# K9 + K10: pattern-probability weighting + global add pool
# ---------------------------------------------------------------------------

def pattern_pool(
    x: torch.Tensor,                  # [N,H]
    pattern_probs: torch.Tensor,      # [N,1] per-node pattern probability
    pattern_num_nodes: torch.Tensor,  # [N,1] nodes in the node's pattern
    batch: torch.Tensor,              # [N] graph id per node
    num_graphs: int,
) -> torch.Tensor:
    """x*p/n -> segment-sum by graph (model.py:106-107)."""
    weighted = x * pattern_probs / pattern_num_nodes
    return scatter_sum(weighted, batch, num_graphs)


# ---------------------------------------------------------------------------
# K12/K13: loss + eval metrics
# ---------------------------------------------------------------------------

def quantile_loss(y: torch.Tensor, y_hat: torch.Tensor, tau: float) -> torch.Tensor:
    """Pinball loss, mean over batch (pert_gnn.py:191-193)."""
    e = y - y_hat
    return torch.mean(torch.maximum(tau * e, (tau - 1) * e))


def eval_metrics(y: torch.Tensor, y_hat: torch.Tensor, tau: float):
    """Returns (sum |err|, sum |err|/y, sum pinball) — reference accumulates
    sums then divides by dataset size (pert_gnn.py:284-289).  Division by y is
    reproduced as-is (quirk 13: MAPE undefined at y==0)."""
    err = y_hat - y
    abs_err = err.abs()
    mae_sum = abs_err.sum()
    mape_sum = (abs_err / y).sum()
    e = y - y_hat
    q_sum = torch.maximum(tau * e, (tau - 1) * e).sum()
    return mae_sum, mape_sum, q_sum
