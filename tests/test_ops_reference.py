"""Oracle-level tests: the eager reference ops against brute-force math."""
import math

import pytest
import torch

from pertgnn.ops import reference as ref


def _rand_graph(n, e, h, seed=0):
    g = torch.Generator().manual_seed(seed)
    src = torch.randint(0, n, (e,), generator=g)
    dst = torch.randint(0, n, (e,), generator=g)
    edge_index = torch.stack([src, dst])
    q = torch.randn(n, h, generator=g)
    k = torch.randn(n, h, generator=g)
    v = torch.randn(n, h, generator=g)
    ee = torch.randn(e, h, generator=g)
    return edge_index, q, k, v, ee


def test_segment_softmax_matches_loop():
    torch.manual_seed(0)
    n, e = 7, 40
    dst = torch.randint(0, n, (e,))
    logits = torch.randn(e) * 5
    alpha = ref.segment_softmax(logits, dst, n)
    for i in range(n):
        m = dst == i
        if m.any():
            expect = torch.softmax(logits[m], dim=0)
            assert torch.allclose(alpha[m], expect, atol=1e-6)
    # each segment sums to 1
    sums = torch.zeros(n).index_add(0, dst, alpha)
    present = torch.zeros(n, dtype=torch.bool)
    present[dst] = True
    assert torch.allclose(sums[present], torch.ones(present.sum()), atol=1e-6)


def test_segment_softmax_empty():
    out = ref.segment_softmax(torch.zeros(0), torch.zeros(0, dtype=torch.long), 5)
    assert out.numel() == 0


def test_edge_attention_bruteforce():
    n, e, h = 9, 30, 16
    edge_index, q, k, v, ee = _rand_graph(n, e, h, seed=3)
    skip = torch.randn(n, h)
    out = ref.edge_attention(q, k, v, ee, edge_index, n, skip)
    # brute force per node
    src, dst = edge_index
    expect = skip.clone()
    for i in range(n):
        eids = (dst == i).nonzero().flatten()
        if eids.numel() == 0:
            continue
        ke = k[src[eids]] + ee[eids]
        ve = v[src[eids]] + ee[eids]
        logits = (ke * q[i]).sum(-1) / math.sqrt(h)
        a = torch.softmax(logits, dim=0)
        expect[i] += (a.unsqueeze(-1) * ve).sum(0)
    assert torch.allclose(out, expect, atol=1e-5)


def test_edge_attention_isolated_nodes_get_skip_only():
    n, h = 4, 8
    edge_index = torch.tensor([[0], [1]])  # single edge 0->1
    q = torch.randn(n, h); k = torch.randn(n, h); v = torch.randn(n, h)
    ee = torch.randn(1, h)
    skip = torch.randn(n, h)
    out = ref.edge_attention(q, k, v, ee, edge_index, n, skip)
    assert torch.allclose(out[2], skip[2])
    assert torch.allclose(out[3], skip[3])
    # node 1's single-edge softmax weight is exactly 1
    assert torch.allclose(out[1], skip[1] + v[0] + ee[0], atol=1e-6)


def test_pattern_pool():
    n, h, b = 10, 4, 3
    x = torch.randn(n, h)
    probs = torch.rand(n, 1)
    nn = torch.randint(1, 5, (n, 1)).float()
    batch = torch.tensor([0, 0, 0, 1, 1, 1, 1, 2, 2, 2])
    out = ref.pattern_pool(x, probs, nn, batch, b)
    for g in range(b):
        m = batch == g
        expect = (x[m] * probs[m] / nn[m]).sum(0)
        assert torch.allclose(out[g], expect, atol=1e-6)


def test_quantile_loss_known_values():
    y = torch.tensor([1.0, 2.0, 3.0])
    y_hat = torch.tensor([0.0, 2.0, 5.0])
    tau = 0.5
    # e = [1, 0, -2]; max(.5e, -.5e) = [.5, 0, 1]; mean = .5
    assert torch.allclose(ref.quantile_loss(y, y_hat, tau), torch.tensor(0.5))
    tau = 0.9
    # [0.9*1, 0, max(-1.8, 0.2)] = [0.9, 0, 0.2] -> mean 1.1/3
    assert torch.allclose(ref.quantile_loss(y, y_hat, tau), torch.tensor(1.1 / 3))


def test_eval_metrics():
    y = torch.tensor([2.0, 4.0])
    y_hat = torch.tensor([1.0, 6.0])
    mae, mape, q = ref.eval_metrics(y, y_hat, 0.5)
    assert torch.allclose(mae, torch.tensor(3.0))
    assert torch.allclose(mape, torch.tensor(0.5 + 0.5))
    assert torch.allclose(q, torch.tensor(0.5 * 1 + 0.5 * 2))


def test_embed_concat():
    n, f, h = 5, 3, 4
    x_raw = torch.randn(n, f)
    table = torch.randn(7, h)
    cat = torch.randint(0, 7, (n, 1))
    out = ref.embed_concat_node(x_raw, cat, [table])
    assert out.shape == (n, f + h)
    assert torch.allclose(out[:, :f], x_raw)
    assert torch.allclose(out[:, f:], table[cat[:, 0]])

    e = 6
    ifc = torch.randn(9, h); rpc = torch.randn(4, h)
    attr = torch.stack([torch.randint(0, 9, (e,)), torch.randint(0, 4, (e,))], dim=1)
    out = ref.embed_concat_edge(attr, ifc, rpc)
    assert torch.allclose(out[:, :h], ifc[attr[:, 0]])
    assert torch.allclose(out[:, h:], rpc[attr[:, 1]])
