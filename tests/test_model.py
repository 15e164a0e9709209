"""Model structure + math tests (reference model.py parity)."""
import torch

from pertgnn.models import SAGEDeterministic


def _make_model(num_layers=1, h=16):
    return SAGEDeterministic(
        in_channels=9, cat_dims=[11], entry_id_max=5, interface_id_max=7,
        rpctype_id_max=3, hidden_channels=h, num_layers=num_layers, dropout=0.0,
    )


def _make_batch(n=20, e=40, b=3, h=16, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, 9, generator=g)
    cat_X = torch.randint(0, 11, (n, 1), generator=g)
    edge_index = torch.stack([
        torch.randint(0, n, (e,), generator=g),
        torch.randint(0, n, (e,), generator=g),
    ])
    edge_attr = torch.stack([
        torch.randint(0, 8, (e,), generator=g),
        torch.randint(0, 4, (e,), generator=g),
    ], dim=1)
    pattern_num_nodes = torch.randint(1, 6, (n, 1), generator=g).float()
    rt_probs = torch.rand(n, 1, generator=g)
    entry_id = torch.randint(0, 6, (b,), generator=g)
    batch = torch.sort(torch.randint(0, b, (n,), generator=g)).values
    return x, cat_X, edge_index, edge_attr, pattern_num_nodes, rt_probs, entry_id, batch


def test_num_layers_1_builds_2_convs():
    """Quirk 1: num_layers=1 still builds 2 convs, 1 bn (model.py:24-52)."""
    m = _make_model(num_layers=1)
    assert len(m.convs) == 2
    assert len(m.bns) == 1
    m4 = _make_model(num_layers=4)
    assert len(m4.convs) == 4
    assert len(m4.bns) == 3


def test_state_dict_keys_match_reference_schema():
    m = _make_model(num_layers=3)
    keys = set(m.state_dict().keys())
    # PyG TransformerConv submodule names (checkpoint compatibility)
    for i in range(3):
        for sub in ("lin_key", "lin_query", "lin_value", "lin_skip"):
            assert f"convs.{i}.{sub}.weight" in keys
            assert f"convs.{i}.{sub}.bias" in keys
        assert f"convs.{i}.lin_edge.weight" in keys
        assert f"convs.{i}.lin_edge.bias" not in keys
    for k in ("local_linear.weight", "global_linear1.weight", "global_linear2.weight",
              "cat_embedding.0.weight", "entry_embeds.weight", "interface_embeds.weight",
              "rpctype_embeds.weight", "bns.0.weight", "bns.0.running_mean"):
        assert k in keys
    # quirk 4: dead edge_linear module present (lazily uninitialized)
    assert "edge_linear.weight" in keys
    assert "edge_linear.bias" in keys


def test_forward_shapes_and_grad():
    h = 16
    m = _make_model(num_layers=2, h=h)
    x, cat_X, edge_index, edge_attr, pnn, probs, entry_id, batch = _make_batch(h=h)
    b = int(batch.max()) + 1
    gp, lp = m(x, cat_X, edge_index, edge_attr, pnn, probs, entry_id[:b], batch)
    assert gp.shape == (b, 1)
    assert lp.shape == (x.shape[0], 1)
    loss = gp.sum() + 0 * lp.sum()
    loss.backward()
    # gradients reach every trainable leaf that participates
    assert m.convs[0].w4.grad is not None  # fused QKVS parameter
    assert m.entry_embeds.weight.grad is not None
    assert m.cat_embedding[0].weight.grad is not None


def test_first_conv_input_width_is_f_plus_h():
    m = _make_model(num_layers=1, h=16)
    assert m.convs[0].lin_query.weight.shape == (16, 9 + 16)
    assert m.convs[1].lin_query.weight.shape == (16, 16)
    assert m.convs[0].lin_edge.weight.shape == (16, 32)


def test_checkpoint_roundtrip(tmp_path):
    m = _make_model(num_layers=2)
    x, cat_X, edge_index, edge_attr, pnn, probs, entry_id, batch = _make_batch()
    b = int(batch.max()) + 1
    m.eval()
    out1, _ = m(x, cat_X, edge_index, edge_attr, pnn, probs, entry_id[:b], batch)
    p = tmp_path / "ckpt.pt"
    torch.save(m.state_dict(), p)
    m2 = _make_model(num_layers=2)
    m2.load_state_dict(torch.load(p, weights_only=False))
    m2.eval()
    out2, _ = m2(x, cat_X, edge_index, edge_attr, pnn, probs, entry_id[:b], batch)
    assert torch.allclose(out1, out2)


def test_training_reduces_loss():
    """A short fit on one random batch must reduce quantile loss."""
    torch.manual_seed(0)
    from pertgnn.ops import functional as F

    h = 16
    m = _make_model(num_layers=1, h=h)
    x, cat_X, edge_index, edge_attr, pnn, probs, entry_id, batch = _make_batch(h=h)
    b = int(batch.max()) + 1
    y = torch.rand(b) * 10
    opt = torch.optim.Adam(m.parameters(), lr=1e-2)
    m.train()
    losses = []
    for _ in range(60):
        opt.zero_grad()
        gp, _ = m(x, cat_X, edge_index, edge_attr, pnn, probs, entry_id[:b], batch)
        loss = F.quantile_loss(y, gp.flatten(), 0.5)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.5
