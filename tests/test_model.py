"""Model tests mirroring reference networks_test.py: output shapes, simplex
outputs, predict == forward, and the banded-attention zero-outside-band
invariant (networks_test.py:128-149)."""
import numpy as np
import pytest
import torch

from deepconsensus_amd.models import config as cfg
from deepconsensus_amd.models import model as M


def _make_inputs(p, batch=2, seed=0):
    rng = np.random.default_rng(seed)
    rows = np.zeros((batch, p.total_rows, p.max_length), dtype=np.float32)
    mp = p.max_passes
    rows[:, 0:mp] = rng.integers(0, 5, size=(batch, mp, p.max_length))
    rows[:, mp:2 * mp] = rng.integers(0, 256, size=(batch, mp, p.max_length))
    rows[:, 2 * mp:3 * mp] = rng.integers(0, 256, size=(batch, mp, p.max_length))
    rows[:, 3 * mp:4 * mp] = rng.integers(0, 3, size=(batch, mp, p.max_length))
    rows[:, 4 * mp] = rng.integers(0, 5, size=(batch, p.max_length))
    rows[:, -4:] = rng.integers(0, 501, size=(batch, 4, p.max_length))
    return torch.from_numpy(rows)


@pytest.fixture(scope="module")
def prod_params():
    p = cfg.get_config("transformer_learn_values+test")
    cfg.modify_params(p)
    return p


def test_learned_values_transformer_shapes(prod_params):
    torch.manual_seed(0)
    net = M.EncoderOnlyLearnedValuesTransformer(prod_params)
    x = _make_inputs(prod_params)
    with torch.no_grad():
        probs = net(x)
    assert probs.shape == (2, 100, 5)
    # Softmax outputs lie on the simplex.
    np.testing.assert_allclose(
        probs.sum(-1).numpy(), np.ones((2, 100)), atol=1e-5
    )
    assert (probs >= 0).all()


def test_predict_equals_forward(prod_params):
    torch.manual_seed(0)
    net = M.EncoderOnlyLearnedValuesTransformer(prod_params)
    x = _make_inputs(prod_params)
    with torch.no_grad():
        a = net(x)
        b = net.predict(x)
    torch.testing.assert_close(a, b)


@pytest.mark.parametrize("win", [6, 12])
def test_banded_attention_zero_outside_band(win):
    p = cfg.get_config("transformer_learn_values+test")
    p.attn_win_size = win
    cfg.modify_params(p)
    torch.manual_seed(0)
    net = M.EncoderOnlyLearnedValuesTransformer(p)
    x = _make_inputs(p)
    with torch.no_grad():
        out = net.encode(x, need_weights=True)
    L = p.max_length
    i = torch.arange(L)
    outside = (i[:, None] - i[None, :]).abs() > win
    for n in range(p.num_hidden_layers):
        w = out[f"attention_scores_{n}"]  # [B, H, L, L]
        assert w.shape == (2, p.num_heads, L, L)
        assert torch.all(w[:, :, outside] == 0.0), f"layer {n} leaks outside band"
        # In-band rows sum to 1.
        torch.testing.assert_close(
            w.sum(-1), torch.ones(2, p.num_heads, L), atol=1e-5, rtol=0
        )


def test_fc_net_shapes():
    p = cfg.get_config("fc+test")
    cfg.modify_params(p)
    net = M.FullyConnectedNet(p)
    x = _make_inputs(p)
    with torch.no_grad():
        probs = net(x)
    assert probs.shape == (2, 100, 5)
    np.testing.assert_allclose(
        probs.sum(-1).numpy(), np.ones((2, 100)), atol=1e-5
    )


def test_conv_net_shapes():
    p = cfg.get_config("transformer_learn_values+test")
    cfg.modify_params(p)
    net = M.ConvNet(p, channels=8, blocks=1)
    x = _make_inputs(p)
    with torch.no_grad():
        probs = net(x)
    assert probs.shape == (2, 100, 5)


def test_embedding_zero_mask(prod_params):
    """id-0 inputs embed to exactly zero (networks.py:57-63)."""
    emb = M.ScaledEmbedding(5, 8)
    ids = torch.tensor([[0, 1, 2, 0]])
    out = emb(ids)
    assert torch.all(out[0, 0] == 0)
    assert torch.all(out[0, 3] == 0)
    assert not torch.all(out[0, 1] == 0)


def test_model_factory(prod_params):
    assert isinstance(
        M.get_model(prod_params), M.EncoderOnlyLearnedValuesTransformer
    )
    p = cfg.get_config("fc+test")
    cfg.modify_params(p)
    assert isinstance(M.get_model(p), M.FullyConnectedNet)


def test_gradients_flow(prod_params):
    """Backward through the whole model reaches every embedding table."""
    torch.manual_seed(0)
    net = M.EncoderOnlyLearnedValuesTransformer(prod_params)
    x = _make_inputs(prod_params)
    probs = net(x, training=False)
    loss = -torch.log(probs + 1e-9).mean()
    loss.backward()
    for name, param in net.named_parameters():
        if "alpha" in name:
            continue  # ReZero alphas start at 0: attn branch grads exist.
        assert param.grad is not None, f"no grad for {name}"


def test_embed_stack_fused_backward_matches_autograd():
    """The fused-backward embedding stack produces the same forward and
    the same table gradients as the plain autograd chain (CPU fallback)."""
    import copy

    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models.model import get_model

    for use_bq in (False, True):
        params = cfg.get_config("transformer_learn_values+custom")
        params.use_ccs_bq = use_bq
        cfg.modify_params(params)
        torch.manual_seed(3)
        m1 = get_model(params)
        m2 = copy.deepcopy(m1)
        rng = np.random.default_rng(5)
        R, L = params.total_rows, params.max_length
        rows = np.zeros((3, R, L), dtype=np.float32)
        mp = params.max_passes
        rows[:, 0:mp] = rng.integers(0, 5, size=(3, mp, L))
        rows[:, mp:3 * mp] = rng.integers(0, 256, size=(3, 2 * mp, L))
        rows[:, 3 * mp:4 * mp] = rng.integers(0, 3, size=(3, mp, L))
        rows[:, 4 * mp] = rng.integers(0, 5, size=(3, L))
        if use_bq:
            rows[:, 4 * mp + 1] = rng.integers(-1, 94, size=(3, L))
        rows[:, -4:] = rng.uniform(3.0, 30.0, size=(3, 4, 1))
        x = torch.from_numpy(rows)

        m1.train()
        torch.manual_seed(77)  # align dropout draws across both paths
        out1 = m1(x, training=True)
        out1.sum().backward()

        # m2: force the plain autograd chain by flipping train mode off
        # during embed (the gate checks self.training).
        m2.train()
        m2_embed_gate = m2.bases_embedding.table.requires_grad
        assert m2_embed_gate
        # Disable the fused path via eval-mode embed call semantics:
        # run forward with the original chain by monkeypatching training.
        import deepconsensus_amd.models.model as model_mod

        orig = model_mod.EncoderOnlyLearnedValuesTransformer.embed

        def plain_embed(self, inputs):
            was = self.training
            self.training = False  # gate off -> original chain
            try:
                return orig(self, inputs)
            finally:
                self.training = was

        model_mod.EncoderOnlyLearnedValuesTransformer.embed = plain_embed
        try:
            torch.manual_seed(77)
            out2 = m2(x, training=True)
            out2.sum().backward()
        finally:
            model_mod.EncoderOnlyLearnedValuesTransformer.embed = orig

        torch.testing.assert_close(out1, out2, atol=1e-6, rtol=1e-6)
        for n in ("bases", "pw", "ip", "strand", "sn") + (
            ("ccs_bq",) if use_bq else ()
        ):
            g1 = getattr(m1, f"{n}_embedding").table.grad
            g2 = getattr(m2, f"{n}_embedding").table.grad
            torch.testing.assert_close(g1, g2, atol=1e-4, rtol=1e-4)


def test_device_table_flat_matches_build_gather_tables():
    """embed_stack's per-step table rebuild reproduces the canonical
    build_gather_tables layout bit-for-bit (CPU check — guards layout
    drift between the training forward and the serving tables)."""
    import torch

    from deepconsensus_amd.models import config as cfg
    from deepconsensus_amd.models.embed_stack import (
        EmbedMeta,
        _device_table_flat,
        _gather_maps,
    )
    from deepconsensus_amd.models.model import get_model

    for use_bq in (False, True):
        params = cfg.get_config("transformer_learn_values+custom")
        params.use_ccs_bq = use_bq
        cfg.modify_params(params, is_training=True)
        torch.manual_seed(3)
        model = get_model(params)
        meta = EmbedMeta(model)
        tables = [
            getattr(model, a).table for a in meta.table_attrs
        ]
        from deepconsensus_amd.models.runner import build_gather_tables

        tf_ref, *_ = build_gather_tables(model)
        _maps, flat_len = _gather_maps(meta, model, "cpu")
        tf_new = _device_table_flat(meta, tables, flat_len, "cpu")
        assert torch.equal(tf_ref, tf_new)
