import math

import pytest
import torch

from rainbowiqn_amd.models import NoisyLinear, RainbowIQN


def test_noisy_linear_noise_properties():
    torch.manual_seed(0)
    nl = NoisyLinear(64, 32, std_init=0.1)
    eps1 = nl.weight_epsilon.clone()
    nl.reset_noise()
    assert not torch.equal(eps1, nl.weight_epsilon)
    # factorised structure: eps_w = eps_out outer eps_in -> rank 1
    # (fp32 outer product has ~1e-7-relative residual singular values)
    sv = torch.linalg.svdvals(nl.weight_epsilon.double())
    assert (sv > 1e-5 * sv[0]).sum() == 1
    # f(x)=sign(x)sqrt(|x|): E[eps]=0-ish, values bounded plausibly
    assert nl.weight_epsilon.abs().mean() < 2.0


def test_noisy_linear_eval_mode_deterministic():
    nl = NoisyLinear(16, 8)
    x = torch.randn(4, 16)
    nl.eval()
    y1 = nl(x)
    nl.reset_noise()
    y2 = nl(x)
    assert torch.equal(y1, y2)  # eval uses mu only
    nl.train()
    y3 = nl(x)
    assert not torch.equal(y1, y3)


def test_model_forward_shapes(cfg):
    m = RainbowIQN(cfg, action_space=6)
    x = torch.rand(3, cfg.history_length, 84, 84)
    z, taus = m(x, num_quantiles=5)
    assert z.shape == (3, 5, 6)
    assert taus.shape == (15, 1)
    assert (taus >= 0).all() and (taus < 1).all()
    q = m.q_values(x, 5)
    assert q.shape == (3, 6)


def test_dueling_identity(cfg):
    """Q = V + A - mean(A) implies mean_a Q == V."""
    m = RainbowIQN(cfg, action_space=4)
    m.eval()
    x = torch.rand(2, cfg.history_length, 84, 84)
    taus = torch.rand(2 * 3, 1)
    z, _ = m(x, 3, taus=taus)
    h = m.encode(x).repeat_interleave(3, dim=0) * m.embed_taus(taus)
    v = m.fc_z_v(torch.relu(m.fc_h_v(h))).view(2, 3)
    assert torch.allclose(z.mean(dim=2), v, atol=1e-5)


def test_cosine_embedding_definition(cfg):
    m = RainbowIQN(cfg, action_space=4)
    taus = torch.tensor([[0.25], [0.5]])
    cos = torch.cos(taus * math.pi * torch.arange(cfg.quantile_embedding_dim))
    expect = torch.relu(m.iqn_fc(cos))
    assert torch.allclose(m.embed_taus(taus), expect)


def test_state_dict_reference_keys(cfg):
    """Checkpoint-format parity: reference key names."""
    m = RainbowIQN(cfg, action_space=4)
    keys = set(m.state_dict().keys())
    for k in [
        "convs.0.weight", "convs.0.bias", "convs.2.weight", "convs.4.weight",
        "iqn_fc.weight", "iqn_fc.bias",
        "fc_h_v.weight_mu", "fc_h_v.weight_sigma", "fc_h_v.bias_mu",
        "fc_h_v.bias_sigma", "fc_h_v.weight_epsilon",
        "fc_h_a.weight_mu", "fc_z_v.weight_mu", "fc_z_a.weight_sigma",
    ]:
        assert k in keys, k


def test_checkpoint_roundtrip(tmp_path, cfg):
    from rainbowiqn_amd.agent import Agent

    a = Agent(cfg, action_space=4, device="cpu")
    a.save(str(tmp_path))
    b = Agent(cfg, action_space=4, device="cpu")
    b.load(str(tmp_path))
    for (k1, p1), (k2, p2) in zip(
        a.online_net.state_dict().items(), b.online_net.state_dict().items()
    ):
        assert k1 == k2
        assert torch.equal(p1, p2), k1


def test_checkpoint_alias_remap(tmp_path, cfg):
    """A reference checkpoint using an alternate IQN-layer name still loads."""
    from rainbowiqn_amd.agent import Agent

    a = Agent(cfg, action_space=4, device="cpu")
    sd = a.online_net.state_dict()
    renamed = {
        (k.replace("iqn_fc.", "quantile_fc.") if k.startswith("iqn_fc.") else k): v
        for k, v in sd.items()
    }
    p = tmp_path / "model.pth"
    torch.save(renamed, str(p))
    b = Agent(cfg, action_space=4, device="cpu")
    b.load(str(p))
    assert torch.equal(b.online_net.iqn_fc.weight, a.online_net.iqn_fc.weight)


def test_param_count_matches_survey(cfg):
    """SURVEY.md 3.3 sizes the full model at ~6.8M params (hidden 512)."""
    from rainbowiqn_amd.config import Config

    m = RainbowIQN(Config(), action_space=18)
    n = sum(p.numel() for p in m.parameters())
    assert 6e6 < n < 8e6, n


def test_fused_noise_path_matches_standard_forward(cfg):
    """enable_fused_noise + manual weight materialization must reproduce the
    standard NoisyLinear forward bit-for-bit (same eps), and grads must route
    to mu/sigma identically (CPU check of the GPU fast-path contract)."""
    import torch

    from rainbowiqn_amd.models.model import NoisyLinear

    torch.manual_seed(0)
    a = NoisyLinear(16, 8)
    b = NoisyLinear(16, 8)
    b.load_state_dict(a.state_dict())
    b.enable_fused_noise()
    b.weight_epsilon.copy_(a.weight_epsilon)
    b.bias_epsilon.copy_(a.bias_epsilon)
    with torch.no_grad():
        b.weight_noisy.copy_(b.weight_mu + b.weight_sigma * b.weight_epsilon)
        b.bias_noisy.copy_(b.bias_mu + b.bias_sigma * b.bias_epsilon)
    x = torch.randn(4, 16)
    ya = a(x)
    yb = b(x)
    assert torch.allclose(ya, yb, atol=1e-6), (ya - yb).abs().max()
    ga = torch.autograd.grad(ya.square().sum(), [a.weight_mu, a.weight_sigma,
                                                 a.bias_mu, a.bias_sigma])
    gb = torch.autograd.grad(yb.square().sum(), [b.weight_mu, b.weight_sigma,
                                                 b.bias_mu, b.bias_sigma])
    for u, v in zip(ga, gb):
        assert torch.allclose(u, v, atol=1e-6)


def test_checkpoint_alien_keys_fail_loud(cfg):
    """A state_dict with unexpected names must RAISE with a key diff, never
    silently backfill."""
    import pytest

    from rainbowiqn_amd.agent import Agent
    from rainbowiqn_amd.utils.checkpoint import (CheckpointKeyError,
                                                 remap_reference_state_dict)

    a = Agent(cfg, action_space=4, device="cpu")
    sd = a.online_net.state_dict()
    alien = {k.replace("fc_h_v.", "value_hidden."): v for k, v in sd.items()}
    with pytest.raises(CheckpointKeyError) as ei:
        remap_reference_state_dict(alien, a.online_net)
    msg = str(ei.value)
    assert "value_hidden.weight_mu" in msg  # unmatched checkpoint key listed
    assert "fc_h_v.weight_mu" in msg  # missing model key listed


def test_checkpoint_missing_param_fails_loud(cfg):
    """A truncated checkpoint (dropped parameter) raises."""
    import pytest

    from rainbowiqn_amd.agent import Agent
    from rainbowiqn_amd.utils.checkpoint import (CheckpointKeyError,
                                                 remap_reference_state_dict)

    a = Agent(cfg, action_space=4, device="cpu")
    sd = dict(a.online_net.state_dict())
    del sd["iqn_fc.weight"]
    with pytest.raises(CheckpointKeyError, match="iqn_fc.weight"):
        remap_reference_state_dict(sd, a.online_net)


def test_checkpoint_shape_mismatch_fails_loud(cfg):
    import pytest
    import torch

    from rainbowiqn_amd.agent import Agent
    from rainbowiqn_amd.utils.checkpoint import (CheckpointKeyError,
                                                 remap_reference_state_dict)

    a = Agent(cfg, action_space=4, device="cpu")
    sd = dict(a.online_net.state_dict())
    sd["iqn_fc.weight"] = torch.zeros(7, 7)
    with pytest.raises(CheckpointKeyError, match="shape"):
        remap_reference_state_dict(sd, a.online_net)


def test_checkpoint_without_epsilon_buffers_loads(cfg):
    """Transient NoisyLinear epsilon buffers are optional in a checkpoint
    (regenerated every step); everything else intact -> loads cleanly."""
    import torch

    from rainbowiqn_amd.agent import Agent
    from rainbowiqn_amd.utils.checkpoint import remap_reference_state_dict

    a = Agent(cfg, action_space=4, device="cpu")
    sd = {k: v for k, v in a.online_net.state_dict().items()
          if not k.endswith("_epsilon")}
    out = remap_reference_state_dict(sd, a.online_net)
    assert torch.equal(out["fc_h_v.weight_mu"], sd["fc_h_v.weight_mu"])
    assert "fc_h_v.weight_epsilon" in out  # backfilled from the model
