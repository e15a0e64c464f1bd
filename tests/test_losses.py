import torch

from melgan_amd.engine import losses as L
from melgan_amd.models.discriminator import MultiScaleDiscriminator


def _disc_outputs():
    torch.manual_seed(0)
    d = MultiScaleDiscriminator(num_D=2, ndf=4, n_layers=2)
    real = torch.randn(1, 1, 1024)
    fake = torch.randn(1, 1, 1024)
    return d(real), d(fake)


def test_adversarial_g_loss():
    d_real, d_fake = _disc_outputs()
    loss = L.adversarial_g_loss(d_fake)
    assert loss > 0 and torch.isfinite(loss)
    # perfect scores -> zero loss
    perfect = [[torch.ones_like(f) for f in feats] for feats in d_fake]
    assert float(L.adversarial_g_loss(perfect)) == 0.0


def test_feature_matching():
    d_real, d_fake = _disc_outputs()
    loss = L.feature_matching_loss(d_fake, d_real)
    assert loss > 0
    assert float(L.feature_matching_loss(d_real, d_real)) == 0.0


def test_discriminator_loss():
    d_real, d_fake = _disc_outputs()
    loss = L.discriminator_loss(d_real, d_fake)
    assert loss > 0
    # ideal discriminator: real->1, fake->0
    ideal_r = [[torch.ones_like(f) for f in feats] for feats in d_real]
    ideal_f = [[torch.zeros_like(f) for f in feats] for feats in d_fake]
    assert float(L.discriminator_loss(ideal_r, ideal_f)) == 0.0


def test_fused_adam_matches_torch_adam():
    from melgan_amd.ops.adam import FusedAdam

    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.randn(64, 32))
    p2 = torch.nn.Parameter(p1.detach().clone())
    o1 = torch.optim.Adam([p1], lr=1e-3, betas=(0.5, 0.9))
    o2 = FusedAdam([p2], lr=1e-3, betas=(0.5, 0.9))
    for _ in range(5):
        g = torch.randn_like(p1)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max().item()


def test_fused_adam_late_grad_bias_correction():
    """a param whose grad first appears later must get its own
    bias-correction step count, not the group's (partition-by-step)."""
    from melgan_amd.ops.adam import FusedAdam

    torch.manual_seed(1)
    a1 = torch.nn.Parameter(torch.randn(8, 4))
    b1 = torch.nn.Parameter(torch.randn(8, 4))
    a2 = torch.nn.Parameter(a1.detach().clone())
    b2 = torch.nn.Parameter(b1.detach().clone())
    o1 = torch.optim.Adam([a1, b1], lr=1e-2, betas=(0.5, 0.9))
    o2 = FusedAdam([a2, b2], lr=1e-2, betas=(0.5, 0.9))
    for i in range(6):
        ga = torch.randn(8, 4)
        a1.grad, a2.grad = ga.clone(), ga.clone()
        if i >= 3:  # b joins 3 steps late (e.g. frozen then unfrozen)
            gb = torch.randn(8, 4)
            b1.grad, b2.grad = gb.clone(), gb.clone()
        else:
            b1.grad = b2.grad = None
        o1.step()
        o2.step()
    assert torch.allclose(a1, a2, atol=1e-6)
    assert torch.allclose(b1, b2, atol=1e-6), \
        (b1 - b2).abs().max().item()


def test_fused_adam_state_dict_interop_with_torch_adam():
    """Optimizer checkpoints are interchangeable with torch.optim.Adam
    (same state layout) — what engine/checkpoint resume relies on."""
    from melgan_amd.ops.adam import FusedAdam

    torch.manual_seed(2)
    pr = torch.nn.Parameter(torch.randn(16, 8))
    pf = torch.nn.Parameter(pr.detach().clone())
    o_ref = torch.optim.Adam([pr], lr=1e-3, betas=(0.5, 0.9))
    grads = [torch.randn(16, 8) for _ in range(6)]

    o_fused = FusedAdam([pf], lr=1e-3, betas=(0.5, 0.9))
    for g in grads[:3]:
        pf.grad = g.clone()
        o_fused.step()
    # hand the state to a plain torch Adam mid-run
    o_torch = torch.optim.Adam([pf], lr=1e-3, betas=(0.5, 0.9))
    o_torch.load_state_dict(o_fused.state_dict())
    for g in grads[3:]:
        pf.grad = g.clone()
        o_torch.step()

    for g in grads:
        pr.grad = g.clone()
        o_ref.step()
    assert torch.allclose(pr, pf, atol=1e-6), (pr - pf).abs().max().item()
