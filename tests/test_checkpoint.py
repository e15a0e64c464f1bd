import os

import torch

from melgan_amd.engine.checkpoint import (
    load_checkpoint,
    load_flexible,
    save_checkpoint,
)
from melgan_amd.models import build_discriminator, build_generator


def test_save_load_roundtrip(tiny_hp, tmp_path):
    g = build_generator(tiny_hp)
    d = build_discriminator(tiny_hp)
    og = torch.optim.Adam(g.parameters())
    od = torch.optim.Adam(d.parameters())
    path = save_checkpoint(str(tmp_path), "run", 42, g, d, og, od,
                           tiny_hp.hp_str)
    assert os.path.basename(path) == "run_0042.pt"
    ck = load_checkpoint(path)
    assert set(ck.keys()) >= {"model_g", "model_d", "optim_g", "optim_d",
                              "step", "hp_str"}
    assert ck["step"] == 42
    assert ck["hp_str"] == tiny_hp.hp_str

    g2 = build_generator(tiny_hp)
    g2.load_state_dict(ck["model_g"])
    mel = torch.randn(1, 80, 8)
    with torch.no_grad():
        assert torch.allclose(g(mel), g2(mel), atol=1e-6)


def test_load_flexible_renamed_paths(tiny_hp):
    """Simulate a reference checkpoint whose module paths differ."""
    g = build_generator(tiny_hp)
    state = g.state_dict()
    renamed = {k.replace("layers.", "generator."): v
               for k, v in state.items()}
    g2 = build_generator(tiny_hp)
    load_flexible(g2, renamed)
    mel = torch.randn(1, 80, 8)
    with torch.no_grad():
        assert torch.allclose(g(mel), g2(mel), atol=1e-6)


def test_load_flexible_shape_mismatch_raises(tiny_hp, tiny_mb_hp):
    g1 = build_generator(tiny_hp)
    g4 = build_generator(tiny_mb_hp)
    try:
        load_flexible(g4, g1.state_dict())
        raised = False
    except RuntimeError:
        raised = True
    assert raised


def test_trainer_resume(tiny_hp, tmp_path):
    from melgan_amd.engine.trainer import Trainer

    tiny_hp.log.chkpt_dir = str(tmp_path)
    tiny_hp.log.log_dir = str(tmp_path / "logs")
    tr = Trainer(tiny_hp, "resume_test", enable_tb=False)
    tr.step = 7
    path = tr.save()
    tr2 = Trainer(tiny_hp, "resume_test2", enable_tb=False)
    tr2.load(path)
    assert tr2.step == 7
    for p1, p2 in zip(tr.generator.parameters(),
                      tr2.generator.parameters()):
        assert torch.equal(p1, p2)
