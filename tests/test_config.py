import os

from melgan_amd.config import HParam, merge_default, _default_ratios


def test_attribute_access(tiny_hp):
    assert tiny_hp.audio.n_mel_channels == 80
    assert tiny_hp.model.generator_ratio == [8, 8, 2, 2]
    assert tiny_hp.train.adam.lr == 1e-4  # default filled
    assert tiny_hp.model.nonexistent_key is None  # family Dotdict behaviour


def test_hp_str_roundtrip(tiny_hp):
    from melgan_amd.config import load_hparam_str

    hp2 = merge_default(load_hparam_str(tiny_hp.hp_str))
    assert hp2.audio.segment_length == tiny_hp.audio.segment_length
    assert hp2.hp_str == tiny_hp.hp_str


def test_yaml_configs_load():
    cfg_dir = os.path.join(os.path.dirname(__file__), "..", "configs")
    for name in os.listdir(cfg_dir):
        hp = merge_default(HParam(os.path.join(cfg_dir, name)))
        # total upsampling * out_channels must equal hop_length
        total = hp.model.out_channels
        for r in hp.model.generator_ratio:
            total *= r
        assert total == hp.audio.hop_length, name
        assert hp.audio.segment_length % hp.audio.hop_length == 0, name


def test_default_ratios():
    assert _default_ratios(256) == [8, 8, 2, 2]
    for total in (256, 64, 32, 75, 300):
        r = _default_ratios(total)
        prod = 1
        for x in r:
            prod *= x
        assert prod == total, (total, r)


def test_reference_config_without_backend_group():
    """Reference configs (no backend: group) must load unchanged."""
    raw = """
data: {train_dir: /tmp/x}
train: {batch_size: 4}
audio: {hop_length: 256}
model: {out_channels: 4}
log: {chkpt_dir: c, log_dir: l}
"""
    hp = merge_default(HParam(raw=raw))
    assert hp.backend.use_hip_kernels is True
    assert hp.model.generator_ratio is not None


def test_geometry_mismatch_raises(tiny_hp):
    """prod(ratios) x out_channels != hop_length must fail LOUDLY at
    model construction, not deep inside the STFT loss."""
    import pytest

    from melgan_amd.models import build_generator

    tiny_hp.model.generator_ratio = [8, 8, 2]  # 128 x bands != hop 256
    with pytest.raises(ValueError, match="geometry mismatch"):
        build_generator(tiny_hp)


def test_unknown_key_detection():
    """§5.6: unknown keys warn (typo check), never reject."""
    from melgan_amd.config import HParam, merge_default, unknown_keys

    raw = """
audio: {segment_lenght: 8192, hop_length: 256}
trian: {batch_size: 4}
model: {out_channels: 4}
"""
    hp = merge_default(HParam(raw=raw))
    unk = unknown_keys(hp)
    assert "audio.segment_lenght" in unk
    assert "trian" in unk
    assert not any(u.startswith("model") for u in unk)

    # every shipped config is clean
    import glob
    import os

    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for cfg in glob.glob(os.path.join(here, "configs", "*.yaml")):
        hp = merge_default(HParam(cfg))
        assert unknown_keys(hp) == [], (cfg, unknown_keys(hp))


def test_hparam_pickles_into_dataloader_workers():
    """num_workers>0 pickles the dataset (hp inside) into worker
    processes — HParam/Dotdict must survive the round trip with hp_str."""
    import pickle

    from melgan_amd.config import HParam, merge_default
    from melgan_amd.data.dataset import create_dataloader

    hp = merge_default(HParam(raw="train: {batch_size: 2, num_workers: 2}\n"
                                  "audio: {segment_length: 4096, "
                                  "hop_length: 256}"))
    hp2 = pickle.loads(pickle.dumps(hp))
    assert hp2.train.batch_size == 2
    assert hp2.hp_str == hp.hp_str
    dl = create_dataloader(hp, train=True, synthetic=True)
    assert dl.num_workers == 2
    mel, audio = next(iter(dl))
    assert mel.shape == (2, 80, 16) and audio.shape == (2, 1, 4096)
