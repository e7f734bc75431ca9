"""Config loader: defaults + the reference's real config.cfg format."""

import os

import pytest

from tskd_amd.config import DEFAULT_CHANNEL_NAMES, get_global_config


class TestDefaults:
    def test_defaults(self, tmp_path, monkeypatch):
        monkeypatch.chdir(tmp_path)  # no config.cfg in cwd
        monkeypatch.delenv("TSKD_CONFIG", raising=False)
        cfg = get_global_config()
        assert cfg.n_channels == 10
        assert cfg.window_size == 120
        assert cfg.record_overlap == 0.4
        assert cfg.channel_names == DEFAULT_CHANNEL_NAMES
        assert cfg.preprocess_window_s == 180.0
        assert cfg.predict_window_s == 600.0

    def test_channel_topic_mapping(self):
        cfg = get_global_config()
        assert cfg.topic_for_channel("PVC Rate per Minute") == \
            "PVC_Rate_per_Minute"
        assert cfg.channel_index("SpO2") == 4  # wire index order

    def test_custom_file(self, tmp_path):
        p = tmp_path / "c.cfg"
        p.write_text("[SETTINGS]\nWINDOWSIZE = 60\n"
                     "CHANNEL_NAMES = HR, RESP\nBATCHSIZE = 8\n")
        cfg = get_global_config(str(p))
        assert cfg.window_size == 60
        assert cfg.channel_names == ["HR", "RESP"]
        assert cfg.batch_size == 8


class TestReferenceConfig:
    def test_parses_reference_config_cfg(self, reference_dir):
        """The upstream config.cfg (with MOUNTPATH interpolation, LOCAL*
        duplicates and the 10-channel list) must parse unchanged."""
        path = os.path.join(reference_dir, "config.cfg")
        if not os.path.exists(path):
            pytest.skip("reference config.cfg missing")
        cfg = get_global_config(path)
        assert cfg.window_size == 120
        assert cfg.record_overlap == 0.4
        assert cfg.batch_size == 16
        assert cfg.n_channels == 10
        assert cfg.channel_names[0] == "HR"
        assert cfg.channel_names[3] == "PVC Rate per Minute"
        assert cfg.patient_records == ["p000194-2112-05-23-14-34n",
                                       "p044083-2112-05-04-19-50n"]
        assert not cfg.use_cuda  # USE_CUDA = 0 in the reference
        # interpolated paths resolve (container tree on linux)
        assert cfg.model_path.endswith("MyCNN5.pth")
        assert cfg.wavef_path.startswith("/volume")
