"""Global configuration: config.cfg-compatible INI loader.

Mirrors the reference's two-tier config system (reference bin/utils.py:17-65 +
config.cfg): an INI file with interpolated ``[PATHS]`` and ``[SETTINGS]``
sections, platform-switched between deployment and local-dev path trees.
Unlike the reference we default every path relative to the repo/data dir and
drop the hard-coded container mount points, but any reference-format
config.cfg parses unchanged.
"""

from __future__ import annotations

import configparser
import os
import sys
from dataclasses import dataclass, field
from typing import List, Optional

# Wire/tensor channel order — the index order IS the channel order on the wire
# and in the (N, C, 120) model tensor (reference config.cfg:23 CHANNEL_NAMES).
DEFAULT_CHANNEL_NAMES: List[str] = [
    "HR", "RESP", "PULSE", "PVC Rate per Minute", "SpO2",
    "CVP", "ST V", "NBP Mean", "NBP Dias", "NBP Sys",
]

DEFAULT_PATIENT_RECORDS: List[str] = [
    "p000194-2112-05-23-14-34n",
    "p044083-2112-05-04-19-50n",
]


@dataclass
class GlobalConfig:
    """Parsed configuration with reference-compatible keys.

    Settings semantics (reference config.cfg:20-27):
      - ``window_size``   : model context length in 5-s grid points (120)
      - ``record_overlap``: sliding-window overlap fraction for batching (0.4)
      - ``channel_names`` : 10 channels; list index = wire channel index
    """

    mount_path: str = "."
    data_path: str = "data"
    wavef_path: str = "data/waveform/physionet.org/files/mimic3wdb-matched/1.0"
    output_path: str = "model"
    explore_path: str = "explore_output"
    model_path: str = "model/MyCNN5.pth"

    use_cuda: bool = False
    num_workers: int = 0
    channel_names: List[str] = field(default_factory=lambda: list(DEFAULT_CHANNEL_NAMES))
    patient_records: List[str] = field(default_factory=lambda: list(DEFAULT_PATIENT_RECORDS))
    window_size: int = 120
    record_overlap: float = 0.4
    batch_size: int = 16

    # Engine constants (reference processStream.py:196-214, predictStream.py:248-263).
    # All durations in seconds at speed=1; every consumer divides them by --speed.
    preprocess_window_s: float = 180.0
    preprocess_slide_s: float = 5.0
    preprocess_trigger_s: float = 60.0
    predict_window_s: float = 600.0
    predict_slide_s: float = 60.0
    watermark_s: float = 10.0

    # Bus defaults (reference utils.py:417-428): acks=all, retries=5,
    # startingOffsets=latest.
    bus_dir: str = ".tskd_bus"
    producer_acks: str = "all"
    producer_retries: int = 5
    consumer_starting_offsets: str = "latest"

    raw: Optional[configparser.ConfigParser] = None

    @property
    def n_channels(self) -> int:
        return len(self.channel_names)

    def channel_index(self, name: str) -> int:
        return self.channel_names.index(name)

    def topic_for_channel(self, name: str) -> str:
        # Topic name = channel name with spaces replaced (reference sendStream.py:59).
        return name.replace(" ", "_")


def _split_csv(v: str) -> List[str]:
    return [s.strip() for s in v.split(",") if s.strip()]


def get_global_config(path: Optional[str] = None) -> GlobalConfig:
    """Load config.cfg if present; fall back to defaults.

    Search order: explicit ``path`` arg, ``$TSKD_CONFIG``, ``./config.cfg``.
    Reference-format files (with MOUNTPATH/LOCAL* interpolation and
    [SETTINGS]) parse directly; the darwin/linux LOCAL* switch of reference
    utils.py:17-65 is honoured.
    """
    cfg = GlobalConfig()
    candidates = [path, os.environ.get("TSKD_CONFIG"), "config.cfg"]
    found = next((c for c in candidates if c and os.path.exists(c)), None)
    if found is None:
        return cfg

    parser = configparser.ConfigParser()
    parser.read(found)
    cfg.raw = parser

    prefix = "LOCAL" if sys.platform == "darwin" else ""
    if parser.has_section("PATHS"):
        p = parser["PATHS"]

        def pget(key: str, default: str) -> str:
            return p.get(prefix + key, p.get(key, default))

        cfg.mount_path = pget("MOUNTPATH", cfg.mount_path)
        cfg.data_path = pget("DATAPATH", cfg.data_path)
        cfg.wavef_path = pget("WAVEFPATH", cfg.wavef_path)
        cfg.output_path = pget("OUTPUTPATH", cfg.output_path)
        cfg.explore_path = pget("EXPLOREPATH", cfg.explore_path)
        cfg.model_path = pget("MODELPATH", cfg.model_path)

    if parser.has_section("SETTINGS"):
        s = parser["SETTINGS"]
        cfg.use_cuda = s.get("USE_CUDA", "0").strip() not in ("0", "", "false", "False")
        cfg.num_workers = int(s.get("NUM_WORKERS", cfg.num_workers))
        if "CHANNEL_NAMES" in s:
            cfg.channel_names = _split_csv(s["CHANNEL_NAMES"])
        if "PATIENTRECORDS" in s:
            cfg.patient_records = _split_csv(s["PATIENTRECORDS"])
        cfg.window_size = int(s.get("WINDOWSIZE", cfg.window_size))
        cfg.record_overlap = float(s.get("RECORDOVERLAP", cfg.record_overlap))
        cfg.batch_size = int(s.get("BATCHSIZE", cfg.batch_size))

    if parser.has_section("BUS"):
        b = parser["BUS"]
        cfg.bus_dir = b.get("DIR", cfg.bus_dir)
        cfg.producer_acks = b.get("ACKS", cfg.producer_acks)
        cfg.producer_retries = int(b.get("RETRIES", cfg.producer_retries))
        cfg.consumer_starting_offsets = b.get("STARTING_OFFSETS", cfg.consumer_starting_offsets)

    return cfg
