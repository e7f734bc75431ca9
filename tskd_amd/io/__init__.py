from tskd_amd.io.wfdb import rdrecord, Record, get_waveform_path  # noqa: F401
