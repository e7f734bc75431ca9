"""tskd_amd — MI355X-native streaming waveform-inference engine.

A brand-new, MI355X-first framework with the capabilities of
``travistangvh/time-series-kafka-demo`` (real-time ICU cardiac-arrest risk
pipeline): waveform stream replay -> event-time sliding-window preprocessing
-> CNN+LSTM risk scoring -> durable prediction store -> dashboard.

Layer map (vs reference):
  - :mod:`tskd_amd.models`    — MyCNN model family + legacy-pickle checkpoints
                                (reference bin/models.py, explore_torch.ipynb cell 26)
  - :mod:`tskd_amd.ops`       — hand-written CDNA4 HIP kernels (fused conv+LSTM
                                inference, fused preprocess, training)
                                (replaces PyTorch ATen CPU kernels)
  - :mod:`tskd_amd.engine`    — streaming window engine: event-time ring
                                buffers, watermarks, 180s/5s + 600s/60s windows
                                (replaces Spark Structured Streaming)
  - :mod:`tskd_amd.bus`       — broker-less keyed topic bus
                                (replaces Kafka + librdkafka)
  - :mod:`tskd_amd.store`     — embedded prediction store + age table
                                (replaces MySQL)
  - :mod:`tskd_amd.io`        — WFDB waveform reader (replaces wfdb-python)
  - :mod:`tskd_amd.parallel`  — patient-shard DP across GPUs, RCCL over xGMI
  - :mod:`tskd_amd.cli`       — sendStream / processStream / predictStream /
                                plotData entrypoints with reference flags
"""

__version__ = "0.1.0"

from tskd_amd.config import get_global_config  # noqa: F401
