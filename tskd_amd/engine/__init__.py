from tskd_amd.engine.stream_engine import StreamEngine  # noqa: F401
from tskd_amd.engine.windowing import (preprocess_series_oracle,  # noqa: F401
                                       sliding_windows)
