from tskd_amd.train.metrics import AverageMeter, compute_batch_accuracy  # noqa: F401
from tskd_amd.train.loop import evaluate, train  # noqa: F401
from tskd_amd.train.data import (create_batch, load_dataset,  # noqa: F401
                                 make_synthetic_labeled_windows,
                                 random_oversample, random_undersample,
                                 record_to_training_frame)
