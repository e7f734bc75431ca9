from tskd_amd.models.mycnn import MyCNN, MyCNN2, MyCNN3, MyCNN4, MyCNN5, build_model  # noqa: F401
from tskd_amd.models.checkpoint import load_checkpoint, save_checkpoint  # noqa: F401
