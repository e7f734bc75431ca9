from tskd_amd.parallel.dist import (DPServing, all_gather_predictions,  # noqa: F401
                                    init_distributed, shard_for_key,
                                    shard_streams)
