from horovod_amd.utils.util import (num_rank_is_power_2,  # noqa: F401
                                    split_list, host_hash)
