from horovod_amd.data.data_loader_base import (AsyncDataLoaderMixin,  # noqa: F401
                                               BaseDataLoader)
