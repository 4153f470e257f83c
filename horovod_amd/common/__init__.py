from .exceptions import HorovodInternalError, HostsUpdatedInterrupt  # noqa: F401
from .process_sets import ProcessSet, global_process_set  # noqa: F401
