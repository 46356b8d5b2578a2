from .comparison import assert_close, assert_close_loose, assert_equal, check_state_dict_equal
from .utils import (
    DummyDataloader,
    clear_cache_before_run,
    free_port,
    parameterize,
    rerun_if_address_is_in_use,
    spawn,
)

__all__ = [
    "assert_close",
    "assert_close_loose",
    "assert_equal",
    "check_state_dict_equal",
    "DummyDataloader",
    "clear_cache_before_run",
    "free_port",
    "parameterize",
    "rerun_if_address_is_in_use",
    "spawn",
]
