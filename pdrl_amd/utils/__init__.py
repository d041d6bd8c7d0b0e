from .utils import (  # noqa: F401
    Params,
    Machines,
    Protocol,
    DataFrameKeyword,
    ExecutionTimer,
    encode,
    decode,
    load_params,
    load_machines,
    refresh_result_dirs,
    to_torch,
    flatten,
    mul,
    make_gpu_batch,
    obs_preprocess,
    extract_file_num,
    select_least_used_gpu,
    save_error_log,
    register_child,
    terminate_children,
)
from .lock import Mutex, Lock  # noqa: F401
from .logger import SummaryWriter  # noqa: F401
