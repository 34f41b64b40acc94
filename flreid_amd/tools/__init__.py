from flreid_amd.tools.logger import Logger
from flreid_amd.tools.utils import (
    clear_cache,
    get_one_hot,
    model_on_device,
    params_state_size,
    same_seeds,
    tensor_reverse_permute,
)
