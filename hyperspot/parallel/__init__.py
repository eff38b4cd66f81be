from .state import (initialize_model_parallel, destroy_model_parallel,
                    get_tp_group, get_tp_rank, get_tp_size,
                    get_ep_group, get_ep_rank, get_ep_size,
                    tensor_model_parallel_all_reduce)
from .layers import (ColumnParallelLinear, RowParallelLinear, QKVParallelLinear,
                     MergedColumnParallelLinear)
