from gllm_amd.parallel.state import (  # noqa: F401
    init_distributed, destroy_distributed,
    get_rank, get_world_size, get_tp_rank, get_tp_size, get_pp_rank,
    get_pp_size, get_dp_rank, get_dp_size, get_ep_rank, get_ep_size,
    get_tp_group, get_dp_group, get_ep_group,
    is_first_pp_rank, is_last_pp_rank,
    tensor_parallel_all_reduce, tensor_parallel_all_gather,
    send_pp_data, recv_pp_data, get_prev_pp_rank, get_next_pp_rank,
    dp_meta_barrier, dp_all_gather, set_dp_forward_counts,
    get_dp_forward_counts,
)
