from .dist import (all_reduce_, barrier, broadcast_obj, env_rank, env_world,
                   init_distributed, is_tp, recv_cmd, send_ints, send_obj,
                   tp_group, tp_rank, tp_size)

__all__ = [
    "all_reduce_", "barrier", "broadcast_obj", "env_rank", "env_world",
    "init_distributed", "is_tp", "recv_cmd", "send_ints", "send_obj",
    "tp_group", "tp_rank", "tp_size",
]
