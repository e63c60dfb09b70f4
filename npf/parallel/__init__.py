"""Distributed (RCCL/xGMI) data-parallel backend."""

from .ddp import (  # noqa: F401
    FlatDDP,
    all_gather_cat,
    barrier,
    get_rank,
    get_world_size,
    init_distributed,
    is_distributed,
)
