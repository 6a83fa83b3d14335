from .tp import TPGroup, init_distributed, make_tp_group, shard_sizes

__all__ = ["TPGroup", "init_distributed", "make_tp_group", "shard_sizes"]
