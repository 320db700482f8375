from .fsdp import ShardedModel, ShardedAdamW, FlatUnit

__all__ = ["ShardedModel", "ShardedAdamW", "FlatUnit"]
