from .fsdp import ShardedModel, ShardedAdamW, FlatUnit, DynamicGradScaler

__all__ = ["ShardedModel", "ShardedAdamW", "FlatUnit", "DynamicGradScaler"]
