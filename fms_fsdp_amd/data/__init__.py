from .dataloader import get_data_loader, get_dummy_loader, causal_lm, parse_data_args

__all__ = ["get_data_loader", "get_dummy_loader", "causal_lm", "parse_data_args"]
