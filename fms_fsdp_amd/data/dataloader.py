"""Dataloader assembly (reference: fms_fsdp/utils/dataloader_utils.py).

get_data_loader builds the full stateful streaming pipeline (data/datasets.py);
get_dummy_loader provides the deterministic synthetic stream used for
benchmarking and plumbing tests (reference: dataloader_utils.py:36-57).
"""

import torch
from torch.utils.data import DataLoader, IterableDataset


class SteadyCounter(IterableDataset):
    """Infinite deterministic token stream: step i yields
    [i*seq_len, ..., (i+1)*seq_len) mod vocab (reference SteadyCounter,
    dataloader_utils.py:40-57)."""

    def __init__(self, seq_len, vocab_size, rank=0, worldsize=1):
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.rank = rank
        self.worldsize = worldsize
        self.i = rank * seq_len

    def __iter__(self):
        while True:
            out = torch.arange(self.i, self.i + self.seq_len + 1) % self.vocab_size
            yield out[:-1], out[1:]
            self.i += (self.seq_len + 1) * self.worldsize

    def state_dict(self):
        return {"i": self.i}

    def load_state_dict(self, sd):
        self.i = sd["i"]


def causal_lm(data_seq, prompt_len=0):
    """(input, label) pair with labels shifted and prompt masked to -100
    (reference: dataloader_utils.py:24-33)."""
    data_seq = torch.tensor(data_seq, dtype=torch.long) \
        if not torch.is_tensor(data_seq) else data_seq.long()
    t = data_seq.clone()
    t[:prompt_len + 1] = -100
    return data_seq[:-1], t[1:]


def get_dummy_loader(cfg, rank, world_size):
    data = SteadyCounter(cfg.seq_length, cfg.vocab_size, rank, world_size)
    return DataLoader(data, batch_size=cfg.batch_size)


def parse_data_args(datas, weights):
    """csv strings -> lists (reference: dataloader_utils.py:149-163)."""
    def splitstrip(x):
        if isinstance(x, str):
            return [item.strip() for item in x.split(",")]
        if isinstance(x, (list, tuple)):
            return list(x)
        if isinstance(x, (int, float, complex)):
            return [x]
        raise ValueError(f"arg input {x} cannot be parsed.")
    datas = splitstrip(datas)
    weights = [float(x) for x in splitstrip(weights)]
    return datas, weights


def get_data_loader(cfg, rank, world_size, postprocess=None):
    """Build the full 8-stage stateful pipeline (reference:
    dataloader_utils.py:60-146)."""
    from fms_fsdp_amd.data import datasets as D

    if postprocess is None:
        postprocess = [torch.IntTensor, causal_lm]

    datas, weights = parse_data_args(cfg.datasets, cfg.weights)

    def _mk_handler():
        if cfg.file_type == "arrow":
            return D.ArrowHandler(cfg.col_name)
        if cfg.file_type == "parquet":
            return D.ParquetHandler(cfg.tokenizer_path, cfg.col_name)
        if cfg.file_type == "auto":
            return D.AutoHandler(cfg.tokenizer_path, cfg.col_name)
        raise ValueError(f"file type {cfg.file_type} not recognized")

    droplist = [int(x.strip()) for x in cfg.strip_tokens.split(",") if x.strip()]
    droplist += [cfg.bos_token, cfg.eos_token, cfg.bol_token, cfg.eol_token]
    droplist = [x for x in droplist if x is not None]

    data = D.StreamingDocDataset(cfg.data_path, rank, world_size, _mk_handler(),
                                 cfg.eos_token, bos_token=cfg.bos_token,
                                 strip_tokens=set(droplist),
                                 min_length=3, seed=cfg.seed,
                                 max_chunksize=cfg.seq_length)
    data = D.ScalableShardDataset(data, n_logical_shards=cfg.logical_shards)
    data = D.SamplingDataset(cfg.data_path, data, cfg.eos_token,
                             datasets=datas, weights=weights,
                             verbose=(rank == 0))
    data = D.BufferDataset(data, cfg.seq_length + 1, bos_token=cfg.bol_token,
                           eos_token=cfg.eol_token, pack_hard=True)
    data = D.PreloadBufferDataset(data, 10000)
    for pp in postprocess:
        data = D.PreprocessDataset(data, pp)
    data = D.CheckpointDataset(
        data,
        cfg.ckpt_load_path if cfg.resuming_dataset else cfg.ckpt_save_path,
        cfg.checkpoint_interval,
        cfg.batch_size,
        cfg.ckpt_save_path,
    )
    return DataLoader(data, num_workers=cfg.num_workers,
                      batch_size=cfg.batch_size)
