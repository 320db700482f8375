"""Checkpointable, rescalable streaming dataset pipeline.

A from-scratch implementation with the behavioral contract of the
reference's stateful dataset stack (fms_fsdp/utils/dataset_utils.py):

1. Workers never communicate: each (rank, worldsize) pair owns a
   deterministic partition of the data.
2. The pipeline is composable wrapper layers over a base disk reader.
3. Every layer checkpoints through recursive state_dict()/
   load_state_dict(), giving token-exact resume.
4. Rescalability: state fields are split into `state_params` (scalars,
   droppable when the world size changes) and `reshard_params` (lists
   redistributed across the new world). ScalableShardDataset keeps
   exactly-once-per-epoch guarantees across world-size changes by
   tracking many small logical shards.

Layer stack (bottom-up, as assembled by data/dataloader.py):
  StreamingDocDataset -> ScalableShardDataset -> SamplingDataset ->
  BufferDataset -> PreloadBufferDataset -> PreprocessDataset(s) ->
  CheckpointDataset -> torch DataLoader
"""

import csv
import logging
import math
import os
import random
import time
from copy import deepcopy
from typing import Any, Callable, List, Optional, Set

import torch
import torch.utils.data as tdata

logger = logging.getLogger(__name__)


def _shard_partition(items: List[Any], rank: int, worldsize: int) -> List[Any]:
    """Contiguous 1/worldsize slice of items for this rank."""
    return items[(rank * len(items)) // worldsize:
                 ((rank + 1) * len(items)) // worldsize]


def _shard_inclusive(items: List[Any], rank: int, worldsize: int) -> List[Any]:
    """Slice covering every item this rank owns at least fractionally."""
    lo = math.floor(len(items) * rank / worldsize)
    hi = math.ceil(len(items) * (rank + 1) / worldsize)
    return items[lo:hi]


class _StatefulDataset(tdata.IterableDataset):
    """Base: rank/world bookkeeping + recursive state save/load with
    rescaling support (state_params dropped, reshard_params redistributed
    when the saved world size differs from the current one)."""

    def __init__(self, datapath: Optional[str], rank: int, worldsize: int):
        assert rank >= 0 and worldsize > rank, \
            f"invalid rank/worldsize {rank}/{worldsize}"
        assert datapath is None or (os.path.isdir(datapath)
                                    and len(os.listdir(datapath)) > 0), \
            f"datapath {datapath} must be a non-empty directory or None"
        self.datapath = datapath
        self.rank = rank
        self.worldsize = worldsize
        self.local_worldsize = -1
        self.load_worldsize = worldsize
        self.state_params: List[str] = []
        self.reshard_params: List[str] = []
        self.is_setup = False

    # -- setup --------------------------------------------------------

    def setup(self):
        """Deferred rank-dependent init. Inflates rank/worldsize once to
        account for torch DataLoader worker processes."""
        if self.is_setup:
            return
        self.is_setup = True
        if self.local_worldsize == -1:
            info = tdata.get_worker_info()
            if info is None or info.num_workers == 1:
                self.local_worldsize = 1
            else:
                self.local_worldsize = info.num_workers
                self.rank = self.rank * info.num_workers + info.id
                self.worldsize = self.worldsize * info.num_workers

    # -- state --------------------------------------------------------

    def _key(self, name: str) -> str:
        # class-qualified key: one instance of each layer type per pipeline
        return f"{type(self).__name__}.{name}"

    def state_dict(self):
        self.setup()
        return {self._key(f): getattr(self, f)
                for f in self.state_params + self.reshard_params}

    def _redistribute(self, shards: List[List[Any]]) -> List[Any]:
        """Given the (inclusively-owned) per-checkpoint-rank list shards of
        one reshard_param, return exactly the flattened span this rank owns
        under the new worldsize."""
        per = len(shards[0])
        for s in shards:
            assert len(s) == per, "reshard list shards must be equal length"
        first_shard = math.floor(self.load_worldsize * self.rank / self.worldsize)
        skipped = per * first_shard
        total = self.load_worldsize * per
        lo = int(total * self.rank / self.worldsize) - skipped
        hi = int(total * (self.rank + 1) / self.worldsize) - skipped
        return [shards[i // per][i % per] for i in range(lo, hi)]

    def load_state_dict(self, state_dicts, sharded_input=False):
        self.setup()
        if not sharded_input:
            self.load_worldsize = len(state_dicts)
            state_dicts = _shard_inclusive(state_dicts, self.rank, self.worldsize)
        if self.load_worldsize == self.worldsize:
            for f in self.state_params + self.reshard_params:
                setattr(self, f, state_dicts[0][self._key(f)])
        else:
            for f in self.reshard_params:
                setattr(self, f, self._redistribute(
                    [sd[self._key(f)] for sd in state_dicts]))
        return state_dicts

    # -- disk ---------------------------------------------------------

    def save_to_path(self, path: str):
        os.makedirs(path, exist_ok=True)
        torch.save(self.state_dict(),
                   os.path.join(path, f"loader_state_{self.rank}.pth"))

    def load_from_path(self, path: str):
        assert os.path.isdir(path), f"{path} must be a checkpoint folder"
        files = sorted([f for f in os.listdir(path) if "loader" in f],
                       key=lambda f: int(f.split("_")[2][:-4]))
        assert files, f"no loader state files in {path}"
        self.load_worldsize = len(files)
        mine = _shard_inclusive(files, self.rank, self.worldsize)
        states = [torch.load(os.path.join(path, f), weights_only=False)
                  for f in mine]
        self.load_state_dict(states, True)


class _WrapperDataset(_StatefulDataset):
    """A layer holding exactly one wrapped sub-dataset; state methods
    recurse. rank/worldsize adjustments propagate downward in setup()."""

    def __init__(self, dataset: _StatefulDataset):
        self.dataset = dataset
        super().__init__(dataset.datapath, dataset.rank, dataset.worldsize)

    def setup(self):
        if self.is_setup:
            return
        super().setup()
        self.dataset.datapath = self.datapath
        self.dataset.rank = self.rank
        self.dataset.worldsize = self.worldsize
        self.dataset.local_worldsize = self.local_worldsize
        self.dataset.setup()

    def state_dict(self):
        self.setup()
        out = self.dataset.state_dict()
        out.update(_StatefulDataset.state_dict(self))
        return out

    def load_state_dict(self, state_dicts, sharded_input=False):
        self.setup()
        sharded = _StatefulDataset.load_state_dict(self, state_dicts, sharded_input)
        self.dataset.load_worldsize = self.load_worldsize
        self.dataset.load_state_dict(sharded, True)
        return sharded


# ------------------------- file handlers -------------------------

class _ShardFileHandler:
    """Format adapter: open / count / fetch-doc / slice-doc."""

    def is_legal(self, filepath: str) -> bool:
        return os.path.isfile(filepath)

    def open(self, path: str):
        raise NotImplementedError

    def length(self, path: str) -> int:
        raise NotImplementedError

    def get(self, reader, index: int, drop_tokens: Set):
        raise NotImplementedError

    def slice(self, doc, index: int, n_pull: int) -> List:
        raise NotImplementedError


class ArrowHandler(_ShardFileHandler):
    """Memory-mapped pyarrow IPC shards; doc = RecordBatch column
    (pre-tokenized). Chunks slice zero-copy."""

    def __init__(self, col_name: str = "tokens"):
        self.col_name = col_name

    def is_legal(self, filepath: str):
        return "arrow" in os.path.splitext(filepath)[1]

    def open(self, path: str):
        import pyarrow as pa
        return pa.ipc.open_file(pa.memory_map(path))

    def length(self, path: str):
        return self.open(path).num_record_batches

    def get(self, reader, index: int, drop_tokens: Set):
        doc = reader.get_batch(index)[self.col_name]
        if len(doc) > 0 and doc[0].as_py() in drop_tokens:
            doc = doc.slice(1, len(doc) - 1)
        if len(doc) > 0 and doc[-1].as_py() in drop_tokens:
            doc = doc.slice(0, len(doc) - 1)
        return doc

    def slice(self, doc, index: int, n_pull: int) -> List:
        return doc.slice(index, n_pull).to_pylist()


class ParquetHandler(_ShardFileHandler):
    """Parquet shards of raw text, tokenized on the fly with an HF
    tokenizer."""

    def __init__(self, tokenizer_path: str, col_name: str = "text"):
        from transformers import AutoTokenizer
        self.tokenizer = AutoTokenizer.from_pretrained(tokenizer_path)
        self.col_name = col_name

    def is_legal(self, filepath: str):
        return "parquet" in os.path.splitext(filepath)[1]

    def open(self, path: str):
        import pyarrow.parquet as pq
        return pq.read_pandas(path, columns=[self.col_name],
                              partitioning=None)[self.col_name]

    def length(self, path: str):
        import pyarrow.parquet as pq
        return pq.read_metadata(path).num_rows

    def get(self, reader, index: int, drop_tokens: Set):
        doc = self.tokenizer(str(reader[index]))["input_ids"]
        if len(doc) > 0 and doc[0] in drop_tokens:
            doc = doc[1:]
        if len(doc) > 0 and doc[-1] in drop_tokens:
            doc = doc[:-1]
        return doc

    def slice(self, doc, index: int, n_pull: int) -> List:
        return doc[index:index + n_pull]


class AutoHandler(_ShardFileHandler):
    """Dispatch to Arrow or Parquet per file extension."""

    def __init__(self, tokenizer_path: str, col_name: str = "text"):
        self.PHandler = ParquetHandler(tokenizer_path, col_name)
        self.AHandler = ArrowHandler()
        self.current: _ShardFileHandler = _ShardFileHandler()

    def is_legal(self, filepath: str):
        ext = os.path.splitext(filepath)[1]
        return "parquet" in ext or "arrow" in ext

    def open(self, path: str):
        self.current = self.AHandler \
            if "arrow" in os.path.splitext(path)[1] else self.PHandler
        return self.current.open(path)

    def length(self, path: str):
        h = self.AHandler if "arrow" in os.path.splitext(path)[1] else self.PHandler
        return h.length(path)

    def get(self, reader, index: int, drop_tokens: Set):
        return self.current.get(reader, index, drop_tokens)

    def slice(self, doc, index: int, n_pull: int) -> List:
        return self.current.slice(doc, index, n_pull)


# ------------------------- base reader -------------------------

class StreamingDocDataset(_StatefulDataset):
    """Reads documents from shard files in a dataset directory, owned
    partition determined by (rank, worldsize).

    - Shard files are over-split into worldsize fragments each; this rank
      takes a contiguous run of fragments (minimizes file opens).
    - Per-shard doc counts come from ``<parent>/meta/*counts*.csv`` when
      present, else by touching each owned file.
    - Within a shard, docs are visited in an LCG-shuffled order (a=5,
      c = 2*(rank+seed)+1, modulus = next pow2) so no doc list is ever
      materialized.
    - Docs are emitted in chunks of <= max_chunksize, delimiter appended
      after the final chunk (and optional BOS before the first).
    - Mid-doc resume: chunk_index persists; on iteration restart the
      already-delivered chunks of the current doc are skipped, then
      replayed at epoch end so the epoch is exactly-once.
    """

    def __init__(self, datapath: str, rank: int, worldsize: int,
                 filehandler: _ShardFileHandler, delimiter_token: Any,
                 bos_token: Optional[Any] = None,
                 strip_tokens: Optional[Set[Any]] = None,
                 seed: int = 42, min_length: int = 1, max_chunksize: int = 1024,
                 verbose: bool = False):
        super().__init__(datapath, rank, worldsize)
        assert max_chunksize > 0
        self.filehandler = filehandler
        self.eos = delimiter_token
        self.bos = bos_token
        self.drop = strip_tokens or set()
        self.seed = seed
        self.min_length = min_length
        self.chunksize = max_chunksize
        self.verbose = verbose

        # owned docs: list of (shard relpath, first docid, last docid)
        self.docset: List[Any] = []
        self._len = 0
        self.dataset = ""

        # position + stats (all droppable on rescale)
        self.docset_index = 0
        self.chunk_index = -1
        self.epochs_seen = -1
        self.tokens_seen = 0
        self.docs_seen = 0
        self.percent_seen = 0
        self.lcg_state = 0
        self.state_params = ["dataset", "docset_index", "chunk_index",
                             "epochs_seen", "tokens_seen", "docs_seen",
                             "percent_seen", "lcg_state"]

    # -- setup --------------------------------------------------------

    def setup(self):
        if self.is_setup:
            return
        super().setup()
        datapath = self.datapath
        # dataset name = last path component (robust to trailing slashes)
        head, tail = os.path.split(datapath)
        while not tail:
            head, tail = os.path.split(head)
        pardir, self.dataset = head, tail

        shards = sorted(
            os.path.join(root, name)[len(datapath) + 1:]
            for root, _, files in os.walk(datapath, topdown=False)
            for name in files
            if self.filehandler.is_legal(os.path.join(root, name)))

        # fragment partition: worldsize fragments per shard, contiguous run
        nfrag_start = (self.rank * self.worldsize * len(shards)) // self.worldsize
        nfrag_end = ((self.rank + 1) * self.worldsize * len(shards)) // self.worldsize
        frags = [(shards[i // self.worldsize], i % self.worldsize)
                 for i in range(nfrag_start, nfrag_end)]

        counts = self._doc_counts(pardir, datapath, frags)

        # aggregate owned fragments into per-shard [min docid, max docid]
        ranges = {}
        for shard, frag in frags:
            n = counts[shard]
            lo = (n * frag) // self.worldsize
            hi = (n * frag + n) // self.worldsize - 1  # inclusive
            if shard in ranges:
                ranges[shard][0] = min(ranges[shard][0], lo)
                ranges[shard][1] = max(ranges[shard][1], hi)
            else:
                ranges[shard] = [lo, hi]
        self.docset = [(s, r[0], r[1]) for s, r in ranges.items()]
        self._len = sum(r[1] - r[0] + 1 for _, r in ranges.items())

        # deterministic per-rank shuffle of shard visit order
        rng = random.Random(self.seed + self.rank)
        rng.shuffle(self.docset)
        self.lcg_state = self.seed + self.rank

    def _doc_counts(self, pardir, datapath, frags):
        """Per-shard doc counts: meta csv if present, else touch files."""
        metadir = os.path.join(pardir, "meta")
        if os.path.exists(metadir):
            csvs = [f for f in os.listdir(metadir)
                    if "counts" in f and "csv" in f]
            if csvs:
                out = {}
                with open(os.path.join(metadir, csvs[0]), "r") as fh:
                    for row in csv.DictReader(fh):
                        full = row["dataset/filename"]
                        cut = full.find("/" + self.dataset) + 1
                        if cut > 0:
                            out[full[cut + len(self.dataset) + 1:]] = \
                                int(row["documents"])
                return out
        return {shard: self.filehandler.length(os.path.join(datapath, shard))
                for shard in set(s for s, _ in frags)}

    # -- iteration ----------------------------------------------------

    def _locate(self, i):
        """Owned-doc index -> (shard relpath, range size, range start)."""
        assert i <= self._len, f"doc index {i} out of range {self._len}"
        seen = 0
        for shard, lo, hi in self.docset:
            seen += hi - lo + 1
            if seen > i:
                return shard, hi - lo + 1, lo

    def _lcg_next(self, size):
        """Advance the LCG until it lands inside [0, size): a random
        bijection over the shard's doc range (Knuth 3.2.1.3 params)."""
        m = 2 ** math.ceil(math.log2(size)) if size > 1 else 1
        a, c = 5, (self.rank + self.seed) * 2 + 1
        state = self.lcg_state
        while True:
            state = (a * state + c) % m if m > 1 else 0
            if state < size:
                return state

    def _emit_chunk(self, j, doc, n_chunks):
        """Chunk j of a doc, with BOS on the first and EOS after the last."""
        start = j * self.chunksize
        pull = self.chunksize
        if self.bos is not None:
            if j == 0:
                pull -= 1
            else:
                start -= 1
        chunk = self.filehandler.slice(doc, start, pull)
        self.tokens_seen += len(chunk)
        if self.bos is not None and j == 0:
            chunk = [self.bos] + chunk
        if j == n_chunks - 1:
            chunk = chunk + [self.eos]
        return chunk

    def __iter__(self):
        self.setup()
        start_doc = self.docset_index
        start_lcg = self.lcg_state
        residual = self.chunk_index + 1  # chunks already delivered pre-ckpt
        path, reader = "", None
        while True:
            for i in range(self._len):
                idx = (start_doc + i) % self._len
                if idx == 0:
                    self.epochs_seen += 1
                self.docset_index = idx
                shard, size, lo = self._locate(idx)
                newpath = os.path.join(self.datapath, shard)
                if newpath != path:
                    reader = self.filehandler.open(newpath)
                    path = newpath
                pos = self._lcg_next(size)
                doc = self.filehandler.get(reader, pos + lo, self.drop)
                if len(doc) != 0:
                    doclen = len(doc) + (1 if self.bos is None else 2)
                    if doclen >= self.min_length:
                        n_chunks = math.ceil(doclen / self.chunksize)
                        for j in range(n_chunks):
                            if i == 0 and j < residual:
                                continue  # already delivered before resume
                            self.chunk_index = j
                            if j == n_chunks - 1:
                                self.docs_seen += 1
                                self.percent_seen = \
                                    self.docs_seen * 100 / (self._len + 1e-9)
                            yield self._emit_chunk(j, doc, n_chunks)
                self.lcg_state = pos

            # epoch wrap: replay the chunks skipped at resume time
            self.docset_index = start_doc
            self.lcg_state = start_lcg
            shard, size, lo = self._locate(start_doc)
            pos = self._lcg_next(size)
            newpath = os.path.join(self.datapath, shard)
            if newpath != path:
                reader = self.filehandler.open(newpath)
                path = newpath
            doc = self.filehandler.get(reader, pos + lo, self.drop)
            if len(doc) == 0:
                continue
            doclen = len(doc) + (1 if self.bos is None else 2)
            if doclen >= self.min_length:
                n_chunks = math.ceil(doclen / self.chunksize)
                for j in range(residual):
                    self.chunk_index = j
                    yield self._emit_chunk(j, doc, n_chunks)

    def load_state_dict(self, state_dicts, sharded_input=False):
        self.setup()
        assert self.load_worldsize == self.worldsize, \
            "StreamingDocDataset cannot rescale; wrap in ScalableShardDataset"
        prev = self.dataset
        out = super().load_state_dict(state_dicts, sharded_input)
        assert prev == self.dataset, \
            f"checkpoint is for dataset {self.dataset}, expected {prev}"
        return out


# ------------------------- rescalable sharding -------------------------

class ScalableShardDataset(_WrapperDataset):
    """Rescalability layer: owns total_shards/worldsize logical shards,
    each a deep-copied StreamingDocDataset with rank in [0, total_shards).
    Documents are drawn from logical shards weighted by docs-remaining, so
    partially-consumed epochs survive world-size changes without repeats."""

    def __init__(self, dataset: StreamingDocDataset, n_logical_shards: int = 2048,
                 verbose: bool = False):
        super().__init__(dataset)
        assert n_logical_shards % self.worldsize == 0, \
            f"worldsize {self.worldsize} must divide {n_logical_shards}"
        self.total_shards = n_logical_shards
        self.verbose = verbose
        self.delimiter = dataset.eos

        self.data: List[StreamingDocDataset] = []
        self.n_logicals = 0
        self.n_docs_remaining: List[int] = []
        self.generator = None

        self.current_reader = None
        self.logical_shard_states = None
        self.g_state = None
        self.state_params = ["current_reader", "g_state"]
        self.reshard_params = ["n_docs_remaining", "logical_shard_states"]

    def setup(self):
        if self.is_setup:
            return
        _StatefulDataset.setup(self)
        owned = _shard_partition(list(range(self.total_shards)),
                                 self.rank, self.worldsize)
        self.n_logicals = self.total_shards // self.worldsize
        assert len(owned) == self.n_logicals
        for i, logical in enumerate(owned):
            d = deepcopy(self.dataset)
            d.worldsize = self.total_shards
            d.load_worldsize = self.total_shards
            d.rank = logical
            d.local_worldsize = 1
            d.datapath = self.datapath
            d.verbose = self.rank == 0
            d.setup()
            self.data.append(d)
        self.n_docs_remaining = [d._len for d in self.data]
        self.generator = torch.Generator().manual_seed(self.rank)

    def __iter__(self):
        self.setup()
        iters = [iter(d) for d in self.data]
        while True:
            if self.current_reader is not None:
                ind = self.current_reader
            else:
                assert sum(self.n_docs_remaining) > 0, \
                    f"no documents found under {self.datapath}"
                ind = torch.multinomial(
                    torch.tensor(self.n_docs_remaining, dtype=torch.float),
                    1, generator=self.generator).item()
            self.current_reader = ind
            out = next(iters[ind])
            while out[-1] != self.delimiter:   # mid-doc: stay on this shard
                yield out
                out = next(iters[ind])
            self.current_reader = None
            self.n_docs_remaining[ind] -= 1
            if sum(self.n_docs_remaining) == 0:   # epoch boundary
                self.n_docs_remaining = [d._len for d in self.data]
                self.generator.manual_seed(self.rank)
            yield out

    def state_dict(self):
        self.setup()
        self.g_state = self.generator.get_state()
        self.logical_shard_states = [d.state_dict() for d in self.data]
        return _StatefulDataset.state_dict(self)

    def load_state_dict(self, state_dicts, sharded_input=False):
        self.setup()
        sharded = _StatefulDataset.load_state_dict(self, state_dicts, sharded_input)
        if self.g_state is not None:
            self.generator.set_state(self.g_state)
        for i in range(self.n_logicals):
            self.data[i].load_state_dict([self.logical_shard_states[i]], True)
        return sharded


# ------------------------- corpus mixing -------------------------

class SamplingDataset(_WrapperDataset):
    """Weighted multi-corpus mixing: one cloned sub-pipeline per corpus;
    the next document comes from whichever corpus is furthest below its
    target token share (greedy deficit). Documents are never split across
    corpus switches (delimiter detection)."""

    def __init__(self, datapath: str, dataset: _StatefulDataset,
                 delimiter_token: Any, datasets=None, weights=None,
                 verbose: bool = False):
        super().__init__(dataset)
        self.datapath = datapath
        self.delimiter = delimiter_token
        self.verbose = verbose
        self.datasets = datasets if datasets is not None else [
            f for f in os.listdir(datapath)
            if not os.path.isfile(os.path.join(datapath, f)) and "meta" not in f]
        assert len(self.datasets) > 0, "need at least one dataset"
        if weights is not None:
            assert len(weights) == len(self.datasets), \
                f"{len(weights)} weights vs {len(self.datasets)} datasets"
            assert all(w > 0 for w in weights)
        w = [1] * len(self.datasets) if weights is None else list(weights)
        self.weights = [x / sum(w) for x in w]
        self.tokens_seen = [0] * len(self.datasets)
        self.current_iterator = -1
        self.state_params = ["tokens_seen", "current_iterator"]

    def setup(self):
        if self.is_setup:
            return
        _StatefulDataset.setup(self)
        self.data = []
        for d in self.datasets:
            sub = deepcopy(self.dataset)
            sub.datapath = os.path.join(self.datapath, d)
            sub.rank = self.rank
            sub.worldsize = self.worldsize
            sub.local_worldsize = self.local_worldsize
            sub.setup()
            self.data.append(sub)

    def __iter__(self):
        self.setup()
        iters = [iter(d) for d in self.data]
        while True:
            if self.current_iterator != -1:
                out = next(iters[self.current_iterator])
                self.tokens_seen[self.current_iterator] += len(out)
                if out[-1] == self.delimiter:
                    self.current_iterator = -1
                yield out
            else:
                total = sum(self.tokens_seen) + 1e-9
                deficit = [self.weights[i] - self.tokens_seen[i] / total
                           for i in range(len(self.datasets))]
                self.current_iterator = max(
                    (d, i) for i, d in enumerate(deficit))[1]

    def state_dict(self):
        self.setup()
        out = {self._key("sample_iterator_states"):
               [d.state_dict() for d in self.data]}
        out.update(_StatefulDataset.state_dict(self))
        return out

    def load_state_dict(self, state_dicts, sharded_input=False):
        self.setup()
        sharded = _StatefulDataset.load_state_dict(self, state_dicts, sharded_input)
        for i, sub in enumerate(self.data):
            sub.load_worldsize = self.load_worldsize
            sub.load_state_dict(
                [sd[self._key("sample_iterator_states")][i] for sd in sharded],
                True)
        return sharded


# ------------------------- packing -------------------------

class BufferDataset(_WrapperDataset):
    """Packs variable-length chunk streams into fixed seq_len lines.
    pack_hard splits documents across lines (carrying the displaced token);
    otherwise lines are padded. Optional per-line BOS/EOS injection with
    dedup against tokens already in place."""

    def __init__(self, dataset: _StatefulDataset, seq_len: int, pack_hard: bool,
                 bos_token=None, eos_token=None, pad_token=None):
        super().__init__(dataset)
        self.len = seq_len
        self.buffer: List[Any] = []
        self.bos = bos_token
        self.eos = eos_token
        self.pad = pad_token
        self.pack_hard = pack_hard
        if not pack_hard:
            assert pad_token is not None, "pad mode requires a pad_token"
        self.state_params = ["buffer"]

    def _next_line(self, stream):
        buffer = self.buffer
        new = []
        while len(buffer) + len(new) < self.len:
            buffer = buffer + new
            new = next(stream)
        if self.bos is not None and (not buffer or buffer[0] != self.bos):
            buffer = [self.bos] + buffer
        if len(buffer) >= self.len:
            out, buffer = buffer[:self.len], buffer[self.len:]
            if self.eos is not None and out[-1] != self.eos:
                buffer = [out[-1]] + buffer
                out[-1] = self.eos
            buffer = buffer + new
        elif self.pack_hard:
            buffer = buffer + new
            out, buffer = buffer[:self.len], buffer[self.len:]
            if self.eos is not None and out[-1] != self.eos:
                buffer = [out[-1]] + buffer
                out[-1] = self.eos
        else:
            if self.eos is not None and buffer[-1] != self.eos:
                buffer = buffer + [self.eos]
            out = buffer + [self.pad] * (self.len - len(buffer)) \
                if self.pad is not None else buffer
            buffer = new
        self.buffer = buffer
        return out

    def __iter__(self):
        stream = iter(self.dataset)
        while True:
            yield self._next_line(stream)


# ------------------------- shuffling -------------------------

class PreloadBufferDataset(_WrapperDataset):
    """Swap-sample shuffle buffer: grows to window_size, then each step
    emits a uniformly-sampled slot and refills it with the next input.
    Expected distance between consecutive emissions ~= window_size.
    The buffer is a reshard_param: rescaling re-grows/shrinks it."""

    def __init__(self, dataset: _StatefulDataset, window_size: int):
        super().__init__(dataset)
        assert window_size > 1
        self.window_size = window_size
        self.g_state = None
        self.generator = torch.Generator().manual_seed(self.rank)
        self.buffer: List[Any] = []
        self.buffer_size = 0
        self.state_params = ["g_state"]
        self.reshard_params = ["buffer"]

    def __iter__(self):
        stream = iter(self.dataset)
        while True:
            if self.buffer_size < self.window_size:
                # grow phase: backfill empty slots
                if len(self.buffer) < self.window_size:
                    self.buffer += [[]] * (self.window_size - len(self.buffer))
                self.buffer[self.buffer_size] = next(stream)
                self.buffer_size += 1
            i = torch.randint(self.buffer_size, (1,),
                              generator=self.generator).item()
            out = self.buffer[i]
            if self.buffer_size > self.window_size:
                # shrink phase (after downscale-grown buffer): pop last
                self.buffer[i] = self.buffer[self.buffer_size - 1]
                self.buffer_size -= 1
            else:
                self.buffer[i] = next(stream)
            yield out

    def state_dict(self):
        self.g_state = self.generator.get_state()
        self.buffer = self.buffer[:self.buffer_size]
        return super().state_dict()

    def load_state_dict(self, state_dicts, sharded_input=False):
        sharded = super().load_state_dict(state_dicts, sharded_input)
        if self.g_state is not None:
            self.generator.set_state(self.g_state)
        self.buffer_size = len(self.buffer)
        return sharded


# ------------------------- mapping + auto-checkpoint -------------------------

class PreprocessDataset(_WrapperDataset):
    """Applies fn to every emitted item."""

    def __init__(self, dataset: _StatefulDataset, aug_fn: Callable):
        super().__init__(dataset)
        self.aug_fn = aug_fn

    def __iter__(self):
        stream = iter(self.dataset)
        while True:
            yield self.aug_fn(next(stream))


class CheckpointDataset(_WrapperDataset):
    """Auto-saves the recursive loader state every `interval` batches from
    inside each worker process (no inter-process communication needed).
    On setup, a checkpoint in the save dir (job resume) wins over the load
    dir (continued training, step reset to 0)."""

    def __init__(self, dataset: _StatefulDataset, load_path: str, interval: int,
                 steps_per_batch: int = 1, save_path: str = ""):
        super().__init__(dataset)
        self.interval = interval
        self.spb = steps_per_batch
        self.load_path = os.path.join(load_path, "checkpoints")
        self.path = os.path.join(save_path, "checkpoints") if save_path \
            else self.load_path
        self.step = 0
        self.ministep = 0

    def setup(self):
        if self.is_setup:
            return
        super().setup()
        self._restore()

    def __iter__(self):
        self.setup()
        stream = iter(self.dataset)
        while True:
            yield next(stream)
            self.ministep += 1
            if self.ministep == self.spb:
                self.ministep = 0
                self.step += 1
                if self.step % self.interval == 0:
                    self.save_to_path(
                        os.path.join(self.path, f"step_{self.step}_ckp"))

    def _report(self, msg):
        if self.rank == 0:
            print(msg)

    def _find_ckpt(self, root, verbose=False):
        """Newest valid loader checkpoint under root, else ''. Also sets
        self.step from the folder name."""
        if not os.path.exists(root) or not os.listdir(root):
            if verbose:
                self._report(f"  Dataset: no checkpoint at {root}; "
                             "starting from scratch.")
            return ""
        subdirs = [d for d in os.listdir(root) if "_ckp" in d]
        best, best_step = "", -1
        for d in subdirs:
            try:
                step = int(d.split("_")[-2])
            except (ValueError, IndexError):
                continue
            full = os.path.join(root, d)
            if os.path.isdir(full) and step > best_step and \
                    any("loader" in f for f in os.listdir(full)):
                best, best_step = full, step
        if best:
            self.step = best_step
        return best

    def save_to_path(self, path: str):
        self._report(f"Saving dataset to {path}")
        t0 = time.time()
        super().save_to_path(path)
        self._report(f"Dataset saved to {path} in {time.time() - t0:.2f}s")

    def _restore(self):
        target = self._find_ckpt(self.path)
        if target:
            self._report(f"  Dataset: resuming from save-dir checkpoint {target}")
        else:
            target = self._find_ckpt(self.load_path, verbose=True)
            if not target:
                return
            self.step = 0  # continued pretraining: new run counts from 0
        t0 = time.time()
        self.dataset.load_from_path(target)
        self._report(f"Dataset checkpoint loaded in {time.time() - t0:.2f}s")
