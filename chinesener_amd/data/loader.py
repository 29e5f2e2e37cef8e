"""L2 input pipeline: cached npz -> torch batches; multi-task interleave.

Parity with the reference's dataset.py: NerDataset (:11-68) including the
``params`` property that propagates data_params + derived
step_per_epoch/num_train_steps into TRAIN_PARAMS (:57-68, consumed at
main.py:24-25), and MultiDataset (:71-127) whose round-robin sample
interleave makes every batch a ~50/50 task mix (:89-100).

Data parallel: ``rank``/``world_size`` shard samples (the reference is
single-process; sharding is new per SURVEY.md §2.7).
"""
from __future__ import annotations

import os
import pickle
from typing import Dict, Iterator, List, Optional

import numpy as np
import torch

from .datasets import get_spec
from .preprocess import build_cache, cache_name, extract_prefix_surfix

TENSOR_KEYS_INT = ("token_ids", "label_ids", "mask", "softword_ids",
                   "softlexicon_ids", "bichar_ids", "task_ids")
TENSOR_KEYS_FLOAT = ("ex_softword_ids", "softlexicon_weights")


class DevicePrefetcher:
    """One-batch-ahead host->device staging (replaces the reference's
    tf.data prefetch, dataset.py:43,52). Pins each host batch and issues
    its H2D copies on a dedicated HIP stream while the previous batch
    computes; the compute stream waits on a recorded event before the
    batch is handed out. Pass-through on CPU."""

    def __init__(self, it, device):
        self.it = iter(it)
        self.device = torch.device(device)
        self.use_stream = self.device.type == "cuda"
        if self.use_stream:
            self.stream = torch.cuda.Stream(device=self.device)
        self._next = None
        self._event = None
        self._host = None          # keep pinned source alive until copied
        # 2-deep ring of PERSISTENT device buffers per batch signature:
        # the training loop must stay allocation-free between hipGraph
        # replays (fresh per-batch device tensors are foreign allocator
        # activity — see Trainer._stage), and reuse also spares the
        # caching-allocator cross-stream bookkeeping
        self._ring: list = [None, None]
        self._ring_sig: list = [None, None]
        self._slot = 0
        self._preload()

    def _ring_buffers(self, host):
        sig = tuple(sorted((k, tuple(v.shape), str(v.dtype))
                           for k, v in host.items()
                           if isinstance(v, torch.Tensor)))
        s = self._slot
        self._slot = 1 - s
        if self._ring_sig[s] != sig:
            self._ring[s] = {k: torch.empty_like(v, device=self.device)
                             for k, v in host.items()
                             if isinstance(v, torch.Tensor)}
            self._ring_sig[s] = sig
        return self._ring[s]

    def _preload(self):
        try:
            host = next(self.it)
        except StopIteration:
            self._next = None
            return
        if not self.use_stream:
            self._next = host
            return
        try:
            sig = tuple(sorted((k, tuple(v.shape), str(v.dtype))
                               for k, v in host.items()
                               if isinstance(v, torch.Tensor)))
            if sig == getattr(self, "_last_host_sig", None):
                self._stable_run = getattr(self, "_stable_run", 0) + 1
            else:
                self._stable_run = 0
            self._last_host_sig = sig
            # require a SUSTAINED stable run: ragged streams repeat a
            # shape by chance, and a single chance repeat must not pay
            # a fresh cudaHostAlloc (10-1000 ms)
            stable = self._stable_run >= 8
            if not stable:
                # ragged stream (MRC padded batching): hand the host
                # batch through untouched — fresh pinned allocations
                # (cudaHostAlloc misses the block cache every time) and
                # side-stream pageable copies both cost 10-100x the
                # model step; the consumer's _stage does the H2D
                self._next = host
                self._event = None
                self._host = host
                return
            host = {k: (v.pin_memory()
                        if isinstance(v, torch.Tensor)
                        and v.device.type == "cpu" else v)
                    for k, v in host.items()}
            bufs = self._ring_buffers(host)
            # _preload(t+2) is issued after the consumer ENQUEUED all
            # reads of batch t (the for-loop calls __next__ after
            # train_step returns), so an event on the compute stream here
            # orders the slot overwrite after those reads
            done = torch.cuda.Event()
            done.record(torch.cuda.current_stream(self.device))
            self.stream.wait_event(done)
            with torch.cuda.stream(self.stream):
                for k, v in host.items():
                    if isinstance(v, torch.Tensor):
                        bufs[k].copy_(v, non_blocking=True)
                self._next = dict(bufs)
                self._event = torch.cuda.Event()
                self._event.record(self.stream)
            self._host = host
        except Exception:
            # degrade to synchronous hand-off (consumer moves to device)
            self.use_stream = False
            self._next = host

    def __iter__(self):
        return self

    def __next__(self):
        if self._next is None:
            raise StopIteration
        batch = self._next
        if self.use_stream and self._event is not None:
            # ring buffers are persistent (no allocator hand-off), so a
            # stream wait on the copy event is all that's needed
            torch.cuda.current_stream(self.device).wait_event(self._event)
        self._preload()
        return batch


def _to_tensors(arrays: Dict[str, np.ndarray], idx: np.ndarray) -> Dict[str, torch.Tensor]:
    out = {}
    for k, v in arrays.items():
        if k == "raw":
            continue
        sel = v[idx]
        if k in TENSOR_KEYS_FLOAT:
            out[k] = torch.from_numpy(np.ascontiguousarray(sel, dtype=np.float32))
        elif k == "seq_len":
            out[k] = torch.from_numpy(np.ascontiguousarray(sel, dtype=np.int64))
        else:
            out[k] = torch.from_numpy(np.ascontiguousarray(sel, dtype=np.int64))
    return out


class NerDataset:
    """One corpus for one model; builds/loads the npz cache lazily."""

    def __init__(self, data_dir: str, dataset_name: str, batch_size: int,
                 epochs: int, model_name: str, rank: int = 0, world_size: int = 1):
        self.data_dir = os.path.join(data_dir, dataset_name)
        self.dataset_name = dataset_name
        self.batch_size = batch_size
        self.epochs = epochs
        self.model_name = model_name
        self.rank, self.world_size = rank, world_size
        self.word_enhance, self.tokenizer_type = extract_prefix_surfix(model_name)
        self.spec = get_spec(dataset_name)
        self._params: Optional[Dict] = None
        self._splits: Dict[str, Dict[str, np.ndarray]] = {}

    # ------------------------------------------------------------ params
    @property
    def params(self) -> Dict:
        if self._params is None:
            pkl = os.path.join(self.data_dir, f"{cache_name(self.tokenizer_type, 'data', self.word_enhance)}_params.pkl")
            if not os.path.exists(pkl):
                build_cache(self.dataset_name, self.data_dir, self.model_name)
            with open(pkl, "rb") as f:
                p = pickle.load(f)
            p = dict(p)
            shard = max(1, p["n_sample"] // self.world_size)
            p["step_per_epoch"] = max(1, shard // self.batch_size)
            p["num_train_steps"] = p["step_per_epoch"] * self.epochs
            self._params = p
        return self._params

    # ------------------------------------------------------------ arrays
    def arrays(self, split: str) -> Dict[str, np.ndarray]:
        if split not in self._splits:
            _ = self.params  # ensure cache exists
            path = os.path.join(
                self.data_dir, cache_name(self.tokenizer_type, split, self.word_enhance) + ".npz")
            if not os.path.exists(path):
                build_cache(self.dataset_name, self.data_dir, self.model_name,
                            splits=(split,))
            z = np.load(path, allow_pickle=True)
            self._splits[split] = {k: z[k] for k in z.files}
        return self._splits[split]

    def raw_sentences(self, split: str) -> List[str]:
        return list(self.arrays(split)["raw"])

    # ----------------------------------------------------------- batches
    def iter_batches(self, split: str = "train", shuffle: bool = True,
                     epochs: Optional[int] = None, seed: int = 1234,
                     drop_last: Optional[bool] = None) -> Iterator[Dict[str, torch.Tensor]]:
        arrays = self.arrays(split)
        n = arrays["token_ids"].shape[0]
        epochs = epochs if epochs is not None else (self.epochs if split == "train" else 1)
        drop_last = drop_last if drop_last is not None else (split == "train")
        for ep in range(epochs):
            order = np.arange(n)
            if shuffle:
                rng = np.random.default_rng(seed + ep)
                rng.shuffle(order)
            order = order[self.rank::self.world_size]   # DP shard
            stop = len(order) - (len(order) % self.batch_size) if drop_last else len(order)
            for i in range(0, stop, self.batch_size):
                yield _to_tensors(arrays, order[i:i + self.batch_size])


class MultiDataset:
    """Two NerDatasets interleaved sample-wise with task_ids (reference
    dataset.py:71-127): strict alternation -> each batch ~50/50 task mix;
    step_per_epoch = max over tasks."""

    def __init__(self, data_dir: str, dataset_names: List[str], batch_size: int,
                 epochs: int, model_name: str, rank: int = 0, world_size: int = 1):
        assert len(dataset_names) == 2, "multi-task supports exactly 2 datasets"
        self.task_list = list(dataset_names)
        self.batch_size = batch_size
        self.epochs = epochs
        self.pipes = [NerDataset(data_dir, d, batch_size, epochs, model_name,
                                 rank, world_size) for d in dataset_names]

    @property
    def params(self) -> Dict:
        p0, p1 = self.pipes[0].params, self.pipes[1].params
        params = dict(p0)
        params["task_list"] = self.task_list
        params[self.task_list[0]] = p0
        params[self.task_list[1]] = p1
        params["label_size"] = None  # per-task label sizes live in sub-dicts
        params["step_per_epoch"] = max(p0["step_per_epoch"], p1["step_per_epoch"]) * 2
        params["num_train_steps"] = params["step_per_epoch"] * self.epochs
        params["max_seq_len"] = p0["max_seq_len"]
        return params

    def iter_batches(self, split: str = "train", shuffle: bool = True,
                     epochs: Optional[int] = None, seed: int = 1234
                     ) -> Iterator[Dict[str, torch.Tensor]]:
        """Alternate samples task0,task1,task0,... then batch (reference
        choose_from_datasets round robin, dataset.py:89-100)."""
        epochs = epochs if epochs is not None else (self.epochs if split == "train" else 1)

        def sample_stream(pipe, task_id):
            passes = 0
            while True:   # repeat like reference .repeat()
                # reshuffle each pass over the corpus (the single-task path
                # uses seed+ep; without this every repeat replays the same
                # order for the shorter task)
                for batch in pipe.iter_batches(split, shuffle, epochs=1,
                                               seed=seed + passes):
                    bsz = batch["token_ids"].shape[0]
                    L = batch["token_ids"].shape[1]
                    batch = dict(batch)
                    batch["task_ids"] = torch.full((bsz, L), task_id, dtype=torch.int64)
                    for b in range(bsz):
                        yield {k: v[b] for k, v in batch.items()}
                passes += 1

        streams = [sample_stream(p, i) for i, p in enumerate(self.pipes)]
        n_steps = max(p.params["step_per_epoch"] for p in self.pipes) * 2 * epochs
        # union of feature keys: fill task-specific missing keys with zeros
        for _ in range(n_steps):
            samples = []
            for b in range(self.batch_size):
                samples.append(next(streams[b % 2]))
            keys = sorted(set().union(*[s.keys() for s in samples]))
            batch = {}
            for k in keys:
                vals = []
                for s in samples:
                    if k in s:
                        vals.append(s[k])
                    else:
                        ref = next(x[k] for x in samples if k in x)
                        vals.append(torch.zeros_like(ref))
                batch[k] = torch.stack(vals)
            yield batch

    def build_predict_pipe(self, task: str) -> NerDataset:
        return self.pipes[self.task_list.index(task)]


def make_synthetic_batch(batch_size: int, seq_len: int, label_size: int = 10,
                         vocab_size: int = 21128, word_enhance: Optional[str] = None,
                         seed: int = 0, device: str = "cpu",
                         is_bert: bool = True) -> Dict[str, torch.Tensor]:
    """MSRA-shaped random batch for bench.py (synthetic data per BASELINE.json)."""
    g = torch.Generator().manual_seed(seed)
    L = seq_len
    lens = torch.randint(max(4, L // 2), L + 1, (batch_size,), generator=g)
    token_ids = torch.randint(106, vocab_size, (batch_size, L), generator=g)
    label_ids = torch.randint(1, label_size - 2, (batch_size, L), generator=g)
    mask = torch.arange(L).unsqueeze(0) < lens.unsqueeze(1)
    token_ids = token_ids * mask
    label_ids = label_ids * mask
    if is_bert:
        token_ids[:, 0] = 2           # CLS
        label_ids[:, 0] = label_size - 2
        token_ids[torch.arange(batch_size), lens - 1] = 3   # SEP
        label_ids[torch.arange(batch_size), lens - 1] = label_size - 1
    batch = {"token_ids": token_ids, "label_ids": label_ids,
             "mask": mask.long(), "seq_len": lens.long()}
    if word_enhance == "softlexicon":
        batch["softlexicon_ids"] = torch.randint(0, 5000, (batch_size, L, 40), generator=g)
        w = torch.rand(batch_size, L, 40, generator=g)
        batch["softlexicon_weights"] = w / w.sum(-1, keepdim=True)
    elif word_enhance == "softword":
        batch["softword_ids"] = torch.randint(0, 5, (batch_size, L), generator=g)
    elif word_enhance == "ex_softword":
        batch["ex_softword_ids"] = (torch.rand(batch_size, L, 5, generator=g) > 0.7).float()
    elif word_enhance == "bichar":
        batch["bichar_ids"] = torch.randint(0, 50000, (batch_size, L), generator=g)
    return {k: v.to(device) for k, v in batch.items()}
