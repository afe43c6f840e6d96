"""Memory-mapped token dataset + deterministic resumable loader.

The reference trains nothing, so its profile schema just has a
``batch_generator_time_ms`` slot (README.md:79); this is the component
that fills it with real data. Format: a flat binary file of token ids
(uint16 by default — GPT/Llama vocabs < 65536 — or uint32), no header;
``np.memmap`` keeps the working set in page cache, so the loader reads
only the microbatches it serves (288 GB HBM nodes routinely train from
multi-TB token files).

Determinism contract (matches the synthetic path's):
- every TP/PP rank of one DP replica sees the SAME microbatch sequence;
- different DP replicas see disjoint samples (strided sharding of a
  per-epoch shuffle);
- the sequence is a pure function of (seed, epoch, dp_rank), and the
  loader fast-forwards by sample count — checkpoint/resume replays
  exactly the stream an uninterrupted run would have drawn.
"""

from __future__ import annotations

import numpy as np
import torch


class TokenDataset:
    def __init__(self, path: str, seq_length: int, dtype: str = "uint16",
                 split=(0.0, 1.0)):
        """``split``: fraction range of the sample space this dataset
        covers — e.g. train (0, 0.98) / eval (0.98, 1.0) on one file."""
        self.tokens = np.memmap(path, dtype=np.dtype(dtype), mode="r")
        self.seq_length = seq_length
        # sample i = tokens[i*S : i*S + S + 1] (inputs + shifted labels)
        total = (len(self.tokens) - 1) // seq_length
        self._first = int(total * split[0])
        self.num_samples = int(total * split[1]) - self._first
        if self.num_samples <= 0:
            raise ValueError(f"{path}: too short for seq_length {seq_length}")

    def sample(self, idx: int) -> np.ndarray:
        s = self.seq_length
        idx += self._first
        return np.asarray(self.tokens[idx * s: idx * s + s + 1])


class TokenLoader:
    """Yields (tokens, labels) microbatches of shape [mbs, seq_length]."""

    def __init__(self, dataset: TokenDataset, mbs: int, dp: int, dp_rank: int,
                 seed: int = 1234, device=None):
        self.ds = dataset
        self.mbs = mbs
        self.dp = dp
        self.dp_rank = dp_rank
        self.seed = seed
        self.device = device or torch.device("cpu")
        self.per_rank = dataset.num_samples // dp
        if self.per_rank < mbs:
            raise ValueError("dataset too small for mbs x dp")
        self._consumed = 0          # microbatches served (resume cursor)
        self._epoch = -1
        self._order: np.ndarray = np.empty(0, dtype=np.int64)

    @property
    def microbatches_per_epoch(self) -> int:
        return self.per_rank // self.mbs

    def _ensure_epoch(self, epoch: int) -> None:
        if epoch == self._epoch:
            return
        rng = np.random.RandomState(self.seed + epoch)
        perm = rng.permutation(self.ds.num_samples)
        self._order = perm[self.dp_rank::self.dp]   # disjoint per replica
        self._epoch = epoch

    def state(self) -> int:
        return self._consumed

    def load_state(self, consumed: int) -> None:
        self._consumed = consumed

    def next_batch(self):
        epoch = self._consumed // self.microbatches_per_epoch
        within = self._consumed % self.microbatches_per_epoch
        self._ensure_epoch(epoch)
        idxs = self._order[within * self.mbs:(within + 1) * self.mbs]
        rows = np.stack([self.ds.sample(int(i)) for i in idxs])
        t = torch.from_numpy(rows.astype(np.int64)).to(self.device)
        self._consumed += 1
        return t[:, :-1].contiguous(), t[:, 1:].contiguous()
