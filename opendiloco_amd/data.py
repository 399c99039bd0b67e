"""Synthetic data pipeline.

Re-implements the reference's fake-data path with explicit, resumable RNG:

  - ``FakeTokenizedDataset`` semantics: random token ids in [3, vocab),
    attention mask all ones (reference: open_diloco/utils.py:155-167).
  - LM collation: labels = input_ids (DataCollatorForLanguageModeling with
    mlm=False, reference: open_diloco/train_fsdp.py:161 and
    train_diloco_torch.py:227).
  - Stateful iteration: the reference wraps its loader in torchdata's
    StatefulDataLoader so that checkpoint resume replays the stream from the
    same point (open_diloco/ckpt_utils.py:83-87,141-144).  Here the stream
    state is just (seed, n_batches_yielded): the generator is re-seeded and
    fast-forwarded on resume, which reproduces the stream exactly.
"""

from __future__ import annotations

import torch


class FakeTokenizedDataLoader:
    """Infinite loader of seeded fake LM batches.

    Yields dicts with int64 tensors ``input_ids`` [B, S], ``attention_mask``
    [B, S] (all ones), ``labels`` [B, S] (= input_ids), matching what the
    reference's collated FakeTokenizedDataset produces.
    """

    def __init__(self, seq_len: int, vocab_size: int, batch_size: int, seed: int, rank: int = 0):
        assert vocab_size > 3, "Vocab size must be greater than 3"
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.batch_size = batch_size
        self.seed = seed
        self.rank = rank
        self._gen = torch.Generator()
        self._gen.manual_seed(seed + 1337 * rank)
        self._n_yielded = 0

    def _next_batch(self) -> dict[str, torch.Tensor]:
        ids = torch.randint(
            3, self.vocab_size, (self.batch_size, self.seq_len), generator=self._gen, dtype=torch.int64
        )
        return {
            "input_ids": ids,
            "attention_mask": torch.ones_like(ids),
            "labels": ids.clone(),
        }

    def __iter__(self):
        while True:
            batch = self._next_batch()
            self._n_yielded += 1
            yield batch

    # -- StatefulDataLoader-compatible state (ckpt_utils.py:83-87 analogue) --
    def state_dict(self) -> dict:
        return {"seed": self.seed, "rank": self.rank, "n_yielded": self._n_yielded}

    def load_state_dict(self, state: dict) -> None:
        self.seed = state["seed"]
        self.rank = state.get("rank", self.rank)
        self._gen.manual_seed(self.seed + 1337 * self.rank)
        # fast-forward: regenerate and discard to reach the same stream point
        self._n_yielded = 0
        for _ in range(state["n_yielded"]):
            self._next_batch()
            self._n_yielded += 1
