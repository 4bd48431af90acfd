"""Dynamic request batcher: coalesce concurrent synthesis requests into
padded GPU batches.

The reference serves one ort session.run per sentence (rayon threads,
synth/src/lib.rs:316-320).  On an MI355X the GPU wants batched work:
this scheduler queues incoming sentences and a worker drains up to
`max_batch` of them (or whatever arrived within `max_wait_ms`) into ONE
`speak_batch` call — the serving-side counterpart of SURVEY.md §7 step 5
("scheduler balancing 512 concurrent utterances").  Per-utterance
seeding keeps every result independent of batch composition, so
batching is invisible to callers.
"""

from __future__ import annotations

import queue
import threading
from concurrent.futures import Future
from typing import List, Optional, Tuple

from ..core import Audio, SonataModel


class DynamicBatcher:
    def __init__(self, model: SonataModel, max_batch: int = 64,
                 max_wait_ms: float = 3.0):
        self.model = model
        self.max_batch = max_batch
        self.max_wait = max_wait_ms / 1000.0
        self._q: "queue.Queue[Optional[Tuple[str, Future]]]" = queue.Queue()
        self._worker = threading.Thread(target=self._run, daemon=True,
                                        name="sonata_batcher")
        self._closed = False
        self._worker.start()

    def submit(self, phonemes: str) -> "Future[Audio]":
        """Queue one sentence; the future resolves with its Audio."""
        if self._closed:
            raise RuntimeError("batcher closed")
        f: "Future[Audio]" = Future()
        self._q.put((phonemes, f))
        return f

    def synthesize(self, phonemes: str) -> Audio:
        return self.submit(phonemes).result()

    def close(self) -> None:
        self._closed = True
        self._q.put(None)
        self._worker.join(timeout=10)

    # ------------------------------------------------------------------ #
    def _run(self) -> None:
        while True:
            item = self._q.get()
            if item is None:
                return
            batch: List[Tuple[str, Future]] = [item]
            # drain whatever arrives within the wait window
            deadline = None
            while len(batch) < self.max_batch:
                try:
                    timeout = self.max_wait if deadline is None else deadline
                    nxt = self._q.get(timeout=timeout)
                except queue.Empty:
                    break
                if nxt is None:
                    self._flush(batch)
                    return
                batch.append(nxt)
                deadline = 0.0  # after the first wait, drain non-blocking
            self._flush(batch)

    def _flush(self, batch: List[Tuple[str, Future]]) -> None:
        # Length-bucketed sub-batches: padded batching costs compute
        # proportional to the LONGEST utterance, so a 3-word request
        # coalesced with a 40-word one would pay 10x.  Sort by length
        # and cut where the next item is >2x the bucket's minimum.
        batch = sorted(batch, key=lambda it: len(it[0]))
        start = 0
        for i in range(1, len(batch) + 1):
            if i == len(batch) or (
                    len(batch[i][0]) > 2 * max(len(batch[start][0]), 8)):
                self._run_bucket(batch[start:i])
                start = i

    def _run_bucket(self, bucket: List[Tuple[str, Future]]) -> None:
        phonemes = [p for p, _ in bucket]
        try:
            audios = self.model.speak_batch(phonemes)
            for (_, f), audio in zip(bucket, audios):
                f.set_result(audio)
        except BaseException as e:  # noqa: BLE001 - propagate to callers
            for _, f in bucket:
                if not f.done():
                    f.set_exception(e)
