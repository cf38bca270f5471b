"""Request batching for the serving engines.

The reference served each request's LLM call as its own HTTP round trip to
a provider that batched internally (reference src/core/llm/providers/
openai.py:117).  On-device, concurrent `/chat` requests must be batched by
US: decode is weight-bandwidth-bound, so 64 rows of decode cost barely
more than 1 (the 16 GB of weights stream once per token either way).

Three coalescing layers live here:

* `ContinuousBatcher` (default generation frontend) — TRUE continuous
  batching: requests join a persistent slot-session decode loop at step
  boundaries (admission prefill on a side stream, per-row seq_lens and
  per-row-temperature sampling), stream per step, and free their slot the
  moment they finish — no wave barrier.
* `DynamicBatcher` (CONTINUOUS_BATCHING=0 fallback, and the mock-engine
  path) — wave batching: group by sampling params, keep collecting while
  the engine runs the previous batch, run ONE batched generate.
* `MicroBatcher` + `BatchedEncoder`/`BatchedReranker` — concurrent
  single-query embeds / rerank pair scores coalesce into one engine
  forward each.
"""

from __future__ import annotations

import queue
import threading
import time
from concurrent.futures import Future
from dataclasses import dataclass, field
from typing import Any


@dataclass
class _Item:
    prompt: str
    max_new_tokens: int
    temperature: float
    stop_on_eos: bool
    future: Future = field(default_factory=Future)
    # streaming requests get a per-request token queue instead of blocking
    # on the future: the worker pushes this row's token id after every
    # decode step and a None sentinel at the end, so a slow SSE consumer
    # never holds the engine (VERDICT r1 item 5 — the old engine.stream
    # held the generation lock for the stream's whole life)
    stream_q: "queue.Queue[int | None] | None" = None

    @property
    def group_key(self) -> tuple:
        # EXACT temperature: sampling is per-batch in the engine, so a
        # bucketed key would silently run a request at batch[0]'s
        # temperature (up to ~0.05 off, arrival-order dependent)
        return (self.temperature, self.max_new_tokens, self.stop_on_eos)


class DynamicBatcher:
    def __init__(self, generator, max_batch: int = 32,
                 max_wait_ms: float = 8.0):
        self.generator = generator
        self.max_batch = max_batch
        self.max_wait_s = max_wait_ms / 1e3
        self._q: queue.Queue[_Item] = queue.Queue()
        self._stop = threading.Event()
        self._thread: threading.Thread | None = None
        self._start_lock = threading.Lock()
        self.stats = {"requests": 0, "batches": 0, "max_batch_seen": 0}

    # ----- caller side -----
    def generate(self, prompt: str, max_new_tokens: int = 128,
                 temperature: float = 0.3, stop_on_eos: bool = True,
                 timeout_s: float = 300.0) -> str:
        self.start()
        item = _Item(prompt, max_new_tokens, float(temperature), stop_on_eos)
        self._q.put(item)
        return item.future.result(timeout=timeout_s)

    def generate_stream(self, prompt: str, max_new_tokens: int = 128,
                        temperature: float = 0.3, timeout_s: float = 300.0):
        """Yield text deltas token-by-token.  The request JOINS batched
        decode (continuous-batching-lite: it occupies a batch slot and its
        token ids stream out per decode step); concurrent non-stream chat
        requests coalesce into the same engine batches, so neither starves
        the other."""
        self.start()
        item = _Item(prompt, max_new_tokens, float(temperature), True,
                     stream_q=queue.Queue())
        self._q.put(item)
        from sentio_amd.engines.tokenizer import EOS_ID

        tok = self.generator.tokenizer
        generated: list[int] = []
        emitted = ""
        deadline = time.monotonic() + timeout_s
        while True:
            try:
                t = item.stream_q.get(timeout=max(0.1,
                                                  deadline - time.monotonic()))
            except queue.Empty:
                raise TimeoutError(
                    f"stream starved for {timeout_s:.0f}s (engine stalled "
                    "or request never scheduled)") from None
            if t is None or t == EOS_ID:
                break
            generated.append(t)
            text = tok.decode(generated)
            if len(text) > len(emitted):
                yield text[len(emitted):]
                emitted = text

    # ----- worker side -----
    def start(self) -> None:
        # locked check-then-act: concurrent FIRST requests must not each
        # spawn a worker — multiple workers would fragment batching
        with self._start_lock:
            if self._thread is not None and self._thread.is_alive():
                return
            self._stop.clear()
            self._thread = threading.Thread(target=self._loop, daemon=True,
                                            name="sentio-batcher")
            self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None
        # fail anything still queued so callers unblock immediately
        # instead of waiting out their own timeout
        while True:
            try:
                item = self._q.get_nowait()
            except queue.Empty:
                break
            if not item.future.done():
                item.future.set_exception(RuntimeError("batcher stopped"))

    def _engine_busy(self) -> bool:
        lock = getattr(self.generator, "_gen_lock", None)
        return lock is not None and lock.locked()

    def _loop(self) -> None:
        while not self._stop.is_set():
            try:
                first = self._q.get(timeout=0.1)
            except queue.Empty:
                continue
            batch = [first]
            deadline = time.monotonic() + self.max_wait_s
            # while the engine is still decoding the PREVIOUS batch, keep
            # collecting past the wait window — that wait costs nothing
            # (batches run serially through the engine lock), and it is
            # what turns staggered closed-loop arrivals into full batches
            # (measured: avg batch 4.2/32 with a fixed 8 ms window)
            hard_stop = time.monotonic() + 30.0
            leftovers: list[_Item] = []
            while len(batch) < self.max_batch:
                now = time.monotonic()
                busy = self._engine_busy() and now < hard_stop
                if not busy and now >= deadline:
                    break
                try:
                    nxt = self._q.get(
                        timeout=(0.005 if busy
                                 else max(deadline - now, 0.001)))
                except queue.Empty:
                    if busy:
                        continue
                    break
                if nxt.group_key == first.group_key:
                    batch.append(nxt)
                else:
                    leftovers.append(nxt)   # different params: next batch
            for item in leftovers:
                self._q.put(item)
            self._run_batch(batch)

    def _run_batch(self, batch: list[_Item]) -> None:
        self.stats["requests"] += len(batch)
        self.stats["batches"] += 1
        self.stats["max_batch_seen"] = max(self.stats["max_batch_seen"],
                                           len(batch))
        streams = [it for it in batch if it.stream_q is not None]
        on_token = None
        if streams:
            def on_token(_step: int, toks: list[int]) -> None:
                for i, it in enumerate(batch):
                    if it.stream_q is not None:
                        it.stream_q.put(toks[i])
        try:
            outs = self.generator.generate(
                [it.prompt for it in batch],
                max_new_tokens=batch[0].max_new_tokens,
                temperature=batch[0].temperature,
                stop_on_eos=batch[0].stop_on_eos,
                on_token=on_token,
            )
            for it, out in zip(batch, outs):
                if it.stream_q is not None:
                    it.stream_q.put(None)
                it.future.set_result(out)
        except Exception as exc:
            for it in batch:
                if it.stream_q is not None:
                    it.stream_q.put(None)
                if not it.future.done():
                    it.future.set_exception(exc)

    def health(self) -> dict[str, Any]:
        return {"queued": self._q.qsize(),
                "running": self._thread is not None and self._thread.is_alive(),
                **self.stats}


class ContinuousBatcher:
    """TRUE continuous batching: requests JOIN the decode loop at step
    boundaries (admission prefill into free KV-cache slots), stream their
    tokens out per step, and leave when done.  Unlike the wave-batched
    DynamicBatcher, a request's latency is its own prefill + its own
    tokens — it never waits for co-tenants' remaining tokens, and
    max_new_tokens / temperature / stop_on_eos may differ per request
    (per-row sampling).  This is what the decode engine's per-row
    seq_lens design exists for.

    Slot safety: free rows keep riding the (graph-replayed) decode step
    with garbage inputs — their seq_lens are re-zeroed every step so the
    KV write position can never run off the cache end, and their outputs
    are never read."""

    def __init__(self, generator, slots: int = 32, admit_max: int = 8):
        self.generator = generator
        self.n_slots = slots
        self.admit_max = admit_max
        self._q: queue.Queue[_Item] = queue.Queue()
        self._stop = threading.Event()
        self._thread: threading.Thread | None = None
        self._start_lock = threading.Lock()
        self.stats = {"requests": 0, "completed": 0, "steps": 0,
                      "max_concurrent": 0, "admissions": 0}

    # ----- caller side -----
    def generate(self, prompt: str, max_new_tokens: int = 128,
                 temperature: float = 0.3, stop_on_eos: bool = True,
                 timeout_s: float = 300.0) -> str:
        self.start()
        item = _Item(prompt, max_new_tokens, float(temperature), stop_on_eos)
        self._q.put(item)
        return item.future.result(timeout=timeout_s)

    def generate_stream(self, prompt: str, max_new_tokens: int = 128,
                        temperature: float = 0.3, timeout_s: float = 300.0):
        self.start()
        item = _Item(prompt, max_new_tokens, float(temperature), True,
                     stream_q=queue.Queue())
        self._q.put(item)
        from sentio_amd.engines.tokenizer import EOS_ID

        tok = self.generator.tokenizer
        generated: list[int] = []
        emitted = ""
        deadline = time.monotonic() + timeout_s
        while True:
            try:
                t = item.stream_q.get(timeout=max(0.1,
                                                  deadline - time.monotonic()))
            except queue.Empty:
                raise TimeoutError("stream starved (engine stalled)") from None
            if t is None or t == EOS_ID:
                break
            generated.append(t)
            text = tok.decode(generated)
            if len(text) > len(emitted):
                yield text[len(emitted):]
                emitted = text

    # ----- worker side -----
    def start(self) -> None:
        with self._start_lock:
            if self._thread is not None and self._thread.is_alive():
                return
            self._stop.clear()
            # admissions prefill on their own thread (+ side HIP stream):
            # bounded to 2 in flight so temp KV caches stay small
            self._ready_q = queue.Queue(maxsize=2)
            self._adm_thread = threading.Thread(
                target=self._admitter, daemon=True, name="sentio-admit")
            self._adm_thread.start()
            self._thread = threading.Thread(target=self._loop, daemon=True,
                                            name="sentio-continuous")
            self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        for t in (self._thread, getattr(self, "_adm_thread", None)):
            if t is not None:
                t.join(timeout=5.0)
        self._thread = None
        self._adm_thread = None
        for q_ in (self._q, getattr(self, "_ready_q", None)):
            if q_ is None:
                continue
            while True:
                try:
                    got = q_.get_nowait()
                except queue.Empty:
                    break
                items = got[0] if isinstance(got, tuple) else [got]
                for item in items:
                    if item.stream_q is not None:
                        item.stream_q.put(None)
                    if not item.future.done():
                        item.future.set_exception(
                            RuntimeError("batcher stopped"))

    def _admitter(self) -> None:
        """Collect queued requests and prefill them OFF the decode loop:
        on GPU the prefill runs on a dedicated side stream, concurrent
        with the loop's decode-graph replays; the loop only pays the
        KV splice (integrate_admission) between steps."""
        import torch

        gen = self.generator
        side = (torch.cuda.Stream()
                if gen.device != "cpu" and torch.cuda.is_available() else None)
        while not self._stop.is_set():
            want: list[_Item] = []
            try:
                want.append(self._q.get(timeout=0.05))
            except queue.Empty:
                continue
            while len(want) < self.admit_max:
                try:
                    want.append(self._q.get_nowait())
                except queue.Empty:
                    break
            try:
                from sentio_amd.observability.kernel_timer import get_timer

                if side is not None:
                    with torch.cuda.stream(side), \
                            get_timer("admission_prefill").measure():
                        pre = gen.prefill_admission(
                            [it.prompt for it in want],
                            [it.max_new_tokens for it in want])
                    ev = torch.cuda.Event()
                    ev.record(side)
                else:
                    pre = gen.prefill_admission(
                        [it.prompt for it in want],
                        [it.max_new_tokens for it in want])
                    ev = None
                placed = False
                while not self._stop.is_set():
                    try:
                        self._ready_q.put((want, pre, ev), timeout=0.2)
                        placed = True
                        break
                    except queue.Full:
                        continue
                if not placed:     # shut down while the ready queue was
                    raise RuntimeError("batcher stopped")   # full: fail, not drop
            except Exception as exc:
                for it in want:
                    if it.stream_q is not None:
                        it.stream_q.put(None)
                    if not it.future.done():
                        it.future.set_exception(exc)

    def _loop(self) -> None:
        try:
            self._loop_inner()
        except Exception as exc:   # a dead loop must not strand callers:
            for s in getattr(self, "_live_state", []):   # admitted slots
                if s is None:
                    continue
                it = s["item"]
                if it.stream_q is not None:
                    it.stream_q.put(None)
                if not it.future.done():
                    it.future.set_exception(exc)
            while True:            # still-queued requests
                try:
                    item = self._q.get_nowait()
                except queue.Empty:
                    break
                if item.stream_q is not None:
                    item.stream_q.put(None)
                if not item.future.done():
                    item.future.set_exception(exc)
            raise

    def _loop_inner(self) -> None:
        import torch

        from sentio_amd.engines.tokenizer import EOS_ID

        from sentio_amd.observability.kernel_timer import get_timer

        _step_timer = get_timer("decode_step")
        gen = self.generator
        dev = gen.device
        sess = gen.make_slot_session(self.n_slots)
        state: list[dict | None] = [None] * self.n_slots
        self._live_state = state            # _loop's crash handler fails these
        cur = torch.zeros(self.n_slots, dtype=torch.int64, device=dev)
        temps = torch.zeros(self.n_slots, device=dev)
        free_mask = torch.ones(self.n_slots, dtype=torch.bool, device=dev)

        def finish(r: int, reason: str) -> None:
            s = state[r]
            state[r] = None
            free_mask[r] = True
            self.stats["completed"] += 1
            it = s["item"]
            if it.stream_q is not None:
                it.stream_q.put(None)
            ids = s["ids"]
            if it.stop_on_eos and EOS_ID in ids:
                ids = ids[: ids.index(EOS_ID)]
            if not it.future.done():
                it.future.set_result(gen.tokenizer.decode(ids))

        def feed(r: int, tok: int) -> None:
            s = state[r]
            s["ids"].append(tok)
            it = s["item"]
            if it.stream_q is not None:
                it.stream_q.put(tok)
            s["remaining"] -= 1
            if (s["remaining"] <= 0
                    or (it.stop_on_eos and tok == EOS_ID)):
                finish(r, "done")

        pending: tuple | None = None      # an admission waiting for slots
        while not self._stop.is_set():
            n_active = sum(s is not None for s in state)
            # ---- integrate a ready admission (prefilled off-loop) ----
            if pending is None:
                try:
                    want_, pre_, ev_ = self._ready_q.get(
                        timeout=0.05 if n_active == 0 else 0)
                    pending = (want_, pre_, ev_, list(range(len(want_))))
                except queue.Empty:
                    pending = None
            if pending is not None:
                want, pre, ev, left = pending
                free_rows = [i for i, st_ in enumerate(state) if st_ is None]
                take = left[: len(free_rows)]   # partial: as many as fit
                if take:
                    rows = free_rows[: len(take)]
                    left = left[len(take):]
                    pending = (want, pre, ev, left) if left else None
                    try:
                        if ev is not None:   # admission stream -> loop stream
                            torch.cuda.current_stream().wait_event(ev)
                        logits = gen.integrate_admission(sess, rows, pre,
                                                         idx=take)
                        items = [want[j] for j in take]
                        lens_host = [pre["lens_host"][j] for j in take]
                        t_adm = torch.tensor([it.temperature for it in items],
                                             device=dev)
                        tok0 = gen.sample_rows(logits, t_adm).cpu().tolist()
                        for j, (r, it) in enumerate(zip(rows, items)):
                            # decode may never write past the slot cache:
                            # remaining steps clamp to Smax - prompt_len - 1
                            cap = max(1, sess.cache.max_seq
                                      - lens_host[j] - 1)
                            state[r] = {"item": it, "ids": [],
                                        "remaining": min(it.max_new_tokens,
                                                         cap)}
                            free_mask[r] = False
                            temps[r] = it.temperature
                            cur[r] = tok0[j]
                            feed(r, tok0[j])
                        self.stats["requests"] += len(items)
                        self.stats["admissions"] += 1
                        self.stats["max_concurrent"] = max(
                            self.stats["max_concurrent"],
                            sum(st_ is not None for st_ in state))
                    except Exception as exc:
                        pending = None
                        for it in want:
                            if it.stream_q is not None:
                                it.stream_q.put(None)
                            if not it.future.done():
                                it.future.set_exception(exc)
            if not any(st_ is not None for st_ in state):
                continue
            # ---- one decode step over every slot ----
            # free rows must never advance their KV write position off the
            # cache end: re-zero their seq_lens each step
            sess.cache.seq_lens.masked_fill_(free_mask, 0)
            with _step_timer.measure():   # lazy HIP events, no sync
                logits = gen.decode_step_session(sess, cur)
            toks = gen.sample_rows(logits, temps)
            cur.copy_(toks)
            self.stats["steps"] += 1
            toks_host = toks.cpu().tolist()
            for r in range(self.n_slots):
                if state[r] is not None:
                    feed(r, toks_host[r])

        # graceful stop mid-traffic: active slots and any un-integrated
        # admission must fail fast, not block callers until their timeout
        err = RuntimeError("batcher stopped")
        for r in range(self.n_slots):
            if state[r] is None:
                continue
            it = state[r]["item"]
            state[r] = None
            if it.stream_q is not None:
                it.stream_q.put(None)
            if not it.future.done():
                it.future.set_exception(err)
        if pending is not None:
            want, _pre, _ev, left = pending
            for j in left:
                it = want[j]
                if it.stream_q is not None:
                    it.stream_q.put(None)
                if not it.future.done():
                    it.future.set_exception(err)

    def health(self) -> dict[str, Any]:
        return {"queued": self._q.qsize(), "mode": "continuous",
                "slots": self.n_slots,
                "running": self._thread is not None and self._thread.is_alive(),
                **self.stats}


class ContinuousGenerator:
    """Generator frontend for continuous batching: single-prompt calls and
    streams join the slot loop; multi-prompt (bench-style) calls pass
    through to the raw engine (they interleave with the loop between
    steps via the engine lock)."""

    def __init__(self, raw, slots: int = 32):
        self.raw = raw
        self.batcher = ContinuousBatcher(raw, slots=slots)

    def generate(self, prompts: list[str], max_new_tokens: int = 128,
                 temperature: float = 0.3, stop_on_eos: bool = True,
                 **kwargs) -> list[str]:
        if len(prompts) != 1 or kwargs.get("on_token") is not None:
            return self.raw.generate(prompts, max_new_tokens=max_new_tokens,
                                     temperature=temperature,
                                     stop_on_eos=stop_on_eos, **kwargs)
        return [self.batcher.generate(prompts[0],
                                      max_new_tokens=max_new_tokens,
                                      temperature=temperature,
                                      stop_on_eos=stop_on_eos)]

    def stream(self, prompt: str, max_new_tokens: int = 128,
               temperature: float = 0.3):
        return self.batcher.generate_stream(prompt,
                                            max_new_tokens=max_new_tokens,
                                            temperature=temperature)

    def __getattr__(self, name):
        return getattr(self.raw, name)


@dataclass
class _MicroItem:
    units: list
    future: Future = field(default_factory=Future)


class MicroBatcher:
    """Coalesce concurrent small ENGINE calls (single-query embeds,
    per-request rerank scores) into one batched forward.  The GPU load test
    showed generation batching alone is not enough: 16 concurrent /chat
    requests each ran their own tiny encoder/reranker forward between
    generation batches, serializing ~0.8 s of retrieval per wave.  Each
    submit() contributes a list of units; the worker concatenates queued
    units, runs batch_fn(all_units) ONCE, and fans the per-item slices
    back out."""

    def __init__(self, batch_fn, max_units: int = 64, max_wait_ms: float = 3.0,
                 name: str = "micro"):
        self.batch_fn = batch_fn
        self.max_units = max_units
        self.max_wait_s = max_wait_ms / 1e3
        self.name = name
        self._q: queue.Queue[_MicroItem] = queue.Queue()
        self._stop = threading.Event()
        self._thread: threading.Thread | None = None
        self._start_lock = threading.Lock()
        self.stats = {"calls": 0, "batches": 0, "max_units_seen": 0}

    def submit(self, units: list, timeout_s: float = 120.0):
        """Blocking: returns batch_fn(flat)[i0:i1] for this item's units."""
        self._ensure_worker()
        item = _MicroItem(units)
        self._q.put(item)
        return item.future.result(timeout=timeout_s)

    def _ensure_worker(self) -> None:
        with self._start_lock:
            if self._thread is not None and self._thread.is_alive():
                return
            self._stop.clear()
            self._thread = threading.Thread(
                target=self._loop, daemon=True,
                name=f"sentio-micro-{self.name}")
            self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None
        while True:
            try:
                item = self._q.get_nowait()
            except queue.Empty:
                break
            if not item.future.done():
                item.future.set_exception(RuntimeError("micro-batcher stopped"))

    def _loop(self) -> None:
        while not self._stop.is_set():
            try:
                first = self._q.get(timeout=0.1)
            except queue.Empty:
                continue
            batch = [first]
            n_units = len(first.units)
            deadline = time.monotonic() + self.max_wait_s
            while n_units < self.max_units:
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    break
                try:
                    nxt = self._q.get(timeout=remaining)
                except queue.Empty:
                    break
                batch.append(nxt)
                n_units += len(nxt.units)
            self._run(batch, n_units)

    def _run(self, batch: list[_MicroItem], n_units: int) -> None:
        self.stats["calls"] += len(batch)
        self.stats["batches"] += 1
        self.stats["max_units_seen"] = max(self.stats["max_units_seen"],
                                           n_units)
        try:
            flat: list = []
            for it in batch:
                flat.extend(it.units)
            res = self.batch_fn(flat)
            off = 0
            for it in batch:
                it.future.set_result(res[off: off + len(it.units)])
                off += len(it.units)
        except Exception as exc:
            for it in batch:
                if not it.future.done():
                    it.future.set_exception(exc)

    def health(self) -> dict[str, Any]:
        return {"queued": self._q.qsize(), **self.stats}


class BatchedEncoder:
    """Encoder frontend: small embed calls (a request's single query)
    coalesce through a MicroBatcher; bulk calls (ingest, bench batches)
    pass straight through."""

    _BULK = 17   # > this many texts: caller already batches

    def __init__(self, raw, max_units: int = 64, max_wait_ms: float = 3.0):
        self.raw = raw
        self.micro = MicroBatcher(lambda texts: raw.embed(texts),
                                  max_units=max_units,
                                  max_wait_ms=max_wait_ms, name="encoder")

    def embed(self, texts, batch_size: int = 64):
        if not texts or len(texts) >= self._BULK:
            return self.raw.embed(texts, batch_size)
        return self.micro.submit(list(texts))

    def embed_one(self, text: str):
        return self.embed([text])[0]

    def __getattr__(self, name):
        return getattr(self.raw, name)


class BatchedReranker:
    """Reranker frontend: concurrent requests' pair scores coalesce into
    one cross-encoder forward (pair text carries its own query, so mixed
    queries batch fine — same packing the bench uses)."""

    _BULK = 33

    def __init__(self, raw, max_units: int = 64, max_wait_ms: float = 3.0):
        self.raw = raw
        self.micro = MicroBatcher(
            raw.score_packed,
            max_units=max_units, max_wait_ms=max_wait_ms, name="reranker")

    def score_pairs(self, query: str, texts: list[str],
                    batch_size: int = 32) -> list[float]:
        if not texts or len(texts) >= self._BULK:
            return self.raw.score_pairs(query, texts, batch_size)
        return list(self.micro.submit([f"{query}\n{t}" for t in texts]))

    def rerank(self, query: str, docs, top_k: int):
        if not docs:
            return []
        top_n = min(len(docs), 2 * top_k)
        cand = docs[:top_n]
        scores = self.score_pairs(query, [d.text for d in cand])
        order = sorted(range(len(cand)), key=lambda i: scores[i], reverse=True)
        out = []
        for i in order[:top_k]:
            d = cand[i]
            d.metadata["rerank_score"] = float(scores[i])
            d.metadata["score"] = float(scores[i])
            out.append(d)
        return out

    def __getattr__(self, name):
        return getattr(self.raw, name)


class BatchedGenerator:
    """Drop-in generator frontend: single-prompt calls route through the
    shared DynamicBatcher; already-batched calls pass straight through."""

    def __init__(self, raw, max_batch: int = 32, max_wait_ms: float = 8.0):
        self.raw = raw
        self.batcher = DynamicBatcher(raw, max_batch=max_batch,
                                      max_wait_ms=max_wait_ms)

    def generate(self, prompts: list[str], max_new_tokens: int = 128,
                 temperature: float = 0.3, stop_on_eos: bool = True,
                 **kwargs) -> list[str]:
        if len(prompts) != 1 or kwargs.get("on_token") is not None:
            return self.raw.generate(prompts, max_new_tokens=max_new_tokens,
                                     temperature=temperature,
                                     stop_on_eos=stop_on_eos, **kwargs)
        return [self.batcher.generate(prompts[0],
                                      max_new_tokens=max_new_tokens,
                                      temperature=temperature,
                                      stop_on_eos=stop_on_eos)]

    def stream(self, prompt: str, max_new_tokens: int = 128,
               temperature: float = 0.3):
        """Streams JOIN batched decode via the DynamicBatcher — no request
        holds the engine's generation lock for its stream's lifetime."""
        return self.batcher.generate_stream(prompt,
                                            max_new_tokens=max_new_tokens,
                                            temperature=temperature)

    def __getattr__(self, name):
        return getattr(self.raw, name)
