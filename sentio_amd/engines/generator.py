"""On-device LLM generation engine (replaces the reference's OpenAI-compatible
HTTP calls, reference src/core/llm/providers/openai.py:117-121; K6/K7 in
SURVEY §2.3).

Llama-3-class decoder: batched prefill (flash-style MFMA attention) + KV-cache
decode (single-pass online-softmax kernel), on-device sampling, optional
per-token streaming callback (the reference streamed SSE chunks).
Temperature-per-mode mirrors reference generator.py:262-267
(fast=0.0, balanced=0.3, quality=0.2, creative=0.7).
"""

from __future__ import annotations

from typing import Callable, Iterator

import torch

from sentio_amd import ops
from sentio_amd.engines.configs import get_model_config
from sentio_amd.engines.bpe import get_tokenizer
from sentio_amd.engines.tokenizer import EOS_ID
from sentio_amd.engines.transformer import KVCache, Transformer

MODE_TEMPERATURE = {"fast": 0.0, "balanced": 0.3, "quality": 0.2, "creative": 0.7}


class _DecodeSession:
    """One (batch, cache-size) decode context: preallocated KV cache plus an
    optional hipGraph capture of the whole per-token decode step."""

    def __init__(self, engine: "GeneratorEngine", batch: int, cache_len: int,
                 use_graphs: bool):
        self.engine = engine
        self.cache = engine._new_cache(batch, cache_len)
        self.graph = None
        self.static_tok = None
        self.static_logits = None
        self.use_graphs = use_graphs
        self.batch = batch

    def prefill(self, tokens: torch.Tensor,
                lens: torch.Tensor | None = None) -> torch.Tensor:
        self.cache.seq_lens.zero_()
        return self.engine.model.prefill(tokens, self.cache, lens=lens)

    def prefill_with_prefix(self, prefix_ids: list[int],
                            suffix_tokens: torch.Tensor,
                            suffix_lens: torch.Tensor | None = None
                            ) -> torch.Tensor:
        """Copy the (cached) prefix KV into every batch slot, then prefill
        only the suffix (prefix-KV caching: the shared system-prompt +
        instruction header is prefilled ONCE per prefix, not per request)."""
        engine = self.engine
        pk, pv = engine._prefix_kv(tuple(prefix_ids))
        P = len(prefix_ids)
        for i in range(engine.cfg.n_layers):
            self.cache.k[i][:, :, :P].copy_(pk[i])   # [1,...] broadcasts
            self.cache.v[i][:, :, :P].copy_(pv[i])
        return engine.model.prefill_suffix(suffix_tokens, self.cache, P,
                                           suffix_lens=suffix_lens)

    def _capture(self):
        model = self.engine.model
        dev = self.engine.device
        self.static_tok = torch.zeros(self.batch, 1, dtype=torch.int64,
                                      device=dev)
        # warmup on a side stream (required before capture), then capture.
        # seq_lens must be restored: warmup/capture each advance it.
        saved = self.cache.seq_lens.clone()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                model.decode_step(self.static_tok, self.cache)
        torch.cuda.current_stream().wait_stream(s)
        self.cache.seq_lens.copy_(saved)
        self.graph = torch.cuda.CUDAGraph()
        # relaxed capture: concurrent engine work on OTHER threads
        # (pipelined retrieval, encoder graph replays, allocator traffic)
        # must not invalidate this capture — global/thread_local modes abort
        # on foreign allocator calls; the captured region itself only
        # contains this stream's decode ops
        with torch.cuda.graph(self.graph, capture_error_mode="relaxed"):
            self.static_logits = model.decode_step(self.static_tok, self.cache)
        self.cache.seq_lens.copy_(saved)

    def decode_step(self, tok: torch.Tensor) -> torch.Tensor:
        if not self.use_graphs:
            return self.engine.model.decode_step(tok.view(self.batch, 1),
                                                 self.cache)
        if self.graph is None:
            self._capture()
        self.static_tok.copy_(tok.view(self.batch, 1))
        self.graph.replay()
        return self.static_logits


class GeneratorEngine:
    def __init__(self, model: str = "llama3-8b", device: str = "cpu",
                 dtype: str = "bf16", max_seq: int = 4096, seed: int = 303,
                 tp=None):
        self.cfg = get_model_config(model)
        self.device = device
        self.max_seq = min(max_seq, self.cfg.max_seq)
        self.tokenizer = get_tokenizer()
        self.model = Transformer(self.cfg, device=device, dtype=dtype,
                                 seed=seed, tp=tp)
        self._step_seed = 0
        import threading

        self._gen_lock = threading.Lock()   # decode sessions hold static state

    def _new_cache(self, batch: int, max_seq: int) -> KVCache:
        return KVCache(self.cfg, batch, max_seq, self.device, self.model.dtype,
                       n_kv_heads=self.model.hkv_local)

    # ----- prefix-KV cache (shared prompt prefixes prefilled once) -----
    _PREFIX_MIN_TOKENS = 64

    def _split_shared_prefix(self, clipped: list[str], prompt_budget: int
                             ) -> tuple[list[int], list[str]]:
        """Split the batch's longest common prompt prefix from the per-prompt
        suffixes such that prefix_ids + encode(suffix, add_bos=False) is
        EXACTLY encode(full prompt) for every prompt.  The cap is computed in
        TOKENS via tokenizer.prefix_split — a char-based cap overshoots the
        budget for multi-byte UTF-8 prefixes (len(prefix_ids) > chars) and
        crashed prefill with negative suffix budgets."""
        import os as _os

        if len(clipped) < 2:
            return [], clipped
        prefix_txt = _os.path.commonprefix(clipped)
        # leave >= 1 suffix char per prompt and >= 8 suffix tokens of budget
        limit_chars = min(len(prefix_txt), min(len(c) for c in clipped) - 1)
        if limit_chars <= 0 or prompt_budget <= 9:
            return [], clipped
        n, prefix_ids = self.tokenizer.prefix_split(
            prefix_txt[:limit_chars], prompt_budget - 8)
        if n <= 0 or not prefix_ids:
            return [], clipped
        return prefix_ids, [c[n:] for c in clipped]

    def _prefix_kv(self, prefix_ids: tuple):
        """Per-prefix KV, computed once with a batch-1 forward and kept in
        HBM (a few tens of MB per distinct prefix; LRU of 4)."""
        store = getattr(self, "_prefix_store", None)
        if store is None:
            store = self._prefix_store = {}
        hit = store.get(prefix_ids)
        if hit is not None:
            return hit
        P = len(prefix_ids)
        tmp = KVCache(self.cfg, 1, P, self.device, self.model.dtype,
                      n_kv_heads=self.model.hkv_local)
        toks = torch.tensor([list(prefix_ids)], dtype=torch.int64,
                            device=self.device)
        self.model.forward_hidden(toks, cache=tmp)
        entry = ([k.clone() for k in tmp.k], [v.clone() for v in tmp.v])
        if len(store) >= 4:
            store.pop(next(iter(store)))
        store[prefix_ids] = entry
        return entry

    # ----- hipGraph-captured decode session -----
    # The per-token decode step launches ~10 kernels per layer; on MI355X the
    # wall time at small batch is launch-bound, so the whole step is captured
    # once as a hipGraph and replayed per token (sampling stays outside: its
    # seed changes per step).  Sessions are pooled by (batch, cache size).
    _GRAPHS_ENABLED = True

    def _decode_session(self, batch: int, cache_len: int):
        import os

        use_graphs = (
            self._GRAPHS_ENABLED
            and self.device != "cpu"
            # TP decode: the chunked async all-reduces (parallel/tp.py)
            # are stream-ordered and RCCL supports graph capture, but this
            # environment has no multi-GPU lease to validate the captured
            # collective schedule on real xGMI — graphed TP decode is
            # therefore opt-in (SENTIO_TP_HIPGRAPH=1); the default stays
            # eager, correct by construction.
            and (not self.model.tp.enabled
                 or os.environ.get("SENTIO_TP_HIPGRAPH", "0") == "1")
            and os.environ.get("SENTIO_DISABLE_HIPGRAPH", "0") != "1"
        )
        key = (batch, cache_len)
        pool = getattr(self, "_sessions", None)
        if pool is None:
            pool = self._sessions = {}
        if key in pool:
            sess = pool[key]
            sess.cache.seq_lens.zero_()
            return sess
        sess = _DecodeSession(self, batch, cache_len, use_graphs)
        pool[key] = sess
        return sess

    # ----- continuous-batching support (serving.ContinuousBatcher) -----
    def make_slot_session(self, slots: int) -> "_DecodeSession":
        """A PRIVATE decode session for the continuous batcher: its KV
        slots persist across requests, so it must never come from (or be
        reset by) the shared _decode_session pool."""
        import os

        use_graphs = (
            self._GRAPHS_ENABLED
            and self.device != "cpu"
            and (not self.model.tp.enabled
                 or os.environ.get("SENTIO_TP_HIPGRAPH", "0") == "1")
            and os.environ.get("SENTIO_DISABLE_HIPGRAPH", "0") != "1"
        )
        return _DecodeSession(self, slots, self.max_seq, use_graphs)

    @torch.inference_mode()
    def prefill_admission(self, prompts: list[str],
                          max_new_list: list[int]) -> dict:
        """Admission prefill (continuous batching), LOCK-FREE: encode +
        prefill the prompts into a temporary cache on the CALLER's stream.
        Touches only the temp cache and read-only weights, so it may run
        on a side stream CONCURRENTLY with the decode loop's graph
        replays — a joining request's prefill never stalls resident
        requests.  integrate_admission() splices the result into slots."""
        ids_list = []
        for p, mn in zip(prompts, max_new_list):
            budget = max(self.max_seq - mn - 1, 8)
            ids_list.append(self.tokenizer.encode(p[-4 * budget:], budget))
        lens = [len(i) for i in ids_list]
        S = max(lens)
        padded = [i + [0] * (S - len(i)) for i in ids_list]
        tokens = torch.tensor(padded, dtype=torch.int64, device=self.device)
        lens_t = torch.tensor(lens, dtype=torch.int32, device=self.device)
        tmp = KVCache(self.cfg, len(prompts), S, self.device,
                      self.model.dtype, n_kv_heads=self.model.hkv_local)
        logits = self.model.prefill(tokens, tmp, lens=lens_t)
        return {"tmp": tmp, "lens": lens_t, "logits": logits, "S": S,
                "lens_host": lens}

    @torch.inference_mode()
    def integrate_admission(self, sess: "_DecodeSession", rows: list[int],
                            pre: dict,
                            idx: list[int] | None = None) -> torch.Tensor:
        """Copy a prefill_admission's KV into sess.cache rows and set their
        seq_lens (run on the decode loop's stream, AFTER syncing with the
        admission stream).  `idx` selects a SUBSET of the admission's
        requests (partial integration when fewer slots are free than the
        admission holds).  Returns those requests' logits."""
        tmp, lens_t, S = pre["tmp"], pre["lens"], pre["S"]
        logits = pre["logits"]
        with self._gen_lock:
            rows_t = torch.tensor(rows, dtype=torch.int64, device=self.device)
            if idx is not None:
                sel = torch.tensor(idx, dtype=torch.int64, device=self.device)
                lens_t = lens_t[sel]
                logits = logits[sel]
            for li in range(self.cfg.n_layers):
                src_k, src_v = tmp.k[li], tmp.v[li]
                if idx is not None:
                    src_k, src_v = src_k[sel], src_v[sel]
                sess.cache.k[li][rows_t, :, :S] = src_k
                sess.cache.v[li][rows_t, :, :S] = src_v
            sess.cache.seq_lens[rows_t] = lens_t
        return logits

    @torch.inference_mode()
    def prefill_into_slots(self, sess: "_DecodeSession", rows: list[int],
                           prompts: list[str],
                           max_new_list: list[int]) -> torch.Tensor:
        """prefill_admission + integrate_admission in one call (same-thread
        admission path / tests)."""
        with self._gen_lock:
            pre = self.prefill_admission(prompts, max_new_list)
        return self.integrate_admission(sess, rows, pre)

    @torch.inference_mode()
    def decode_step_session(self, sess: "_DecodeSession",
                            cur: torch.Tensor) -> torch.Tensor:
        """One decode step over ALL slots (hipGraph replay when captured)."""
        with self._gen_lock:
            return sess.decode_step(cur)

    @torch.inference_mode()
    def sample_rows(self, logits: torch.Tensor,
                    temps: torch.Tensor) -> torch.Tensor:
        """Per-ROW-temperature sampling — continuous batches mix requests
        with different temperatures: greedy rows take argmax, the rest
        Gumbel-argmax (torch eager; the split sampling kernel takes one
        scalar temperature and stays on the uniform-batch path).  Noise
        comes from a SEEDED per-engine generator so stochastic decodes are
        reproducible, matching the wave path's seeded sample_token."""
        self._step_seed += 1
        rng = getattr(self, "_sample_rng", None)
        if rng is None or str(rng.device) != str(logits.device):
            rng = self._sample_rng = torch.Generator(device=logits.device)
            rng.manual_seed(909)
        t = temps.to(logits.device).view(-1, 1)
        u = torch.rand(logits.shape, device=logits.device, generator=rng)
        g = -torch.log(-torch.log(u.clamp_min(1e-20)).clamp_min(1e-20))
        scores = torch.where(t > 0, logits / t.clamp_min(1e-6) + g, logits)
        return scores.argmax(-1)

    @torch.inference_mode()
    def generate(
        self,
        prompts: list[str],
        max_new_tokens: int = 128,
        temperature: float = 0.3,
        stop_on_eos: bool = True,
        on_token: Callable[[int, list[int]], None] | None = None,
    ) -> list[str]:
        """Batched generation.  Returns decoded completions (prompt excluded).

        on_token(step, token_ids_per_batch) fires after every decode step —
        the serving layer turns it into SSE streaming.
        """
        if not prompts:
            return []
        # a request cannot generate past the cache: clamp so the prompt
        # budget stays positive and decode never writes beyond max_seq
        max_new_tokens = max(1, min(max_new_tokens, self.max_seq - 9))
        import time as _time

        from sentio_amd.observability.kernel_timer import get_timer

        self._gen_lock.acquire()
        try:
            with get_timer("generate").measure():
                return self._generate_locked(prompts, max_new_tokens,
                                             temperature, stop_on_eos,
                                             on_token, _time)
        finally:
            self._gen_lock.release()

    # Decode-session batch buckets: serving batches arrive at arbitrary
    # sizes, and every DISTINCT (batch, cache) shape costs a fresh KV-cache
    # allocation + hipGraph capture (~1 s) — the GPU load test spent more
    # wall on captures than on decoding.  Padding rows are near-free
    # (decode streams the same weights either way), so round up and slice.
    _BATCH_BUCKETS = (1, 2, 4, 8, 16, 24, 32, 48, 64)

    def _bucket_batch(self, b: int) -> int:
        for cap in self._BATCH_BUCKETS:
            if b <= cap:
                return cap
        return b

    def _generate_locked(self, prompts, max_new_tokens, temperature,
                         stop_on_eos, on_token, _time) -> list[str]:
        B_req = len(prompts)
        if self.device != "cpu":
            pad = self._bucket_batch(B_req) - B_req
            if pad:
                prompts = list(prompts) + [prompts[-1]] * pad
        B = len(prompts)
        prompt_budget = self.max_seq - max_new_tokens - 1
        # prefix-KV caching: requests share the system-prompt + instruction
        # header; prefill it once per distinct prefix and only forward each
        # request's suffix (byte tokenizer → char-exact prefix alignment)
        import os as _os

        clipped = [p[-4 * prompt_budget:] for p in prompts]
        prefix_ids, suffixes = self._split_shared_prefix(clipped, prompt_budget)
        use_prefix = (len(prefix_ids) >= self._PREFIX_MIN_TOKENS
                      and _os.environ.get("SENTIO_PREFIX_KV", "1") != "0")
        sess = self._decode_session(B, self.max_seq)
        _t0 = _time.perf_counter()
        if use_prefix:
            P = len(prefix_ids)
            # suffixes continue the stream: NO second BOS
            padded, lens = self.tokenizer.encode_batch(
                suffixes, prompt_budget - P, add_bos=False)
            tokens = torch.tensor(padded, dtype=torch.int64,
                                  device=self.device)
            lens_t = torch.tensor(lens, dtype=torch.int32, device=self.device)
            sess.cache.seq_lens.zero_()
            logits = sess.prefill_with_prefix(prefix_ids, tokens, lens_t)
            n_prompt = B_req * P + sum(lens[:B_req])
        else:
            padded, lens = self.tokenizer.encode_batch(clipped, prompt_budget)
            tokens = torch.tensor(padded, dtype=torch.int64,
                                  device=self.device)
            lens_t = torch.tensor(lens, dtype=torch.int32, device=self.device)
            logits = sess.prefill(tokens, lens_t)
            n_prompt = sum(lens[:B_req])
        if self.device != "cpu":
            torch.cuda.synchronize()
        self.last_prefill_s = _time.perf_counter() - _t0
        self.last_prompt_tokens = n_prompt
        _t0 = _time.perf_counter()

        # tokens accumulate on-device; the host syncs only for EOS checks
        # (every 16 steps) or the streaming callback — not per token.
        toks_buf = torch.zeros(B, max_new_tokens, dtype=torch.int64,
                               device=self.device)
        cur = self._sample(logits, temperature)
        n_steps = max_new_tokens
        for step in range(max_new_tokens):
            toks_buf[:, step] = cur
            if on_token is not None:
                on_token(step, cur.cpu().tolist()[:B_req])
            if stop_on_eos and (step % 16 == 15 or on_token is not None):
                done = (toks_buf[:, : step + 1] == EOS_ID).any(dim=1)
                if bool(done.all()):
                    n_steps = step + 1
                    break
            if step == max_new_tokens - 1:
                break
            logits = sess.decode_step(cur)
            cur = self._sample(logits, temperature)
        if self.device != "cpu":
            torch.cuda.synchronize()
        self.last_decode_s = _time.perf_counter() - _t0
        self.last_decode_steps = n_steps
        try:
            from sentio_amd.observability.metrics import metrics_collector

            metrics_collector.inc("rag_llm_tokens_total",
                                  float(n_prompt), kind="prompt")
            metrics_collector.inc("rag_llm_tokens_total",
                                  float(B_req * n_steps), kind="completion")
        except Exception:
            pass
        rows = toks_buf[:B_req, :n_steps].cpu().tolist()
        out = []
        for row in rows:
            ids = []
            for t in row:
                if stop_on_eos and t == EOS_ID:
                    break
                ids.append(t)
            out.append(self.tokenizer.decode(ids))
        return out

    def _sample(self, logits: torch.Tensor, temperature: float) -> torch.Tensor:
        self._step_seed += 1
        return ops.sample_token(logits, temperature, seed=self._step_seed)

    @torch.inference_mode()
    def stream(self, prompt: str, max_new_tokens: int = 128,
               temperature: float = 0.3) -> Iterator[str]:
        """Yield text deltas token-by-token (single prompt) — the on-device
        equivalent of the reference's SSE streaming (openai.py:149-157).
        Holds the generation lock for the stream's duration (decode sessions
        and the model's decode-attention selection are engine-level state)."""
        with self._gen_lock:
            yield from self._stream_locked(prompt, max_new_tokens, temperature)

    def _stream_locked(self, prompt: str, max_new_tokens: int,
                       temperature: float) -> Iterator[str]:
        prompt_budget = self.max_seq - max_new_tokens - 1
        ids = self.tokenizer.encode(prompt[-4 * prompt_budget:], prompt_budget)
        tokens = torch.tensor([ids], dtype=torch.int64, device=self.device)
        sess = self._decode_session(1, self.max_seq)
        logits = sess.prefill(tokens)
        cur = self._sample(logits, temperature)
        generated: list[int] = []
        emitted = ""
        for step in range(max_new_tokens):
            t = int(cur.item())
            if t == EOS_ID:
                break
            generated.append(t)
            text = self.tokenizer.decode(generated)
            if len(text) > len(emitted):
                delta = text[len(emitted):]
                emitted = text
                yield delta
            if step == max_new_tokens - 1:
                break
            logits = sess.decode_step(cur)
            cur = self._sample(logits, temperature)
