"""Continuous-batching serving engine.

The decode sweep (profiles/r02_decode_sweep.log) showed 2.5x the B=1
latency buys 256x the throughput, so the way to serve many concurrent
generations is ONE batched decode step for all in-flight requests, with
requests joining and leaving mid-flight:

    admit: run the new request's prompt (B=1) through the model, then
           CONCAT its (h, c) rows into the live batch state
    step:  one forward of (1, B_active) -> greedy next token per slot
    retire: finished slots' rows are index_select'ed out of the state

Everything is a row operation on the engine's native state layout (a
list of (h, c) tensors with batch as dim 0), so no request ever pays for
another's prompt or waits for the batch to drain.  Per-request decoding
knobs (temperature/top-k/top-p with an OWN seeded generator, EOS
early-stop) keep every request REPRODUCIBLE regardless of batch
composition -- the test oracle is exact token-for-token parity with a
solo `model.generate()` call, for greedy and seeded-sampled requests
alike.

This engine targets multi-tenant serving latency; the fixed-batch
hipGraph decode (`model.generate(use_graph=True)`) remains the peak
single-tenant throughput path (the graph is shape-specialized, a
dynamic batch cannot replay it).

SURVEY.md §2.1 has no counterpart row (the 2016 reference does not
serve); this is beyond-parity serving infrastructure.
"""
from __future__ import annotations

import itertools
import threading
from collections import deque
from typing import Dict, List, Optional

import torch


class _Slot:
    __slots__ = ("rid", "tokens", "remaining", "prompt_len", "eos",
                 "temperature", "top_k", "top_p", "gen")

    def __init__(self, rid: int, tokens: List[int], remaining: int,
                 eos: int = -1, temperature: float = 0.0, top_k: int = 0,
                 top_p: float = 0.0, gen=None):
        self.rid = rid
        self.tokens = list(tokens)
        self.prompt_len = len(tokens)
        self.remaining = remaining
        self.eos = eos
        self.temperature = temperature
        self.top_k = top_k
        self.top_p = top_p
        self.gen = gen

    def finished(self) -> bool:
        return (self.remaining <= 0
                or (self.eos >= 0 and self.tokens[-1] == self.eos))


class ContinuousBatcher:
    """Thread-safe continuous batcher around an LM-head LSTMModel.

    Usage (inline, deterministic -- how the tests drive it):
        eng = ContinuousBatcher(model)
        rid = eng.submit([1, 2, 3], steps=16)
        eng.drain()                      # run decode steps until idle
        out = eng.result(rid)            # full token list (prompt + gen)

    Usage (server): eng.start() spawns an engine thread; submit()/result()
    are safe from any number of request threads.
    """

    def __init__(self, model, max_batch: int = 64, max_steps: int = 4096):
        assert model.cfg.head == "lm", "serving engine requires the LM head"
        assert not model.cfg.bidirectional, "decode is causal: fwd-only"
        self.model = model
        self.device = next(iter(model.weight_dict().values())).device
        self.max_batch = max_batch
        self.max_steps = max_steps
        self._ids = itertools.count()
        self._queue: deque = deque()
        self._slots: List[_Slot] = []
        self._state: Optional[list] = None   # list[(h, c)], batch dim 0
        self._results: Dict[int, List[int]] = {}
        self._cv = threading.Condition()
        self._thread: Optional[threading.Thread] = None
        self._running = False
        self._decode_steps = 0      # batched decode steps executed
        self._slot_steps = 0        # sum of active slots over those steps
        self._cancelled: set = set()  # rids retired by cancel()

    # ------------------------------------------------------------ requests
    def submit(self, tokens: List[int], steps: int,
               eos_token: int = -1, temperature: float = 0.0,
               top_k: int = 0, top_p: float = 0.0,
               seed: Optional[int] = None) -> int:
        """eos_token >= 0: the request finishes early when it emits that
        token (same semantics as model.generate(eos_token=...)).

        temperature > 0 samples with this request's OWN generator (seeded
        with `seed` when given), so a sampled request is reproducible and
        token-identical to a solo model.generate() with the same seed --
        regardless of what else shares the batch."""
        V = self.model.cfg.vocab_size
        if not tokens or any(t < 0 or t >= V for t in tokens):
            raise ValueError(f"tokens must be in [0, {V})")
        if steps < 1 or steps > self.max_steps:
            raise ValueError(f"steps must be in [1, {self.max_steps}]")
        gen = None
        if temperature > 0 and seed is not None:
            gen = torch.Generator(device=self.device)
            gen.manual_seed(seed)
        with self._cv:
            rid = next(self._ids)
            self._queue.append(_Slot(rid, tokens, steps, eos_token,
                                     temperature, top_k, top_p, gen))
            self._cv.notify_all()
        return rid

    def peek(self, rid: int):
        """(tokens_so_far, done) without claiming the result -- the
        polling primitive for streaming responses."""
        with self._cv:
            if rid in self._results:
                return list(self._results[rid]), True
            for s in self._slots:
                if s.rid == rid:
                    return list(s.tokens), False
            for s in self._queue:
                if s.rid == rid:
                    return list(s.tokens), False
        raise KeyError(f"unknown request id {rid}")

    def cancel(self, rid: int) -> bool:
        """Abort a request: dequeue it, retire its live slot at the next
        step boundary, or drop its unclaimed result.  Returns True if the
        rid was known.  (The SSE endpoint calls this when the client
        disconnects so abandoned generations stop consuming batch slots.)
        """
        with self._cv:
            for i, s in enumerate(self._queue):
                if s.rid == rid:
                    del self._queue[i]
                    return True
            for s in self._slots:
                if s.rid == rid:
                    s.remaining = 0     # retires at the next step boundary
                    self._cancelled.add(rid)
                    return True
            return self._results.pop(rid, None) is not None

    def try_result(self, rid: int) -> Optional[List[int]]:
        with self._cv:
            return self._results.pop(rid, None)

    def result(self, rid: int, timeout: Optional[float] = None) -> List[int]:
        """Block until request `rid` finishes (engine thread must be
        running, or call drain() first)."""
        with self._cv:
            ok = self._cv.wait_for(lambda: rid in self._results,
                                   timeout=timeout)
            if not ok:
                raise TimeoutError(f"request {rid} not done in {timeout}s")
            return self._results.pop(rid)

    def stats(self) -> dict:
        with self._cv:
            occ = (self._slot_steps / self._decode_steps
                   if self._decode_steps else 0.0)
            return {"active": len(self._slots), "queued": len(self._queue),
                    "done_unclaimed": len(self._results),
                    "decode_steps": self._decode_steps,
                    "mean_occupancy": round(occ, 2)}

    # ------------------------------------------------------------- engine
    @torch.no_grad()
    def _admit_locked(self):
        """Move queued requests into live slots (runs their prompts)."""
        m = self.model
        while self._queue and len(self._slots) < self.max_batch:
            slot = self._queue.popleft()
            prompt = torch.tensor(slot.tokens, dtype=torch.long,
                                  device=self.device).unsqueeze(1)
            out, st = m.forward(prompt)              # (Tp, 1, V)
            tok = int(self._pick(out[-1], slot)[0])
            slot.tokens.append(tok)
            slot.remaining -= 1
            if self._state is None or not self._slots:
                self._state = [(h, c) for h, c in st]
            else:
                self._state = [(torch.cat([h0, h1]), torch.cat([c0, c1]))
                               for (h0, c0), (h1, c1)
                               in zip(self._state, st)]
            self._slots.append(slot)
        self._retire_locked()

    @torch.no_grad()
    def _decode_step_locked(self):
        """One batched decode step for every live slot."""
        if not self._slots:
            return
        self._decode_steps += 1
        self._slot_steps += len(self._slots)
        m = self.model
        x = torch.tensor([[s.tokens[-1] for s in self._slots]],
                         dtype=torch.long, device=self.device)  # (1, B)
        out, self._state = m.forward(x, self._state)            # (1, B, V)
        logits = out[0]                                         # (B, V)
        greedy = logits.argmax(dim=-1)                          # (B,)
        for i, s in enumerate(self._slots):
            if s.temperature > 0:
                s.tokens.append(int(self._pick(logits[i:i + 1], s)[0]))
            else:
                s.tokens.append(int(greedy[i]))
            s.remaining -= 1
        self._retire_locked()

    def _pick(self, logits, slot: _Slot):
        """Next token for one slot from its (1, V) logits row -- the same
        _sample used by model.generate, fed the slot's own generator."""
        return self.model._sample(logits, slot.temperature, slot.gen,
                                  slot.top_k, slot.top_p)

    def _retire_locked(self):
        keep, done = [], []
        for i, s in enumerate(self._slots):
            (done if s.finished() else keep).append(i)
        if not done:
            return
        for i in done:
            s = self._slots[i]
            if s.rid in self._cancelled:
                self._cancelled.discard(s.rid)   # drop, don't store
            else:
                self._results[s.rid] = s.tokens
        if keep:
            idx = torch.tensor(keep, device=self.device)
            self._state = [(h.index_select(0, idx), c.index_select(0, idx))
                           for h, c in self._state]
            self._slots = [self._slots[i] for i in keep]
        else:
            self._state, self._slots = None, []
        self._cv.notify_all()

    def step(self) -> int:
        """One engine iteration: admit, then one batched decode step.
        Returns the number of live slots after the step."""
        was_training = self.model.training
        self.model.eval()
        try:
            with self._cv:
                self._admit_locked()
                self._decode_step_locked()
                return len(self._slots)
        finally:
            if was_training:
                self.model.train()

    def drain(self):
        """Run engine steps inline until no work remains (deterministic
        single-threaded mode -- what the tests use)."""
        while True:
            with self._cv:
                idle = not self._queue and not self._slots
            if idle:
                return
            self.step()

    # ------------------------------------------------------------- thread
    def start(self):
        if self._thread is not None:
            return
        self._running = True
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def stop(self):
        with self._cv:
            self._running = False
            self._cv.notify_all()
        if self._thread is not None:
            self._thread.join(timeout=10)
            self._thread = None

    def _loop(self):
        while True:
            with self._cv:
                if not self._running:
                    return
                has_work = bool(self._queue or self._slots)
                if not has_work:
                    self._cv.wait(timeout=0.05)
                    continue
            self.step()
