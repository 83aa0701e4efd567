"""Streaming token-generation service: one request up, a token stream down.

The RPC shape mirrors the reference's streaming lifecycle + single-up/
multiple-down client (nvrpc life_cycle_streaming.h:61,
client_single_up_multiple_down.h:43); what runs behind it is
beyond-reference: a continuous-batching engine loop over an incremental
DecodeSession — every replayed step advances ALL live streams one token,
new requests claim idle slots mid-flight (their prompt tokens are fed
one per step while other slots keep generating), finished slots park via
idle_slot() and return to the free list.

The engine only needs the DecodeSession surface (batch, step(ids) ->
logits, reset_slot, idle_slot), so tests drive it with a deterministic
fake on CPU and the same service serves GPT-2 or LLaMA sessions on GPU
(examples/generation_server.py).
"""
from __future__ import annotations

import asyncio
from typing import List, Optional

import numpy as np
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_T = descriptor_pb2.FieldDescriptorProto

_f = descriptor_pb2.FileDescriptorProto()
_f.name = "trtlab_amd/rpc/generation.proto"
_f.package = "trtlab.gen"
_f.syntax = "proto3"

_req = _f.message_type.add()
_req.name = "GenerateRequest"
for name, num, ft, rep in (("prompt", 1, _T.TYPE_INT32, True),
                           ("max_tokens", 2, _T.TYPE_INT32, False),
                           ("temperature", 3, _T.TYPE_FLOAT, False),
                           ("top_k", 4, _T.TYPE_INT32, False),
                           ("top_p", 5, _T.TYPE_FLOAT, False),
                           ("seed", 6, _T.TYPE_INT32, False)):
    fd = _req.field.add()
    fd.name = name
    fd.number = num
    fd.type = ft
    fd.label = _T.LABEL_REPEATED if rep else _T.LABEL_OPTIONAL

_tok = _f.message_type.add()
_tok.name = "GenerateToken"
for name, num, ft in (("token", 1, _T.TYPE_INT32),
                      ("slot", 2, _T.TYPE_INT32),
                      ("index", 3, _T.TYPE_INT32),
                      ("done", 4, _T.TYPE_BOOL)):
    fd = _tok.field.add()
    fd.name = name
    fd.number = num
    fd.type = ft
    fd.label = _T.LABEL_OPTIONAL

_pool = descriptor_pool.Default()
_fd = _pool.Add(_f)
GenerateRequest = message_factory.GetMessageClass(
    _fd.message_types_by_name["GenerateRequest"])
GenerateToken = message_factory.GetMessageClass(
    _fd.message_types_by_name["GenerateToken"])


class GenerationEngine:
    """Continuous-batching loop over a lockstep decode session."""

    def __init__(self, session, eos: int = -1, inline_step: bool = False,
                 max_waiting: int = 0):
        """inline_step=True runs session.step() directly on the event
        loop instead of a worker thread. MEASURED COUNTERPRODUCTIVE
        (tools/gen_load.py, MI355X, 16 streams/8 slots): 5,491 -> 1,229
        tok/s. The worker-thread hop YIELDS the event loop for the whole
        GPU step — exactly when new streams get admitted — so inlining
        starves admission and the loop burns steps on half-empty batches
        (1,185 steps vs the 159 ideal). Kept as an off-default knob for
        single-stream latency experiments."""
        import collections

        self.session = session
        self.B = session.batch
        self.eos = eos
        self.inline_step = inline_step
        # admission control: > 0 bounds the wait queue — beyond it,
        # submit() raises OverflowError (the service maps it to
        # RESOURCE_EXHAUSTED) instead of queueing unboundedly
        self.max_waiting = int(max_waiting)
        self._slots: List[Optional[dict]] = [None] * self.B
        self._free: List[int] = list(range(self.B))
        self._pending: collections.deque = collections.deque()
        self._wake = asyncio.Event()
        self._task: Optional[asyncio.Task] = None
        self.steps = 0       # total engine steps (observability / tests)
        self.tokens_out = 0  # total tokens emitted across all streams
        self.streams = 0     # total streams admitted
        # a fresh DecodeSession has every slot ACTIVE at pos 0; park them
        # all so unclaimed slots cost nothing and never hit the sequence
        # limit while other slots generate (submit() re-activates)
        for b in range(self.B):
            self.session.idle_slot(b)

    def ensure_started(self) -> None:
        if self._task is None or self._task.done():
            self._task = asyncio.get_running_loop().create_task(self._loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None

    async def submit(self, prompt: List[int], max_tokens: int,
                     temperature: float = 0.0, top_k: int = 0,
                     top_p: float = 0.0, seed: int = 0) -> tuple:
        """Claim a slot (waiting for one to free if the batch is full),
        prime it with the prompt, and return (slot, token queue). The
        queue yields ints and a final None sentinel.

        temperature <= 0 decodes greedily; otherwise logits/T are
        softmaxed and sampled, restricted to the top_k highest (0 = all)
        and the top_p nucleus (0 = all), from a per-slot RNG seeded by
        `seed` so streams are reproducible."""
        if not prompt:
            raise ValueError("empty prompt")
        smax = getattr(self.session, "smax", 1 << 30)
        if len(prompt) + max_tokens >= smax:
            raise ValueError(
                f"prompt+max_tokens ({len(prompt)}+{max_tokens}) exceeds "
                f"the session window ({smax})")
        if self.max_waiting > 0 and len(self._pending) >= self.max_waiting:
            raise OverflowError(
                f"admission queue full ({self.max_waiting} waiting)")
        # the LOOP is the sole owner of session state: submissions queue
        # and are admitted between steps (reset_slot must never race an
        # in-flight replay that is reading this slot's position)
        fut = asyncio.get_running_loop().create_future()
        self._pending.append((dict(q=asyncio.Queue(), prompt=list(prompt),
                                   pi=0, remaining=int(max_tokens), cur=0,
                                   temperature=float(temperature),
                                   top_k=int(top_k), top_p=float(top_p),
                                   seed=int(seed),
                                   rng=np.random.RandomState(seed or None)),
                              fut))
        self._wake.set()
        return await fut

    def stats(self) -> dict:
        """Engine counters for observability (tokens/steps ratio ==
        continuous-batching packing efficiency x batch)."""
        return dict(steps=self.steps, tokens_out=self.tokens_out,
                    streams=self.streams,
                    active=sum(1 for s_ in self._slots if s_ is not None),
                    free=len(self._free), pending=len(self._pending))

    def _finish(self, b: int) -> None:
        self._slots[b]["q"].put_nowait(None)
        self._slots[b] = None
        self.session.idle_slot(b)
        self._free.append(b)

    def cancel(self, b: int, q: asyncio.Queue) -> None:
        """Mark slot b for release (client disconnected mid-stream);
        the loop reclaims it between steps. The q identity guards
        against cancelling a slot that already finished and was handed
        to another stream."""
        st = self._slots[b]
        if st is not None and st["q"] is q:
            st["dead"] = True
            self._wake.set()

    @staticmethod
    def _sample(logits: np.ndarray, temperature: float, top_k: int,
                top_p: float, rng) -> int:
        if temperature <= 0.0:
            return int(np.argmax(logits))
        x = logits.astype(np.float64) / temperature
        x -= x.max()
        p = np.exp(x)
        p /= p.sum()
        order = np.argsort(-p)
        keep = len(order)
        if top_k > 0:
            keep = min(keep, top_k)
        if 0.0 < top_p < 1.0:
            c = np.cumsum(p[order])
            # smallest prefix whose mass reaches top_p (always >= 1)
            keep = min(keep, int(np.searchsorted(c, top_p) + 1))
        idx = order[:keep]
        pk = p[idx] / p[idx].sum()
        return int(rng.choice(idx, p=pk))

    def _emit(self, b: int, logits_row, tok: int = -1) -> None:
        st = self._slots[b]
        if tok < 0:
            tok = self._sample(logits_row, st["temperature"], st["top_k"],
                               st["top_p"], st["rng"])
        st["cur"] = tok
        st["q"].put_nowait(tok)
        self.tokens_out += 1
        st["remaining"] -= 1
        if st["remaining"] <= 0 or tok == self.eos:
            self._finish(b)

    async def _loop(self) -> None:
        loop = asyncio.get_running_loop()
        while True:
            # reclaim cancelled slots, then admit queued submissions —
            # all session mutations happen HERE, between steps
            for b in range(self.B):
                st = self._slots[b]
                if st is not None and st.get("dead"):
                    self._finish(b)
            while self._pending and self._free:
                st, fut = self._pending.popleft()
                if fut.cancelled():
                    continue
                b = self._free.pop()
                self.session.reset_slot(b)
                self._slots[b] = st
                self.streams += 1
                fut.set_result((b, st["q"]))
            active = [b for b in range(self.B) if self._slots[b]]
            if not active:
                self._wake.clear()
                if self._pending:  # waiting on a free slot, not on work
                    self._wake.set()
                await self._wake.wait()
                continue
            ids = np.zeros(self.B, np.int32)
            snap = {}
            for b in active:
                st = snap[b] = self._slots[b]
                ids[b] = (st["prompt"][st["pi"]]
                          if st["pi"] < len(st["prompt"]) else st["cur"])
            # device fast path: greedy uses the in-graph argmax head;
            # plain-temperature sampling uses the Gumbel-max kernel on
            # the resident logits — either way only B ints cross PCIe.
            # top-k / nucleus streams still need their logits rows.
            ids_only = (getattr(self.session, "supports_ids", False) and
                        all(snap[b]["top_k"] == 0 and
                            snap[b]["top_p"] == 0.0 for b in active))
            if ids_only:
                sampled = any(snap[b]["temperature"] > 0.0 for b in active)
                temps = np.zeros(self.B, np.float32)
                seeds = np.zeros(self.B, np.int32)
                for b in active:
                    temps[b] = max(snap[b]["temperature"], 0.0)
                    seeds[b] = snap[b].get("seed", 0)

                def _dev_step():
                    g = self.session.step(ids, return_ids=True)
                    if sampled:
                        g = self.session.sample_tokens(temps, seeds)
                    return g

                if self.inline_step:
                    gids = _dev_step()
                else:
                    gids = await loop.run_in_executor(None, _dev_step)
                logits = None
            else:
                if self.inline_step:
                    logits = self.session.step(ids)
                else:
                    logits = await loop.run_in_executor(
                        None, self.session.step, ids)
            self.steps += 1
            for b in active:
                st = self._slots[b]
                # identity check: a stream cancelled during the step may
                # have freed the slot AND a new submit() re-claimed it —
                # this step's logits belong to the OLD stream's token
                if st is None or st is not snap[b]:
                    continue
                row = None if ids_only else logits[b]
                tok = int(gids[b]) if ids_only else -1
                if st["pi"] < len(st["prompt"]):
                    st["pi"] += 1
                    if st["pi"] == len(st["prompt"]):
                        self._emit(b, row, tok)  # first generated token
                else:
                    self._emit(b, row, tok)


class GenerationService:
    """`trtlab.gen.Generation/Generate` — single request up, greedy token
    stream down, over the shared continuous-batching engine."""

    def __init__(self, session, eos: int = -1,
                 name: str = "trtlab.gen.Generation",
                 inline_step: bool = False, max_waiting: int = 0):
        from trtlab_amd.rpc.server import StreamingService

        self.engine = GenerationEngine(session, eos=eos,
                                       inline_step=inline_step,
                                       max_waiting=max_waiting)
        svc = StreamingService(name)
        svc.register_streaming("Generate", self._generate, GenerateRequest,
                               GenerateToken)
        self.service = svc

    async def _generate(self, request_iter, context, resources):
        self.engine.ensure_started()
        req = await request_iter.__anext__()  # single-up
        try:
            b, q = await self.engine.submit(
                list(req.prompt), req.max_tokens or 16,
                temperature=req.temperature, top_k=req.top_k,
                top_p=req.top_p, seed=req.seed)
        except ValueError as e:
            import grpc

            await context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(e))
            return
        except OverflowError as e:
            import grpc

            await context.abort(grpc.StatusCode.RESOURCE_EXHAUSTED, str(e))
            return
        i = 0
        try:
            while True:
                tok = await q.get()
                if tok is None:
                    yield GenerateToken(slot=b, index=i, done=True)
                    return
                yield GenerateToken(token=tok, slot=b, index=i)
                i += 1
        finally:
            # client gone mid-stream (cancellation / disconnect): free
            # the slot instead of generating into a dead queue
            self.engine.cancel(b, q)
