"""Continuous-batching serving engine for the in-house model family.

Iteration-level scheduling (the vLLM/Orca idea, built MI355X-first on
our own kernels — no reference counterpart, SURVEY §2.4 note):

* a fixed pool of KV-cache SLOTS (one preallocated batched cache, full
  stride — the varlen flash-decode kernel reads each slot at its own
  valid length, decode.hip);
* new requests are admitted BETWEEN decode steps: the prompt prefills
  into a free slot through the flash fwd kernel (a [slot:slot+1] view
  of the batched cache — zero copies);
* every step decodes ONE token for all active slots in a single batched
  kernel pass (slots at different positions — per-slot RoPE rows +
  per-slot cache scatter, models/llama.py decode_step);
* finished slots free immediately and the queue refills them, so
  throughput tracks the arrival rate instead of the slowest member of a
  static batch.

Greedy decoding is token-exact with ``model.generate`` per request.
"""

from collections import deque


class Request(object):
    _next_id = 0

    def __init__(self, prompt_tokens, max_new_tokens):
        self.id = Request._next_id
        Request._next_id += 1
        self.prompt = list(prompt_tokens)
        self.max_new = int(max_new_tokens)
        self.generated = []
        self.done = False


class _SlotView(object):
    """Single-slot view of the batched KV cache: the model's prefill
    path writes through it in place (views, no copies)."""

    def __init__(self, cache, slot):
        self.k = [t[slot:slot + 1] for t in cache.k]
        self.v = [t[slot:slot + 1] for t in cache.v]
        self.pos = 0


class ContinuousBatcher(object):
    def __init__(self, model, max_batch=8, max_len=2048, graph="auto",
                 prefill_chunk=None):
        """graph: capture the whole decode step in a hipGraph and replay
        it per token (decode is LAUNCH-bound — ~200 small kernels per
        step; replay collapses them to one submit). "auto" captures on
        GPU and falls back silently; the step stays host-read-free
        because the varlen kernel chunks from device lengths
        (decode.hip) and splits come from the cache capacity.

        prefill_chunk: bound per-step latency under long prompts — a
        newly admitted request prefills at most this many prompt tokens
        per step() instead of all at once, so active slots keep
        decoding every step instead of stalling behind a monolithic
        prefill (a slot mid-prefill sits at cache length
        positions[slot] and its decode row is masked by the varlen
        kernel's per-slot lengths). None = whole-prompt prefill at
        admission (lowest total latency when prompts are short)."""
        import torch

        from .models.llama import KVCache

        self.model = model
        self.max_batch = max_batch
        self.max_len = max_len
        device = next(model.parameters()).device
        self.device = device
        self.cache = KVCache(model.cfg, max_batch, max_len, device,
                             dtype=model.embed.weight.dtype)
        self.positions = [0] * max_batch    # cached tokens per slot
        self.slots = [None] * max_batch     # Request or None
        self.next_token = [0] * max_batch   # token to feed next step
        self.prefill_chunk = prefill_chunk
        self._views = [None] * max_batch    # _SlotView while prefilling
        self._prefill_done = [0] * max_batch  # prompt tokens cached
        self.queue = deque()
        self._torch = torch
        self._graph = None
        if graph is True and not getattr(model, "graph_safe_decode",
                                         False):
            raise ValueError(
                "model's decode_step is not hipGraph-safe (data-"
                "dependent shapes, e.g. MoE routing)")
        if graph and device.type == "cuda" and \
                getattr(model, "graph_safe_decode", False):
            try:
                self._capture_graph()
            except Exception:
                if graph is True:
                    raise
                self._graph = None  # auto: eager decode

    def _capture_graph(self):
        torch = self._torch
        self._tok_buf = torch.zeros(self.max_batch, 1, dtype=torch.long,
                                    device=self.device)
        self._pos_buf = torch.zeros(self.max_batch, dtype=torch.long,
                                    device=self.device)
        # warmup on a side stream (allocator primes its graph pool)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.model.decode_step(self._tok_buf, self.cache,
                                       self._pos_buf,
                                       max_len=self.max_len)
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._logits_buf = self.model.decode_step(
                self._tok_buf, self.cache, self._pos_buf,
                max_len=self.max_len)
        self._graph = g

    def _decode(self, tokens, positions):
        if self._graph is not None:
            torch = self._torch
            self._tok_buf.copy_(tokens)
            self._pos_buf.copy_(torch.as_tensor(positions,
                                                device=self.device))
            self._graph.replay()
            return self._logits_buf
        return self.model.decode_step(tokens, self.cache, positions)

    def submit(self, prompt_tokens, max_new_tokens):
        req = Request(prompt_tokens, max_new_tokens)
        self.queue.append(req)
        return req

    # ------------------------------------------------------------- internals
    def _admit(self):
        torch = self._torch
        for slot in range(self.max_batch):
            if self.slots[slot] is not None or not self.queue:
                continue
            req = self.queue.popleft()
            assert len(req.prompt) + req.max_new <= self.max_len, \
                "request longer than the slot capacity"
            self.slots[slot] = req
            if self.prefill_chunk is not None:
                # chunked: cache nothing yet; step() feeds chunks
                self._views[slot] = _SlotView(self.cache, slot)
                self._prefill_done[slot] = 0
                self.positions[slot] = 0
                continue
            prompt = torch.tensor([req.prompt], device=self.device)
            view = _SlotView(self.cache, slot)
            with torch.no_grad():
                logits = self.model(prompt, cache=view)
            nxt = int(logits[0, -1].float().argmax())
            req.generated.append(nxt)
            self.positions[slot] = len(req.prompt)
            self.next_token[slot] = nxt
            if req.max_new <= 1:
                self._finish(slot)

    def _prefill_step(self):
        """Advance every mid-prefill slot by one prompt chunk; a slot
        whose final chunk just ran emits its first token and joins the
        decode batch next step."""
        torch = self._torch
        for slot in range(self.max_batch):
            view = self._views[slot]
            if view is None:
                continue
            req = self.slots[slot]
            a = self._prefill_done[slot]
            b = min(a + self.prefill_chunk, len(req.prompt))
            chunk = torch.tensor([req.prompt[a:b]], device=self.device)
            with torch.no_grad():
                logits = self.model(chunk, cache=view)
            self._prefill_done[slot] = b
            if b < len(req.prompt):
                continue
            self._views[slot] = None
            nxt = int(logits[0, -1].float().argmax())
            req.generated.append(nxt)
            self.positions[slot] = len(req.prompt)
            self.next_token[slot] = nxt
            if req.max_new <= 1:
                self._finish(slot)

    def _finish(self, slot):
        self.slots[slot].done = True
        self.slots[slot] = None
        self.positions[slot] = 0
        self._views[slot] = None
        self._prefill_done[slot] = 0

    def step(self):
        """Admit waiting requests, advance mid-prefill slots by one
        chunk, then decode one token for every active slot in a single
        batched pass. Returns the number of active slots decoded."""
        torch = self._torch
        self._admit()
        if self.prefill_chunk is not None:
            self._prefill_step()
        active = [i for i, r in enumerate(self.slots) if r is not None
                  and self._views[i] is None]
        # the batched decode scatters a (garbage) K/V row for EVERY
        # slot at its position; a mid-prefill slot must aim that write
        # at its next-unwritten row — the following prefill chunk
        # overwrites it, so the cached prefix stays intact
        pos_list = list(self.positions)
        for i in range(self.max_batch):
            if self._views[i] is not None:
                pos_list[i] = self._prefill_done[i]
        if not active:
            return 0
        tokens = torch.zeros(self.max_batch, 1, dtype=torch.long,
                             device=self.device)
        for i in active:
            tokens[i, 0] = self.next_token[i]
        logits = self._decode(tokens, pos_list)
        for i in active:
            req = self.slots[i]
            nxt = int(logits[i, -1].float().argmax())
            req.generated.append(nxt)
            self.positions[i] += 1
            self.next_token[i] = nxt
            if len(req.generated) >= req.max_new:
                self._finish(i)
        return len(active)

    def run(self):
        """Drain the queue; returns {request_id: generated tokens}."""
        results = {}
        pending = list(self.queue)
        while self.queue or any(r is not None for r in self.slots):
            self.step()
        for req in pending:
            results[req.id] = req.generated
        return results
