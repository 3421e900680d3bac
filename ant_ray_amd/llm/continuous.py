"""Continuous (token-level) batching for the native LLM engine.

Role parity: vLLM's continuous batching scheduler, which the reference
delegates all serving to (reference python/ray/llm/_internal/serve/
engines/vllm/vllm_engine.py:1). MI355X-first design: a fixed-slot
decoder over ONE shared KV cache — the ragged flash-decode kernel
(csrc/kernels/attention_decode.hip) already takes per-sequence lens from
a device buffer, so slots at different positions decode together in one
kernel, and the whole token step stays hipGraph-captured (the graph is
batch-shape-fixed; admission/retire happen eagerly between replays).
New requests are prefilled into a free slot's rows of the cache (a
row-sliced view) while the other slots keep decoding; no request waits
for a batch-mate to finish.

Greedy decoding only (the captured step argmaxes on-device), like
GraphedDecoder. CPU runs the same path eagerly via the ops fallbacks,
so the scheduler is fully testable without a GPU.
"""
from __future__ import annotations

import itertools
import threading
from concurrent.futures import Future
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch


class _RowCache:
    """One slot's rows of the shared KVCache (write-through views)."""

    def __init__(self, cache, row: int):
        self.k = cache.k[:, row : row + 1]
        self.v = cache.v[:, row : row + 1]
        self.max_seq = cache.max_seq
        self.pos = 0

    def layer(self, i: int):
        return self.k[i], self.v[i]


@dataclass
class _Request:
    rid: int
    prompt: List[int]
    max_new: int
    future: Future = field(default_factory=Future)
    row: int = -1
    start: int = -1        # trace index of the first emitted token
    emitted: int = 0
    stop_ids: Optional[frozenset] = None  # terminate early on these
    on_token: Optional[object] = None     # streaming callback(token)
    # host-side tokens, filled only when stop_ids/on_token force a
    # per-tick readback for this request
    tokens: List[int] = field(default_factory=list)

    @property
    def eager_host(self) -> bool:
        return self.stop_ids is not None or self.on_token is not None


class SlotDecoder:
    """Fixed-slot token stepper over one shared KV cache.

    step() advances EVERY slot by one token (idle slots decode garbage
    into their own rows — clamped in-bounds and never surfaced). On GPU
    the step is hipGraph-captured after two eager warmups; admission
    (prefill + lens/cur writes) happens eagerly between replays.
    """

    def __init__(self, model, slots: int, max_seq: int, device):
        from ant_ray_amd.models.llama import KVCache

        self.model = model
        self.slots = slots
        self.max_seq = max_seq
        self.device = device
        self.cache = KVCache(model.cfg, slots, max_seq, device)
        self.cache.k.zero_()  # idle slots attend over zeros, not NaNs
        self.cache.v.zero_()
        self.lens = torch.ones(slots, dtype=torch.int32, device=device)
        self.cur = torch.zeros(slots, 1, dtype=torch.long, device=device)
        self.graph = None
        self._eager_steps = 0

    def prefill(self, row: int, tokens: List[int],
                prefix_cache=None) -> None:
        toks = torch.tensor([tokens], dtype=torch.long, device=self.device)
        rv = _RowCache(self.cache, row)
        hit = 0
        if prefix_cache is not None:
            hit, blocks = prefix_cache.lookup(tokens[: len(tokens) - 1])
            if hit:
                prefix_cache.seed_row(self.cache, blocks, row)
        with torch.no_grad():
            logits = self.model.forward(
                toks[:, hit:] if hit else toks, cache=rv, pos=hit)
        self.cur[row] = logits.argmax(-1)
        self.lens[row] = len(tokens)
        if prefix_cache is not None:
            prefix_cache.insert(tokens, self.cache, row=row)

    def _step(self):
        lg = self.model.forward(self.cur, cache=self.cache, lens=self.lens)
        self.cur.copy_(lg.argmax(dim=-1, keepdim=True))

    @torch.no_grad()
    def step(self) -> None:
        use_graph = (str(self.device).startswith("cuda")
                     and self.model.cfg.head_dim == 128)
        if self.graph is not None:
            self.graph.replay()
            return
        if use_graph and self._eager_steps >= 2:
            g = torch.cuda.CUDAGraph()
            torch.cuda.synchronize()
            with torch.cuda.graph(g):
                # idle slots keep stepping; clamp keeps their reads
                # in-bounds until they are re-admitted
                self.lens.add_(1).clamp_(max=self.max_seq)
                self._step()
            self.graph = g
            g.replay()
            return
        self._eager_steps += 1
        self.lens.add_(1).clamp_(max=self.max_seq)
        self._step()


class ContinuousLLMEngine:
    """submit() returns a Future; a pump (caller-driven or background
    thread) admits queued requests into free slots and steps all slots
    together. Token traces stay on-device; one D2H gather per request
    at retire time."""

    def __init__(self, model_name: str, slots: int = 8, max_seq: int = 4096,
                 seed: int = 0, device: str = "cuda",
                 start_thread: bool = False):
        from ant_ray_amd.models import build_model, setup_tunableop
        from ant_ray_amd.models.llama import LlamaForCausalLM

        if str(device).startswith("cuda"):
            setup_tunableop()
        torch.manual_seed(seed)
        self.model = build_model(model_name, device=device, seq_len=max_seq)
        if not isinstance(self.model, LlamaForCausalLM):
            raise ValueError("continuous batching requires a KV-cache "
                             "model family (llama)")
        self.model.eval()
        self.device = device
        self.max_seq = max_seq
        self.dec = SlotDecoder(self.model, slots, max_seq, device)
        from ant_ray_amd.llm.prefix_cache import prefix_cache_from_env

        self.prefix_cache = prefix_cache_from_env()
        self._ids = itertools.count()
        self._lock = threading.Lock()
        self._queue: List[_Request] = []
        self._active: Dict[int, _Request] = {}   # row -> request
        self._trace: List[torch.Tensor] = []     # per-step cur clones
        self._trace_base = 0                     # steps pruned off
        self._wake = threading.Event()
        self._stop = False
        self._thread: Optional[threading.Thread] = None
        self.steps = 0
        if (str(device).startswith("cuda")
                and self.model.cfg.head_dim == 128):
            # pre-capture the step graph (the process's first hipGraph
            # instantiation costs ~0.7 s — pay it at init, not on the
            # first request)
            try:
                self.dec.prefill(0, [1, 2, 3])
                for _ in range(3):
                    self.dec.step()
                self.dec.lens.fill_(1)
            except Exception:
                pass
        if start_thread:
            self._thread = threading.Thread(target=self._pump_loop,
                                            daemon=True)
            self._thread.start()

    # ------------------------------------------------------------ submit
    def submit(self, prompt_ids: List[int], max_new_tokens: int,
               stop_token_ids=None, on_token=None) -> Future:
        """stop_token_ids: terminate the request early when one is
        produced (the stop token is included in the output). on_token:
        per-token streaming callback, called from the pump thread.
        Either option switches the request to per-tick host readback
        (one batched D2H per step while any such request is active)."""
        if len(prompt_ids) + max_new_tokens > self.max_seq:
            raise ValueError("prompt + max_new_tokens exceeds max_seq")
        req = _Request(next(self._ids), list(prompt_ids),
                       int(max_new_tokens),
                       stop_ids=(frozenset(stop_token_ids)
                                 if stop_token_ids else None),
                       on_token=on_token)
        with self._lock:
            self._queue.append(req)
        self._wake.set()
        return req.future

    # -------------------------------------------------------------- pump
    def _admit(self) -> None:
        while True:
            with self._lock:
                if not self._queue:
                    return
                free = [r for r in range(self.dec.slots)
                        if r not in self._active]
                if not free:
                    return
                req = self._queue.pop(0)
            row = free[0]
            self.dec.prefill(row, req.prompt, self.prefix_cache)
            req.row = row
            req.start = self._trace_base + len(self._trace)
            req.emitted = 0
            self._active[row] = req

    def _retire(self, req: _Request) -> None:
        del self._active[req.row]
        if len(req.tokens) == req.emitted:  # host copy already complete
            req.future.set_result(list(req.tokens))
        else:
            lo = req.start - self._trace_base
            cols = torch.cat(self._trace[lo : lo + req.emitted], dim=1)
            req.future.set_result(cols[req.row].tolist())
        # prune trace entries no active request still needs
        if self._active:
            keep = min(r.start for r in self._active.values())
        else:
            keep = self._trace_base + len(self._trace)
        drop = keep - self._trace_base
        if drop > 0:
            del self._trace[:drop]
            self._trace_base = keep

    def pump(self) -> bool:
        """One scheduler tick: admit, record, retire, step. Returns True
        if any work remains."""
        self._admit()
        if not self._active:
            return False
        self._trace.append(self.dec.cur.clone())
        host_cur = None
        if any(r.eager_host for r in self._active.values()):
            # one batched D2H for every streaming/stop-scanning request
            host_cur = self._trace[-1].squeeze(1).tolist()
        for req in list(self._active.values()):
            req.emitted += 1
            stop_hit = False
            if req.eager_host:
                tok = host_cur[req.row]
                req.tokens.append(tok)
                if req.on_token is not None:
                    try:
                        req.on_token(tok)
                    except Exception:
                        pass
                stop_hit = (req.stop_ids is not None
                            and tok in req.stop_ids)
            if stop_hit or req.emitted >= req.max_new:
                self._retire(req)
        # freed slots admit at the NEXT tick's _admit: a row admitted
        # here would lose its first token (step() advances cur before
        # the next trace append records it)
        if self._active:
            self.dec.step()
            self.steps += 1
        return bool(self._active or self._queue)

    def run_until_idle(self) -> None:
        while self.pump():
            pass

    def _pump_loop(self) -> None:
        while not self._stop:
            if not self.pump():
                self._wake.wait(timeout=0.005)
                self._wake.clear()

    def shutdown(self) -> None:
        self._stop = True
        self._wake.set()
        if self._thread is not None:
            self._thread.join(timeout=5)

    def stats(self) -> dict:
        with self._lock:
            return {"steps": self.steps, "active": len(self._active),
                    "queued": len(self._queue),
                    "graph_captured": self.dec.graph is not None}
