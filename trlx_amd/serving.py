"""Continuous batching for generation serving.

The RL rollout engine (models/nn/generation.py DecodeEngine) generates a
fixed, aligned batch — the right shape for PPO experience collection.  A
serving workload is different: requests arrive at different times with
different lengths, and a fixed-batch engine either waits to fill a batch or
runs underfilled.  `ContinuousBatcher` keeps a pool of KV-cache SLOTS and
admits/retires requests per token step:

  - each slot owns a row of the shared KV cache and decodes at its OWN depth
    (AttentionContext.cache_rows — per-row cache append positions);
  - admission runs a single-row prefill into the slot's cache rows (no
    padding: serving prompts are never left-padded);
  - every step is ONE batched forward over all slots (the same fused
    decode kernels as the RL path: qkv_prep + attention_decode with per-row
    seq_lens), finished slots free up immediately for queued requests.

The reference has no serving story (its hh example calls an external Triton
server); this plus serve.py closes the deploy loop for `save_pretrained`
checkpoints.  The per-step forward runs eagerly (not graph-captured: slot
state changes shape-free but admission rewrites host state between steps);
graph capture of the steady-state step is an optimization left open.
"""

import collections
import os
import threading
from concurrent.futures import Future
from typing import List, Optional

import torch

from . import ops
from .models.nn.generation import GenerateConfig, _ENGINES


class _SlotView:
    """KV-cache view of one slot row: lets a [1, T] prefill write into the
    shared [B, ...] cache through the normal KVCache.update protocol."""

    def __init__(self, kv, b: int):
        self.kv = kv
        self.b = b
        self.max_len = kv.max_len

    def update(self, layer: int, k, v, start: int):
        T = k.shape[2]
        self.kv.k[layer][self.b : self.b + 1, :, start : start + T] = k
        self.kv.v[layer][self.b : self.b + 1, :, start : start + T] = v
        return (self.kv.k[layer][self.b : self.b + 1],
                self.kv.v[layer][self.b : self.b + 1])


class _Slot:
    __slots__ = ("active", "future", "tokens", "seq_len", "remaining", "temperature",
                 "greedy")

    def __init__(self):
        self.active = False
        self.future = None
        self.tokens = []
        self.seq_len = 1
        self.remaining = 0
        self.temperature = 1.0
        self.greedy = True


class ContinuousBatcher:
    """Slot-pool continuous batching over a CausalTransformer.

    submit() returns a Future resolving to the generated token ids (prompt
    excluded).  Drive with step()/run_until_idle() (tests, synchronous use)
    or start()/close() for a background pump thread (the HTTP server).
    """

    def __init__(self, model, slots: int = 8, cache_len: int = 1024,
                 gen: Optional[GenerateConfig] = None, device=None):
        self.model = model.eval()
        self.B = slots
        self.cache_len = cache_len
        self.gen = gen or GenerateConfig(do_sample=False)
        p = next(model.parameters())
        self.device = device or p.device
        self.kv = model.new_kv_cache(slots, cache_len, device=self.device)
        self.slots = [_Slot() for _ in range(slots)]
        self.pending = collections.deque()
        self.cur_tok = torch.zeros(slots, 1, dtype=torch.long, device=self.device)
        self.lock = threading.Lock()
        self.step_count = 0
        self.seed = self.gen.seed if self.gen.seed is not None else 0
        self._thread = None
        self._stop = False
        # graph-captured step state (GPU): the whole per-token step — forward
        # + sampling + per-slot position/row/length advance — replays as ONE
        # hipGraph; admission rewrites these persistent tensors between
        # replays (same discipline as DecodeEngine)
        self.pos = torch.ones(slots, 1, dtype=torch.int32, device=self.device)
        self.rows = torch.zeros(slots, dtype=torch.long, device=self.device)
        self.lens = torch.ones(slots, dtype=torch.int32, device=self.device)
        self.active_i = torch.zeros(slots, dtype=torch.int32, device=self.device)
        self.ks = torch.zeros(slots, dtype=torch.int32, device=self.device)
        self.rng_offset = torch.zeros(1, dtype=torch.long, device=self.device)
        # per-request sampling state (per-row temperature + greedy mask stay
        # graph-safe: one elementwise scale + one where-select)
        self.inv_temp = torch.ones(slots, 1, device=self.device)
        self.greedy_m = torch.ones(slots, 1, dtype=torch.bool, device=self.device)
        self.graph = None
        self._use_graph = (self.device.type == "cuda" and ops.extension_available()
                           and os.environ.get("TRLX_AMD_NO_GRAPHS") != "1")
        _ENGINES.add(self)  # trlx_amd.release_graphs() tears the step graph down

    # ---- client API -------------------------------------------------------

    def submit(self, input_ids, max_new_tokens: int = 40,
               temperature: Optional[float] = None,
               do_sample: Optional[bool] = None) -> Future:
        """``temperature``/``do_sample`` default to the batcher's
        GenerateConfig; per-request overrides ride the same batched step."""
        ids = torch.as_tensor(input_ids, dtype=torch.long).view(-1)
        if ids.numel() + max_new_tokens > self.cache_len:
            raise ValueError(
                f"prompt ({ids.numel()}) + max_new_tokens ({max_new_tokens}) exceeds "
                f"the slot cache length {self.cache_len}")
        temp = self.gen.temperature if temperature is None else float(temperature)
        sample = self.gen.do_sample if do_sample is None else bool(do_sample)
        fut = Future()
        with self.lock:
            self.pending.append((ids, max_new_tokens, temp, sample, fut))
        if self._thread is not None:
            self._wake.set()
        return fut

    def run_until_idle(self):
        """Synchronously pump until every submitted request resolved."""
        while self._pump():
            pass

    def start(self):
        self._wake = threading.Event()
        self._stop = False
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def close(self):
        if self._thread is not None:
            self._stop = True
            self._wake.set()
            self._thread.join(timeout=10)
            self._thread = None

    def _loop(self):
        while not self._stop:
            if not self._pump():
                self._wake.wait(timeout=0.05)
                self._wake.clear()

    # ---- engine -----------------------------------------------------------

    def _free_slot(self):
        for i, s in enumerate(self.slots):
            if not s.active:
                return i
        return None

    def _admit(self, b: int, ids: torch.Tensor, max_new: int, temp: float, sample: bool,
               fut: Future):
        slot = self.slots[b]
        ids = ids.to(self.device).unsqueeze(0)
        T = ids.shape[1]
        with torch.no_grad():
            out = self.model(ids, attention_mask=torch.ones_like(ids),
                             kv_cache=_SlotView(self.kv, b), start_pos=0,
                             return_logits=False)
            logits = self.model.lm_head(out.last_hidden_state[:, -1:, :])[:, 0].float()
        slot.temperature = temp
        slot.greedy = (not sample) or temp == 0.0
        self.inv_temp[b, 0] = 1.0 if slot.greedy else 1.0 / max(temp, 1e-6)
        self.greedy_m[b, 0] = slot.greedy
        if slot.greedy:
            tok = logits.argmax(dim=-1)[0]
        else:
            self.step_count += 1
            tok = ops.sample_token(logits / max(temp, 1e-6), 1.0, self.gen.top_k,
                                   self.gen.top_p, seed=self.seed,
                                   offset=self.step_count)[0]
        slot.active = True
        slot.future = fut
        slot.tokens = [int(tok)]
        slot.seq_len = T
        slot.remaining = max_new - 1
        self.cur_tok[b, 0] = tok
        self._maybe_finish(b)

    def _sample(self, logits):
        if not self.gen.do_sample:
            return logits.argmax(dim=-1)
        self.step_count += 1
        return ops.sample_token(logits, self.gen.temperature, self.gen.top_k,
                                self.gen.top_p, seed=self.seed, offset=self.step_count)

    def _maybe_finish(self, b: int):
        slot = self.slots[b]
        eos = self.gen.eos_token_id
        if slot.remaining <= 0 or (eos is not None and slot.tokens[-1] == eos):
            slot.active = False
            fut, toks = slot.future, slot.tokens
            slot.future = None
            fut.set_result(toks)

    def _pump(self) -> bool:
        """Admit what fits, then one decode step.  Returns True if work
        remains (active slots or queued requests).

        The lock guards only the pending queue: every slot mutation happens
        in the single pump thread, so the (long) model forwards run outside
        the lock and submit() never waits on a decode step."""
        while True:
            with self.lock:
                b = self._free_slot()
                if not self.pending or b is None:
                    break
                ids, max_new, temp, sample, fut = self.pending.popleft()
            self._admit(b, ids, max_new, temp, sample, fut)
        active = [i for i, s in enumerate(self.slots) if s.active]
        if not active:
            with self.lock:
                return bool(self.pending)
        self._step(active)
        return True

    def _sync_slot_state(self, active: List[int]):
        """Host -> persistent device tensors (between graph replays)."""
        pos = torch.ones(self.B, dtype=torch.int32)
        rows = torch.zeros(self.B, dtype=torch.long)
        lens = torch.ones(self.B, dtype=torch.int32)
        act = torch.zeros(self.B, dtype=torch.int32)
        for i in active:
            pos[i] = self.slots[i].seq_len
            rows[i] = self.slots[i].seq_len
            lens[i] = self.slots[i].seq_len + 1
            act[i] = 1
        self.pos.copy_(pos.unsqueeze(1))
        self.rows.copy_(rows)
        self.lens.copy_(lens)
        self.active_i.copy_(act)

    def _step_body(self):
        """One decode step over all slots, expressed entirely in device ops
        (hipGraph-capturable): forward + sample + per-slot state advance."""
        out = self.model(self.cur_tok, kv_cache=self.kv, start_pos=0,
                         position_ids=self.pos, seq_lens=self.lens,
                         key_starts=self.ks, cache_rows=self.rows,
                         return_logits=False)
        logits = self.model.lm_head(out.last_hidden_state[:, -1:, :])[:, 0].float()
        # per-request sampling: row temperature folds into the logits, the
        # greedy mask selects argmax rows — both graph-safe
        sampled = ops.sample_token(logits * self.inv_temp, 1.0, self.gen.top_k,
                                   self.gen.top_p, seed=self.seed,
                                   offset=self.rng_offset)
        toks = torch.where(self.greedy_m[:, 0], logits.argmax(dim=-1), sampled)
        self.cur_tok.copy_(toks.unsqueeze(1))
        adv = self.active_i
        self.pos.add_(adv.unsqueeze(1))
        self.rows.add_(adv.long())
        self.lens.add_(adv)
        self.rng_offset.add_(1)

    def _capture(self):
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                with torch.no_grad():
                    self._step_body()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph), torch.no_grad():
            self._step_body()

    def _step(self, active: List[int]):
        # per-slot depths: token at position seq_len, cache row seq_len,
        # attention over rows [0, seq_len]
        self._sync_slot_state(active)
        if self._use_graph:
            if self.graph is None:
                # warmup/capture run the step 3x for real state: they write
                # FUTURE cache rows (overwritten before ever read) and
                # clobber cur_tok/pos — restore both, then replay this step
                # for real
                self._capture()
                self._sync_slot_state(active)
                cur = torch.zeros(self.B, dtype=torch.long)
                for i in active:
                    cur[i] = self.slots[i].tokens[-1]
                self.cur_tok.copy_(cur.unsqueeze(1))
            self.graph.replay()
        else:
            with torch.no_grad():
                self._step_body()
        toks = self.cur_tok[:, 0].tolist()
        for i in active:
            slot = self.slots[i]
            slot.tokens.append(int(toks[i]))
            slot.seq_len += 1
            slot.remaining -= 1
            self._maybe_finish(i)
