"""Continuous-batching serving engine (linear or paged KV, with
prefix caching on the paged path).

The reference defers serving orchestration to vLLM and ships the model-side
hooks (continuous batching seq_ids, block KV, chunked prefill — SURVEY
§2.4/§2.6); this engine closes the loop natively: a request queue, slot
allocation over the KV cache's seq_id rows, and a step loop that
interleaves one prefill (optionally chunked through the windowed-CTE path)
with full-batch decode steps.

Scheduling policy (deliberately simple, latency-fair):
- if a slot is free and a request is queued: run that request's context
  encoding (chunked when `chunk_size` is set — each chunk is one step, so
  decode latency of running requests is bounded by one chunk);
- otherwise: one token-generation step for every active slot, batched and
  padded to the engine batch (seq_id -1 pads; the captured decode graph
  serves every step regardless of how many slots are live).

All state (tokens, positions) stays on the model's device between steps.
"""

from __future__ import annotations

import collections
import time
from dataclasses import dataclass, field
from typing import Deque, Dict, List, Optional

import torch


@dataclass
class Request:
    req_id: int
    prompt: torch.Tensor            # (S,) input ids
    max_new_tokens: int = 32
    top_k: int = 1
    top_p: float = 1.0
    temperature: float = 1.0
    adapter_id: int = 0             # multi-LoRA slot (0 = no adapter)
    output: List[int] = field(default_factory=list)
    slot: int = -1
    pos: int = 0                    # next position to fill
    prefill_done: int = 0           # prompt tokens already encoded
    last_token: Optional[int] = None
    done: bool = False
    blocks: List[int] = field(default_factory=list)  # paged: block chain
    n_cached: int = 0               # prompt tokens served from prefix cache
    t_submit: float = 0.0           # metrics timestamps (monotonic)
    t_first_token: float = 0.0
    t_finish: float = 0.0


class ServingEngine:
    def __init__(self, causal_lm, max_slots: Optional[int] = None,
                 chunk_size: Optional[int] = None,
                 eos_token_id: Optional[int] = None,
                 capture_decode: Optional[bool] = None,
                 on_token=None, lora_state=None, bucket_chunks: bool = True,
                 async_depth: Optional[int] = None,
                 mixed_chunking: bool = False):
        self.model = causal_lm
        ec = causal_lm.config.neuron_config
        self.batch = ec.batch_size
        self.max_slots = max_slots or ec.batch_size
        self.chunk_size = chunk_size
        self.eos = eos_token_id
        self.device = next(causal_lm.parameters()).device
        self.kv_mgr = causal_lm.model.kv_mgr
        self.paged = bool(getattr(self.kv_mgr, "is_paged", False))
        if self.paged:
            from neuronx_distributed_inference_amd.modules.kvcache.prefix_cache import (
                BlockAllocator,
                PrefixCache,
            )

            self.block_size = self.kv_mgr.block_size
            self.allocator = BlockAllocator(self.kv_mgr.num_blocks)
            self.prefix_cache = PrefixCache(self.allocator, self.block_size)
            self.max_blocks = (ec.max_length + self.block_size - 1) // self.block_size
        self.free_slots = list(range(self.max_slots))[::-1]
        self.active: Dict[int, Request] = {}    # slot -> request
        self.queue: Deque[Request] = collections.deque()
        self.finished: Dict[int, Request] = {}
        self._next_id = 0
        self.on_token = on_token        # callback(req_id, token) per token
        self.lora_state = lora_state    # multi-LoRA: per-request adapter rows
        # pad prefill chunks up a power-of-2 ladder (pad rows carry
        # position -1, so the pad-skip invariant keeps them out of the KV
        # cache): bounds the set of [active, prefix] shapes the engine
        # presents, the analog of the reference's 2-D prefix buckets
        # (ref model_wrapper.py:923) — and what makes tail-prefill
        # autotune/capture shapes finite
        self.bucket_chunks = bucket_chunks
        # hipGraph-captured decode step (paged serving would otherwise run
        # its full-batch decode eagerly; linear serving uses the model's
        # own captured wrappers)
        if capture_decode is None:
            capture_decode = self.paged and self.device.type == "cuda"
        self._capture_decode = capture_decode and self.device.type == "cuda"
        self._decode_graph = None
        # async double-buffered decode (ref modules/async_execution.py +
        # _get_model_outputs_async, model_base.py:3737): the decode step's
        # token feedback stays ON DEVICE (self._slot_tok), host readback
        # of step i's tokens happens after step i+1 is launched, so the
        # per-token host<->device round trip hides under device execution.
        from neuronx_distributed_inference_amd.modules.async_execution import (
            AsyncExecutor,
        )

        if async_depth is None:
            async_depth = 2 if self.device.type == "cuda" else 1
        self.async_depth = max(1, async_depth)
        self._async = AsyncExecutor(max_inflight=self.async_depth,
                                    device=self.device)
        self._slot_tok = torch.zeros(self.max_slots, dtype=torch.long,
                                     device=self.device)
        self._inflight: Deque = collections.deque()  # (slots, AsyncResult)
        # mixed chunked prefill (ref chunked-prefill tile buckets,
        # autobucketing.py:101-146 / kvcache/utils.py:209-506): one
        # multi-token TKG step carries a prefill CHUNK row alongside the
        # decoding rows (their single token + -1 pads), so decode latency
        # never stalls behind a prefill chunk. Greedy engines only (the
        # mixed pass samples per position greedily).
        self.mixed_chunking = bool(mixed_chunking) and chunk_size is not None

    # ------------------------------------------------------------------
    def add_request(self, prompt_ids: torch.Tensor,
                    max_new_tokens: int = 32, top_k: int = 1,
                    top_p: float = 1.0, temperature: float = 1.0,
                    adapter_id: int = 0) -> int:
        # per-request on-device sampling params ride in the (B,3)
        # sampling_params rows; per-request LoRA adapters in the LoraState
        # adapter_ids rows (ref per-request sampling_params / adapter_ids)
        rid = self._next_id
        self._next_id += 1
        # a static-greedy sampler (no on-device sampling config, or one with
        # do_sample=False and dynamic=False) ignores the per-request (B,3)
        # sampling_params rows — refuse silently-greedy sampling requests
        if top_k != 1 or top_p != 1.0 or temperature != 1.0:
            sc = self.model.config.neuron_config.on_device_sampling_config
            if sc is None or (not getattr(sc, "do_sample", False)
                              and not getattr(sc, "dynamic", False)):
                raise ValueError(
                    "request asks for sampling (top_k=%s top_p=%s temperature=%s) "
                    "but the engine's sampler is static-greedy; configure "
                    "OnDeviceSamplingConfig(do_sample=True) or dynamic=True"
                    % (top_k, top_p, temperature))
        self.queue.append(Request(rid, prompt_ids.view(-1).cpu(),
                                  max_new_tokens, top_k, top_p, temperature,
                                  adapter_id))
        self.queue[-1].t_submit = time.monotonic()
        return rid

    def _set_adapters(self, reqs):
        if self.lora_state is None:
            return
        ids = torch.zeros(self.batch, dtype=torch.long)
        for i, r in enumerate(reqs):
            ids[i] = r.adapter_id
        self.lora_state.set(ids)

    def _params_for(self, reqs) -> Optional[torch.Tensor]:
        if self.model.config.neuron_config.on_device_sampling_config is None:
            return None
        # sized to the live requests: the wrapper pads to its batch rung
        B = max(len(reqs), 1)
        p = torch.ones(B, 3, dtype=torch.float32, device=self.device)
        for i, r in enumerate(reqs):
            p[i, 0] = r.top_k
            p[i, 1] = r.top_p
            p[i, 2] = r.temperature
        return p

    def has_work(self) -> bool:
        return bool(self.queue or self.active or self._inflight)

    # ------------------------------------------------------------------
    def _ensure_blocks(self, req: Request, upto_pos: int):
        """Grow the request's block chain to cover position `upto_pos`,
        evicting cache-only blocks under pressure."""
        need = (upto_pos + self.block_size) // self.block_size - len(req.blocks)
        if need > 0:
            if self.allocator.available() < need:
                self.prefix_cache.evict(want_blocks=need)
            req.blocks.extend(self.allocator.alloc(need))

    def _block_table(self, reqs) -> torch.Tensor:
        bt = torch.full((len(reqs), self.max_blocks), -1, dtype=torch.long,
                        device=self.device)
        for i, r in enumerate(reqs):
            if r.blocks:
                bt[i, : len(r.blocks)] = torch.tensor(r.blocks,
                                                      device=self.device)
        return bt

    def _prefill_chunk(self, req: Request):
        S = req.prompt.shape[0]
        if self.paged and req.prefill_done == 0 and not req.blocks:
            # prefix-cache hit: shared FULL blocks skip their prefill
            req.blocks = self.prefix_cache.match(req.prompt.tolist())
            req.n_cached = len(req.blocks) * self.block_size
            # never skip the whole prompt: the last token must run to
            # produce logits
            if req.n_cached >= S:
                drop = req.blocks.pop()
                self.allocator.deref([drop])
                req.n_cached -= self.block_size
            req.prefill_done = req.n_cached
        lo = req.prefill_done
        hi = S if self.chunk_size is None else min(S, lo + self.chunk_size)
        ids = req.prompt[lo:hi].view(1, -1).to(self.device)
        positions = torch.arange(lo, hi, device=self.device).view(1, -1)
        L = hi - lo
        # only INTERMEDIATE chunks are padded: their logits are discarded
        # (just the pad-skipped KV writes matter). The final chunk must
        # stay ragged — the CTE last-token gather reads the last row
        if self.bucket_chunks and hi < S:
            Lp = 1
            while Lp < L:
                Lp *= 2
            if self.chunk_size is not None:
                Lp = min(Lp, self.chunk_size)
            if Lp > L:
                pad = Lp - L
                ids = torch.cat(
                    [ids, torch.zeros(1, pad, dtype=ids.dtype,
                                      device=self.device)], dim=1)
                positions = torch.cat(
                    [positions, torch.full((1, pad), -1, dtype=positions.dtype,
                                           device=self.device)], dim=1)
        seq_ids = torch.tensor([req.slot], device=self.device)
        sp = self._params_for([req])
        self._set_adapters([req])
        kw = {}
        if self.paged:
            self._ensure_blocks(req, hi - 1)
            kw["block_table"] = self._block_table([req])
            if lo > 0:
                # continuation chunk over prior (possibly shared) KV must
                # run the multi-token TKG path, like windowed CTE
                kw["phase"] = "token_generation"
        out = self.model(ids, None, positions, seq_ids,
                         None if sp is None else sp[:1], **kw)
        req.prefill_done = hi
        if hi == S:
            lg = out.logits if out.logits is not None else out.tokens
            tok = int(lg[0, -1].argmax()) if lg.dim() == 3 \
                else int(lg.view(1, -1)[0, -1])
            req.last_token = tok
            self._slot_tok[req.slot] = tok
            req.pos = S
            req.output.append(tok)
            req.t_first_token = time.monotonic()
            if self.on_token is not None:
                self.on_token(req.req_id, tok)
            self._maybe_finish(req, tok)

    def _decode_step(self):
        """Launch one batched decode step. Token feedback for the NEXT
        step is read from the device-side self._slot_tok (no host sync);
        host bookkeeping of this step's tokens is deferred until after
        the next launch (async double-buffering — ref
        modules/async_execution.py, model_base.py:3737)."""
        slots = sorted(self.active)
        # linear path: the batch stays RAGGED (len(slots) rows) so the
        # ModelWrapper pads only to the smallest captured batch rung
        # (partial occupancy pays sub-bucket cost). The paged captured
        # graph has a fixed batch: pad to it there.
        B = self.batch if (self._capture_decode and self.paged) else len(slots)
        pos = torch.full((B, 1), -1, dtype=torch.long, device=self.device)
        seq_ids = torch.full((B,), -1, dtype=torch.long, device=self.device)
        for i, slot in enumerate(slots):
            r = self.active[slot]
            pos[i, 0] = r.pos
            seq_ids[i] = slot
        slot_idx = torch.tensor(slots, dtype=torch.long, device=self.device)
        reqs = [self.active[s] for s in slots]
        self._set_adapters(reqs)
        kw = {}
        if self.paged:
            for r in reqs:
                self._ensure_blocks(r, r.pos)
            full = torch.full((B, self.max_blocks), -1, dtype=torch.long,
                              device=self.device)
            full[: len(reqs)] = self._block_table(reqs)
            kw["block_table"] = full
        sp = self._params_for(reqs)
        if sp is not None and sp.shape[0] < B:
            # paged captured graph has a fixed batch: pad the live rows
            sp = torch.cat([sp, torch.ones(B - sp.shape[0], 3,
                                           dtype=sp.dtype,
                                           device=sp.device)], dim=0)

        def run():
            # the _slot_tok gather MUST happen on the executor stream:
            # the previous step's scatter into it ran there, and the two
            # are ordered only within that stream
            ids = self._slot_tok.index_select(0, slot_idx).view(-1, 1)
            if ids.shape[0] < B:
                ids = torch.cat([ids, torch.zeros(B - ids.shape[0], 1,
                                                  dtype=torch.long,
                                                  device=self.device)], dim=0)
            if self._capture_decode and self.paged:
                toks = self._run_decode_graph(ids, pos, seq_ids, sp,
                                              kw["block_table"])
            else:
                out = self.model(ids, None, pos, seq_ids, sp, **kw)
                toks = out.logits if out.logits is not None else out.tokens
            if toks.dim() > 2:
                toks = toks.argmax(-1)
            toks = toks.view(B, -1)[:, -1].clone()
            # device-side feedback for the next step
            self._slot_tok.scatter_(0, slot_idx, toks[: len(slots)])
            return toks

        res = self._async.submit(run)
        for r in reqs:
            r.pos += 1  # launch-time advance (host-known)
        self._inflight.append((slots, res))
        self._drain_inflight(keep=self.async_depth - 1)

    def _drain_inflight(self, keep: int = 0):
        """Process deferred decode results (host bookkeeping: outputs,
        callbacks, finish checks) down to `keep` outstanding."""
        while len(self._inflight) > keep:
            slots, res = self._inflight.popleft()
            toks = res.result().tolist()
            for i, slot in enumerate(slots):
                r = self.active.get(slot)
                if r is None or r.done:
                    continue  # finished meanwhile; extra token discarded
                tok = int(toks[i])
                r.last_token = tok
                r.output.append(tok)
                if self.on_token is not None:
                    self.on_token(r.req_id, tok)
                self._maybe_finish(r, tok)

    def _run_decode_graph(self, ids, pos, seq_ids, sp, block_table):
        from neuronx_distributed_inference_amd.runtime.graph_engine import (
            GraphRunner,
        )

        dm = self.model.model

        if self._decode_graph is None:
            if sp is None:
                def fn(i, p, s, bt):
                    t, lg = dm(i, None, p, s, None,
                               phase="token_generation", block_table=bt)
                    return t if t is not None else lg
                ex = (ids, pos, seq_ids, block_table)
            else:
                def fn(i, p, s, smp, bt):
                    t, lg = dm(i, None, p, s, smp,
                               phase="token_generation", block_table=bt)
                    return t if t is not None else lg
                ex = (ids, pos, seq_ids, sp, block_table)
            self._decode_graph = GraphRunner(fn, ex)
            self._decode_graph_has_sp = sp is not None
        args = (ids, pos, seq_ids) + ((sp,) if self._decode_graph_has_sp else ()) \
            + (block_table,)
        return self._decode_graph(*args).clone()

    def _maybe_finish(self, req: Request, tok: int):
        if len(req.output) >= req.max_new_tokens or (
            self.eos is not None and tok == self.eos
        ):
            req.done = True
            req.t_finish = time.monotonic()
            self.active.pop(req.slot)
            self.free_slots.append(req.slot)
            self.finished[req.req_id] = req
            if self.paged and req.blocks:
                # register the prompt's FULL blocks for future prefix hits,
                # then drop this request's references
                n_prompt_blocks = req.prompt.shape[0] // self.block_size
                self.prefix_cache.insert(req.prompt.tolist(),
                                         req.blocks[:n_prompt_blocks])
                self.allocator.deref(req.blocks)
                req.blocks = []

    # ------------------------------------------------------------------
    def step(self):
        """One scheduling step: admit + prefill one chunk, or decode.
        With `mixed_chunking`, a prefill chunk rides INSIDE the decode
        step (one batch: chunk row + decode rows) instead of stalling it."""
        # admit (deferred decode results must land first: a finished slot
        # may be freed and re-assigned, and pending tokens reference it)
        if self.queue:
            self._drain_inflight(0)
        if self.queue and self.free_slots:
            req = self.queue.popleft()
            req.slot = self.free_slots.pop()
            self.active[req.slot] = req
        # a request mid-prefill takes priority (its chunks must finish
        # before it can decode); pick the first such
        for slot in sorted(self.active):
            r = self.active[slot]
            if r.prefill_done < r.prompt.shape[0]:
                self._drain_inflight(0)
                decoding = [s for s in sorted(self.active)
                            if self.active[s] is not r
                            and self.active[s].prefill_done
                            >= self.active[s].prompt.shape[0]]
                if self.mixed_chunking and decoding and not self.paged:
                    self._mixed_step(r, decoding)
                else:
                    self._prefill_chunk(r)
                return
        if self.active:
            self._decode_step()

    def _mixed_step(self, pre: Request, decode_slots):
        """One multi-token TKG pass: row 0 carries `pre`'s next prefill
        chunk, the other rows carry one decode token + -1 pads (the
        pad-skip invariant keeps pads out of the KV cache and attention).
        Decode rows emit their token from column 0; the chunk row emits
        only when it completes the prompt."""
        S = pre.prompt.shape[0]
        lo = pre.prefill_done
        hi = S if self.chunk_size is None else min(S, lo + self.chunk_size)
        L = hi - lo
        B = self.batch
        ids = torch.zeros(B, L, dtype=torch.long, device=self.device)
        pos = torch.full((B, L), -1, dtype=torch.long, device=self.device)
        seq_ids = torch.full((B,), -1, dtype=torch.long, device=self.device)
        ids[0, :L] = pre.prompt[lo:hi].to(self.device)
        pos[0, :L] = torch.arange(lo, hi, device=self.device)
        seq_ids[0] = pre.slot
        reqs = [pre] + [self.active[s] for s in decode_slots]
        for i, slot in enumerate(decode_slots, start=1):
            if i >= B:
                decode_slots = decode_slots[: i - 1]
                reqs = reqs[: i]
                break
            r = self.active[slot]
            ids[i, 0] = r.last_token
            pos[i, 0] = r.pos
            seq_ids[i] = slot
        self._set_adapters(reqs)
        out = self.model(ids, None, pos, seq_ids, None,
                         phase="token_generation")
        toks = out.logits if out.logits is not None else out.tokens
        if toks.dim() > 2:
            toks = toks.argmax(-1)
        toks = toks.view(B, -1)
        pre.prefill_done = hi
        host = toks.cpu()
        if hi == S:  # chunk completes the prompt: emit its first token
            tok = int(host[0, L - 1])
            pre.last_token = tok
            self._slot_tok[pre.slot] = tok
            pre.pos = S
            pre.output.append(tok)
            pre.t_first_token = time.monotonic()
            if self.on_token is not None:
                self.on_token(pre.req_id, tok)
            self._maybe_finish(pre, tok)
        for i, slot in enumerate(decode_slots, start=1):
            r = self.active.get(slot)
            if r is None or r.done:
                continue
            tok = int(host[i, 0])
            r.last_token = tok
            self._slot_tok[slot] = tok
            r.pos += 1
            r.output.append(tok)
            if self.on_token is not None:
                self.on_token(r.req_id, tok)
            self._maybe_finish(r, tok)

    def run(self, max_steps: int = 100000) -> Dict[int, List[int]]:
        steps = 0
        while self.has_work() and steps < max_steps:
            self.step()
            steps += 1
        self._drain_inflight(0)
        return {rid: r.output for rid, r in self.finished.items()}

    def metrics(self) -> Dict[str, float]:
        """Serving metrics over finished requests (ref utils/benchmark.py
        p50/p90/p99 latency report): TTFT = submit -> first token, TPOT =
        mean inter-token time after the first, plus prefix-cache stats."""
        def pct(vals, q):
            if not vals:
                return 0.0
            vals = sorted(vals)
            return vals[min(len(vals) - 1, int(q * len(vals)))]

        ttft, tpot, cached, prompt = [], [], 0, 0
        for r in self.finished.values():
            if r.t_first_token and r.t_submit:
                ttft.append(r.t_first_token - r.t_submit)
            if r.t_finish and r.t_first_token and len(r.output) > 1:
                tpot.append((r.t_finish - r.t_first_token)
                            / (len(r.output) - 1))
            cached += r.n_cached
            prompt += int(r.prompt.shape[0])
        return {
            "requests": float(len(self.finished)),
            "ttft_p50_s": pct(ttft, 0.50),
            "ttft_p90_s": pct(ttft, 0.90),
            "ttft_p99_s": pct(ttft, 0.99),
            "tpot_p50_s": pct(tpot, 0.50),
            "tpot_p90_s": pct(tpot, 0.90),
            "output_tokens": float(sum(len(r.output)
                                       for r in self.finished.values())),
            "prefix_cached_tokens": float(cached),
            "prefix_hit_rate": cached / prompt if prompt else 0.0,
        }
