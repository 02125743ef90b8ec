"""LocalEngine: on-node prefill+decode for one opponent model.

This is the MI355X replacement for the reference's remote `completion()`
(SURVEY.md §2.4 K1): chat templating, single-shot long-context prefill,
sampled decode with early stop on the protocol close tag, and exact token
accounting from our own tokenizer (replacing the len//4 estimate,
reference models.py:444-447).

Engines are cached per (model name, weights, device): weights stay resident
in HBM3E between rounds, so round k+1 pays only prefill+decode.
"""

from __future__ import annotations

import os
import threading
import time
from typing import Any, Optional

import torch

from .. import ops
from ..models import LlamaModel, get_config
from ..utils.timing import PhaseTimer
from .tokenizer import build_tokenizer

# `[/SPEC]` seen in the decoded tail stops decode early — the critique is
# complete by protocol.
_STOP_SUBSTR = "[/SPEC]"

# Serializes HIP-graph captures across opponent threads (capture itself is
# rare — once per engine — but must not run concurrently with another capture).
_CAPTURE_LOCK = threading.Lock()

# decode steps captured per HIP graph (amortizes the per-launch host cost)
_SPG = 4


class _StopScan:
    """Sliding-window [/SPEC] detector over DECODED text.

    Decodes the last few token ids with the engine's own tokenizer each
    push, so the close tag is detected for byte-level AND BPE tokenizers
    alike. (Round 1 built the tail only from ids < 256 — with an
    HFTokenizer-backed checkpoint every BPE piece id is >= 256 and the tag
    was never seen, burning up to 8000 wasted tokens per critique.)
    """

    __slots__ = ("_tok", "_ids")
    WINDOW = 16  # [/SPEC] is 7 chars; 16 tokens always covers it

    def __init__(self, tokenizer) -> None:
        self._tok = tokenizer
        self._ids: list[int] = []

    def push(self, tok: int) -> bool:
        self._ids.append(tok)
        if len(self._ids) > self.WINDOW:
            del self._ids[0]
        return _STOP_SUBSTR in self._tok.decode(list(self._ids))


def _seed_from_name(name: str) -> int:
    h = 2166136261
    for ch in name:
        h = ((h ^ ord(ch)) * 16777619) & 0x7FFFFFFF
    return h


class LocalEngine:
    def __init__(self, spec: dict[str, Any], device: Optional[str] = None,
                 tp=None) -> None:
        self.name = spec.get("name", spec["arch"])
        self.arch = spec["arch"]
        self.config = get_config(self.arch)
        if device is None:
            gpu = spec.get("gpu")
            if torch.cuda.is_available():
                device = f"cuda:{gpu}" if gpu is not None else "cuda:0"
            else:
                device = "cpu"
        self.device = torch.device(device)
        self.model = LlamaModel(
            self.config, device=self.device, seed=_seed_from_name(self.name),
            tp=tp,
        )
        # Per-engine HIP stream: co-resident opponents on one GPU issue all
        # their work (prefill GEMMs, decode GEMV/attention, graph replays)
        # on their own stream, so concurrent rounds of N opponents overlap
        # on-device instead of serializing through the default stream.
        self.stream = (
            torch.cuda.Stream(device=self.device)
            if self.device.type == "cuda" else None
        )
        weights = spec.get("weights")
        if weights:
            self.model.load_safetensors(weights)
            self.tokenizer = build_tokenizer(self.config.vocab_size, weights)
        else:
            self.model.init_random()
            self.tokenizer = build_tokenizer(self.config.vocab_size)
        if spec.get("dtype") == "fp8":
            # BASELINE config 5: e4m3 MFMA prefill + fp8 weight streaming
            self.model.quantize_fp8()
        # ORDER the engine stream after weight init: weights/tables are
        # initialized on the CREATION stream (usually the default stream),
        # but all generate() work runs on self.stream. Without this event
        # dependency the first prefill races the init kernels and reads
        # unwritten weight memory (garbage bf16 ~1e33 -> NaN logits; flaky
        # by device load — the round-1 "NaN first prefill" bug).
        if self.stream is not None:
            self.stream.wait_stream(torch.cuda.current_stream(self.device))
        self._gen_lock = threading.Lock()
        self._sample_gen = torch.Generator().manual_seed(_seed_from_name(self.name) ^ 0x5EED)
        self._seed_counter = _seed_from_name(self.name) ^ 0x5EED
        self._cache = None  # persistent KV cache (see _get_cache)
        self._graph_state: Optional[dict] = None  # captured decode graph + state

    def _pick_split_blocks(self) -> int:
        """Decode-attention split-grid target from live concurrency.

        The split kernel is occupancy/latency-bound, not bandwidth-bound,
        so the right block count depends on the decode regime (all
        A/B-measured, profiles/r02_measurements.md): bf16 wants 256 (solo
        step 3.772 ms at 256 vs 3.789 at 512; 3-opponent rounds 0.594 vs
        0.581 critiques/s), fp8 wants 512 (0.779 vs 0.763 — its faster
        GEMVs raise the attention share). Baked into the decode graph at
        capture; ADVSPEC_SPLIT_BLOCKS still overrides in the dispatch."""
        return 512 if self.model.fp8 else 256

    def _fit_prompt(self, ids: list[int], reserve: int) -> list[int]:
        """Clamp a prompt into the context window, dropping the middle."""
        budget = self.config.max_seq_len - reserve - 8
        if len(ids) <= budget:
            return ids
        head = budget * 2 // 3
        tail = budget - head
        return ids[:head] + ids[len(ids) - tail :]

    def generate(
        self,
        system_prompt: str,
        user_message: str,
        max_tokens: int = 8000,
        temperature: float = 0.7,
        timeout: float = 600.0,
        top_p: float = 1.0,
    ) -> tuple[str, int, int, dict[str, float]]:
        """Returns (text, input_tokens, output_tokens, phase timings ms)."""
        with self._gen_lock:
            return self._generate_locked(
                system_prompt, user_message, max_tokens, temperature, timeout, top_p
            )

    def _generate_locked(self, system_prompt, user_message, max_tokens,
                         temperature, timeout, top_p):
        deadline = time.monotonic() + timeout
        timer = PhaseTimer(sync_cuda=self.device.type == "cuda")
        ids = self.tokenizer.render_chat(system_prompt, user_message)
        reserve = min(max_tokens, self.config.max_seq_len // 2)
        ids = self._fit_prompt(ids, reserve)
        max_new = min(max_tokens, self.config.max_seq_len - len(ids) - 1)
        stop_ids = self.tokenizer.stop_ids()

        prev_device = None
        if self.device.type == "cuda":
            prev_device = torch.cuda.current_device()
            torch.cuda.set_device(self.device)
        try:
            import contextlib

            stream_ctx = (
                torch.cuda.stream(self.stream)
                if self.stream is not None else contextlib.nullcontext()
            )
            with stream_ctx:
                return self._generate_on_stream(
                    ids, max_new, temperature, top_p, stop_ids, deadline, timer
                )
        finally:
            if prev_device is not None:
                torch.cuda.set_device(prev_device)

    def _get_cache(self, need: int):
        """Persistent KV cache, bucketed to 2k tokens: the same cache tensors
        (and therefore the same captured decode graph) serve every round
        whose prompt+decode fits. Growth reallocates and invalidates the
        graph."""
        cache = self._cache
        if cache is None or cache.max_seq < need:
            old = cache
            bucket = (need + 2047) // 2048 * 2048
            cache = self.model.new_cache(bucket)
            self._cache = cache
            # Drop the captured graph BEFORE releasing the old cache's
            # decode-attention scratch: once _graph_state is gone no replay
            # can reference the retired scratch tensors, so the extension
            # may erase the per-cache workspace map entries (otherwise a
            # long-lived process leaks scratch per growth and can inherit a
            # stale entry when the allocator reuses the freed address).
            self._graph_state = None
            if old is not None and self.device.type == "cuda" and ops.hip_available():
                from ..ops import _load_hip

                _load_hip().ws_release(old.k)
        cache.seq_len = 0
        return cache

    def _generate_on_stream(self, ids, max_new, temperature, top_p, stop_ids,
                            deadline, timer):
        # concurrency-aware attention split target (see _pick_split_blocks);
        # a change invalidates any captured graph (geometry is baked in)
        pick = self._pick_split_blocks()
        if pick != self.model.split_blocks:
            self.model.split_blocks = pick
            self._graph_state = None
        cache = self._get_cache(len(ids) + max_new + 8)
        tokens = torch.tensor(ids, device=self.device, dtype=torch.long)
        with timer.phase("prefill"):
            logits = self.model.prefill(tokens, cache)
        if os.environ.get("ADVSPEC_DEBUG_NAN"):
            n = int(torch.isnan(logits.float()).sum().item())
            print(f"[nan-debug] {self.name} prefill({len(ids)}): {n} NaNs",
                  flush=True)

        out_ids: list[int] = []
        with timer.phase("decode"):
            # TP decode runs the SAME allocation-free workspace loop as
            # single-GPU decode, just ungraphed: RCCL collectives cannot be
            # captured in HIP graphs on this stack, and per-segment graphs
            # between the all-reduces would not pay — the measured graph
            # replay floor (~10-16 us host per replay) matches the eager
            # launch cost of the ~5 kernels a segment holds. Ranks sample
            # identically (deterministic kernel + identical all-reduced
            # activations), and the stop decision is rank-coordinated in
            # _decode_graphed so no rank leaves a collective early.
            use_async = (
                self.device.type == "cuda"
                and top_p >= 1.0
                and ops.hip_available()
            )
            if use_async:
                # same device-state step either way; capture only pays off
                # for longer generations (ADVSPEC_NO_GRAPH forces eager —
                # same kernels, bitwise-identical tokens)
                use_graph = (max_new >= 8
                             and self.model.tp is None
                             and not os.environ.get("ADVSPEC_NO_GRAPH"))
                out_ids = self._decode_graphed(
                    logits, cache, max_new, temperature, stop_ids, deadline,
                    use_graph=use_graph,
                )
            else:
                out_ids = self._decode_sync(
                    logits, cache, max_new, temperature, top_p, stop_ids,
                    deadline,
                )

        text = self.tokenizer.decode(out_ids)
        return text, len(ids), len(out_ids), timer.as_dict()

    def _next_seed(self) -> int:
        self._seed_counter = (
            self._seed_counter * 6364136223846793005 + 1
        ) & 0x7FFFFFFF
        return self._seed_counter

    def _decode_sync(self, logits, cache, max_new, temperature, top_p,
                     stop_ids, deadline) -> list[int]:
        """Host-stepped decode (CPU path and nucleus sampling)."""
        out_ids: list[int] = []
        scan = _StopScan(self.tokenizer)
        for _ in range(max_new):
            tok = ops.sample(
                logits, temperature=temperature, top_p=top_p,
                generator=self._sample_gen, seed=self._next_seed(),
            )
            if tok in stop_ids:
                break
            out_ids.append(tok)
            if scan.push(tok):
                break
            if time.monotonic() > deadline:
                # deadline: return the partial critique rather than erroring
                # the opponent (a partial decode is still a usable critique).
                break
            logits = self.model.decode_one(tok, cache)
        return out_ids

    def _decode_graphed(self, logits, cache, max_new, temperature, stop_ids,
                        deadline, use_graph: bool = True) -> list[int]:
        """Device-state decode loop; HIP-graph capture optional.

        The whole per-token step (sample -> n-layer forward -> logits copy
        -> position bump) keeps all dynamic state (position, RNG, token ids)
        in device words read in-kernel. With use_graph the step is captured
        once and replayed per token — eager decode is launch-gap bound
        (~420 kernel launches/token; rocprof: dispatch span ~2x kernel
        busy), graph replay collapses that to one hipGraphLaunch. Without
        use_graph (ADVSPEC_NO_GRAPH, short generations) the SAME step runs
        eagerly, so both modes execute identical kernels with identical
        split geometry and produce bitwise-identical tokens.
        """
        from ..ops import _load_hip

        hip = _load_hip()
        dev = self.device
        prompt_len = cache.seq_len
        # replay-invariant attention bound: the split geometry is sized once
        # for the whole cache, so the SAME captured graph serves every round
        # that reuses this cache (splits past the live position contribute
        # nothing in the combine).
        max_total = cache.max_seq

        gs = self._graph_state if use_graph else None
        # The key is the CACHE identity only: max_new is covered by sizing
        # tok_hist to the cache's full capacity, and temperature lives in a
        # device-side f32 word read in-kernel — so one captured graph serves
        # every request shape on a warm cache (rounds that alternate
        # max_new/temperature no longer recapture, and capture's device-wide
        # synchronize no longer stalls co-resident opponents per round).
        key = (id(cache), cache.max_seq)
        if gs is not None and gs["key"] == key:
            # Round k+1 on a warm engine: reset the device-side state words
            # and replay the graph captured in round 1 — no re-capture.
            pos_state = gs["pos_state"]
            step_state = gs["step_state"]
            rng_state = gs["rng_state"]
            temp_state = gs["temp_state"]
            tok_hist = gs["tok_hist"]
            W = gs["ws"]
            graph = gs["graph"]
            pos_state.fill_(prompt_len)
            step_state.zero_()
            rng_state.fill_(self._next_seed() | 1)
            temp_state.fill_(float(temperature))
            tok_hist.fill_(-1)
            W.tok_long.zero_()
            W.logits.copy_(logits.reshape(1, -1))
            warm = 0
        else:
            pos_state = torch.tensor([prompt_len], dtype=torch.int32, device=dev)
            step_state = torch.zeros(1, dtype=torch.int32, device=dev)
            rng_state = torch.tensor([self._next_seed() | 1], dtype=torch.int32,
                                     device=dev)
            temp_state = torch.tensor([float(temperature)], dtype=torch.float32,
                                      device=dev)
            # sized to the cache capacity (not max_new) so the same graph
            # state serves any decode length that fits this cache; + _SPG
            # slack: multi-step replays may overshoot by _SPG-1 steps
            tok_hist = torch.full((cache.max_seq + _SPG + 2,), -1,
                                  dtype=torch.int32, device=dev)
            W = self.model.new_decode_ws()
            W.logits.copy_(logits.reshape(1, -1))
            logits_buf = W.logits.view(-1)

            def step():
                # ZERO allocations (capture-safe): sample from W.logits
                # writes W.tok_long (the embedding index) in-kernel, the
                # forward writes W.logits back.
                hip.sample_state(logits_buf, temp_state, rng_state, tok_hist,
                                 step_state, W.tok_long)
                self.model.decode_step_ws(cache, pos_state, max_total, W)
                hip.bump(pos_state, step_state)

            warm = 0
            graph = None
            if use_graph:
                # Warm up ON THE ENGINE STREAM (real decode steps — tokens
                # land in tok_hist[0..1]); warming loads the decode kernels'
                # code objects before capture. The usual warm-on-a-side-
                # stream recipe exists to keep allocator state out of the
                # capture, but our step is allocation-free — and side-stream
                # warms measured as the trigger for caching-allocator
                # aliasing on this torch/ROCm combo (flaky NaN prefills in
                # LATER engines).
                warm = min(2, max_new)
                for _ in range(warm):
                    step()
                if os.environ.get("ADVSPEC_DEBUG_NAN"):
                    print(f"[nan-debug] {self.name} warm toks:",
                          tok_hist[:warm].cpu().tolist(), "logits_buf NaNs:",
                          int(torch.isnan(logits_buf.float()).sum().item()),
                          flush=True)

                # thread_local capture mode + a process-wide capture lock:
                # other opponents' threads keep replaying their own graphs on
                # their own streams while this engine captures (global mode
                # would abort the capture on any concurrent allocator
                # traffic). Capture on the ENGINE's stream (not torch's
                # shared default capture stream): the decode-attention kernel
                # uses per-stream scratch, so two graphs captured on one
                # stream would share buffers and race when replayed
                # concurrently.
                # Capture on a FRESH dedicated stream, never the engine's
                # live eager stream: capturing on a stream that also carries
                # eager work corrupts the caching allocator's bookkeeping
                # (measured: later engines' prefills read aliased memory ->
                # NaN logits, flaky by allocation layout). The decode
                # attention scratch is keyed by the KV-cache pointer, not the
                # stream, so the capture-stream choice carries no aliasing
                # risk.
                # capture _SPG steps per graph: one hipGraphLaunch of a
                # ~230-node graph costs ~0.1-0.2 ms host-side, measurably
                # pacing single-opponent decode; 4 steps/launch cuts that 4x
                graph = torch.cuda.CUDAGraph()
                with _CAPTURE_LOCK:
                    with torch.cuda.graph(graph,
                                          stream=torch.cuda.Stream(device=dev),
                                          capture_error_mode="thread_local"):
                        for _ in range(_SPG):
                            step()
                if graph is not None:
                    self._graph_state = {
                        "key": key, "graph": graph, "pos_state": pos_state,
                        "step_state": step_state, "rng_state": rng_state,
                        "temp_state": temp_state, "tok_hist": tok_hist,
                        "ws": W,
                    }

        CHECK = 32
        done = False
        i = warm
        out_ids: list[int] = []
        scan = _StopScan(self.tokenizer)
        scanned = 0

        def scan_until(upto: int) -> bool:
            """Pull tokens [scanned, upto) to host, extend out_ids; True if a
            stop condition fired."""
            nonlocal scanned
            chunk = tok_hist[scanned:upto].cpu().tolist()
            scanned = upto
            for tok in chunk:
                if tok in stop_ids or tok < 0:
                    return True
                out_ids.append(tok)
                if scan.push(tok):
                    return True
            return False

        tp = self.model.tp
        while i < max_new and not done:
            n = min(CHECK, max_new - i)
            if graph is not None:
                # replays advance _SPG steps each; overshoot past max_new
                # (< _SPG steps) lands in the tok_hist slack and is never
                # scanned
                for _ in range((n + _SPG - 1) // _SPG):
                    graph.replay()
                i += (n + _SPG - 1) // _SPG * _SPG
            else:
                for _ in range(n):
                    step()
                i += n
            done = scan_until(min(i, max_new))
            if time.monotonic() > deadline:
                done = True
            if tp is not None and tp.size > 1:
                # rank-coordinate the stop decision: token content is
                # identical across ranks (deterministic sampling on
                # identical all-reduced activations) but the wall-clock
                # deadline is not — an uncoordinated early exit would
                # leave the other ranks blocked in the next all-reduce
                done = tp.any_flag(done)
        if not done:
            scan_until(min(i, max_new))

        cache.seq_len = prompt_len + i  # device-side pos advanced i times
        return out_ids

_ENGINES: dict[tuple, LocalEngine] = {}
_ENGINES_LOCK = threading.Lock()


def get_engine(spec: dict[str, Any], device: Optional[str] = None) -> LocalEngine:
    """Process-wide engine cache: weights stay HBM-resident across rounds."""
    key = (spec.get("name"), spec.get("arch"), spec.get("weights"),
           spec.get("dtype"), device or spec.get("gpu"))
    with _ENGINES_LOCK:
        eng = _ENGINES.get(key)
        if eng is None:
            eng = LocalEngine(spec, device=device)
            _ENGINES[key] = eng
        return eng


def clear_engines() -> None:
    with _ENGINES_LOCK:
        for eng in _ENGINES.values():
            # drop captured graphs first, then the per-cache attention
            # scratch the extension holds for this engine's KV cache
            eng._graph_state = None
            cache = eng._cache
            if (cache is not None and eng.device.type == "cuda"
                    and ops.hip_available()):
                from ..ops import _load_hip

                _load_hip().ws_release(cache.k)
        _ENGINES.clear()
