"""LocalEngine: continuous-batching inference over the HIP/CDNA4 ops.

MI355X-first design (replaces the reference's HTTP model-access layer,
reference: lib/quoracle/models/model_query.ex):

  * One LocalEngine per GPU hosts one or more models (288 GB HBM3E holds
    several 8B-class pool members comfortably).
  * Every pending agent turn across the whole spawn tree lands in one
    per-model run queue; each engine step runs ONE mixed forward per model:
    all decode sequences (1 token each) + chunked prefill tokens, exactly
    the ForwardBatch the model/kernels were designed around.
  * A sequence is unified prefill/decode: each step advances the KV cache by
    a chunk; a 1-token chunk takes the specialized decode-attention kernel.
  * Conversations are prefix-cached Sessions — consecutive consensus cycles
    of the same (agent, model) only prefill the newly appended history.
  * The engine loop runs on its own thread so host-side orchestration
    (asyncio actor runtime) overlaps GPU compute; results complete asyncio
    futures via call_soon_threadsafe.
"""

from __future__ import annotations

import asyncio
import os
import queue
import sys
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..models import LlamaModel, get_config
from .api import Engine, GenerateRequest, GenerateResult
from .graphs import DecodeGraphs
from .kv_cache import BlockManager, OutOfBlocks, Session, SessionCache
from .sampler import FORCED, ActionGrammar, Sampler, SamplingParams
from .tokenizer import ByteTokenizer, EOS
from ..models.llama import ForwardBatch, KVCache

BLOCK_SIZE = 16
PREFILL_CHUNK = 2048         # max prefill tokens per model per engine step
QT = 16                      # prefill tile height (must match attention.hip)

# $/1M tokens (input, output) per preset — budget plumbing needs real-ish
# numbers (reference records per-call costs: lib/quoracle/costs/recorder.ex)
_PRICES = {
    "tiny": (0.01, 0.02), "gpt2s": (0.02, 0.05),
    "llama3-8b": (0.2, 0.8), "llama3-70b": (0.9, 3.6),
    "mixtral-8x7b": (0.5, 2.0), "embed-small": (0.01, 0.01),
}


@dataclass
class _Seq:
    request: GenerateRequest
    prompt: List[int]
    session: Session
    params: SamplingParams
    grammar: Optional[ActionGrammar]
    generator: Optional[torch.Generator]
    emitted: List[int] = field(default_factory=list)
    finished: bool = False
    t_start: float = field(default_factory=time.monotonic)
    _complete: Optional[Tuple] = None        # (loop, future) | ("sync", event)
    result: Optional[GenerateResult] = None

    @property
    def known(self) -> List[int]:
        return self.prompt + self.emitted


class _HostedModel:
    def __init__(self, key: str, device: torch.device, kv_blocks: int,
                 tp=None):
        self.key = key
        self.cfg = get_config(key)
        self.model = LlamaModel(key, device, tp=tp)
        self.kv: KVCache = self.model.new_kv_cache(kv_blocks, BLOCK_SIZE)
        self.mgr = BlockManager(kv_blocks, BLOCK_SIZE)
        self.sessions = SessionCache(self.mgr)
        self.active: List[_Seq] = []
        self.sampler = Sampler(self.cfg.vocab_size, device)
        # each hosted model launches on its own HIP stream so pool members
        # overlap on the GPU instead of serializing through the engine loop
        self.stream = (torch.cuda.Stream(device)
                       if device.type == "cuda" else None)
        # hipGraph-captured decode step (scratch block absorbs pad-row writes)
        scratch = self.mgr.alloc(1)[0]
        maxb = min(kv_blocks, (self.cfg.max_context + BLOCK_SIZE - 1)
                   // BLOCK_SIZE)
        self.graphs = DecodeGraphs(self.model, self.kv, device, maxb, scratch)
        if tp is not None and tp.world > 1:
            # RCCL all-reduce under hipGraph capture is unvalidated on this
            # stack; TP decode stays eager until proven
            self.graphs.enabled = False

    def stream_ctx(self):
        import contextlib
        return (torch.cuda.stream(self.stream) if self.stream is not None
                else contextlib.nullcontext())


class LocalEngine(Engine):
    def __init__(self, model_keys: Sequence[str],
                 device: Optional[torch.device] = None,
                 kv_gb_per_model: float = 4.0,
                 embed_model_key: Optional[str] = "embed-small",
                 kv_blocks_override: Optional[int] = None,
                 prefill_chunk: int = PREFILL_CHUNK,
                 tp=None):
        self.device = device or torch.device(
            "cuda:0" if torch.cuda.is_available() else "cpu")
        self.tokenizer = ByteTokenizer()
        self.prefill_chunk = prefill_chunk
        self.models: Dict[str, _HostedModel] = {}
        for key in model_keys:
            cfg = get_config(key)
            if kv_blocks_override is not None:
                blocks = kv_blocks_override
            else:
                per_block = cfg.kv_bytes_per_token() * BLOCK_SIZE
                blocks = max(8, int(kv_gb_per_model * (1 << 30) / per_block))
                # small models would otherwise allocate millions of blocks
                # for the same byte budget; 2M tokens of KV is plenty
                blocks = min(blocks, (2 << 20) // BLOCK_SIZE)
            self.models[key] = _HostedModel(key, self.device, blocks, tp=tp)
        self.embed_model: Optional[LlamaModel] = None
        if embed_model_key:
            self.embed_model = LlamaModel(embed_model_key, self.device)
        self._inbox: "queue.Queue[_Seq]" = queue.Queue()
        self._gpu_lock = threading.Lock()
        self._thread: Optional[threading.Thread] = None
        self._running = False
        self._wake = threading.Event()
        # telemetry (read by bench/monitor; written only by the engine thread)
        self.stats = {"engine_steps": 0, "forward_tokens": 0,
                      "decode_tokens": 0, "prefill_tokens": 0,
                      "requests_done": 0, "prefix_hit_tokens": 0}
        self._log_every = int(os.environ.get("QUORACLE_ENGINE_LOG", "0"))
        self._roctx = bool(os.environ.get("QUORACLE_ROCTX"))
        # QUORACLE_CHECK=1: synchronize after every launch phase so async
        # HIP faults surface at their source (SURVEY.md §5.2 debug mode)
        self._strict_sync = bool(os.environ.get("QUORACLE_CHECK"))

    # -- lifecycle -----------------------------------------------------------

    def start(self) -> "LocalEngine":
        if self._thread is None:
            if self.device.type == "cuda":
                torch.cuda.synchronize(self.device)   # weight init done
                self.precapture_graphs()
            self._running = True
            self._thread = threading.Thread(target=self._loop, daemon=True,
                                            name="quoracle-engine")
            self._thread.start()
        return self

    def precapture_graphs(self, buckets=(1, 2, 4, 8, 16)) -> None:
        """hipGraph-capture each model's decode step per batch bucket while
        the GPU is quiescent; decode steps then replay instead of
        relaunching ~260 kernels eagerly."""
        t0 = time.monotonic()
        for hm in self.models.values():
            hm.graphs.precapture(buckets)
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
            print(f"[engine] graph precapture done in "
                  f"{time.monotonic() - t0:.1f}s "
                  f"({sum(len(h.graphs.graphs) for h in self.models.values())}"
                  f" graphs)", file=sys.stderr, flush=True)

    def stop(self) -> None:
        self._running = False
        self._wake.set()
        if self._thread:
            self._thread.join(timeout=30)
            self._thread = None

    # -- Engine protocol -----------------------------------------------------

    async def generate(self, request: GenerateRequest) -> GenerateResult:
        loop = asyncio.get_running_loop()
        fut: asyncio.Future = loop.create_future()
        err = self._submit(request, ("async", loop, fut))
        if err is not None:
            return err
        return await fut

    def generate_sync(self, request: GenerateRequest,
                      timeout: Optional[float] = None) -> GenerateResult:
        """Blocking path for tests / non-async callers; drives the engine
        inline when the loop thread is not running."""
        ev = threading.Event()
        box: List[GenerateResult] = []
        err = self._submit(request, ("sync", ev, box))
        if err is not None:
            return err
        if self._thread is None:
            deadline = None if timeout is None else time.monotonic() + timeout
            while not ev.is_set():
                self.step()
                if deadline and time.monotonic() > deadline:
                    raise TimeoutError("generate_sync timed out")
        else:
            ev.wait(timeout)
        if not box:
            raise TimeoutError("generate_sync timed out")
        return box[0]

    async def embed(self, texts: List[str]) -> Sequence[Sequence[float]]:
        return await asyncio.get_running_loop().run_in_executor(
            None, self.embed_sync, list(texts))

    def embed_sync(self, texts: List[str]) -> Sequence[Sequence[float]]:
        if self.embed_model is None:
            raise RuntimeError("no embedding model hosted")
        batches = []
        for t in texts:
            ids = self.tokenizer.encode(t)[:512] or [EOS]
            batches.append(torch.tensor(ids, dtype=torch.int32,
                                        device=self.device))
        with self._gpu_lock:
            hid = self.embed_model.embed_texts_hidden(batches)
        norm = hid.norm(dim=-1, keepdim=True).clamp_min(1e-8)
        return (hid / norm).cpu().tolist()

    def similarity_matrix(self, texts: List[str]):
        """Pairwise cosine matrix of the texts' embeddings — embedding
        forward + the fused cosine_sim_matrix kernel in one GPU pass
        (the consensus vote's hot path)."""
        if self.embed_model is None:
            raise RuntimeError("no embedding model hosted")
        batches = []
        for t in texts:
            ids_ = self.tokenizer.encode(t)[:512] or [EOS]
            batches.append(torch.tensor(ids_, dtype=torch.int32,
                                        device=self.device))
        from ..ops import dispatch as D
        with self._gpu_lock:
            hid = self.embed_model.embed_texts_hidden(batches).float()
            out = torch.empty((hid.shape[0], hid.shape[0]),
                              dtype=torch.float32, device=hid.device)
            D.cosine_sim_matrix(out, hid)
        return out.cpu().tolist()

    def count_tokens(self, text: str) -> int:
        return self.tokenizer.count(text)

    def context_limit(self, model_key: str) -> int:
        return get_config(model_key).max_context

    def output_limit(self, model_key: str) -> int:
        return get_config(model_key).max_output

    def drop_session(self, model_key: str, session_id: str) -> None:
        hm = self.models.get(model_key)
        if hm:
            hm.sessions.drop(session_id)

    # -- request intake ------------------------------------------------------

    def _submit(self, request: GenerateRequest, completion,
                enqueue: bool = True) -> Optional[GenerateResult]:
        hm = self.models.get(request.model_key)
        if hm is None:
            return GenerateResult(model_key=request.model_key,
                                  error=f"model_not_hosted:{request.model_key}")
        prompt = self.tokenizer.render_messages(request.messages)
        if len(prompt) > hm.cfg.max_context:
            return GenerateResult(model_key=request.model_key,
                                  error="context_overflow",
                                  input_tokens=len(prompt))
        max_tokens = min(request.max_tokens, hm.cfg.max_output,
                         hm.cfg.max_context - len(prompt))
        if request.action_grammar:
            # the grammar terminates by itself (~600 tokens worst case);
            # never truncate it into unparseable JSON — but stay inside the
            # model's remaining window or positions run past max_context
            max_tokens = min(max(max_tokens, 1024),
                             hm.cfg.max_context - len(prompt))
        params = SamplingParams(
            temperature=request.temperature, top_p=request.top_p,
            max_tokens=max_tokens, seed=request.seed)
        grammar = None
        if request.action_grammar:
            grammar = ActionGrammar(
                request.allowed_actions
                or ["orient", "send_message", "todo", "wait"],
                context=request.grammar_context)
        gen = None
        if request.seed is not None:
            gen = torch.Generator(device=self.device)
            gen.manual_seed(request.seed)
        sid = request.session_id or f"anon-{id(request)}"
        seq = _Seq(request=request, prompt=prompt, session=None,  # type: ignore
                   params=params, grammar=grammar, generator=gen,
                   _complete=completion)
        seq._session_id = sid                                     # type: ignore
        if not enqueue:
            self._made_seq = seq
            return None
        self._inbox.put(seq)
        self._wake.set()
        return None

    # -- engine loop ---------------------------------------------------------

    def _loop(self) -> None:
        while self._running:
            worked = self.step()
            if not worked:
                self._wake.wait(timeout=0.005)
                self._wake.clear()

    def step(self) -> bool:
        """One engine step: admit new work, then for every model with pending
        sequences LAUNCH one mixed forward on that model's own stream (phase
        1, async) and only then sample (phase 2, syncs per stream) — pool
        members overlap on the GPU.  Returns True if any work was done."""
        self._admit()
        pending = []
        for hm in self.models.values():
            if not hm.active:
                continue
            try:
                with self._gpu_lock:
                    ctxt = self._launch_model(hm)
            except Exception as exc:  # noqa: BLE001 — fail loud, not hang
                self._crash_model(hm, exc)
                continue
            if ctxt is not None:
                if self._strict_sync and self.device.type == "cuda":
                    torch.cuda.synchronize(self.device)
                pending.append((hm, ctxt))
        for hm, ctxt in pending:
            try:
                with self._gpu_lock:
                    self._sample_model(hm, *ctxt)
            except Exception as exc:  # noqa: BLE001
                self._crash_model(hm, exc)
        return bool(pending)

    def _crash_model(self, hm: _HostedModel, exc: Exception) -> None:
        import traceback
        traceback.print_exc()
        print(f"[engine {hm.key}] step failed: {exc}; failing "
              f"{len(hm.active)} in-flight requests",
              file=sys.stderr, flush=True)
        crashed = list(hm.active)
        for seq in crashed:
            seq.result = GenerateResult(
                model_key=hm.key, error=f"engine_error:{exc}")
        self._finish(hm, crashed)
        # Watchdog: repeated failures mean corrupted
        # device state — rebuild the model's KV world.  Conversation
        # history lives in the orchestrator, so sessions re-prefill
        # lazily via the prefix cache on the next request; the failed
        # shard simply drops out of votes until then (consensus already
        # tolerates partial pools).
        hm.crash_count = getattr(hm, "crash_count", 0) + 1
        if hm.crash_count >= 2:
            self.reset_model(hm.key)

    def reset_model(self, key: str) -> None:
        """Rebuild a hosted model's KV cache, block manager and sessions
        after a device fault; weights stay resident (they are immutable)."""
        hm = self.models.get(key)
        if hm is None:
            return
        print(f"[engine {key}] watchdog reset: rebuilding KV state",
              file=sys.stderr, flush=True)
        crashed = list(hm.active)
        for seq in crashed:
            if seq.result is None:
                seq.result = GenerateResult(model_key=key,
                                            error="engine_reset")
        self._finish(hm, crashed)
        blocks = hm.mgr.num_blocks
        bs = hm.kv.block_size
        for t in hm.kv.k + hm.kv.v:
            t.zero_()
        hm.mgr = BlockManager(blocks, bs)
        hm.sessions = SessionCache(hm.mgr)
        scratch = hm.mgr.alloc(1)[0]
        hm.graphs.scratch_block = scratch
        hm.crash_count = 0

    def _admit(self) -> None:
        if getattr(self, "_tp_direct", False):
            return      # lockstep engines admit via the admit-broadcast only
        while True:
            try:
                seq = self._inbox.get_nowait()
            except queue.Empty:
                return
            self._admit_seq(seq)

    def _admit_seq(self, seq: "_Seq") -> None:
        hm = self.models[seq.request.model_key]
        sess = hm.sessions.get_or_create(seq._session_id)   # type: ignore
        hit = hm.sessions.match_prefix(sess, seq.prompt)
        self.stats["prefix_hit_tokens"] += hit
        seq.session = sess
        hm.active.append(seq)

    def _expand_forced(self, hm: _HostedModel, seq: _Seq) -> None:
        """Append the grammar's next run of FORCED tokens in one go: known
        structural JSON needs no per-token decode — the run becomes a
        chunked prefill, collapsing ~60% of generated tokens into a handful
        of forwards."""
        g = seq.grammar
        if g is None or seq.finished:
            return
        while not seq.finished and not g.done and g.current()[0] == FORCED:
            tok = g.advance(0)
            seq.emitted.append(tok)
            if (tok == EOS or g.done
                    or len(seq.emitted) >= seq.params.max_tokens):
                seq.finished = True
                seq.result = self._make_result(hm, seq)
                self.stats["requests_done"] += 1

    def _launch_model(self, hm: _HostedModel):
        dev = self.device
        decode: List[_Seq] = []
        prefill: List[Tuple[_Seq, int]] = []       # (seq, n_new_tokens)
        early_done: List[_Seq] = []
        budget = self.prefill_chunk
        for seq in hm.active:
            self._expand_forced(hm, seq)
            if seq.finished:
                early_done.append(seq)
                continue
            cached = len(seq.session.token_ids)
            remaining = len(seq.known) - cached
            if remaining <= 0:
                remaining = 1      # shouldn't happen; treat as decode resample
            if remaining == 1:
                decode.append(seq)
            elif budget > 0:
                n = min(remaining, budget)
                budget -= n
                prefill.append((seq, n))

        if early_done:
            self._finish(hm, early_done)
        if not decode and not prefill:
            return None

        tokens: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        bt_rows: List[List[int]] = []
        ctx_lens: List[int] = []
        failed: List[_Seq] = []
        active_ids = [s.session.session_id for s in hm.active]

        def _extend(seq: _Seq, toks: List[int]) -> Optional[List[int]]:
            try:
                return hm.sessions.extend(seq.session, toks, active=active_ids)
            except OutOfBlocks as exc:
                seq.result = GenerateResult(model_key=hm.key,
                                            error=f"kv_exhausted:{exc}")
                failed.append(seq)
                return None

        for seq in decode:
            cached = len(seq.session.token_ids)
            tok = seq.known[cached]
            s = _extend(seq, [tok])
            if s is None:
                continue
            tokens.append(tok)
            positions.append(cached)
            slots.extend(s)
            bt_rows.append(list(seq.session.blocks))
            ctx_lens.append(cached + 1)
        n_decode = len(ctx_lens)

        # pure-decode steps replay the hipGraph-captured step when possible.
        # Replay + its input copies run on the DEFAULT stream: hipGraph
        # replay launched from a side stream deadlocks on ROCm 7.2 (observed
        # on MI355X; graph kernels never complete), so graph steps serialize
        # across models while eager prefill keeps per-model streams.
        if n_decode and not prefill:
            sample_seqs = [s for s in decode if s not in failed]
            if hm.stream is not None:
                # order after this model's eager (prefill) stream, and make
                # that stream wait for the replay before later prefills
                torch.cuda.current_stream(self.device).wait_stream(hm.stream)
            logits = hm.graphs.run(
                tokens, positions, slots, bt_rows, ctx_lens,
                # the full block tuple participates in the staging key:
                # after eviction + re-prefill a session can reuse the same
                # length and tail block with different middle blocks
                bt_keys=[(s.session.session_id,
                          hash(tuple(s.session.blocks)))
                         for s in sample_seqs])
            if logits is not None and hm.stream is not None:
                hm.stream.wait_stream(torch.cuda.current_stream(self.device))
            if logits is not None:
                st = self.stats
                st["engine_steps"] += 1
                st["graph_steps"] = st.get("graph_steps", 0) + 1
                st["forward_tokens"] += n_decode
                st["decode_tokens"] += n_decode
                return (sample_seqs, logits, failed, True)

        tile_q0: List[int] = []
        tile_qn: List[int] = []
        tile_seq: List[int] = []
        tile_pos0: List[int] = []
        t32_q0: List[int] = []
        t32_qn: List[int] = []
        t32_seq: List[int] = []
        t32_pos0: List[int] = []
        # decode rows sample at their own row index
        sample_rows: List[int] = list(range(n_decode))
        sample_seqs: List[_Seq] = [s for s in decode if s not in failed]

        row = n_decode
        max_kv = 0
        # 128-row tiles match the default T12 prefill kernel (352 TF,
        # validated r2); QUORACLE_MFMA64/QUORACLE_MFMA32 fall back
        if os.environ.get("QUORACLE_MFMA32"):
            big_step = 32
        elif os.environ.get("QUORACLE_MFMA64"):
            big_step = 64
        else:
            big_step = 128
        for seq, n in prefill:
            cached = len(seq.session.token_ids)
            chunk = seq.known[cached:cached + n]
            s = _extend(seq, chunk)
            if s is None:
                continue
            bt_rows.append(list(seq.session.blocks))
            seq_row = len(bt_rows) - 1
            tokens.extend(chunk)
            positions.extend(range(cached, cached + n))
            slots.extend(s)
            for off in range(0, n, QT):
                tile_q0.append(row + off)
                tile_qn.append(min(QT, n - off))
                tile_seq.append(seq_row)
                tile_pos0.append(cached + off)
            for off in range(0, n, big_step):
                t32_q0.append(row + off)
                t32_qn.append(min(big_step, n - off))
                t32_seq.append(seq_row)
                t32_pos0.append(cached + off)
            if cached + n == len(seq.known):
                sample_rows.append(row + n - 1)
                sample_seqs.append(seq)
            max_kv = max(max_kv, cached + n)
            row += n

        if not tokens:
            self._finish(hm, failed)
            return None

        maxb = max(len(r) for r in bt_rows)
        # numpy staging: per-row torch.tensor() creation measured ~20us
        # each on the eager step path
        bt_np = np.zeros((len(bt_rows), maxb), dtype=np.int32)
        for i, r in enumerate(bt_rows):
            bt_np[i, :len(r)] = r
        bt = torch.from_numpy(bt_np)

        def t32(x):
            return torch.tensor(x, dtype=torch.int32, device=dev)

        t_fwd = time.monotonic()
        # rocprof/omnitrace attribution: roctx ranges around each eager
        # forward; QUORACLE_ROCTX=1 enables
        if self._roctx:
            torch.cuda.nvtx.range_push(
                f"fwd:{hm.key}:T{len(tokens)}:d{n_decode}")
        # batch tensors are H2D copies: build them on the model's stream so
        # the forward (same stream) is ordered after them without a sync
        with hm.stream_ctx():
            batch = ForwardBatch(
                tokens=t32(tokens), positions=t32(positions), slots=t32(slots),
                block_tables=bt.to(dev, non_blocking=False),
                n_decode=n_decode,
                ctx_lens=t32(ctx_lens) if n_decode else None,
                max_ctx=max(ctx_lens) if ctx_lens else 0,
                max_kv=max_kv,
                tile_q0=t32(tile_q0), tile_qn=t32(tile_qn),
                tile_seq=t32(tile_seq), tile_pos0=t32(tile_pos0),
                tile32_q0=t32(t32_q0), tile32_qn=t32(t32_qn),
                tile32_seq=t32(t32_seq), tile32_pos0=t32(t32_pos0))
            hidden = hm.model.forward(batch, hm.kv)
            logits = None
            if sample_rows:
                rows = torch.tensor(sample_rows, dtype=torch.long, device=dev)
                logits = hm.model.compute_logits(hidden, rows)
        if self._roctx:
            torch.cuda.nvtx.range_pop()
        st = self.stats
        st["engine_steps"] += 1
        st["forward_tokens"] += len(tokens)
        st["decode_tokens"] += n_decode
        st["prefill_tokens"] += len(tokens) - n_decode
        if self._log_every and st["engine_steps"] % self._log_every == 0:
            print(f"[engine {hm.key}] step={st['engine_steps']} "
                  f"T={len(tokens)} dec={n_decode} active={len(hm.active)} "
                  f"launch_ms={(time.monotonic() - t_fwd) * 1e3:.1f}",
                  file=sys.stderr, flush=True)
        return (sample_seqs, logits, failed, False)

    def _sample_model(self, hm: _HostedModel, sample_seqs, logits,
                      failed, on_default_stream=False) -> None:
        import contextlib
        done: List[_Seq] = list(failed)
        if logits is not None:
            with (contextlib.nullcontext() if on_default_stream
                  else hm.stream_ctx()):
                ids = hm.sampler.sample(
                    logits, [s.params for s in sample_seqs],
                    [s.grammar for s in sample_seqs],
                    [s.generator for s in sample_seqs])
            for seq, tok in zip(sample_seqs, ids):
                seq.emitted.append(tok)
                if (tok == EOS or len(seq.emitted) >= seq.params.max_tokens
                        or (seq.grammar is not None and seq.grammar.done)):
                    seq.finished = True
                    seq.result = self._make_result(hm, seq)
                    self.stats["requests_done"] += 1
                    done.append(seq)
        self._finish(hm, done)

    def _make_result(self, hm: _HostedModel, seq: _Seq) -> GenerateResult:
        out_ids = [t for t in seq.emitted if t != EOS]
        text = self.tokenizer.decode(out_ids)
        n_in, n_out = len(seq.prompt), len(seq.emitted)
        pin, pout = _PRICES.get(hm.cfg.name, (0.1, 0.4))
        return GenerateResult(
            model_key=hm.key, text=text, input_tokens=n_in,
            output_tokens=n_out,
            latency_ms=(time.monotonic() - seq.t_start) * 1e3,
            cost=(n_in * pin + n_out * pout) / 1e6)

    def _finish(self, hm: _HostedModel, done: List[_Seq]) -> None:
        for seq in done:
            if seq in hm.active:
                hm.active.remove(seq)
            # one-shot (anonymous) sessions release their KV immediately;
            # named sessions persist for prefix reuse
            if not seq.request.session_id and seq.session is not None:
                hm.sessions.drop(seq.session.session_id)
            comp = seq._complete
            if comp is None:
                continue
            if comp[0] == "async":
                _, loop, fut = comp
                loop.call_soon_threadsafe(
                    lambda f=fut, r=seq.result: f.done() or f.set_result(r))
            elif comp[0] == "cb":
                comp[1](seq.result)
            else:
                _, ev, box = comp
                box.append(seq.result)
                ev.set()
