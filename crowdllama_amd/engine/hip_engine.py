"""HIP-backed EngineBase: the worker's real compute backend (replaces the
reference's WorkerAPIHandler -> Ollama HTTP shell-out, api.go:45-160)."""

from __future__ import annotations

import asyncio
import threading
import time
from concurrent.futures import ThreadPoolExecutor

from ..quant.gguf import GGUFReader
from ..tokenizer import NativeTokenizer, Tokenizer
from .api import EngineBase, GenerateResult, RollingRate


class HipEngine(EngineBase):
    """One GGUF model resident on one MI355X, served request-at-a-time
    (matching the reference's per-worker request granularity)."""

    def __init__(self, model_name: str, gguf_path: str, device: int = 0,
                 max_seq: int = 4096, use_graph: bool = True):
        from ..ops import get_core
        core = self._core = get_core()
        if core.device_count() == 0:
            raise RuntimeError("HipEngine requires a GPU (none visible); "
                               "use MockEngine for CPU-only meshes")
        self.model_name = model_name
        cfg = core.EngineConfig()
        cfg.batch = 1
        cfg.max_seq = max_seq
        cfg.gen_cap = max_seq  # ring >= any admissible max_new
        cfg.device = device
        cfg.use_graph = use_graph
        self.eng = core.Engine(gguf_path, cfg)
        self.gen_cap = cfg.gen_cap
        with GGUFReader(gguf_path) as r:
            try:
                self.tok = NativeTokenizer.from_gguf(r)
            except Exception:
                import logging
                logging.getLogger("crowdllama_amd.engine").warning(
                    "native tokenizer unavailable for %s; falling back to "
                    "the Python tokenizer", gguf_path, exc_info=True)
                self.tok = Tokenizer.from_gguf(r)
        self._props = core.device_props(device)
        self._rate = RollingRate()
        self._lock = threading.Lock()
        self._pool = ThreadPoolExecutor(max_workers=1,
                                        thread_name_prefix="hipengine")
        self._active = 0
        self.max_seq = max_seq

    # ------------------------------------------------------------ generate

    def _budget(self, prompt: str, max_new_tokens: int):
        """Clamp max_new to the sequence/ring budget (matching
        BatchingHipEngine); reject only when the PROMPT itself does not fit
        — never truncate it silently (round-1 advisor finding)."""
        ids = self.tok.encode(prompt)
        max_new = min(max_new_tokens, self.max_seq - len(ids) - 1,
                      self.gen_cap)
        if max_new < 1:
            raise ValueError(
                f"prompt ({len(ids)} tokens) exceeds max_seq={self.max_seq}")
        return ids or [self.tok.bos_id], max_new

    def _generate_sync(self, prompt: str, max_new_tokens: int,
                       temperature: float = 0.0) -> GenerateResult:
        import numpy as np
        with self._lock:
            t0 = time.monotonic_ns()
            ids, max_new_tokens = self._budget(prompt, max_new_tokens)
            self.eng.reset()
            self.eng.prefill(np.asarray([ids], dtype=np.int32))
            n_new = max(1, max_new_tokens)
            if temperature > 0.0:
                # host sampling path: re-sample each step from the logits
                from .sampling import sample
                rng = np.random.default_rng()
                out = []
                tok_id = sample(self.eng.logits(0), temperature=temperature,
                                rng=rng)
                out.append(tok_id)
                self.eng.set_cur_token(0, tok_id)
                for _ in range(n_new - 1):
                    if tok_id == self.tok.eos_id:
                        break
                    self.eng.decode(1)
                    tok_id = sample(self.eng.logits(0),
                                    temperature=temperature, rng=rng)
                    out.append(tok_id)
                    self.eng.set_cur_token(0, tok_id)
            else:
                if n_new > 1:
                    self.eng.decode(n_new - 1)
                out = list(self.eng.gen_tokens(0))
            if self.tok.eos_id in out:
                out = out[: out.index(self.tok.eos_id)]
                reason = "stop"
            else:
                reason = "length"
            text = self.tok.decode(out)
            dur = time.monotonic_ns() - t0
            self._rate.add(len(out))
            return GenerateResult(text=text, tokens_generated=len(out),
                                  duration_ns=dur, done_reason=reason)

    async def generate(self, prompt: str, max_new_tokens: int = 256,
                       temperature: float = 0.0) -> GenerateResult:
        loop = asyncio.get_running_loop()
        self._active += 1
        try:
            return await loop.run_in_executor(
                self._pool, self._generate_sync, prompt, max_new_tokens,
                temperature)
        finally:
            self._active -= 1

    # ------------------------------------------------------------ streaming

    STREAM_STRIDE = 8  # decode steps between emitted deltas (graph replays)

    def _stream_sync(self, prompt: str, max_new_tokens: int,
                     temperature: float, emit) -> None:
        """Greedy streaming: decode in strides, emit text deltas. Runs on the
        engine executor with the lock held for the whole sequence; `emit` is
        thread-safe."""
        import numpy as np
        with self._lock:
            t0 = time.monotonic_ns()
            ids, max_new_tokens = self._budget(prompt, max_new_tokens)
            self.eng.reset()
            self.eng.prefill(np.asarray([ids], dtype=np.int32))
            n_new = max(1, max_new_tokens)
            sent = ""
            done_reason = "length"
            while True:
                toks = list(self.eng.gen_tokens(0))
                finished = False
                if self.tok.eos_id in toks:
                    toks = toks[: toks.index(self.tok.eos_id)]
                    done_reason = "stop"
                    finished = True
                if len(toks) >= n_new:
                    toks = toks[:n_new]
                    finished = True
                text = self.tok.decode(toks)
                delta = text[len(sent):]
                if finished:
                    self._rate.add(len(toks))
                    emit(GenerateResult(
                        text=delta, tokens_generated=len(toks),
                        duration_ns=time.monotonic_ns() - t0,
                        done_reason=done_reason))
                    return
                if delta:
                    sent = text
                    emit(GenerateResult(text=delta,
                                        tokens_generated=len(toks),
                                        done_reason=""))
                self.eng.decode(min(self.STREAM_STRIDE,
                                    n_new - len(toks)))

    async def generate_stream(self, prompt: str, max_new_tokens: int = 256,
                              temperature: float = 0.0):
        if temperature > 0.0:
            # sampled path is host-driven; stream it as one final chunk
            yield await self.generate(prompt, max_new_tokens, temperature)
            return
        loop = asyncio.get_running_loop()
        q: asyncio.Queue = asyncio.Queue()

        def emit(chunk):
            loop.call_soon_threadsafe(q.put_nowait, chunk)

        def job():
            try:
                self._stream_sync(prompt, max_new_tokens, temperature, emit)
            except Exception as e:  # noqa: BLE001 — surface to the consumer
                emit(e)

        self._active += 1
        fut = loop.run_in_executor(self._pool, job)
        try:
            while True:
                chunk = await q.get()
                if isinstance(chunk, Exception):
                    raise chunk
                yield chunk
                if chunk.done_reason:
                    break
            await fut
        finally:
            self._active -= 1

    # ------------------------------------------------------------ metadata

    def throughput(self) -> float:
        r = self._rate.rate()
        return r if r > 0 else 100.0  # pre-first-request estimate

    def load(self) -> float:
        return min(1.0, float(self._active))

    def vram_gb(self) -> float:
        return float(self._props["total_mem_gb"])

    def gpu_model(self) -> str:
        return str(self._props["name"])

    async def close(self) -> None:
        self._pool.shutdown(wait=False)
