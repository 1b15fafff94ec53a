"""Engine seam between the mesh worker and the compute backend.

Reference parity: pkg/crowdllama/api.go — `UnifiedAPIHandler` is the
function the worker's stream handler calls (peer.go:231); the reference's
`WorkerAPIHandler` POSTs to Ollama, `DefaultAPIHandler` echoes. Here the
seam is an EngineBase, with the HIP engine as the real backend and
MockEngine replacing the integration tests' MockOllamaServer
(integration_test.go:31-135)."""

from __future__ import annotations

import asyncio
import threading
import time
from dataclasses import dataclass


@dataclass
class GenerateResult:
    text: str
    tokens_generated: int = 0
    duration_ns: int = 0
    done_reason: str = "stop"


class EngineBase:
    """One loaded model served by a worker."""

    model_name: str = ""

    async def generate(self, prompt: str, max_new_tokens: int = 256,
                       temperature: float = 0.0) -> GenerateResult:
        raise NotImplementedError

    async def generate_stream(self, prompt: str, max_new_tokens: int = 256,
                              temperature: float = 0.0):
        """Async iterator of GenerateResult chunks; `text` is the DELTA since
        the previous chunk, and exactly the last chunk has done_reason set
        (capability extension over the reference — SURVEY.md §2.2 notes the
        reference carries `stream` but never streams, gateway.go:243-293).
        Default: one final chunk from the non-streaming path."""
        yield await self.generate(prompt, max_new_tokens, temperature)

    def throughput(self) -> float:
        """Measured rolling tokens/sec (advertised in metadata)."""
        return 0.0

    def load(self) -> float:
        """Current load in [0,1] (advertised; scheduler input)."""
        return 0.0

    def vram_gb(self) -> float:
        return 0.0

    def gpu_model(self) -> str:
        return ""

    async def close(self) -> None:
        pass


class MockEngine(EngineBase):
    """Deterministic mock backend (test seam; reference MockOllamaServer)."""

    def __init__(self, model_name: str = "mock", response: str | None = None,
                 delay: float = 0.0, throughput: float = 100.0):
        self.model_name = model_name
        self._response = response
        self._delay = delay
        self._throughput = throughput
        self.calls = 0

    async def generate(self, prompt: str, max_new_tokens: int = 256,
                       temperature: float = 0.0) -> GenerateResult:
        self.calls += 1
        if self._delay:
            await asyncio.sleep(self._delay)
        text = self._response or (
            f"This is a mock response from {self.model_name} "
            f"to: {prompt[:64]}")
        return GenerateResult(text=text, tokens_generated=len(text.split()),
                              duration_ns=int(self._delay * 1e9))

    async def generate_stream(self, prompt: str, max_new_tokens: int = 256,
                              temperature: float = 0.0):
        """Word-at-a-time streaming (test seam for the wire/gateway path)."""
        full = await self.generate(prompt, max_new_tokens, temperature)
        words = full.text.split(" ")
        for i, w in enumerate(words[:-1]):
            yield GenerateResult(text=w + " ", tokens_generated=i + 1,
                                 done_reason="")
        yield GenerateResult(text=words[-1] if words else "",
                             tokens_generated=full.tokens_generated,
                             duration_ns=full.duration_ns,
                             done_reason=full.done_reason)

    def throughput(self) -> float:
        return self._throughput

    def gpu_model(self) -> str:
        return "MockGPU"

    def vram_gb(self) -> float:
        return 24.0


class RollingRate:
    """Rolling tokens/sec over a sliding window (replaces the reference's
    hardcoded 150 tok/s advertisement, peer.go:323)."""

    def __init__(self, window: float = 60.0):
        self.window = window
        self._events: list[tuple[float, int]] = []
        self._lock = threading.Lock()

    def add(self, n_tokens: int) -> None:
        with self._lock:
            now = time.time()
            self._events.append((now, n_tokens))
            cutoff = now - self.window
            while self._events and self._events[0][0] < cutoff:
                self._events.pop(0)

    def rate(self) -> float:
        with self._lock:
            if not self._events:
                return 0.0
            now = time.time()
            cutoff = now - self.window
            toks = sum(n for t, n in self._events if t >= cutoff)
            span = max(1e-3, min(self.window, now - self._events[0][0]))
            return toks / span
