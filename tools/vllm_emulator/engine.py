"""Discrete-event continuous-batching engine.

Models one vLLM-on-MI355X server: a running queue stepped once per decode
iteration, a FIFO waiting queue, a KV-cache memory ledger over 288 GB
HBM3E, admission gated on max batch size + memory, eviction of the
youngest running request under memory pressure.

Timing laws (per iteration with batch n and a set P of newly admitted
requests):

    step_ms = alpha + beta*n  +  [gamma + delta*sum(inTokens in P)  if P]

i.e. decode time grows linearly in batch size and admissions pay a prefill
cost — exactly the laws the autoscaler's queue analyzer assumes, so the
controller's ITL/TTFT predictions can be validated closed-loop against
this emulator.  (The reference emulator uses constant step times and skips
prefill entirely: /root/reference/tools/vllm-emulator/vllm_model.py:67-77.)
"""

from __future__ import annotations

import asyncio
import os
from dataclasses import dataclass
from typing import List, Optional


@dataclass
class EmulatorSettings:
    model: str = os.getenv("MODEL_NAME", "default/llama-8b")
    # CDNA4-profiled linear timing laws (ms)
    decode_alpha: float = float(os.getenv("DECODE_ALPHA", "6.958"))
    decode_beta: float = float(os.getenv("DECODE_BETA", "0.042"))
    prefill_gamma: float = float(os.getenv("PREFILL_GAMMA", "20.0"))
    prefill_delta: float = float(os.getenv("PREFILL_DELTA", "0.002"))
    # memory model: MI355X 288 GB HBM3E
    mem_size_mb: int = int(os.getenv("MEM_SIZE", str(288 * 1024)))
    model_size_mb: int = int(os.getenv("MODEL_SIZE", "16000"))  # llama-8b bf16
    kv_mb_per_token: float = float(os.getenv("KVC_PER_TOKEN", "0.125"))
    usable_ratio: float = float(os.getenv("USABLE_RATIO", "0.9"))
    max_batch_size: int = int(os.getenv("MAX_BATCH_SIZE", "256"))
    avg_generated_len: int = int(os.getenv("AVG_TOKENS", "100"))
    tokens_distribution: str = os.getenv("TOKENS_DISTRIBUTION", "uniform")
    realtime: bool = os.getenv("REALTIME", "true").lower() == "true"
    # waiting-queue admission policy: "fifo" or "sorted_by_token_len"
    # (shortest sequences first — the reference's vLLM_varitaion_sorted_wq)
    waiting_queue_policy: str = os.getenv("WAITING_QUEUE_POLICY", "fifo")


class Clock:
    """Decode-step clock; virtual (no sleeping) or realtime."""

    def __init__(self, realtime: bool = True) -> None:
        self.realtime = realtime
        self.now_ms: float = 0.0

    async def advance(self, ms: float) -> float:
        # always yield to the event loop so submitters/waiters can run even
        # in virtual-time mode
        await asyncio.sleep(ms / 1000.0 if self.realtime else 0)
        self.now_ms += ms
        return self.now_ms


class DeviceState:
    """KV-cache memory ledger for one MI355X."""

    def __init__(self, settings: EmulatorSettings, metrics) -> None:
        self.capacity_mb = settings.mem_size_mb * settings.usable_ratio
        self.used_mb = float(settings.model_size_mb)
        if self.used_mb > self.capacity_mb:
            raise ValueError("model does not fit on the device")
        self._metrics = metrics

    @property
    def available_mb(self) -> float:
        return self.capacity_mb - self.used_mb

    def reserve(self, mb: float) -> None:
        if mb > self.available_mb:
            raise MemoryError(f"device OOM: want {mb}, have {self.available_mb}")
        self.used_mb += mb
        self._update()

    def release(self, mb: float) -> None:
        self.used_mb -= mb
        assert self.used_mb >= 0
        self._update()

    def _update(self) -> None:
        self._metrics.l(self._metrics.kv_cache_usage).set(self.used_mb / self.capacity_mb)


@dataclass
class RequestElement:
    req_id: str
    input_tokens: int
    output_tokens: int  # total target sequence length (input + generated)
    arrival_ms: float = 0.0
    first_token_ms: Optional[float] = None
    completion_ms: Optional[float] = None
    entered_waiting_ms: Optional[float] = None
    token_len: int = 0  # current sequence length
    stage: str = "new"  # new -> waiting? -> running -> finished
    event: Optional[asyncio.Event] = None

    def __post_init__(self) -> None:
        self.token_len = self.input_tokens

    @property
    def generated(self) -> int:
        return self.token_len - self.input_tokens

    @property
    def done(self) -> bool:
        return self.token_len >= self.output_tokens


class EmulatedVLLM:
    def __init__(self, settings: EmulatorSettings, metrics, clock: Optional[Clock] = None) -> None:
        self.settings = settings
        self.metrics = metrics
        self.clock = clock or Clock(realtime=settings.realtime)
        self.device = DeviceState(settings, metrics)
        self.running: List[RequestElement] = []
        self.waiting: List[RequestElement] = []
        self._newly_admitted: List[RequestElement] = []
        self.stop = False

    # ---------------------------------------------------------------- memory
    def _kv_mb(self, req: RequestElement) -> float:
        return req.token_len * self.settings.kv_mb_per_token

    def _can_admit(self, req: RequestElement) -> bool:
        if len(self.running) + 1 > self.settings.max_batch_size:
            return False
        # admit only if the request's KV plus one new token for every
        # running request fits
        need = self._kv_mb(req) + self.settings.kv_mb_per_token * (len(self.running) + 1)
        return need <= self.device.available_mb

    # ------------------------------------------------------------- admission
    def submit(self, req: RequestElement) -> None:
        req.arrival_ms = self.clock.now_ms
        m = self.metrics
        m.l(m.request_arrival).inc()
        m.l(m.prompt_tokens).observe(req.input_tokens)
        if not self.waiting and self._can_admit(req):
            self._admit(req)
        else:
            self._enqueue(req)

    async def submit_and_wait(self, req: RequestElement) -> RequestElement:
        req.event = asyncio.Event()
        self.submit(req)
        await req.event.wait()
        m = self.metrics
        m.l(m.tokens_total).inc(req.token_len)
        m.l(m.generation_tokens).observe(req.generated)
        return req

    def _admit(self, req: RequestElement) -> None:
        self.device.reserve(self._kv_mb(req))
        if req.entered_waiting_ms is not None:
            wait_s = (self.clock.now_ms - req.entered_waiting_ms) / 1000.0
            self.metrics.l(self.metrics.queue_time).observe(max(wait_s, 0.0))
        req.stage = "running"
        self.running.append(req)
        self._newly_admitted.append(req)
        self.metrics.l(self.metrics.running).inc()

    def _enqueue(self, req: RequestElement) -> None:
        req.entered_waiting_ms = self.clock.now_ms
        req.stage = "waiting"
        self.waiting.append(req)
        if self.settings.waiting_queue_policy == "sorted_by_token_len":
            self.waiting.sort(key=lambda r: r.token_len)
        self.metrics.l(self.metrics.waiting).inc()

    def _remove_running(self, req: RequestElement) -> None:
        self.device.release(self._kv_mb(req))
        self.running.remove(req)
        self.metrics.l(self.metrics.running).dec()

    def _evict_youngest(self) -> None:
        victim = self.running[-1]
        self._remove_running(victim)
        victim.entered_waiting_ms = self.clock.now_ms
        victim.stage = "waiting"
        self.waiting.insert(0, victim)
        self.metrics.l(self.metrics.waiting).inc()

    # ------------------------------------------------------------- main loop
    def _step_time_ms(self) -> float:
        s = self.settings
        n = len(self.running)
        ms = s.decode_alpha + s.decode_beta * n
        prefill_tokens = sum(r.input_tokens for r in self._newly_admitted)
        if prefill_tokens > 0:
            ms += s.prefill_gamma + s.prefill_delta * prefill_tokens
        return ms

    async def one_iteration(self) -> None:
        m = self.metrics
        step_ms = self._step_time_ms()
        self._newly_admitted.clear()
        now = await self.clock.advance(step_ms)

        if self.running:
            m.l(m.time_per_output_token).observe(step_ms / 1000.0)

        finished = []
        for req in self.running:
            self.device.reserve(self.settings.kv_mb_per_token)
            req.token_len += 1
            if req.first_token_ms is None:
                req.first_token_ms = now
                m.l(m.time_to_first_token).observe((now - req.arrival_ms) / 1000.0)
            if req.done:
                finished.append(req)
        for req in finished:
            req.stage = "finished"
            req.completion_ms = now
            self._remove_running(req)
            m.l(m.request_success).inc()
            if req.event is not None:
                req.event.set()

        if not self._evict_for_next_iteration():
            self._admit_from_waiting()

    def _evict_for_next_iteration(self) -> bool:
        """Evict tail requests until next iteration's +1-token growth fits."""
        evicted = False
        need = len(self.running) * self.settings.kv_mb_per_token
        while self.running and need > self.device.available_mb:
            self._evict_youngest()
            evicted = True
            need = len(self.running) * self.settings.kv_mb_per_token
        return evicted

    def _admit_from_waiting(self) -> None:
        while self.waiting and self._can_admit(self.waiting[0]):
            req = self.waiting.pop(0)
            self.metrics.l(self.metrics.waiting).dec()
            self._admit(req)

    async def run(self) -> None:
        while not self.stop:
            await self.one_iteration()
            if not self.running and not self.waiting and not self.clock.realtime:
                # virtual-time idle: yield so submitters can run
                await asyncio.sleep(0)
