"""OpenAI-compatible FastAPI server around the emulator engine.

Counterpart of /root/reference/tools/vllm-emulator/server.py: POST
/v1/chat/completions admits a request to the continuous-batching loop and
awaits completion; /metrics exposes the Prometheus series.  App-factory
style so tests can run several instances in-process via ASGI transport.
"""

from __future__ import annotations

import asyncio
import math
import random
import time
import uuid
from contextlib import asynccontextmanager
from typing import List, Optional

from fastapi import FastAPI, Response
from pydantic import BaseModel

from .engine import EmulatedVLLM, EmulatorSettings, RequestElement
from .metrics import EmulatorMetrics


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatCompletionRequest(BaseModel):
    model: str = "emulated-model"
    messages: List[ChatMessage]
    max_tokens: Optional[int] = 512
    temperature: Optional[float] = 0.1
    stream: Optional[bool] = False


class CompletionRequest(BaseModel):
    model: str = "emulated-model"
    prompt: str = ""
    max_tokens: Optional[int] = 512
    temperature: Optional[float] = 0.1
    stream: Optional[bool] = False


class OutputLengthSampler:
    def __init__(self, avg_generated: int, distribution: str) -> None:
        self.avg = avg_generated
        self.distribution = distribution

    def sample(self) -> int:
        if self.distribution == "uniform":
            return random.randint(0, 2 * self.avg)
        if self.distribution == "uniform-narrow":
            return random.randint(self.avg // 2, 3 * self.avg // 2)
        if self.distribution == "sharegpt":
            # heavy-tailed ShareGPT-like lengths: lognormal with CV~0.95
            # (sigma 0.8), mean pinned to avg, capped at 4x to keep a
            # single request from starving the batch
            sigma = 0.8
            mu = math.log(self.avg) - sigma * sigma / 2.0
            return max(1, min(int(random.lognormvariate(mu, sigma)), 4 * self.avg))
        return self.avg  # deterministic


def create_app(settings: Optional[EmulatorSettings] = None) -> FastAPI:
    settings = settings or EmulatorSettings()
    metrics = EmulatorMetrics(settings.model)
    engine = EmulatedVLLM(settings, metrics)
    sampler = OutputLengthSampler(settings.avg_generated_len, settings.tokens_distribution)

    @asynccontextmanager
    async def lifespan(app: FastAPI):
        task = asyncio.create_task(engine.run())
        yield
        engine.stop = True
        task.cancel()

    app = FastAPI(title="MI355X vLLM emulator", lifespan=lifespan)
    app.state.engine = engine
    app.state.metrics = metrics
    app.state.settings = settings

    @app.post("/v1/chat/completions")
    async def chat_completions(request: ChatCompletionRequest):
        input_len = max(len(request.messages[-1].content.split()), 1)
        output_len = input_len + sampler.sample()
        req = RequestElement(
            req_id=str(uuid.uuid4()),
            input_tokens=input_len,
            output_tokens=output_len,
        )
        await engine.submit_and_wait(req)
        ttft_ms = (req.first_token_ms or 0) - req.arrival_ms
        return {
            "id": req.req_id,
            "object": "chat.completion",
            "created": int(time.time()),
            "model": request.model,
            "choices": [
                {
                    "index": 0,
                    "message": {
                        "role": "assistant",
                        "content": (
                            f"arrival={req.arrival_ms:.1f}ms "
                            f"completion={req.completion_ms:.1f}ms ttft={ttft_ms:.1f}ms "
                            f"in={req.input_tokens} out={req.generated}"
                        ),
                    },
                }
            ],
            "usage": {
                "prompt_tokens": req.input_tokens,
                "completion_tokens": req.generated,
                "total_tokens": req.token_len,
            },
        }

    @app.post("/v1/completions")
    async def completions(request: CompletionRequest):
        input_len = max(len(request.prompt.split()), 1)
        req = RequestElement(
            req_id=str(uuid.uuid4()),
            input_tokens=input_len,
            output_tokens=input_len + sampler.sample(),
        )
        await engine.submit_and_wait(req)
        return {
            "id": req.req_id,
            "object": "text_completion",
            "created": int(time.time()),
            "model": request.model,
            "choices": [{"index": 0, "text": f"emulated {req.generated} tokens", "finish_reason": "stop"}],
            "usage": {
                "prompt_tokens": req.input_tokens,
                "completion_tokens": req.generated,
                "total_tokens": req.token_len,
            },
        }

    @app.get("/metrics")
    async def metrics_endpoint():
        return Response(content=metrics.expose(), media_type="text/plain; version=0.0.4")

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app


def main() -> None:
    import uvicorn

    uvicorn.run(create_app(), host="0.0.0.0", port=8000)


if __name__ == "__main__":
    main()
