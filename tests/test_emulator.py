"""Emulator engine tests: continuous batching, memory ledger, metric
emission (counterpart of the reference's emulator behavior, §2c)."""

import asyncio

import pytest

from vllm_emulator.engine import EmulatedVLLM, EmulatorSettings, RequestElement
from vllm_emulator.metrics import EmulatorMetrics


def make_engine(**kw):
    defaults = dict(
        model="m",
        decode_alpha=5.0,
        decode_beta=0.1,
        prefill_gamma=10.0,
        prefill_delta=0.01,
        mem_size_mb=1000,
        model_size_mb=100,
        kv_mb_per_token=1.0,
        usable_ratio=1.0,
        max_batch_size=4,
        realtime=False,
    )
    defaults.update(kw)
    settings = EmulatorSettings(**defaults)
    metrics = EmulatorMetrics(settings.model)
    return EmulatedVLLM(settings, metrics), metrics


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def sample(metrics, name, labels=None):
    labels = labels if labels is not None else {"model_name": "m"}
    return metrics.registry.get_sample_value(name, labels)


class TestEngine:
    def test_single_request_lifecycle(self):
        engine, metrics = make_engine()

        async def scenario():
            req = RequestElement("r1", input_tokens=10, output_tokens=13)
            task = asyncio.ensure_future(engine.submit_and_wait(req))
            while not task.done():
                await engine.one_iteration()
            return await task

        req = run(scenario())
        assert req.stage == "finished"
        assert req.generated == 3
        assert req.first_token_ms is not None and req.first_token_ms < req.completion_ms
        assert sample(metrics, "vllm:request_success_total") == 1.0
        assert sample(metrics, "vllm:request_arrival_total") == 1.0
        assert sample(metrics, "vllm:request_prompt_tokens_sum") == 10.0
        assert sample(metrics, "vllm:request_generation_tokens_sum") == 3.0
        assert sample(metrics, "vllm:time_to_first_token_seconds_count") == 1.0
        assert sample(metrics, "vllm:num_requests_running") == 0.0

    def test_step_time_laws(self):
        engine, _ = make_engine()

        async def scenario():
            r1 = RequestElement("r1", input_tokens=100, output_tokens=200)
            engine.submit(r1)
            # admission pays prefill: 5 + 0.1*1 + 10 + 0.01*100 = 16.1
            assert engine._step_time_ms() == pytest.approx(5.0 + 0.1 + 10.0 + 1.0)
            await engine.one_iteration()
            # steady decode: alpha + beta*n only
            assert engine._step_time_ms() == pytest.approx(5.0 + 0.1)

        run(scenario())

    def test_batch_limit_queues(self):
        engine, metrics = make_engine(max_batch_size=2)

        async def scenario():
            reqs = [RequestElement(f"r{i}", 5, 50) for i in range(4)]
            for r in reqs:
                engine.submit(r)
            assert len(engine.running) == 2
            assert len(engine.waiting) == 2
            assert sample(metrics, "vllm:num_requests_waiting") == 2.0
            for _ in range(120):
                await engine.one_iteration()
            assert all(r.done for r in reqs)

        run(scenario())

    def test_memory_pressure_evicts_youngest(self):
        # capacity 1000, model 100 -> 900 usable KV; two requests of 300
        # tokens each grow 1/iter: evict when +2 would exceed the pool
        engine, _ = make_engine(mem_size_mb=730, model_size_mb=100, max_batch_size=8)

        async def scenario():
            a = RequestElement("a", 300, 10_000)
            b = RequestElement("b", 300, 10_000)
            engine.submit(a)
            engine.submit(b)
            assert len(engine.running) == 2
            evicted = False
            for _ in range(40):
                await engine.one_iteration()
                if len(engine.waiting) == 1:
                    evicted = True
                    break
            assert evicted
            # the youngest (b, submitted last) was evicted
            assert engine.waiting[0].req_id == "b"

        run(scenario())

    def test_oversized_model_rejected(self):
        with pytest.raises(ValueError):
            make_engine(model_size_mb=2000, mem_size_mb=1000)

    def test_mi355x_defaults(self):
        settings = EmulatorSettings()
        assert settings.mem_size_mb == 288 * 1024  # 288 GB HBM3E
        assert settings.max_batch_size == 256


class TestHTTPServer:
    def test_chat_completion_and_metrics(self):
        import threading
        import time as _time

        import httpx
        import uvicorn

        from vllm_emulator.server import create_app

        settings = EmulatorSettings(
            model="http-model",
            decode_alpha=1.0,
            decode_beta=0.01,
            prefill_gamma=1.0,
            prefill_delta=0.001,
            avg_generated_len=5,
            tokens_distribution="deterministic",
            realtime=True,
        )
        app = create_app(settings)
        config = uvicorn.Config(app, host="127.0.0.1", port=0, log_level="error")
        server = uvicorn.Server(config)
        thread = threading.Thread(target=server.run, daemon=True)
        thread.start()
        for _ in range(100):
            if server.started:
                break
            _time.sleep(0.05)
        assert server.started
        port = server.servers[0].sockets[0].getsockname()[1]
        base = f"http://127.0.0.1:{port}"
        try:
            resp = httpx.post(
                f"{base}/v1/chat/completions",
                json={"model": "http-model", "messages": [{"role": "user", "content": "a b c"}]},
                timeout=30.0,
            )
            assert resp.status_code == 200
            body = resp.json()
            assert body["usage"]["prompt_tokens"] == 3
            assert body["usage"]["completion_tokens"] == 5
            comp = httpx.post(
                f"{base}/v1/completions",
                json={"model": "http-model", "prompt": "one two three four"},
                timeout=30.0,
            )
            assert comp.status_code == 200
            assert comp.json()["usage"]["prompt_tokens"] == 4
            metrics_text = httpx.get(f"{base}/metrics", timeout=5.0).text
            assert 'vllm:request_success_total{model_name="http-model"} 2.0' in metrics_text
            assert "vllm:time_to_first_token_seconds_sum" in metrics_text
            assert "vllm:request_prompt_tokens_sum" in metrics_text
        finally:
            server.should_exit = True
            thread.join(timeout=5.0)


class TestWaitingQueuePolicy:
    def test_sorted_by_token_len(self):
        engine, _ = make_engine(max_batch_size=1, waiting_queue_policy="sorted_by_token_len")

        async def scenario():
            engine.submit(RequestElement("running", 5, 500))  # occupies the slot
            engine.submit(RequestElement("long", 100, 200))
            engine.submit(RequestElement("short", 10, 20))
            engine.submit(RequestElement("mid", 50, 60))
            assert [r.req_id for r in engine.waiting] == ["short", "mid", "long"]

        run(scenario())

    def test_fifo_default(self):
        engine, _ = make_engine(max_batch_size=1)

        async def scenario():
            engine.submit(RequestElement("running", 5, 500))
            engine.submit(RequestElement("long", 100, 200))
            engine.submit(RequestElement("short", 10, 20))
            assert [r.req_id for r in engine.waiting] == ["long", "short"]

        run(scenario())


class TestExperiment:
    def test_virtual_time_simulation(self):
        import asyncio as aio

        from vllm_emulator.experiment import simulate

        settings = EmulatorSettings(
            model="exp", decode_alpha=5.0, decode_beta=0.1, prefill_gamma=5.0,
            prefill_delta=0.01, avg_generated_len=20, max_batch_size=16, realtime=False,
        )
        result = aio.new_event_loop().run_until_complete(
            simulate(settings, rate_rps=10.0, duration_s=5.0, in_tokens=32)
        )
        assert result["completed"] == result["submitted"] > 10
        assert result["ttft_ms"]["mean"] > 0
        assert result["latency_ms"]["mean"] > result["ttft_ms"]["mean"]
