"""Closed-loop model validation: the queue analyzer's predictions against
the discrete-event emulator's measurements under Poisson load.

The emulator implements the same linear service laws the analyzer assumes
(decode = alpha + beta*batch, prefill = gamma + delta*tokens) but as an
actual continuous-batching event loop — so agreement here validates the
state-dependent M/M/1/K model itself, not just the arithmetic.  This tier
has no counterpart in the reference (its emulator uses constant step
times the analyzer does not assume, so no closed loop is possible there).
"""

import asyncio

import pytest

from vllm_emulator.engine import EmulatorSettings
from vllm_emulator.experiment import simulate
from wva_amd.analyzer import (
    Configuration,
    DecodeParms,
    PrefillParms,
    QueueAnalyzer,
    RequestSize,
    ServiceParms,
)

ALPHA, BETA, GAMMA, DELTA = 10.0, 0.5, 5.0, 0.02
IN_TOKENS, OUT_TOKENS, MAX_BATCH = 128, 50, 16


def predict(rate_rps):
    qa = QueueAnalyzer(
        Configuration(
            max_batch_size=MAX_BATCH,
            max_queue_size=10 * MAX_BATCH,
            service_parms=ServiceParms(
                prefill=PrefillParms(GAMMA, DELTA), decode=DecodeParms(ALPHA, BETA)
            ),
        ),
        RequestSize(IN_TOKENS, OUT_TOKENS),
    )
    return qa, qa.analyze(rate_rps)


def measure(rate_rps, duration_s=90.0, seed=3):
    settings = EmulatorSettings(
        model="validation",
        decode_alpha=ALPHA,
        decode_beta=BETA,
        prefill_gamma=GAMMA,
        prefill_delta=DELTA,
        avg_generated_len=OUT_TOKENS,
        tokens_distribution="deterministic",
        max_batch_size=MAX_BATCH,
        realtime=False,
        mem_size_mb=300_000,
        model_size_mb=1_000,
        kv_mb_per_token=0.01,
    )
    return asyncio.new_event_loop().run_until_complete(
        simulate(settings, rate_rps=rate_rps, duration_s=duration_s, in_tokens=IN_TOKENS, seed=seed)
    )


@pytest.mark.parametrize("utilization", [0.3, 0.6])
def test_predictions_match_emulator(utilization):
    qa, _ = predict(1.0)
    rate = qa.rate_range.max * utilization
    metrics = qa.analyze(rate)
    result = measure(rate)

    measured_itl = (result["latency_ms"]["mean"] - result["ttft_ms"]["mean"]) / (OUT_TOKENS - 1)
    measured_ttft = result["ttft_ms"]["mean"]
    measured_tput = result["throughput_rps"]

    assert metrics.avg_token_time == pytest.approx(measured_itl, rel=0.25)
    assert metrics.avg_wait_time + metrics.avg_prefill_time == pytest.approx(
        measured_ttft, rel=0.5
    )
    assert metrics.throughput == pytest.approx(measured_tput, rel=0.10)


def test_sizing_keeps_emulator_within_slo():
    """size() for an ITL target, then drive the emulator at rate*: the
    measured ITL must honor the target."""
    from wva_amd.analyzer import TargetPerf

    qa, _ = predict(1.0)
    target_itl = 16.0  # between alpha+beta and alpha+beta*N
    _, metrics, achieved = qa.size(TargetPerf(target_itl=target_itl))
    rate_star = metrics.throughput
    result = measure(rate_star)
    measured_itl = (result["latency_ms"]["mean"] - result["ttft_ms"]["mean"]) / (OUT_TOKENS - 1)
    # the real system driven at rate* stays at (or tolerably near) the SLO
    assert measured_itl <= target_itl * 1.25
    assert achieved.target_itl <= target_itl * 1.001
