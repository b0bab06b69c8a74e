"""MI355X-profiled vLLM emulator (test/dev backend).

A discrete-event model of a vLLM server used by the e2e tier and the
KEDA-ramp scenarios.  Counterpart of /root/reference/tools/vllm-emulator/
(X1 layer), redesigned rather than translated:

- device defaults are MI355X: 294,912 MB (288 GB HBM3E) with a 0.9 usable
  ratio, CDNA4-profiled KV-cache bytes per token;
- step timing follows the same linear laws the autoscaler assumes
  (decode = alpha + beta*batch; prefill = gamma + delta*inTokens*batch), so
  closed-loop behavior of controller predictions can be validated;
- the emulator emits the FULL vLLM metric set the collector queries —
  including ``vllm:request_prompt_tokens_*`` and
  ``vllm:time_to_first_token_seconds_*``, which the reference emulator
  omits (its collector needs a DISABLING_TTFT flag).
"""

from .engine import Clock, DeviceState, EmulatedVLLM, EmulatorSettings, RequestElement
from .metrics import EmulatorMetrics

__all__ = [
    "Clock",
    "DeviceState",
    "EmulatedVLLM",
    "EmulatorSettings",
    "RequestElement",
    "EmulatorMetrics",
]
