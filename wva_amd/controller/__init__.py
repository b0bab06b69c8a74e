"""Controller / orchestration layers (L2/L3): reconciler, collector, model
analyzer, optimizer engine, actuator, metrics emitter, utils.

Parity with /root/reference/internal/ — restructured into one coherent
package (the reference declares three different directories all as
``package controller`` and disambiguates by import alias).
"""
