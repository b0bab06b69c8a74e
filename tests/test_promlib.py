"""promlib tests: scraping, PromQL-subset evaluation, PromAPI adapter."""

import math

import pytest

from wva_amd.controller.promclient import PromQueryError
from wva_amd.promlib import PromlibAPI, PromQLError, Scraper, TimeSeriesStore, evaluate


def make_store():
    return TimeSeriesStore()


class TestStore:
    def test_select_by_labels(self):
        store = make_store()
        store.add_sample("m", {"a": "1", "b": "x"}, 1.0, ts=100.0)
        store.add_sample("m", {"a": "2", "b": "x"}, 2.0, ts=100.0)
        assert len(store.select("m", {"b": "x"})) == 2
        assert len(store.select("m", {"a": "1"})) == 1
        assert store.select("m", {"a": "3"}) == []

    def test_retention(self):
        store = TimeSeriesStore(retention=10.0)
        store.add_sample("m", {}, 1.0, ts=0.0)
        store.add_sample("m", {}, 2.0, ts=100.0)
        (series,) = store.select("m", {})
        assert len(series.samples) == 1


class TestPromQL:
    def test_instant_selector(self):
        store = make_store()
        store.add_sample("up_x", {"job": "j"}, 1.0, ts=50.0)
        out = evaluate('up_x{job="j"}', store, now=60.0)
        assert len(out) == 1
        assert out[0].value == 1.0
        assert out[0].timestamp == 50.0  # original sample timestamp

    def test_lookback_staleness(self):
        store = make_store()
        store.add_sample("m", {}, 1.0, ts=0.0)
        assert evaluate("m", store, now=301.0) == []
        assert len(evaluate("m", store, now=299.0)) == 1

    def test_rate_counter(self):
        store = make_store()
        for t, v in [(0, 0.0), (30, 30.0), (60, 60.0)]:
            store.add_sample("c_total", {"x": "1"}, v, ts=float(t))
        out = evaluate("rate(c_total[1m])", store, now=60.0)
        assert out[0].value == pytest.approx(1.0)

    def test_rate_needs_two_samples(self):
        store = make_store()
        store.add_sample("c_total", {}, 5.0, ts=59.0)
        assert evaluate("rate(c_total[1m])", store, now=60.0) == []

    def test_counter_reset(self):
        store = make_store()
        store.add_sample("c_total", {}, 100.0, ts=0.0)
        store.add_sample("c_total", {}, 10.0, ts=50.0)
        out = evaluate("rate(c_total[1m])", store, now=50.0)
        assert out[0].value == pytest.approx(10.0 / 50.0)

    def test_sum_rate_over_series(self):
        store = make_store()
        for pod in ("p1", "p2"):
            store.add_sample("c_total", {"pod": pod, "ns": "d"}, 0.0, ts=0.0)
            store.add_sample("c_total", {"pod": pod, "ns": "d"}, 60.0, ts=60.0)
        out = evaluate('sum(rate(c_total{ns="d"}[1m]))', store, now=60.0)
        assert out[0].value == pytest.approx(2.0)

    def test_ratio_query(self):
        store = make_store()
        for name, vals in (("s_sum", (0.0, 120.0)), ("s_count", (0.0, 60.0))):
            store.add_sample(name, {"m": "x"}, vals[0], ts=0.0)
            store.add_sample(name, {"m": "x"}, vals[1], ts=60.0)
        out = evaluate('sum(rate(s_sum{m="x"}[1m]))/sum(rate(s_count{m="x"}[1m]))', store, now=60.0)
        assert out[0].value == pytest.approx(2.0)

    def test_div_by_zero_is_nan(self):
        store = make_store()
        store.add_sample("a", {}, 1.0, ts=0.0)
        store.add_sample("b", {}, 0.0, ts=0.0)
        out = evaluate("sum(a)/sum(b)", store, now=1.0)
        assert math.isnan(out[0].value)

    def test_empty_side_returns_empty(self):
        store = make_store()
        store.add_sample("a", {}, 1.0, ts=0.0)
        assert evaluate("sum(a)/sum(zzz)", store, now=1.0) == []

    def test_avg_min_max_count(self):
        store = make_store()
        for i, v in enumerate((1.0, 3.0, 5.0)):
            store.add_sample("g", {"i": str(i)}, v, ts=10.0)
        assert evaluate("avg(g)", store, now=10.0)[0].value == pytest.approx(3.0)
        assert evaluate("min(g)", store, now=10.0)[0].value == 1.0
        assert evaluate("max(g)", store, now=10.0)[0].value == 5.0
        assert evaluate("count(g)", store, now=10.0)[0].value == 3.0

    def test_parse_errors(self):
        store = make_store()
        for bad in ("sum(", "m{x=1}", "rate(m)", "m / ", "m garbage"):
            with pytest.raises(PromQLError):
                evaluate(bad, store, now=0.0)

    def test_collector_query_shapes_evaluate(self):
        from wva_amd.controller import collector

        store = make_store()
        labels = {"model_name": "llama", "namespace": "default"}
        for name in (
            "vllm:request_success_total",
            "vllm:request_prompt_tokens_sum",
            "vllm:request_prompt_tokens_count",
        ):
            store.add_sample(name, labels, 0.0, ts=0.0)
            store.add_sample(name, labels, 60.0, ts=60.0)
        out = evaluate(collector.arrival_query("llama", "default"), store, now=60.0)
        assert out[0].value == pytest.approx(1.0)
        out = evaluate(collector.avg_prompt_tokens_query("llama", "default"), store, now=60.0)
        assert out[0].value == pytest.approx(1.0)


class TestScraper:
    def test_scrape_prometheus_text(self):
        store = make_store()
        scraper = Scraper(store)
        payload = (
            '# TYPE vllm:request_success counter\n'
            'vllm:request_success_total{model_name="m"} 7.0\n'
            '# TYPE vllm:num_requests_running gauge\n'
            'vllm:num_requests_running{model_name="m"} 3.0\n'
        )
        scraper.add_target(lambda: payload, extra_labels={"namespace": "default"})
        scraper.scrape_once(ts=100.0)
        out = evaluate('vllm:num_requests_running{namespace="default"}', store, now=100.0)
        assert out[0].value == 3.0
        (series,) = store.select("vllm:request_success_total", {"model_name": "m"})
        assert series.samples[0] == (100.0, 7.0)

    def test_failing_target_skipped(self):
        store = make_store()
        scraper = Scraper(store)

        def boom():
            raise RuntimeError("down")

        scraper.add_target(boom)
        scraper.add_target(lambda: "ok_metric 1.0\n")
        scraper.scrape_once(ts=1.0)
        assert store.select("ok_metric", {})


class TestPromlibAPI:
    def test_up_synthetic(self):
        api = PromlibAPI(make_store())
        out = api.query("up")
        assert out[0].value == 1.0

    def test_error_wrapped(self):
        api = PromlibAPI(make_store())
        with pytest.raises(PromQueryError):
            api.query("sum(")


class TestScraperLoop:
    def test_background_interval_scraping(self):
        import time

        store = make_store()
        scraper = Scraper(store)
        calls = {"n": 0}

        def fetch():
            calls["n"] += 1
            return f"loop_metric {calls['n']}\n"

        scraper.add_target(fetch)
        scraper.start(interval=0.05)
        time.sleep(0.4)
        scraper.stop()
        assert calls["n"] >= 3  # several periodic scrapes happened
        (series,) = store.select("loop_metric", {})
        assert len(series.samples) == calls["n"]
        # stop() joins the thread: no further scrapes
        n = calls["n"]
        time.sleep(0.15)
        assert calls["n"] == n


class TestRangeScanRewrite:
    def test_inclusive_bounds_and_order(self):
        # regression for the reversed-scan rewrite: inclusive [start, end],
        # ascending output, early exit must not drop edge samples
        from wva_amd.promlib.store import TimeSeriesStore

        st = TimeSeriesStore(retention=1000)
        for i in range(10):
            st.add_sample("m", {}, float(i), ts=100.0 + i)
        s = st.select("m", {})[0]
        assert s.range(102.0, 105.0) == [(102.0, 2.0), (103.0, 3.0), (104.0, 4.0), (105.0, 5.0)]
        assert s.range(108.5, 200.0) == [(109.0, 9.0)]
        assert s.range(0.0, 99.0) == []
        assert s.range(109.0, 109.0) == [(109.0, 9.0)]
