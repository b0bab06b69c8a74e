"""Structured JSON logger (internal/logger/logger.go analog):
level-from-env table (GetZapLevelFromEnv, logger.go:40-54) and the JSON
line contract every other test's captured stdout relies on."""

import json
import logging

import pytest

from wva_amd.controller.logger import _Logger, level_from_env


class TestLevelFromEnv:
    @pytest.mark.parametrize(
        "raw,expected",
        [
            ("debug", logging.DEBUG),
            ("info", logging.INFO),
            ("warn", logging.WARNING),
            ("error", logging.ERROR),
            ("DEBUG", logging.DEBUG),  # case-insensitive, like the reference
            ("Error", logging.ERROR),
            ("", logging.INFO),        # unset/unknown -> info default
            ("verbose", logging.INFO),
        ],
    )
    def test_table(self, monkeypatch, raw, expected):
        monkeypatch.setenv("LOG_LEVEL", raw)
        assert level_from_env() == expected

    def test_absent_env_defaults_info(self, monkeypatch):
        monkeypatch.delenv("LOG_LEVEL", raising=False)
        assert level_from_env() == logging.INFO


class TestJsonLineContract:
    def _capture(self, capsys, level, emit):
        lg = _Logger()
        lg.init(level=level)
        emit(lg)
        out = capsys.readouterr().out.strip()
        return [json.loads(line) for line in out.splitlines() if line]

    def test_line_shape_and_kv(self, capsys):
        lines = self._capture(
            capsys, logging.INFO, lambda lg: lg.info("hello", variant="va-1", n=3)
        )
        assert len(lines) == 1
        entry = lines[0]
        assert entry["level"] == "info"
        assert entry["msg"] == "hello"
        assert entry["variant"] == "va-1" and entry["n"] == 3
        assert isinstance(entry["ts"], float)

    def test_level_filtering(self, capsys):
        lines = self._capture(
            capsys,
            logging.WARNING,
            lambda lg: (lg.debug("d"), lg.info("i"), lg.warn("w"), lg.error("e")),
        )
        assert [e["level"] for e in lines] == ["warning", "error"]

    def test_every_level_emits_valid_json(self, capsys):
        lines = self._capture(
            capsys,
            logging.DEBUG,
            lambda lg: (lg.debug("d"), lg.info("i"), lg.warn("w"), lg.error("e")),
        )
        assert len(lines) == 4  # and json.loads above already proved validity
