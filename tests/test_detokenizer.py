"""Incremental detokenization: UTF-8 multibyte boundaries must never leak
replacement characters into the stream (reference detokenizer semantics)."""

import pytest

from kserve_amd.engine.detokenizer import Detokenizer
from kserve_amd.engine.request import Request
from kserve_amd.engine.sampling_params import SamplingParams


class ByteTokenizer:
    """1 token = 1 byte (forces multibyte chars to split across tokens)."""

    eos_token_id = None

    def decode(self, ids, skip_special_tokens=True):
        return bytes(ids).decode("utf-8", errors="replace")

    def encode(self, text):
        return list(text.encode("utf-8"))


def test_multibyte_stream_no_replacement_chars():
    tok = ByteTokenizer()
    det = Detokenizer(tok)
    req = Request("r", [1], SamplingParams(max_tokens=32), eos_token_id=None)
    text = "héllo — 日本語 🙂 end"
    deltas = []
    for b in text.encode("utf-8"):
        req.output_token_ids.append(b)
        deltas.append(det.decode_new(req))
    assert "�" not in "".join(deltas)
    assert "".join(deltas) == text
    assert req.output_text == text


def test_stop_string_truncates_text():
    tok = ByteTokenizer()
    det = Detokenizer(tok)
    req = Request(
        "r", [1], SamplingParams(max_tokens=32, stop=["STOP"]),
        eos_token_id=None,
    )
    for b in "abc STOP def".encode("utf-8"):
        req.output_token_ids.append(b)
        det.decode_new(req)
    idx = det.check_stop_strings(req)
    assert idx is not None
    assert req.output_text == "abc "
