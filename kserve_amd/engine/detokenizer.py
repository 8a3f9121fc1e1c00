"""Incremental detokenization + stop-string handling.

Standard incremental scheme: keep (prefix_offset, read_offset) per request;
re-decode a small trailing window and emit only the stable delta (avoids
emitting replacement chars mid-multibyte-token). Fresh implementation.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

from kserve_amd.engine.request import Request


class Detokenizer:
    def __init__(self, tokenizer):
        self.tokenizer = tokenizer

    def decode_new(self, req: Request) -> str:
        """Decode tokens appended since last call; returns the text delta."""
        if self.tokenizer is None:
            return ""
        all_out = req.output_token_ids
        prefix_text = self.tokenizer.decode(
            all_out[req.prefix_offset : req.read_offset],
            skip_special_tokens=True,
        )
        new_text = self.tokenizer.decode(
            all_out[req.prefix_offset :], skip_special_tokens=True
        )
        if len(new_text) > len(prefix_text) and not new_text.endswith("�"):
            delta = new_text[len(prefix_text) :]
            req.prefix_offset = req.read_offset
            req.read_offset = len(all_out)
            req.output_text += delta
            return delta
        return ""

    @staticmethod
    def check_stop_strings(req: Request) -> Optional[int]:
        """If a stop string appears in output_text, truncate and return the
        index where output should be cut; else None."""
        for s in req.sampling_params.stop:
            idx = req.output_text.find(s)
            if idx != -1:
                req.output_text = req.output_text[:idx]
                return idx
        return None
