"""Request/sequence state tracked by the continuous-batching scheduler.

Fresh design (the reference has no engine; vLLM semantics studied per
SURVEY.md §7 stage 4 — "study vLLM semantics, not its code").
"""

from __future__ import annotations

import enum
import time
from dataclasses import dataclass, field
from typing import List, Optional

from kserve_amd.engine.sampling_params import SamplingParams


class RequestStatus(enum.Enum):
    WAITING = enum.auto()
    RUNNING = enum.auto()
    PREEMPTED = enum.auto()
    FINISHED_STOPPED = enum.auto()   # hit stop token / stop string / EOS
    FINISHED_LENGTH = enum.auto()    # hit max_tokens or max_model_len
    FINISHED_ABORTED = enum.auto()   # client abort / engine shutdown

    @property
    def finished(self) -> bool:
        return self in (
            RequestStatus.FINISHED_STOPPED,
            RequestStatus.FINISHED_LENGTH,
            RequestStatus.FINISHED_ABORTED,
        )


FINISH_REASON = {
    RequestStatus.FINISHED_STOPPED: "stop",
    RequestStatus.FINISHED_LENGTH: "length",
    RequestStatus.FINISHED_ABORTED: "abort",
}


class Request:
    """One generation stream (n=1; n>1 is fanned out by the engine front)."""

    def __init__(
        self,
        request_id: str,
        prompt_token_ids: List[int],
        sampling_params: SamplingParams,
        arrival_time: Optional[float] = None,
        eos_token_id: Optional[int] = None,
    ):
        self.request_id = request_id
        self.prompt_token_ids = list(prompt_token_ids)
        self.output_token_ids: List[int] = []
        self.sampling_params = sampling_params
        self.status = RequestStatus.WAITING
        self.arrival_time = arrival_time if arrival_time is not None else time.monotonic()
        self.eos_token_id = eos_token_id
        self.is_finished = False  # cached; see maybe_finish / set_status
        # paging state
        self.block_table: List[int] = []
        # number of prompt tokens whose KV is already computed (chunked prefill)
        self.num_computed_tokens = 0
        # multi-LoRA adapter id (0 = base model)
        self.lora_id = 0
        # guided decoding (engine/guided.py): per-request JSON machine
        self.guided_machine = None
        # scheduler arrival sequence (set by Scheduler.add_request)
        self.arrival_seq = 0
        # prefix caching: tokens whose KV was found resident at allocate
        self.num_cached_tokens = 0
        self.prompt_block_hashes = []
        # metrics
        self.first_token_time: Optional[float] = None
        self.finish_time: Optional[float] = None
        # incremental detokenization state
        self.prefix_offset = 0
        self.read_offset = 0
        self.output_text = ""
        # per-step {token_id: logprob} maps when sampling_params.logprobs
        self.logprobs: List[dict] = []

    # -- lengths -----------------------------------------------------------
    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def num_output_tokens(self) -> int:
        return len(self.output_token_ids)

    @property
    def num_tokens(self) -> int:
        return self.num_prompt_tokens + self.num_output_tokens

    @property
    def all_token_ids(self) -> List[int]:
        return self.prompt_token_ids + self.output_token_ids

    @property
    def finished(self) -> bool:
        return self.is_finished

    @property
    def finish_reason(self) -> Optional[str]:
        return FINISH_REASON.get(self.status)

    # -- lifecycle ----------------------------------------------------------
    def append_output_token(self, token_id: int):
        """Record a sampled token. ``num_computed_tokens`` is NOT advanced
        here — the new token's KV enters the cache on the step that feeds it
        through the model (engine advances the counter per executed token)."""
        if self.first_token_time is None:
            self.first_token_time = time.monotonic()
        self.output_token_ids.append(token_id)

    def maybe_finish(self, max_model_len: int) -> bool:
        """Apply stop conditions after a new output token. Returns finished."""
        sp = self.sampling_params
        out = self.output_token_ids
        last = out[-1] if out else None
        n_out = len(out)
        status = None
        if n_out >= sp.min_tokens and last is not None:
            if not sp.ignore_eos and self.eos_token_id is not None and last == self.eos_token_id:
                status = RequestStatus.FINISHED_STOPPED
            elif sp.stop_token_ids and last in sp.stop_token_ids:
                status = RequestStatus.FINISHED_STOPPED
        if status is None:
            if sp.max_tokens is not None and n_out >= sp.max_tokens:
                status = RequestStatus.FINISHED_LENGTH
            elif len(self.prompt_token_ids) + n_out >= max_model_len:
                status = RequestStatus.FINISHED_LENGTH
        if status is not None:
            self.status = status
            self.is_finished = True
            if self.finish_time is None:
                self.finish_time = time.monotonic()
            return True
        return False


@dataclass
class RequestOutput:
    """Per-step output for one request."""

    request_id: str
    new_token_ids: List[int]
    finished: bool
    finish_reason: Optional[str] = None
    output_token_ids: List[int] = field(default_factory=list)
    num_prompt_tokens: int = 0
    text_delta: str = ""
    output_text: str = ""
    logprobs: Optional[list] = None
