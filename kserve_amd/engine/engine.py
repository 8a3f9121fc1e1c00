"""LLMEngine: the synchronous continuous-batching core.

add_request() -> step() loop -> RequestOutputs. This is the native engine
replacing the reference's vLLM delegation (vllm_model.py:109-111 start_engine)
— SURVEY.md §7 stage 4.
"""

from __future__ import annotations

import time
import uuid
from typing import Dict, List, Optional, Union

import torch

from kserve_amd import ops
from kserve_amd.engine.config import EngineConfig
from kserve_amd.engine.detokenizer import Detokenizer
from kserve_amd.engine.model_runner import ModelRunner
from kserve_amd.engine.request import Request, RequestOutput, RequestStatus
from kserve_amd.engine.sampling_params import SamplingParams
from kserve_amd.engine.scheduler import Scheduler
from kserve_amd.logging import logger
from kserve_amd.metrics import (
    LLM_E2E_HIST,
    LLM_SPEC_ACCEPTED,
    LLM_GENERATION_TOKENS,
    LLM_KV_USAGE,
    LLM_PROMPT_TOKENS,
    LLM_TTFT_HIST,
)
from kserve_amd.models.llama import LlamaForCausalLM
from kserve_amd.parallel import comm


class Sampler:
    """Builds per-batch parameter tensors and invokes the sampling ops."""

    def __init__(self, device: torch.device, seed: int = 0):
        self.device = device
        self.generator = None
        if device.type == "cpu":
            self.generator = torch.Generator()
            self.generator.manual_seed(seed)
        self._step = 0

    @staticmethod
    def _apply_penalties(logits: torch.Tensor, requests: List[Request]) -> torch.Tensor:
        """Presence/frequency/repetition penalties (OpenAI semantics) for the
        requests that use them; no-op rows untouched."""
        rows = [
            i
            for i, r in enumerate(requests)
            if r.sampling_params.presence_penalty != 0.0
            or r.sampling_params.frequency_penalty != 0.0
            or r.sampling_params.repetition_penalty != 1.0
            or r.sampling_params.logit_bias
        ]
        if not rows:
            return logits
        logits = logits.clone()
        for i in rows:
            r = requests[i]
            sp = r.sampling_params
            if sp.logit_bias:
                ids = torch.tensor(
                    [t for t in sp.logit_bias if 0 <= t < logits.shape[-1]],
                    device=logits.device,
                    dtype=torch.long,
                )
                if ids.numel():
                    vals = torch.tensor(
                        [sp.logit_bias[int(t)] for t in ids],
                        device=logits.device,
                        dtype=torch.float32,
                    )
                    logits[i, ids] = (
                        logits[i, ids].float() + vals
                    ).to(logits.dtype)
            seen: dict = {}
            for t in r.output_token_ids:
                seen[t] = seen.get(t, 0) + 1
            if not seen:
                continue
            idx = torch.tensor(list(seen.keys()), device=logits.device)
            cnt = torch.tensor(
                list(seen.values()), device=logits.device, dtype=logits.dtype
            )
            row = logits[i].float()
            vals = row[idx]
            if sp.repetition_penalty != 1.0:
                vals = torch.where(
                    vals > 0, vals / sp.repetition_penalty, vals * sp.repetition_penalty
                )
            vals = vals - sp.presence_penalty - sp.frequency_penalty * cnt.float()
            row[idx] = vals
            logits[i] = row.to(logits.dtype)
        return logits

    def sample(self, logits: torch.Tensor, requests: List[Request]) -> List[int]:
        self._step += 1
        logits = self._apply_penalties(logits, requests)
        all_greedy = all(r.sampling_params.greedy for r in requests)
        if all_greedy:
            return ops.greedy_sample(logits).tolist()
        temps = torch.tensor(
            [max(r.sampling_params.temperature, 1e-5) for r in requests],
            dtype=torch.float32,
            device=logits.device,
        )
        top_p = torch.tensor(
            [r.sampling_params.top_p for r in requests],
            dtype=torch.float32,
            device=logits.device,
        )
        top_k = torch.tensor(
            [r.sampling_params.top_k for r in requests],
            dtype=torch.int32,
            device=logits.device,
        )
        seeds = torch.tensor(
            [
                (r.sampling_params.seed if r.sampling_params.seed is not None else hash(r.request_id) & 0x7FFFFFFF)
                + self._step * 0x9E3779B1
                for r in requests
            ],
            dtype=torch.int64,
            device=logits.device,
        )
        min_p = None
        if any(r.sampling_params.min_p > 0 for r in requests):
            min_p = torch.tensor(
                [r.sampling_params.min_p for r in requests],
                dtype=torch.float32,
                device=logits.device,
            )
        out = ops.random_sample(
            logits, temps, top_p, top_k, seeds=seeds,
            generator=self.generator, min_p=min_p,
        )
        # greedy requests in a mixed batch: override with argmax
        if any(r.sampling_params.greedy for r in requests):
            greedy = ops.greedy_sample(logits)
            mask = torch.tensor(
                [r.sampling_params.greedy for r in requests],
                dtype=torch.bool,
                device=logits.device,
            )
            out = torch.where(mask, greedy, out)
        return out.tolist()


class LLMEngine:
    def __init__(
        self,
        config: EngineConfig,
        tokenizer=None,
        model: Optional[LlamaForCausalLM] = None,
    ):
        self.config = config
        self.tokenizer = tokenizer
        self.detokenizer = Detokenizer(tokenizer)
        st = comm.get_state()
        self.device = torch.device(
            st.device if st.device != "cpu" else config.device
            if torch.cuda.is_available()
            else "cpu"
        )
        t0 = time.monotonic()
        if model is None:
            model = self._build_model()
        self.model = model
        self.runner = ModelRunner(config, model, str(self.device))
        # draft-model speculation (vLLM draft-model mode; see draft_worker)
        self.draft = None
        if config.draft_model is not None and config.scheduler.speculative_k > 0:
            from kserve_amd.engine.draft_worker import DraftModelWorker

            self.draft = DraftModelWorker(config, config.draft_model)
            if config.draft_model_path:
                from kserve_amd.engine.weights import (
                    load_safetensors_weights,
                )

                load_safetensors_weights(
                    self.draft.model, config.draft_model_path
                )
        num_blocks = self.runner.profile_and_allocate_kv()
        self.scheduler = Scheduler(
            config.scheduler, config.cache, num_blocks,
            num_cpu_blocks=self.runner.num_cpu_blocks,
        )
        self.sampler = Sampler(self.device, config.seed)
        self.lora_manager = None
        self.eos_token_id = config.eos_token_id
        if tokenizer is not None and getattr(tokenizer, "eos_token_id", None) is not None:
            self.eos_token_id = tokenizer.eos_token_id
        if (
            self.device.type == "cuda"
            and not config.enforce_eager
            and comm.get_state().pp_size == 1
        ):
            # MoE decode captures via the dense-bmm path (static shapes);
            # buckets are capped at the dense/sparse crossover, larger
            # batches run eager sparse steps
            if config.model.num_local_experts > 0:
                from kserve_amd.models.llama import MixtralMoE

                self.runner.capture_decode_graphs(
                    batch_sizes=[
                        b
                        for b in (1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128,
                                  192, 256, 384)
                        if b <= MixtralMoE.DENSE_MAX_TOKENS
                    ]
                )
            else:
                self.runner.capture_decode_graphs()
        logger.info(
            "Engine ready in %.1fs: %s (%.2fB params, tp=%d, device=%s)",
            time.monotonic() - t0,
            config.model.model_name,
            self.model.num_parameters() / 1e9,
            comm.get_state().tp_size,
            self.device,
        )

    def _build_model(self) -> LlamaForCausalLM:
        cfg = self.config
        with torch.device(self.device):
            model = LlamaForCausalLM(cfg.model, device=str(self.device))
        if cfg.model_path:
            from kserve_amd.engine.weights import load_safetensors_weights

            load_safetensors_weights(model, cfg.model_path)
        else:
            model.random_init(seed=cfg.seed)
        return model

    # -- request API ----------------------------------------------------------
    def add_request(
        self,
        prompt: Union[str, List[int]],
        sampling_params: Optional[SamplingParams] = None,
        request_id: Optional[str] = None,
    ) -> str:
        request_id = request_id or str(uuid.uuid4())
        sampling_params = sampling_params or SamplingParams()
        if isinstance(prompt, str):
            if self.tokenizer is None:
                raise ValueError("String prompt requires a tokenizer")
            prompt_token_ids = self.tokenizer.encode(prompt)
        else:
            prompt_token_ids = list(prompt)
        if len(prompt_token_ids) > self.config.scheduler.max_model_len:
            # fail fast: a silently-dropped request would hang its caller
            raise ValueError(
                f"prompt length {len(prompt_token_ids)} exceeds "
                f"max_model_len {self.config.scheduler.max_model_len}"
            )
        req = Request(
            request_id,
            prompt_token_ids,
            sampling_params,
            eos_token_id=self.eos_token_id,
        )
        if sampling_params.lora_name:
            if self.lora_manager is None:
                raise ValueError("no LoRA adapters registered")
            req.lora_id = self.lora_manager.lookup(sampling_params.lora_name)
        if sampling_params.guided_choice:
            from kserve_amd.engine.guided import ChoiceMachine

            req.guided_machine = ChoiceMachine(
                [c.encode("utf-8") for c in sampling_params.guided_choice]
            )
        elif (
            sampling_params.response_format == "json_schema"
            and sampling_params.json_schema
        ):
            from kserve_amd.engine.guided import SchemaMachine

            req.guided_machine = SchemaMachine(sampling_params.json_schema)
        elif sampling_params.response_format in ("json", "json_object",
                                                 "json_schema"):
            # json_schema without a schema degrades to valid-JSON-object
            from kserve_amd.engine.guided import JsonMachine

            req.guided_machine = JsonMachine(object_only=True)
        LLM_PROMPT_TOKENS.inc(len(prompt_token_ids))
        self.scheduler.add_request(req)
        return request_id

    @property
    def guided_json(self):
        """Lazily-built guided-JSON token-mask processor (engine/guided.py)."""
        if getattr(self, "_guided_json", None) is None:
            from kserve_amd.engine.guided import GuidedJsonProcessor

            eos = self.eos_token_id
            if eos is not None and eos < 0:  # -1 = EOS disabled
                eos = None
            self._guided_json = GuidedJsonProcessor.from_tokenizer(
                self.tokenizer, self.config.model.vocab_size, eos
            )
        return self._guided_json

    def register_lora(self, name: str, path: str) -> int:
        """Load a PEFT adapter and make it addressable by name (reference:
        --lora-modules name=path registering adapter model names)."""
        if self.lora_manager is None:
            from kserve_amd.engine.lora import LoRAManager

            st = comm.get_state()
            self.lora_manager = LoRAManager(
                str(self.device),
                self.model.dtype,
                tp_rank=st.tp_rank,
                tp_size=st.tp_size,
            )
            self.runner.lora_manager = self.lora_manager
        return self.lora_manager.register(name, path)

    def abort_request(self, request_id: str):
        self.scheduler.abort_request(request_id)

    def has_unfinished(self) -> bool:
        return self.scheduler.has_unfinished()

    # -- step -----------------------------------------------------------------
    def step(self) -> List[RequestOutput]:
        batch = self.scheduler.schedule()
        # KV tier copies ordered before this batch's kernels
        if batch.swap_out:
            self.runner.swap_blocks(batch.swap_out, to_gpu=False)
        if batch.swap_in:
            self.runner.swap_blocks(batch.swap_in, to_gpu=True)
        if not batch:
            return []
        if batch.is_prefill:
            logits = self.runner.execute_prefill(batch, self.scheduler.block_manager)
        else:
            sched = self.config.scheduler
            # windows need zero host work per token: anything that must see
            # the logits or token history on the host each step disqualifies
            window_ok = all(
                r.sampling_params.logprobs is None
                and r.sampling_params.presence_penalty == 0.0
                and r.sampling_params.frequency_penalty == 0.0
                and r.sampling_params.repetition_penalty == 1.0
                and not r.sampling_params.logit_bias
                and r.sampling_params.min_p == 0.0
                and r.lora_id == 0
                and r.guided_machine is None
                for r in batch.requests
            )
            pure_greedy = window_ok and all(
                r.sampling_params.greedy for r in batch.requests
            )
            # speculative decoding (prompt-lookup): draft from the request's
            # own context, verify k+1 positions in ONE forward through the
            # paged-context prefill path; exact greedy outputs by construction
            pp = comm.get_state().pp_size
            if sched.speculative_ngram > 0 and pure_greedy and pp == 1:
                k_cap = (
                    self.scheduler.reserve_decode_window(
                        batch, sched.speculative_ngram + 1
                    )
                    - 1
                )
                if k_cap > 0:
                    drafts = [
                        self._draft_ngram(r, k_cap) for r in batch.requests
                    ]
                    if any(drafts):
                        return self._run_spec_decode(batch, drafts)
            # draft-MODEL speculation: a small model proposes k tokens,
            # the main model verifies them in one paged-context forward
            if self.draft is not None and pure_greedy and pp == 1:
                k_cap = (
                    self.scheduler.reserve_decode_window(
                        batch, sched.speculative_k + 1
                    )
                    - 1
                )
                if k_cap > 0:
                    drafts = self.draft.propose(batch.requests, k_cap)
                    if any(drafts):
                        return self._run_spec_decode(batch, drafts)
            # multi-step window: decode with no pending scheduling events
            # runs as back-to-back hipGraph replays. temperature>0 batches
            # use the sampled-graph variant (in-graph fused sampler).
            k = 1
            if sched.multi_step > 1 and pp == 1 and window_ok:
                k = self.scheduler.reserve_decode_window(
                    batch, sched.multi_step
                )
            if k > 1:
                return self._run_decode_window(
                    batch, k, sampled=not pure_greedy
                )
            logits = self.runner.execute_decode(batch, self.scheduler.block_manager)
        # advance computed-token counters for executed tokens
        for req, n in zip(batch.requests, batch.num_scheduled_tokens):
            req.num_computed_tokens += n
        if batch.is_prefill and self.scheduler.block_manager.enable_prefix_caching:
            for req in batch.requests:
                self.scheduler.block_manager.register_computed_blocks(req)
        # chunked prefill: requests whose prompt isn't fully computed don't
        # sample this step
        if batch.is_prefill:
            sample_reqs = [
                r for r in batch.requests if r.num_computed_tokens >= r.num_tokens
            ]
            if len(sample_reqs) < len(batch.requests):
                idx = [
                    i
                    for i, r in enumerate(batch.requests)
                    if r.num_computed_tokens >= r.num_tokens
                ]
                if logits is not None:  # None on non-final pipeline stages
                    logits = logits[idx]
                # partial chunks go back to the head of the waiting queue so
                # the next step continues their prefill (not decode)
                for r in reversed(batch.requests):
                    if r.num_computed_tokens < r.num_tokens:
                        self.scheduler.requeue_partial_prefill(r)
        else:
            sample_reqs = batch.requests
        outputs: List[RequestOutput] = []
        finished: List[Request] = []
        if sample_reqs:
            guided_rows = [
                i for i, r in enumerate(sample_reqs)
                if r.guided_machine is not None
            ]
            if guided_rows and logits is not None:
                logits = logits.clone()
                neg = float("-inf")
                for i in guided_rows:
                    allowed = self.guided_json.allowed_tokens(
                        sample_reqs[i].guided_machine
                    )
                    row = torch.full_like(logits[i], neg)
                    if allowed:
                        idx = torch.tensor(allowed, device=logits.device)
                        row[idx] = logits[i, idx]
                    logits[i] = row
            tokens = self._sample_tokens(logits, sample_reqs)
            guided_exhausted = set()
            if guided_rows:
                for i in guided_rows:
                    ok = self.guided_json.advance(
                        sample_reqs[i].guided_machine, int(tokens[i])
                    )
                    if not ok:
                        # the machine has no legal continuation (row fully
                        # masked): stop WITHOUT emitting the bogus token
                        guided_exhausted.add(sample_reqs[i].request_id)
            # top-N logprobs for the (rare) requests that ask for them
            lp_idx = [
                i for i, r in enumerate(sample_reqs)
                if r.sampling_params.logprobs is not None
            ]
            if lp_idx and logits is not None:
                n_top = max(
                    sample_reqs[i].sampling_params.logprobs or 1 for i in lp_idx
                )
                n_top = max(n_top, 1)
                sub = torch.log_softmax(logits[lp_idx].float(), dim=-1)
                topv, topi = sub.topk(n_top, dim=-1)
                topv = topv.cpu()
                topi = topi.cpu()
                for row, i in enumerate(lp_idx):
                    req = sample_reqs[i]
                    entry = {
                        int(topi[row][c]): float(topv[row][c])
                        for c in range(n_top)
                    }
                    tok = int(tokens[i])
                    if tok not in entry:
                        entry[tok] = float(sub[row][tok])
                    req.logprobs.append(entry)
            LLM_GENERATION_TOKENS.inc(len(tokens))
            now = time.monotonic()
            for req, tok in zip(sample_reqs, tokens):
                first = req.first_token_time is None
                if req.request_id in guided_exhausted:
                    req.status = RequestStatus.FINISHED_STOPPED
                    req.is_finished = True
                    req.finish_time = now
                else:
                    req.append_output_token(int(tok))
                if first:
                    LLM_TTFT_HIST.observe(now - req.arrival_time)
                req.maybe_finish(self.config.scheduler.max_model_len)
                delta = self.detokenizer.decode_new(req)
                if not req.finished and req.sampling_params.stop:
                    if self.detokenizer.check_stop_strings(req) is not None:
                        req.status = RequestStatus.FINISHED_STOPPED
                        req.is_finished = True
                        req.finish_time = time.monotonic()
                if req.finished:
                    finished.append(req)
                    LLM_E2E_HIST.observe(now - req.arrival_time)
                outputs.append(
                    RequestOutput(
                        request_id=req.request_id,
                        new_token_ids=(
                            []
                            if req.request_id in guided_exhausted
                            else [int(tok)]
                        ),
                        finished=req.finished,
                        finish_reason=req.finish_reason,
                        # full-list copy only when the request completes
                        # (per-step copies are O(len) x batch)
                        output_token_ids=(
                            list(req.output_token_ids) if req.finished else req.output_token_ids
                        ),
                        num_prompt_tokens=req.num_prompt_tokens,
                        text_delta=delta,
                        output_text=req.output_text,
                        logprobs=req.logprobs if req.sampling_params.logprobs is not None else None,
                    )
                )
        self.scheduler.finish_requests(finished)
        for req in finished:
            self.runner.release_request(req.request_id)
            if self.draft is not None:
                self.draft.release(req.request_id)
        LLM_KV_USAGE.set(self.scheduler.block_manager.usage)
        return outputs

    def _sample_tokens(self, logits, sample_reqs) -> List[int]:
        """Sample on the last pipeline stage and broadcast the token ids so
        every stage's scheduler advances in lockstep (pp_size=1: direct)."""
        st = comm.get_state()
        if st.pp_size == 1:
            return self.sampler.sample(logits, sample_reqs)
        if st.is_last_pp:
            tokens = self.sampler.sample(logits, sample_reqs)
            t = torch.tensor(tokens, dtype=torch.int64, device=self.device)
        else:
            t = torch.empty(
                len(sample_reqs), dtype=torch.int64, device=self.device
            )
        comm.pp_broadcast_from_last(t)
        return t.tolist()

    def _draft_ngram(self, req: Request, k: int) -> List[int]:
        """Prompt-lookup drafting: if the last n-gram (n in [min,max]) occurred
        earlier in the sequence, propose the tokens that followed it."""
        sched = self.config.scheduler
        toks = req.all_token_ids
        L = len(toks)
        for n in range(sched.speculative_ngram_max, sched.speculative_ngram_min - 1, -1):
            if L <= n:
                continue
            suffix = toks[-n:]
            # most recent earlier occurrence wins
            for i in range(L - n - 1, -1, -1):
                if toks[i : i + n] == suffix:
                    cont = toks[i + n : i + n + k]
                    if cont:
                        return list(cont)
                    break
        return []

    def _run_spec_decode(self, batch, drafts) -> List[RequestOutput]:
        logits, cu = self.runner.execute_verify(
            batch, drafts, self.scheduler.block_manager
        )
        sampled = ops.greedy_sample(logits).tolist()
        outputs: List[RequestOutput] = []
        finished: List[Request] = []
        now = time.monotonic()
        max_len = self.config.scheduler.max_model_len
        n_tokens = 0
        n_accepted = 0
        for i, req in enumerate(batch.requests):
            row = cu[i]
            d = drafts[i]
            # emit the model's token at each verified position while the
            # draft agrees; the first disagreement is the model's correction
            emitted: List[int] = []
            for j in range(len(d) + 1):
                t = sampled[row + j]
                emitted.append(t)
                if j < len(d) and t == d[j]:
                    continue
                break
            n_accepted += len(emitted) - 1
            sp = req.sampling_params
            if req.first_token_time is None:
                req.first_token_time = now
                LLM_TTFT_HIST.observe(now - req.arrival_time)
            stop_at = len(emitted)
            if not sp.ignore_eos or sp.stop_token_ids:
                stops = set(sp.stop_token_ids)
                if not sp.ignore_eos and req.eos_token_id is not None:
                    stops.add(req.eos_token_id)
                for j, t in enumerate(emitted):
                    if t in stops and j + 1 >= sp.min_tokens - req.num_output_tokens:
                        stop_at = j + 1
                        break
            emitted = emitted[:stop_at]
            req.output_token_ids.extend(emitted)
            req.num_computed_tokens += len(emitted)
            req.maybe_finish(max_len)
            delta = self.detokenizer.decode_new(req)
            if not req.is_finished and sp.stop:
                if self.detokenizer.check_stop_strings(req) is not None:
                    req.status = RequestStatus.FINISHED_STOPPED
                    req.is_finished = True
                    req.finish_time = now
            if req.is_finished:
                finished.append(req)
                LLM_E2E_HIST.observe(now - req.arrival_time)
            n_tokens += len(emitted)
            outputs.append(
                RequestOutput(
                    request_id=req.request_id,
                    new_token_ids=emitted,
                    finished=req.is_finished,
                    finish_reason=req.finish_reason,
                    output_token_ids=(
                        list(req.output_token_ids)
                        if req.is_finished
                        else req.output_token_ids
                    ),
                    num_prompt_tokens=req.num_prompt_tokens,
                    text_delta=delta,
                    output_text=req.output_text,
                )
            )
        LLM_GENERATION_TOKENS.inc(n_tokens)
        LLM_SPEC_ACCEPTED.inc(n_accepted)
        self.scheduler.finish_requests(finished)
        for req in finished:
            self.runner.release_request(req.request_id)
            if self.draft is not None:
                self.draft.release(req.request_id)
        return outputs

    def _run_decode_window(
        self, batch, k: int, sampled: bool = False
    ) -> List[RequestOutput]:
        tokens_k = self.runner.multi_step_decode(
            batch, self.scheduler.block_manager, k, sampled=sampled
        )
        # ONE RequestOutput per request per window (streaming consumers get
        # k-token chunks); per-token python only for stop conditions
        cols = tokens_k.t().tolist()  # [n][k]
        outputs: List[RequestOutput] = []
        finished: List[Request] = []
        now = time.monotonic()
        max_len = self.config.scheduler.max_model_len
        n_tokens = 0
        for i, req in enumerate(batch.requests):
            col = cols[i]
            sp = req.sampling_params
            if req.first_token_time is None:
                req.first_token_time = now
                LLM_TTFT_HIST.observe(now - req.arrival_time)
            stop_at = k
            if not sp.ignore_eos or sp.stop_token_ids:
                stops = set(sp.stop_token_ids)
                if not sp.ignore_eos and req.eos_token_id is not None:
                    stops.add(req.eos_token_id)
                for j, t in enumerate(col):
                    if t in stops and j + 1 >= sp.min_tokens - req.num_output_tokens:
                        stop_at = j + 1
                        break
            col = col[:stop_at]
            req.output_token_ids.extend(col)
            req.num_computed_tokens += stop_at
            req.maybe_finish(max_len)
            delta = self.detokenizer.decode_new(req)
            if not req.is_finished and sp.stop:
                if self.detokenizer.check_stop_strings(req) is not None:
                    req.status = RequestStatus.FINISHED_STOPPED
                    req.is_finished = True
                    req.finish_time = now
            if req.is_finished:
                finished.append(req)
                LLM_E2E_HIST.observe(now - req.arrival_time)
            n_tokens += len(col)
            outputs.append(
                RequestOutput(
                    request_id=req.request_id,
                    new_token_ids=col,
                    finished=req.is_finished,
                    finish_reason=req.finish_reason,
                    output_token_ids=(
                        list(req.output_token_ids)
                        if req.is_finished
                        else req.output_token_ids
                    ),
                    num_prompt_tokens=req.num_prompt_tokens,
                    text_delta=delta,
                    output_text=req.output_text,
                )
            )
        LLM_GENERATION_TOKENS.inc(n_tokens)
        self.scheduler.finish_requests(finished)
        for req in finished:
            self.runner.release_request(req.request_id)
            if self.draft is not None:
                self.draft.release(req.request_id)
        LLM_KV_USAGE.set(self.scheduler.block_manager.usage)
        return outputs

    # -- convenience -----------------------------------------------------------
    def generate(
        self,
        prompts: List[Union[str, List[int]]],
        sampling_params: Optional[SamplingParams] = None,
    ) -> Dict[str, RequestOutput]:
        """Blocking batch generation (tests / offline use)."""
        ids = [self.add_request(p, sampling_params) for p in prompts]
        results: Dict[str, RequestOutput] = {}
        while self.has_unfinished():
            for out in self.step():
                if out.finished:
                    results[out.request_id] = out
        return {rid: results[rid] for rid in ids if rid in results}
