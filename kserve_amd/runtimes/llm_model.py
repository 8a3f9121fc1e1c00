"""OpenAI-protocol serving model backed by the native LLM engine.

Reference parity: huggingfaceserver vllm/vllm_model.py:55-343 (VLLMModel) —
same surface (completions / chat completions / embeddings stubs, engine
startup hook), engine is ours.
"""

from __future__ import annotations

import time
from typing import AsyncIterator, List, Optional, Union

from kserve_amd.engine.async_engine import AsyncLLMEngine
from kserve_amd.engine.config import EngineConfig
from kserve_amd.engine.sampling_params import SamplingParams
from kserve_amd.errors import InvalidInput
from kserve_amd.logging import logger
from kserve_amd.model import OpenAIModel
from kserve_amd.protocol.rest.openai.types import (
    ChatCompletion,
    ChatCompletionChoice,
    ChatCompletionChoiceMessage,
    ChatCompletionChunk,
    ChatCompletionChunkChoice,
    ChatCompletionChunkDelta,
    ChatCompletionRequest,
    Completion,
    CompletionChoice,
    CompletionLogprobs,
    CompletionRequest,
    UsageInfo,
)


def _to_sampling_params(
    req: Union[CompletionRequest, ChatCompletionRequest],
    max_tokens_default: int = 16,
) -> SamplingParams:
    max_tokens = getattr(req, "max_completion_tokens", None) or req.max_tokens
    stop = req.stop if isinstance(req.stop, list) else ([req.stop] if req.stop else [])
    logprobs = getattr(req, "logprobs", None)
    if isinstance(logprobs, bool):
        logprobs = 1 if logprobs else None
    return SamplingParams(
        n=req.n or 1,
        logprobs=logprobs,
        temperature=req.temperature if req.temperature is not None else 1.0,
        top_p=req.top_p if req.top_p is not None else 1.0,
        top_k=req.top_k if req.top_k is not None else -1,
        min_p=getattr(req, "min_p", 0.0) or 0.0,
        response_format=(
            (req.response_format or {}).get("type")
            if getattr(req, "response_format", None)
            else None
        ),
        json_schema=(
            # OpenAI shape: response_format.json_schema.schema
            ((req.response_format or {}).get("json_schema") or {}).get(
                "schema"
            )
            if getattr(req, "response_format", None)
            else None
        ),
        logit_bias=(
            {int(k): float(v) for k, v in req.logit_bias.items()}
            if getattr(req, "logit_bias", None)
            else None
        ),
        max_tokens=max_tokens or max_tokens_default,
        min_tokens=req.min_tokens or 0,
        stop=stop,
        ignore_eos=bool(req.ignore_eos),
        seed=req.seed,
        presence_penalty=req.presence_penalty or 0.0,
        frequency_penalty=req.frequency_penalty or 0.0,
        repetition_penalty=req.repetition_penalty or 1.0,
    )


class LLMModel(OpenAIModel):
    """The flagship generative model (Llama family) on the native engine."""

    def __init__(
        self,
        name: str,
        engine_config: EngineConfig,
        tokenizer=None,
        lora_modules=None,
    ):
        super().__init__(name)
        self.engine = True  # ModelServer awaits start_engine()
        self.tokenizer = tokenizer
        # name -> adapter dir; each name is served as its own model id
        # (reference registers --lora-modules names with the model server)
        self.lora_modules = dict(lora_modules or {})
        self.async_engine = AsyncLLMEngine(
            engine_config, tokenizer=tokenizer, lora_modules=self.lora_modules
        )

    @property
    def served_names(self):
        # base + registered LoRA adapter names
        return [self.name, *self.lora_modules]

    def _lora_for(self, requested_model):
        return requested_model if requested_model in self.lora_modules else None

    async def start_engine(self):
        await self.async_engine.start()
        self.ready = True
        logger.info("LLM engine for model %s started", self.name)

    def stop(self):
        self.async_engine.stop()
        self.ready = False

    async def healthy(self) -> bool:
        return self.ready and self.async_engine.is_running

    # -- prompt handling -----------------------------------------------------
    def _encode_prompt(self, prompt) -> List[int]:
        if isinstance(prompt, list) and prompt and isinstance(prompt[0], int):
            return prompt
        if isinstance(prompt, str):
            if self.tokenizer is None:
                raise InvalidInput(
                    "String prompts require a tokenizer; pass token ids"
                )
            return self.tokenizer.encode(prompt)
        raise InvalidInput(f"Unsupported prompt type {type(prompt)}")

    def _chat_to_prompt(self, request: ChatCompletionRequest) -> List[int]:
        if self.tokenizer is not None and hasattr(
            self.tokenizer, "apply_chat_template"
        ):
            try:
                return self.tokenizer.apply_chat_template(
                    [m.model_dump(exclude_none=True) for m in request.messages],
                    add_generation_prompt=True,
                )
            except Exception:
                pass
        # template-less fallback (reference: openai_chat_adapter_model.py)
        text = ""
        for m in request.messages:
            content = m.content if isinstance(m.content, str) else ""
            text += f"{m.role}: {content}\n"
        text += "assistant:"
        return self._encode_prompt(text)

    # -- completions -----------------------------------------------------------
    async def create_completion(self, request: CompletionRequest, raw_request=None):
        prompts = request.prompt
        if isinstance(prompts, str):
            prompt_list = [prompts]
        elif prompts and isinstance(prompts[0], int):
            prompt_list = [prompts]
        elif prompts and isinstance(prompts[0], (list, str)):
            prompt_list = list(prompts)
        else:
            raise InvalidInput("Empty prompt")
        sp = _to_sampling_params(request)
        sp.lora_name = self._lora_for(request.model)
        sp.priority = getattr(request, "priority", 0) or 0
        if request.stream:
            if len(prompt_list) != 1:
                raise InvalidInput("Streaming supports a single prompt")
            return self._stream_completion(prompt_list[0], sp, request)
        choices = []
        prompt_tokens = 0
        completion_tokens = 0
        # submit every (prompt, n) sample concurrently: the engine batches
        # them in one continuous-batching schedule instead of serial awaits
        jobs = []
        for i, p in enumerate(prompt_list):
            ids = self._encode_prompt(p)
            prompt_tokens += len(ids)
            for j in range(sp.n):
                sp_j = sp
                if sp.n > 1:
                    import dataclasses

                    sp_j = dataclasses.replace(
                        sp,
                        n=1,
                        seed=(sp.seed + j) if sp.seed is not None else None,
                    )
                jobs.append((i, j, ids, sp_j))
        import asyncio as _asyncio

        try:
            outs = await _asyncio.gather(
                *(
                    self.async_engine.generate_full(ids, sp_j)
                    for (_, _, ids, sp_j) in jobs
                )
            )
        except ValueError as e:
            raise InvalidInput(str(e)) from e
        for (i, j, ids, _), out in zip(jobs, outs):
            completion_tokens += len(out.output_token_ids)
            text = (
                out.output_text
                if getattr(out, "output_text", None)
                else self._decode(out.output_token_ids)
            )
            if request.echo:
                text = self._decode(ids) + text
            lp = None
            if out.logprobs:
                toks = [
                    self._decode([t]) for t in out.output_token_ids
                ]
                lp = CompletionLogprobs(
                    tokens=toks,
                    token_logprobs=[
                        step.get(t)
                        for t, step in zip(out.output_token_ids, out.logprobs)
                    ],
                    top_logprobs=[
                        {self._decode([k]): v for k, v in step.items()}
                        for step in out.logprobs
                    ],
                    text_offset=[],
                )
            choices.append(
                CompletionChoice(
                    index=i * sp.n + j,
                    text=text,
                    logprobs=lp,
                    finish_reason=out.finish_reason or "stop",
                )
            )
        return Completion(
            model=self.name,
            choices=choices,
            usage=UsageInfo(
                prompt_tokens=prompt_tokens,
                completion_tokens=completion_tokens,
                total_tokens=prompt_tokens + completion_tokens,
            ),
        )

    def _decode(self, token_ids: List[int]) -> str:
        if self.tokenizer is None:
            return " ".join(str(t) for t in token_ids)
        return self.tokenizer.decode(token_ids, skip_special_tokens=True)

    async def _stream_completion(
        self, prompt, sp: SamplingParams, request: CompletionRequest
    ) -> AsyncIterator[Completion]:
        ids = self._encode_prompt(prompt)
        completion_tokens = 0
        async for out in self.async_engine.generate(ids, sp):
            text = out.text_delta or (
                ""
                if self.tokenizer is not None
                else "".join(f"{t} " for t in out.new_token_ids)
            )
            completion_tokens += len(out.new_token_ids)
            yield Completion(
                model=self.name,
                choices=[
                    CompletionChoice(
                        index=0,
                        text=text,
                        finish_reason=out.finish_reason if out.finished else None,
                    )
                ],
            )
        # OpenAI stream_options.include_usage: a final usage-only chunk
        opts = getattr(request, "stream_options", None) or {}
        if opts.get("include_usage"):
            yield Completion(
                model=self.name,
                choices=[],
                usage=UsageInfo(
                    prompt_tokens=len(ids),
                    completion_tokens=completion_tokens,
                    total_tokens=len(ids) + completion_tokens,
                ),
            )

    # -- chat ---------------------------------------------------------------
    async def create_chat_completion(
        self, request: ChatCompletionRequest, raw_request=None
    ):
        ids = self._chat_to_prompt(request)
        sp = _to_sampling_params(request, max_tokens_default=256)
        sp.lora_name = self._lora_for(request.model)
        if request.stream:
            return self._stream_chat(ids, sp, request)
        try:
            out = await self.async_engine.generate_full(ids, sp)
        except ValueError as e:
            raise InvalidInput(str(e)) from e
        text = out.output_text if out.output_text else self._decode(out.output_token_ids)
        return ChatCompletion(
            model=self.name,
            choices=[
                ChatCompletionChoice(
                    index=0,
                    message=ChatCompletionChoiceMessage(content=text),
                    finish_reason=out.finish_reason or "stop",
                )
            ],
            usage=UsageInfo(
                prompt_tokens=len(ids),
                completion_tokens=len(out.output_token_ids),
                total_tokens=len(ids) + len(out.output_token_ids),
            ),
        )

    async def _stream_chat(
        self, ids: List[int], sp: SamplingParams, request: ChatCompletionRequest
    ) -> AsyncIterator[ChatCompletionChunk]:
        first = True
        completion_tokens = 0
        async for out in self.async_engine.generate(ids, sp):
            completion_tokens += len(out.new_token_ids)
            delta = ChatCompletionChunkDelta(
                role="assistant" if first else None,
                content=out.text_delta
                or (
                    ""
                    if self.tokenizer is not None
                    else "".join(f"{t} " for t in out.new_token_ids)
                ),
            )
            first = False
            yield ChatCompletionChunk(
                model=self.name,
                choices=[
                    ChatCompletionChunkChoice(
                        index=0,
                        delta=delta,
                        finish_reason=out.finish_reason if out.finished else None,
                    )
                ],
            )
        opts = getattr(request, "stream_options", None) or {}
        if opts.get("include_usage"):
            yield ChatCompletionChunk(
                model=self.name,
                choices=[],
                usage=UsageInfo(
                    prompt_tokens=len(ids),
                    completion_tokens=completion_tokens,
                    total_tokens=len(ids) + completion_tokens,
                ),
            )
