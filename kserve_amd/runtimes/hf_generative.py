"""HF-transformers generative fallback backend.

Reference parity: huggingfaceserver/generative_model.py:158-414
(HuggingfaceGenerativeModel) — the non-vLLM fallback that serves ANY
`AutoModelForCausalLM` architecture with `model.generate()`, request-
serial, streaming via `TextIteratorStreamer`. The native engine keeps the
architectures it implements (Llama/Mistral/Qwen2/Mixtral); everything
else lands here so no HF decoder is unservable.
"""

from __future__ import annotations

import asyncio
import threading
from typing import AsyncIterator, List, Optional

from kserve_amd.errors import InvalidInput
from kserve_amd.logging import logger
from kserve_amd.model import OpenAIModel
from kserve_amd.protocol.rest.openai.types import (
    ChatCompletion,
    ChatCompletionChoice,
    ChatCompletionChoiceMessage,
    ChatCompletionRequest,
    Completion,
    CompletionChoice,
    CompletionRequest,
    UsageInfo,
)


class HFGenerativeModel(OpenAIModel):
    """Request-serial `generate()` wrapper (no continuous batching — this
    is the compatibility fallback, mirroring the reference's)."""

    def __init__(
        self,
        name: str,
        model_dir: Optional[str] = None,
        model=None,
        tokenizer=None,
        device: str = "cpu",
        dtype=None,
    ):
        super().__init__(name)
        self.model_dir = model_dir
        self.model = model
        self.tokenizer = tokenizer
        self.device = device
        self.dtype = dtype
        # request-serial execution (reference: single consumer thread)
        self._lock = asyncio.Lock()
        if model is not None:
            self.ready = True

    def load(self) -> bool:
        if self.model is None:
            import torch
            import transformers

            dtype = self.dtype or (
                torch.bfloat16 if self.device != "cpu" else torch.float32
            )
            self.tokenizer = self.tokenizer or transformers.AutoTokenizer.from_pretrained(
                self.model_dir
            )
            self.model = transformers.AutoModelForCausalLM.from_pretrained(
                self.model_dir, torch_dtype=dtype
            ).to(self.device).eval()
            logger.info(
                "hf fallback loaded %s (%s)",
                self.model.config.architectures,
                self.device,
            )
        self.ready = True
        return self.ready

    # -- helpers -------------------------------------------------------------
    def _encode(self, prompt) -> List[int]:
        if isinstance(prompt, list) and prompt and isinstance(prompt[0], int):
            return prompt
        if isinstance(prompt, str):
            if self.tokenizer is None:
                raise InvalidInput("String prompts require a tokenizer")
            return self.tokenizer.encode(prompt)
        raise InvalidInput(f"Unsupported prompt type {type(prompt)}")

    def _decode(self, ids: List[int]) -> str:
        if self.tokenizer is None:
            return "".join(f"{t} " for t in ids)
        return self.tokenizer.decode(ids, skip_special_tokens=True)

    def _gen_kwargs(self, req, max_default: int) -> dict:
        do_sample = (req.temperature or 0) > 0
        kwargs = dict(
            max_new_tokens=req.max_tokens or max_default,
            do_sample=do_sample,
            pad_token_id=(
                getattr(self.tokenizer, "pad_token_id", None)
                or getattr(self.tokenizer, "eos_token_id", None)
                or 0
            ),
        )
        if do_sample:
            kwargs.update(
                temperature=req.temperature,
                top_p=req.top_p if req.top_p is not None else 1.0,
            )
            if getattr(req, "top_k", -1) and req.top_k > 0:
                kwargs["top_k"] = req.top_k
            if getattr(req, "seed", None) is not None:
                import torch

                torch.manual_seed(req.seed)
        return kwargs

    def _generate_sync(self, ids: List[int], kwargs: dict) -> List[int]:
        import torch

        inp = torch.tensor([ids], device=self.device)
        with torch.no_grad():
            out = self.model.generate(inp, **kwargs)
        return out[0][len(ids):].tolist()

    # -- OpenAI surface --------------------------------------------------------
    async def create_completion(self, request: CompletionRequest, raw_request=None):
        prompts = request.prompt
        if isinstance(prompts, str) or (
            prompts and isinstance(prompts[0], int)
        ):
            prompt_list = [prompts]
        else:
            prompt_list = list(prompts)
        if request.stream:
            if len(prompt_list) != 1:
                raise InvalidInput("Streaming supports a single prompt")
            return self._stream(prompt_list[0], request)
        choices = []
        prompt_tokens = completion_tokens = 0
        async with self._lock:  # request-serial like the reference
            for i, p in enumerate(prompt_list):
                ids = self._encode(p)
                prompt_tokens += len(ids)
                new_ids = await asyncio.get_running_loop().run_in_executor(
                    None, self._generate_sync, ids, self._gen_kwargs(request, 16)
                )
                completion_tokens += len(new_ids)
                choices.append(
                    CompletionChoice(
                        index=i,
                        text=self._decode(new_ids),
                        finish_reason="length",
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

    async def _stream(self, prompt, request) -> AsyncIterator[Completion]:
        """Streaming via transformers TextIteratorStreamer (tokenizer
        required) or chunked id emission without one."""
        ids = self._encode(prompt)
        kwargs = self._gen_kwargs(request, 16)
        async with self._lock:
            if self.tokenizer is not None:
                from transformers import TextIteratorStreamer

                streamer = TextIteratorStreamer(
                    self.tokenizer, skip_prompt=True, skip_special_tokens=True
                )
                kwargs["streamer"] = streamer
                import torch

                inp = torch.tensor([ids], device=self.device)
                t = threading.Thread(
                    target=lambda: self.model.generate(inp, **kwargs),
                    daemon=True,
                )
                t.start()
                loop = asyncio.get_running_loop()
                it = iter(streamer)
                while True:
                    piece = await loop.run_in_executor(
                        None, lambda: next(it, None)
                    )
                    if piece is None:
                        break
                    yield Completion(
                        model=self.name,
                        choices=[
                            CompletionChoice(index=0, text=piece)
                        ],
                    )
                t.join()
            else:
                new_ids = await asyncio.get_running_loop().run_in_executor(
                    None, self._generate_sync, ids, kwargs
                )
                for t_id in new_ids:
                    yield Completion(
                        model=self.name,
                        choices=[
                            CompletionChoice(index=0, text=f"{t_id} ")
                        ],
                    )

    async def create_chat_completion(
        self, request: ChatCompletionRequest, raw_request=None
    ):
        if self.tokenizer is not None and hasattr(
            self.tokenizer, "apply_chat_template"
        ):
            ids = self.tokenizer.apply_chat_template(
                [m.model_dump(exclude_none=True) for m in request.messages],
                add_generation_prompt=True,
            )
        else:
            text = "".join(
                f"{m.role}: {m.content if isinstance(m.content, str) else ''}\n"
                for m in request.messages
            ) + "assistant:"
            ids = self._encode(text)
        async with self._lock:
            new_ids = await asyncio.get_running_loop().run_in_executor(
                None, self._generate_sync, ids, self._gen_kwargs(request, 256)
            )
        return ChatCompletion(
            model=self.name,
            choices=[
                ChatCompletionChoice(
                    index=0,
                    message=ChatCompletionChoiceMessage(
                        content=self._decode(new_ids)
                    ),
                    finish_reason="stop",
                )
            ],
            usage=UsageInfo(
                prompt_tokens=len(ids),
                completion_tokens=len(new_ids),
                total_tokens=len(ids) + len(new_ids),
            ),
        )

    async def healthy(self) -> bool:
        return self.ready
