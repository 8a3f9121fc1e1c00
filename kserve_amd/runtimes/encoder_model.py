"""Encoder serving model (BERT-class): fill-mask, sequence/token
classification, embedding.

Reference parity: python/huggingfaceserver encoder_model.py:71-588 (task
dispatch in preprocess/postprocess) and task.py:33-123 (architecture ->
task inference) — engine is our native BertModel, varlen-batched.
"""

from __future__ import annotations

import json
import os
from enum import Enum
from typing import Dict, List, Union

import numpy as np
import torch

from kserve_amd.errors import InferenceError, InvalidInput
from kserve_amd.logging import logger
from kserve_amd.model import Model
from kserve_amd.models.bert import (
    BertConfig,
    BertForMaskedLM,
    BertForSequenceClassification,
    BertModel,
)
from kserve_amd.protocol.infer_type import InferOutput, InferRequest, InferResponse


class MLTask(str, Enum):
    fill_mask = "fill_mask"
    sequence_classification = "sequence_classification"
    token_classification = "token_classification"
    text_embedding = "text_embedding"


# reference task.py:58-70 ARCHITECTURES_2_TASK
ARCHITECTURES_2_TASK = {
    "ForMaskedLM": MLTask.fill_mask,
    "ForSequenceClassification": MLTask.sequence_classification,
    "ForTokenClassification": MLTask.token_classification,
    "Model": MLTask.text_embedding,
}


def infer_task_from_architecture(architecture: str) -> MLTask:
    for suffix, task in ARCHITECTURES_2_TASK.items():
        if architecture.endswith(suffix):
            return task
    raise ValueError(f"Task cannot be inferred from architecture {architecture}")


class EncoderModel(Model):
    def __init__(
        self,
        name: str,
        model_dir: str,
        task: Union[MLTask, str, None] = None,
        dtype: torch.dtype = None,
        device: str = None,
    ):
        super().__init__(name)
        self.model_dir = model_dir
        self.task = MLTask(task) if task else None
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        self.dtype = dtype or (
            torch.bfloat16 if self.device.startswith("cuda") else torch.float32
        )
        self.tokenizer = None
        self._model = None
        self.id2label = None

    def load(self) -> bool:
        from transformers import AutoTokenizer

        cfg_path = os.path.join(self.model_dir, "config.json")
        with open(cfg_path) as f:
            hf_cfg = json.load(f)
        arch = (hf_cfg.get("architectures") or ["BertModel"])[0]
        if self.task is None:
            self.task = infer_task_from_architecture(arch)
        config = BertConfig.from_hf_config(cfg_path)
        self.id2label = hf_cfg.get("id2label")
        self.tokenizer = AutoTokenizer.from_pretrained(self.model_dir)

        from kserve_amd.engine.weights import _LazySafetensors

        has_st = any(
            f.endswith(".safetensors") for f in os.listdir(self.model_dir)
        )
        if has_st:
            sd = _LazySafetensors(self.model_dir)
            tensors = {k: sd[k] for k in sd._files}
        else:
            bin_path = os.path.join(self.model_dir, "pytorch_model.bin")
            tensors = torch.load(bin_path, map_location="cpu", weights_only=True)

        if self.task == MLTask.fill_mask:
            self._model = BertForMaskedLM(config, dtype=self.dtype, device="cpu")
        elif self.task == MLTask.sequence_classification:
            num_labels = len(self.id2label or {0: "0", 1: "1"})
            self._model = BertForSequenceClassification(
                config, num_labels, dtype=self.dtype, device="cpu"
            )
        elif self.task == MLTask.token_classification:
            num_labels = len(self.id2label or {0: "0", 1: "1"})
            self._model = BertForSequenceClassification(
                config, num_labels, dtype=self.dtype, device="cpu"
            )
        else:
            self._model = BertModel(config, dtype=self.dtype, device="cpu")
        self._model.load_hf_state_dict(tensors)
        self._model = self._model.to(self.device)
        self.ready = True
        return self.ready

    # -- pipeline ----------------------------------------------------------
    def preprocess(self, payload, headers=None):
        if isinstance(payload, InferRequest):
            arr = payload.inputs[0].as_numpy()
            texts = [
                t.decode("utf-8") if isinstance(t, bytes) else str(t)
                for t in arr.flatten()
            ]
            return {"texts": texts, "_v2": payload}
        instances = payload.get("instances") or payload.get("inputs")
        if instances is None:
            raise InvalidInput('Expected "instances"')
        return {"texts": [str(t) for t in instances], "_v2": None}

    def _encode_varlen(self, texts: List[str]):
        ids_list = [
            self.tokenizer.encode(t, truncation=True, max_length=512)
            for t in texts
        ]
        cu = [0]
        flat = []
        for ids in ids_list:
            flat.extend(ids)
            cu.append(cu[-1] + len(ids))
        dev = self.device
        return (
            torch.tensor(flat, dtype=torch.int64, device=dev),
            torch.tensor(cu, dtype=torch.int32, device=dev),
            ids_list,
        )

    @torch.no_grad()
    def predict(self, payload, headers=None):
        texts = payload["texts"]
        input_ids, cu_seqlens, ids_list = self._encode_varlen(texts)
        try:
            if self.task in (
                MLTask.sequence_classification,
                MLTask.token_classification,
            ):
                out = self._model(input_ids, cu_seqlens)
            elif self.task == MLTask.fill_mask:
                out = self._model(input_ids, cu_seqlens)
            else:
                out = self._model(input_ids, cu_seqlens)
        except Exception as e:
            raise InferenceError(str(e))
        return {
            "output": out,
            "ids_list": ids_list,
            "cu": cu_seqlens,
            "_v2": payload["_v2"],
        }

    def postprocess(self, result, headers=None):
        out = result["output"]
        ids_list = result["ids_list"]
        cu = result["cu"].cpu()
        task = self.task
        if task == MLTask.fill_mask:
            # predicted token (argmax) at each [MASK] position, per input
            mask_id = self.tokenizer.mask_token_id
            logits = out.float().cpu()
            preds = []
            for s, ids in enumerate(ids_list):
                a = int(cu[s])
                tokens = []
                for i, tok in enumerate(ids):
                    if tok == mask_id:
                        tokens.append(
                            self.tokenizer.decode([int(logits[a + i].argmax())]).strip()
                        )
                preds.append(tokens[0] if len(tokens) == 1 else tokens)
            predictions = preds
        elif task == MLTask.sequence_classification:
            probs = torch.softmax(out.float().cpu(), dim=-1)
            idx = probs.argmax(-1)
            if self.id2label:
                predictions = [self.id2label[str(int(i))] for i in idx]
            else:
                predictions = idx.tolist()
        elif task == MLTask.text_embedding:
            # mean-pool per sequence
            emb = out.float().cpu()
            predictions = []
            for s in range(len(ids_list)):
                a, b = int(cu[s]), int(cu[s + 1])
                predictions.append(emb[a:b].mean(dim=0).tolist())
        else:
            predictions = out.float().cpu().tolist()
        v2 = result["_v2"]
        if v2 is not None:
            arr = np.array(
                predictions
                if task == MLTask.text_embedding
                else [str(p) for p in predictions],
                dtype=np.float32 if task == MLTask.text_embedding else np.object_,
            )
            o = InferOutput(
                "output-0",
                list(arr.shape),
                "FP32" if task == MLTask.text_embedding else "BYTES",
            )
            o.set_data_from_numpy(arr, binary_data=False)
            return InferResponse(v2.id, self.name, [o])
        return {"predictions": predictions}


class OpenAIEmbeddingAdapter:
    """Exposes an EncoderModel on the OpenAI /v1/embeddings route
    (reference: OpenAIEncoderModel, openai_model.py)."""

    def __init__(self, encoder: "EncoderModel"):
        self.encoder = encoder
        self.name = encoder.name
        self.ready = True
        self.engine = False

    async def create_embedding(self, request, raw_request=None, context=None):
        from kserve_amd.protocol.rest.openai.types import (
            Embedding,
            EmbeddingObject,
            UsageInfo,
        )

        texts = request.input if isinstance(request.input, list) else [request.input]
        texts = [str(t) for t in texts]
        payload = self.encoder.preprocess({"instances": texts})
        result = self.encoder.predict(payload)
        post = self.encoder.postprocess(result)
        vectors = post["predictions"]
        data = [
            EmbeddingObject(index=i, embedding=v) for i, v in enumerate(vectors)
        ]
        total = sum(len(ids) for ids in result["ids_list"])
        return Embedding(
            data=data,
            model=self.name,
            usage=UsageInfo(prompt_tokens=total, total_tokens=total),
        )

    async def create_completion(self, *a, **k):
        raise NotImplementedError("encoder models do not generate")

    async def create_chat_completion(self, *a, **k):
        raise NotImplementedError("encoder models do not generate")

    async def create_rerank(self, request, raw_request=None, context=None):
        from kserve_amd.protocol.rest.openai.types import (
            Rerank,
            RerankResult,
            UsageInfo,
        )

        texts = [request.query] + list(request.documents)
        payload = self.encoder.preprocess({"instances": texts})
        result = self.encoder.predict(payload)
        post = self.encoder.postprocess(result)
        vectors = post["predictions"]
        scores = cosine_rerank(vectors[0], vectors[1:])
        order = sorted(range(len(scores)), key=lambda i: -scores[i])
        if request.top_n:
            order = order[: request.top_n]
        results = [
            RerankResult(
                index=i,
                relevance_score=scores[i],
                document={"text": request.documents[i]}
                if request.return_documents
                else None,
            )
            for i in order
        ]
        total = sum(len(ids) for ids in result["ids_list"])
        return Rerank(results=results, usage=UsageInfo(prompt_tokens=total, total_tokens=total))


def cosine_rerank(query_vec, doc_vecs):
    import numpy as np

    q = np.asarray(query_vec, dtype=np.float64)
    qn = q / (np.linalg.norm(q) + 1e-9)
    scores = []
    for d in doc_vecs:
        dv = np.asarray(d, dtype=np.float64)
        scores.append(float(qn @ (dv / (np.linalg.norm(dv) + 1e-9))))
    return scores
