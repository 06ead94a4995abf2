# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""RAG serving steps: GPU-resident vector index + retrieval step.

Baseline config 5 shape ("gen-AI serving graph: RAG router + Llama
step"): a retrieval step enriches the prompt with top-k documents
before the Llama V2ModelServer step.  The index lives in HBM (288 GB
leaves room for billions of bf16 embedding rows next to the model);
similarity is one skinny-GEMM launch (the same MFMA kernel the decode
path uses) + torch.topk.

The reference has no RAG engine of its own — it routes to external
vector DBs from graph steps; here retrieval is a first-class native
step.  Embeddings come from any encoder; `TokenMeanEmbedder` gives a
dependency-free default (mean-pooled model embedding rows) for
synthetic/offline use.
"""

import typing

import torch

from .. import ops


class VectorIndex:
    """Dense vector index: [N, D] bf16 matrix + payload per row."""

    def __init__(self, dim: int, device=None, normalize: bool = True):
        self.dim = dim
        self.device = torch.device(
            device or ("cuda:0" if torch.cuda.is_available() else "cpu"))
        self.normalize = normalize
        self._embeddings: typing.Optional[torch.Tensor] = None
        self._payloads: typing.List[typing.Any] = []

    def __len__(self):
        return len(self._payloads)

    def _prep(self, emb: torch.Tensor) -> torch.Tensor:
        emb = torch.as_tensor(emb, dtype=torch.float32)
        if emb.dim() == 1:
            emb = emb.unsqueeze(0)
        if self.normalize:
            emb = torch.nn.functional.normalize(emb, dim=-1)
        return emb.to(self.device, dtype=torch.bfloat16)

    def add(self, embeddings, payloads: list):
        emb = self._prep(embeddings)
        if emb.shape[0] != len(payloads):
            raise ValueError("embeddings/payloads length mismatch")
        self._payloads.extend(payloads)
        self._embeddings = emb if self._embeddings is None else \
            torch.cat([self._embeddings, emb], dim=0)

    def search(self, query, k: int = 4):
        """Top-k by inner product.  One skinny-GEMM launch on GPU
        (scores[M, N] = Q @ E^T, M = #queries <= 64), torch matmul on
        CPU."""
        if self._embeddings is None:
            return []
        query = self._prep(query)
        k = min(k, len(self._payloads))
        if query.is_cuda and query.shape[0] <= 64 and \
                self._embeddings.shape[1] % 64 == 0:
            scores = ops.skinny_gemm(query.contiguous(),
                                     self._embeddings.contiguous())
        else:
            scores = query.float() @ self._embeddings.float().t()
        top = torch.topk(scores.float(), k, dim=-1)
        results = []
        for qi in range(query.shape[0]):
            results.append([
                {"score": float(top.values[qi, j]),
                 "payload": self._payloads[int(top.indices[qi, j])]}
                for j in range(k)])
        return results if len(results) > 1 else results[0]


class TokenMeanEmbedder:
    """Dependency-free embedder: mean-pool rows of an embedding
    matrix [vocab, D] over the token ids.  Pass a trained model's
    embed table (e.g. LlamaWeights.embed) or let it build a
    deterministic random one (synthetic pipelines)."""

    def __init__(self, embed_matrix: torch.Tensor = None,
                 vocab_size: int = 1024, dim: int = 256, device=None,
                 seed: int = 0):
        if embed_matrix is None:
            gen = torch.Generator().manual_seed(seed)
            embed_matrix = torch.randn(vocab_size, dim, generator=gen)
        self.embed = embed_matrix.to(
            device or ("cuda:0" if torch.cuda.is_available() else "cpu"))

    def __call__(self, token_ids) -> torch.Tensor:
        ids = torch.as_tensor(token_ids, dtype=torch.long,
                              device=self.embed.device)
        if ids.dim() == 1:
            ids = ids.unsqueeze(0)
        vecs = [self.embed[row[row < self.embed.shape[0]]].float().mean(0)
                for row in ids]
        return torch.stack(vecs)


class RetrievalStep:
    """Serving-graph step: embed the query tokens, search the index,
    PREPEND the top-k documents' tokens to the prompt (token-level
    RAG), and pass the event on to the model step.

    Request body: {"inputs": [[token ids], ...], ...}
    After this step inputs are [doc tokens ... + query tokens].
    """

    def __init__(self, context=None, name=None, index: VectorIndex = None,
                 embedder=None, top_k: int = 2,
                 max_context_tokens: int = 256, **kwargs):
        self.context = context
        self.name = name
        self.index = index
        self.embedder = embedder or TokenMeanEmbedder()
        self.top_k = top_k
        self.max_context_tokens = max_context_tokens

    def do_event(self, event):
        body = event.body if isinstance(event.body, dict) else {}
        inputs = body.get("inputs")
        if not inputs or self.index is None or not len(self.index):
            return event
        queries = self.embedder(inputs)
        hits = self.index.search(queries, k=self.top_k)
        if queries.shape[0] == 1:
            hits = [hits]
        enriched = []
        retrieved_meta = []
        for prompt, prompt_hits in zip(inputs, hits):
            ctx_tokens: typing.List[int] = []
            meta = []
            for hit in prompt_hits:
                doc = hit["payload"]
                tokens = doc.get("tokens", []) if isinstance(doc, dict) \
                    else list(doc)
                ctx_tokens.extend(tokens)
                meta.append({"score": hit["score"],
                             "doc_id": doc.get("id") if
                             isinstance(doc, dict) else None})
            ctx_tokens = ctx_tokens[:self.max_context_tokens]
            enriched.append(ctx_tokens + list(prompt))
            retrieved_meta.append(meta)
        body["inputs"] = enriched
        body["retrieval"] = retrieved_meta
        event.body = body
        return event
