#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Gen-AI RAG serving graph (baseline config 5 shape, single node):

    retrieval step (GPU vector index) -> Llama V2ModelServer

The index holds synthetic token documents; retrieval prepends top-k
document tokens to the prompt before decode.  On an 8-GPU node the
Llama step runs llama-3-70b (TP via scripts/bench_serving_tp.py);
this example stays single-GPU/CPU-sized.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import torch  # noqa: E402

import mlrun_amd  # noqa: E402
from mlrun_amd.models.llama import LlamaServer  # noqa: E402
from mlrun_amd.serving import (  # noqa: E402
    RetrievalStep,
    TokenMeanEmbedder,
    VectorIndex,
)


def main():
    on_gpu = torch.cuda.is_available()
    model = "llama-3-8b" if on_gpu else "tiny"
    vocab = 1000
    embedder = TokenMeanEmbedder(vocab_size=vocab, dim=256, seed=1)
    index = VectorIndex(dim=256)
    gen = torch.Generator().manual_seed(7)
    docs = [{"id": f"doc{i}",
             "tokens": torch.randint(0, vocab, (24,),
                                     generator=gen).tolist()}
            for i in range(64)]
    index.add(torch.stack([embedder(d["tokens"])[0] for d in docs]),
              docs)

    fn = mlrun_amd.new_function(name="rag-serving", kind="serving")
    graph = fn.set_topology("flow", engine="sync")
    graph.add_step(RetrievalStep, name="retrieve", index=index,
                   embedder=embedder, top_k=2)
    graph.add_step(LlamaServer, name="llm", after="retrieve",
                   config=model, batch_size=8, max_new_tokens=16,
                   respond=True)
    server = fn.to_mock_server()
    prompt = torch.randint(0, vocab, (12,), generator=gen).tolist()
    resp = server.test("/v2/models/llm/infer",
                       body={"inputs": [prompt], "max_tokens": 8})
    print("retrieved:", resp.get("retrieval"))
    print("generated:", resp["outputs"])


if __name__ == "__main__":
    main()
