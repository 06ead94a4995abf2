#!/usr/bin/env python3
# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Deploy a Llama serving graph on the local MI355X and query it over
HTTP (baseline config 5 shape, single GPU)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(
    __file__))))

import torch  # noqa: E402

import mlrun_amd  # noqa: E402
from mlrun_amd.models.llama import LlamaServer  # noqa: E402


def main():
    model = "llama-3-8b" if torch.cuda.is_available() else "tiny"
    fn = mlrun_amd.new_function(name="llm-serving", kind="serving")
    # continuous batching: requests admit at token boundaries;
    # weight_dtype="fp8w" / kv_dtype="fp8" / replicas=N stack on top,
    # and fn.deploy(workers=N) scales across processes
    fn.add_model("llama", class_name=LlamaServer, config=model,
                 batch_size=8, max_new_tokens=16,
                 scheduling="continuous",
                 replicas=2 if torch.cuda.is_available() else 1)
    address = fn.deploy()
    print("serving at", address)
    resp = fn.invoke("/v2/models/llama/infer",
                     body={"inputs": [[1, 2, 3, 4]], "max_tokens": 8})
    print("generated:", resp["outputs"])
    # token streaming (ndjson): one line per generated token
    import json
    import requests

    with requests.post(address + "/v2/models/llama/infer",
                       json={"inputs": [[5, 6, 7]], "max_tokens": 4,
                             "stream": True}, stream=True,
                       timeout=120) as stream:
        tokens = [json.loads(line)["token"]
                  for line in stream.iter_lines() if line]
    print("streamed:", tokens)
    fn.stop()


if __name__ == "__main__":
    main()
