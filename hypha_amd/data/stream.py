"""Streaming slice-file dataset + preprocessor registry.

Parity with the reference executor's data path:
  * `SliceStreamDataset` mirrors IterableStreamDataSet
    (executors/accelerate/.../dataset.py:9-34): consume an iterator of
    SafeTensors slice-file paths, optionally run a preprocessor over
    `processor_inputs`, then yield per-sample dicts keyed by `model_inputs`.
  * `infinite` mirrors dataset_wrapper (dataset.py:37-41): endless epochs.
  * `build_preprocessor` mirrors the `PreprocessorType` wire enum
    (crates/messages/src/lib.rs:473-480: tokenizer/feature/image/video/auto)
    resolved against locally-available artifacts only — this environment has
    no network, so hub downloads raise a clear error instead of hanging.
"""

from __future__ import annotations

import os
from typing import Any, Callable, Iterable, Iterator

import torch
from torch.utils.data import IterableDataset

PREPROCESSOR_TYPES = ("tokenizer", "feature", "image", "video", "auto")


class SliceStreamDataset(IterableDataset):
    def __init__(
        self,
        slice_paths: Iterable[str],
        model_inputs: list[str],
        processor_inputs: list[str] | None = None,
        preprocessor: Callable[..., dict[str, Any]] | None = None,
    ) -> None:
        super().__init__()
        self.slice_paths = slice_paths
        self.model_inputs = model_inputs
        self.processor_inputs = processor_inputs or []
        self.preprocessor = preprocessor

    def __iter__(self) -> Iterator[dict[str, torch.Tensor]]:
        from safetensors.torch import load_file

        for path in self.slice_paths:
            data = load_file(path, device="cpu")
            if self.preprocessor is not None:
                fed = {k: data.pop(k) for k in self.processor_inputs}
                data = {**self.preprocessor(**fed), **data}
            for values in zip(*(data[k] for k in self.model_inputs)):
                yield dict(zip(self.model_inputs, values))


def infinite(loader: Iterable) -> Iterator:
    """Endless-epoch wrapper (dataset.py:37-41)."""
    while True:
        yield from loader


def build_preprocessor(task: str, artifact_path: str | None = None):
    """Resolve a wire PreprocessorType to a callable.

    Mirrors the reference's get_preprocessor mapping
    (executors/accelerate/.../utils.py:37-54: tokenizer/feature/image/
    video/auto -> the matching transformers Auto* class loaded from the
    FETCHED artifact path). All types load from LOCAL artifacts
    (`local_files_only=True`): the artifact arrives through the fetch
    connector (scheduler slice / URI / HF), never from the hub at
    preprocess time.
    """
    if task not in PREPROCESSOR_TYPES:
        raise ValueError(
            f"preprocessor {task!r} not supported; available: {PREPROCESSOR_TYPES}"
        )
    if not artifact_path or not os.path.exists(artifact_path):
        raise FileNotFoundError(
            f"preprocessor artifact {artifact_path!r} not found (no network: "
            "artifacts must be local files fetched through the connector)"
        )

    if task == "tokenizer":
        from transformers import AutoTokenizer

        tok = AutoTokenizer.from_pretrained(artifact_path, local_files_only=True)

        def run(**kw):
            (key, texts), = kw.items()
            out = tok(list(texts), return_tensors="pt", padding=True, truncation=True)
            return dict(out)

        return run

    if task in ("feature", "image", "video", "auto"):
        kwargs = {}
        if task == "feature":
            from transformers import AutoFeatureExtractor as Cls
        elif task == "image":
            try:
                import torchvision  # noqa: F401

                from transformers import AutoImageProcessor as Cls
            except ImportError:
                # no torchvision: resolve the PIL-backend processor class
                # named by the artifact config directly (transformers
                # aliases e.g. ViTImageProcessor -> ViTImageProcessorPil)
                import json as _json

                import transformers as _tf

                with open(os.path.join(artifact_path,
                                       "preprocessor_config.json")) as f:
                    cls_name = _json.load(f)["image_processor_type"]
                Cls = getattr(_tf, cls_name)
        elif task == "video":
            try:
                from transformers import AutoVideoProcessor as Cls
            except ImportError as e:  # transformers build without video stack
                raise NotImplementedError(
                    "this transformers build has no AutoVideoProcessor"
                ) from e
        else:
            from transformers import AutoProcessor as Cls

        proc = Cls.from_pretrained(artifact_path, local_files_only=True, **kwargs)

        def run(**kw):
            (key, values), = kw.items()
            if hasattr(values, "numpy"):
                values = [v.numpy() for v in values]
            out = proc(values, return_tensors="pt")
            return dict(out)

        return run
    raise NotImplementedError(task)
