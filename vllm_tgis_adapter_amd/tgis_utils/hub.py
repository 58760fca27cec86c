"""Weight-artifact tooling: hub listing/download, .bin →
.safetensors conversion with duplicate-tensor removal and round-trip
verification, fast-tokenizer conversion helpers.

Mirrors the behavior of the reference's tgis_utils/hub.py; network calls go
through huggingface_hub and simply fail cleanly in air-gapped deployments.
"""

from __future__ import annotations

import concurrent.futures
import datetime
import json
import os
from collections import defaultdict
from pathlib import Path

import torch

from ..logging import init_logger

logger = init_logger(__name__)


def get_model_path(model_name: str, revision: str | None = None) -> str:
    """Resolve a model name to a local path (local dir wins; else HF cache)."""
    if os.path.isdir(model_name):
        return model_name
    from huggingface_hub import snapshot_download

    return snapshot_download(model_name, revision=revision)


def weight_hub_files(
    model_name: str, extension: str = ".safetensors", revision: str | None = None
) -> list[str]:
    """List the repo's weight files with the given extension."""
    from huggingface_hub import HfApi

    api = HfApi()
    info = api.model_info(model_name, revision=revision)
    return [s.rfilename for s in info.siblings if s.rfilename.endswith(extension)]


def download_weights(
    model_name: str,
    extension: str = ".safetensors",
    revision: str | None = None,
    auth_token: str | None = None,
) -> list[str]:
    """Download the model's weight files (threaded)."""
    from huggingface_hub import hf_hub_download

    filenames = weight_hub_files(model_name, extension, revision)
    if not filenames:
        raise EntryNotFoundError(
            f"No {extension} weights found for model {model_name}"
        )

    def _one(fn: str) -> str:
        start = datetime.datetime.now()
        local = hf_hub_download(
            model_name, filename=fn, revision=revision, token=auth_token
        )
        logger.info("Downloaded %s in %s", local, datetime.datetime.now() - start)
        return local

    with concurrent.futures.ThreadPoolExecutor(max_workers=5) as pool:
        return list(pool.map(_one, filenames))


class EntryNotFoundError(FileNotFoundError):
    pass


def local_weight_files(model_path: str, extension: str = ".safetensors") -> list[str]:
    return [str(p) for p in sorted(Path(model_path).glob(f"*{extension}"))]


def _remove_duplicate_tensors(state_dict: dict) -> tuple[dict, dict[str, list[str]]]:
    """Drop tensors that share storage (safetensors requires unique storage)."""
    ptrs = defaultdict(list)
    for name, tensor in state_dict.items():
        ptrs[(tensor.data_ptr(), tensor.shape, tensor.stride())].append(name)
    removed: dict[str, list[str]] = {}
    kept = dict(state_dict)
    for names in ptrs.values():
        if len(names) > 1:
            keeper = names[0]
            removed[keeper] = names[1:]
            for dup in names[1:]:
                kept.pop(dup, None)
    return kept, removed


def convert_file(pt_file: Path, sf_file: Path) -> None:
    """One .bin → .safetensors conversion with round-trip verification."""
    from safetensors.torch import load_file, save_file

    logger.info("Converting %s to %s", pt_file, sf_file)
    state = torch.load(pt_file, map_location="cpu", weights_only=True)
    if "state_dict" in state and isinstance(state["state_dict"], dict):
        state = state["state_dict"]
    state = {k: v for k, v in state.items() if isinstance(v, torch.Tensor)}
    state, removed = _remove_duplicate_tensors(state)
    if removed:
        logger.info("Removed shared tensors: %s", removed)
    state = {k: v.contiguous() for k, v in state.items()}
    sf_file.parent.mkdir(parents=True, exist_ok=True)
    save_file(state, str(sf_file), metadata={"format": "pt"})

    reloaded = load_file(str(sf_file))
    for k, v in state.items():
        if not torch.equal(v, reloaded[k]):
            raise RuntimeError(f"Tensor {k} mismatch after conversion round-trip")


def convert_files(pt_files: list[Path], sf_files: list[Path]) -> None:
    assert len(pt_files) == len(sf_files)
    for i, (pt, sf) in enumerate(zip(pt_files, sf_files)):
        convert_file(pt, sf)
        logger.info("Converted %d/%d files", i + 1, len(pt_files))


def convert_index_file(pt_index: Path, sf_index: Path) -> None:
    """Rewrite a pytorch bin index json to reference the .safetensors names."""
    with open(pt_index) as f:
        index = json.load(f)
    weight_map = {
        k: v.replace("pytorch_model", "model").replace(".bin", ".safetensors")
        for k, v in index.get("weight_map", {}).items()
    }
    with open(sf_index, "w") as f:
        json.dump({"metadata": index.get("metadata", {}), "weight_map": weight_map}, f)
