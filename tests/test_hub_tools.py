"""Weight-artifact tooling tests (reference tests/test_hub.py analog, but
offline: .bin -> .safetensors conversion with verification, index
conversion, local model resolution)."""

from __future__ import annotations

import json
import os

import pytest
import torch

from vllm_tgis_adapter_amd.tgis_utils import hub


def test_convert_bin_to_safetensors(tmp_path):
    st = {
        "a.weight": torch.randn(4, 8),
        "b.weight": torch.arange(6, dtype=torch.float32).reshape(2, 3),
    }
    src = tmp_path / "pytorch_model.bin"
    torch.save(st, src)
    dst = tmp_path / "model.safetensors"
    hub.convert_file(src, dst)
    from safetensors.torch import load_file

    out = load_file(str(dst))
    assert set(out) == set(st)
    for k in st:
        assert torch.equal(out[k], st[k])


def test_convert_removes_shared_duplicates(tmp_path):
    w = torch.randn(4, 4)
    st = {"tied.a": w, "tied.b": w}  # shared storage
    src = tmp_path / "pytorch_model.bin"
    torch.save(st, src)
    dst = tmp_path / "model.safetensors"
    hub.convert_file(src, dst)
    from safetensors.torch import load_file

    out = load_file(str(dst))
    assert len(out) == 1  # duplicate dropped


def test_convert_index_file(tmp_path):
    idx = {
        "metadata": {"total_size": 123},
        "weight_map": {"a": "pytorch_model-00001-of-00001.bin"},
    }
    src = tmp_path / "pytorch_model.bin.index.json"
    src.write_text(json.dumps(idx))
    hub.convert_index_file(src, tmp_path / "model.safetensors.index.json")
    out = json.loads((tmp_path / "model.safetensors.index.json").read_text())
    assert out["weight_map"]["a"].endswith(".safetensors")


def test_get_model_path_local_dir(tmp_path):
    (tmp_path / "config.json").write_text("{}")
    assert hub.get_model_path(str(tmp_path)) == str(tmp_path)
