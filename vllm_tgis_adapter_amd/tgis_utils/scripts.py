"""``model-util`` / ``text-generation-server`` CLIs:
download-weights, convert-to-safetensors, convert-to-fast-tokenizer."""

from __future__ import annotations

import argparse
from pathlib import Path

from ..logging import init_logger
from . import hub

logger = init_logger(__name__)


def download_weights(
    model_name: str,
    extension: str = ".safetensors",
    revision: str | None = None,
    auth_token: str | None = None,
) -> None:
    try:
        hub.download_weights(model_name, extension, revision, auth_token)
        return
    except hub.EntryNotFoundError:
        if extension != ".safetensors":
            raise
    # no safetensors upstream: fetch .bin and convert locally
    logger.info("No .safetensors weights found; downloading .bin and converting")
    pt_files = [Path(p) for p in hub.download_weights(model_name, ".bin", revision, auth_token)]
    sf_files = [p.with_name(p.name.replace("pytorch_model", "model")).with_suffix(".safetensors")
                for p in pt_files]
    hub.convert_files(pt_files, sf_files)


def convert_to_safetensors(model_path: str) -> None:
    pt_files = [Path(p) for p in hub.local_weight_files(model_path, ".bin")]
    if not pt_files:
        raise FileNotFoundError(f"no .bin files under {model_path}")
    sf_files = [p.with_name(p.name.replace("pytorch_model", "model")).with_suffix(".safetensors")
                for p in pt_files]
    hub.convert_files(pt_files, sf_files)
    index = Path(model_path) / "pytorch_model.bin.index.json"
    if index.exists():
        hub.convert_index_file(index, Path(model_path) / "model.safetensors.index.json")


def convert_to_fast_tokenizer(model_path: str, output_path: str | None = None) -> None:
    from transformers import AutoTokenizer

    out = output_path or model_path
    tokenizer = AutoTokenizer.from_pretrained(model_path, use_fast=True)
    if not tokenizer.is_fast:
        raise RuntimeError(f"could not build a fast tokenizer for {model_path}")
    tokenizer.save_pretrained(out)
    logger.info("Saved fast tokenizer to %s", out)


def cli(argv=None) -> None:
    parser = argparse.ArgumentParser(prog="model-util")
    sub = parser.add_subparsers(dest="command", required=True)

    dl = sub.add_parser("download-weights", help="download model weights from the hub")
    dl.add_argument("model_name")
    dl.add_argument("--extension", default=".safetensors")
    dl.add_argument("--revision", default=None)
    dl.add_argument("--auth-token", default=None)

    cv = sub.add_parser("convert-to-safetensors", help=".bin -> .safetensors")
    cv.add_argument("model_path")

    ft = sub.add_parser("convert-to-fast-tokenizer")
    ft.add_argument("model_path")
    ft.add_argument("--output-path", default=None)

    args = parser.parse_args(argv)
    if args.command == "download-weights":
        download_weights(args.model_name, args.extension, args.revision, args.auth_token)
    elif args.command == "convert-to-safetensors":
        convert_to_safetensors(args.model_path)
    elif args.command == "convert-to-fast-tokenizer":
        convert_to_fast_tokenizer(args.model_path, args.output_path)


# TGIS back-compat alias (installed as text-generation-server)
tgis_cli = cli

if __name__ == "__main__":
    cli()
