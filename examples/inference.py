"""Standalone TGIS gRPC client for the fmaas.GenerationService server.

Counterpart of the reference's examples/inference.py (reference:
examples/inference.py:17-209): plaintext / TLS / mTLS channels, unary
Generate, streaming GenerateStream, Tokenize and ModelInfo — built on this
package's runtime-compiled protobuf classes instead of protoc output.

Usage (server: `python -m vllm_tgis_adapter_amd --model-name ... `):
    python examples/inference.py --text "hello" --max-new-tokens 16
    python examples/inference.py --streaming --text "hello"
    python examples/inference.py --tokenize --text "a b c"
    python examples/inference.py --model-info
TLS:
    python examples/inference.py --server-cert ca.pem [--client-cert c.pem --client-key k.pem]
"""

from __future__ import annotations

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import grpc

from vllm_tgis_adapter_amd.grpc import proto
from vllm_tgis_adapter_amd.grpc.stubs import GenerationStub


def make_channel(args) -> grpc.Channel:
    target = f"{args.host}:{args.port}"
    if not args.server_cert:
        return grpc.insecure_channel(target)
    root = Path(args.server_cert).read_bytes()
    if args.client_cert:
        creds = grpc.ssl_channel_credentials(
            root_certificates=root,
            private_key=Path(args.client_key).read_bytes(),
            certificate_chain=Path(args.client_cert).read_bytes(),
        )
    else:
        creds = grpc.ssl_channel_credentials(root_certificates=root)
    return grpc.secure_channel(target, creds)


def build_params(args):
    p = proto.Parameters()
    p.stopping.max_new_tokens = args.max_new_tokens
    p.stopping.min_new_tokens = args.min_new_tokens
    if args.temperature is not None:
        p.sampling.temperature = args.temperature
        p.method = proto.SAMPLE
    if args.seed is not None:
        p.sampling.seed = args.seed
    if args.guided_regex:
        p.decoding.regex = args.guided_regex
    return p


def main() -> int:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--host", default="localhost")
    ap.add_argument("--port", type=int, default=8033)
    ap.add_argument("--text", default="The answer to life is")
    ap.add_argument("--max-new-tokens", type=int, default=16)
    ap.add_argument("--min-new-tokens", type=int, default=0)
    ap.add_argument("--temperature", type=float, default=None)
    ap.add_argument("--seed", type=int, default=None)
    ap.add_argument("--guided-regex", default=None)
    ap.add_argument("--streaming", action="store_true")
    ap.add_argument("--tokenize", action="store_true")
    ap.add_argument("--model-info", action="store_true")
    ap.add_argument("--adapter-id", default=None)
    ap.add_argument("--correlation-id", default=None)
    ap.add_argument("--server-cert", default=None)
    ap.add_argument("--client-cert", default=None)
    ap.add_argument("--client-key", default=None)
    args = ap.parse_args()

    channel = make_channel(args)
    stub = GenerationStub(channel)
    metadata = []
    if args.correlation_id:
        metadata.append(("x-correlation-id", args.correlation_id))

    if args.model_info:
        resp = stub.ModelInfo(proto.ModelInfoRequest(model_id=""), metadata=metadata)
        print(resp)
        return 0

    if args.tokenize:
        req = proto.BatchedTokenizeRequest(
            requests=[proto.TokenizeRequest(text=args.text)], return_tokens=True
        )
        resp = stub.Tokenize(req, metadata=metadata)
        for r in resp.responses:
            print(f"token_count={r.token_count} tokens={list(r.tokens)}")
        return 0

    params = build_params(args)
    if args.streaming:
        req = proto.SingleGenerationRequest(
            request=proto.GenerationRequest(text=args.text), params=params
        )
        if args.adapter_id:
            req.adapter_id = args.adapter_id
        for msg in stub.GenerateStream(req, metadata=metadata):
            if msg.text:
                sys.stdout.write(msg.text)
                sys.stdout.flush()
        print()
        return 0

    req = proto.BatchedGenerationRequest(
        requests=[proto.GenerationRequest(text=args.text)], params=params
    )
    if args.adapter_id:
        req.adapter_id = args.adapter_id
    resp = stub.Generate(req, metadata=metadata)
    for r in resp.responses:
        print(f"[{proto.StopReasonValue.Name(r.stop_reason)}] {r.text!r} "
              f"({r.generated_token_count} tokens)")
    return 0


if __name__ == "__main__":
    sys.exit(main())
