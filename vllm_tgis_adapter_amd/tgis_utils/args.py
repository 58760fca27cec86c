"""Config / flag system.

Triple-layer argparse like the reference (reference tgis_utils/args.py):
engine/server args (ours, replacing vLLM's) → EnvVarArgumentParser giving
every flag an UPPER_SNAKE env-var fallback with correct bool coercion →
TGIS-compat aliases (`--model-name`, `--num-gpus`, ...) mapped onto engine
args by ``postprocess_tgis_args`` with the same consistency errors.
"""

from __future__ import annotations

import argparse
import os

from ..logging import init_logger

logger = init_logger(__name__)

MAX_TOP_N_TOKENS = 10  # == grpc.validation.MAX_TOP_N_TOKENS (import cycle)


class StoreBoolean(argparse.Action):
    def __call__(self, parser, namespace, values, option_string=None):
        v = str(values).lower()
        if v == "true":
            setattr(namespace, self.dest, True)
        elif v == "false":
            setattr(namespace, self.dest, False)
        else:
            raise ValueError(
                f"Invalid boolean value: {values}. Expected 'true' or 'false'."
            )


class FlexibleArgumentParser(argparse.ArgumentParser):
    """Accepts underscores in place of dashes in option names."""

    def parse_known_args(self, args=None, namespace=None):
        if args is None:
            import sys

            args = sys.argv[1:]
        fixed = []
        for a in args:
            if a.startswith("--") and "=" in a:
                key, _, val = a.partition("=")
                fixed.append(key.replace("_", "-") + "=" + val)
            elif a.startswith("--"):
                fixed.append(a.replace("_", "-"))
            else:
                fixed.append(a)
        return super().parse_known_args(fixed, namespace)


def _to_env_var(arg_name: str) -> str:
    return arg_name.upper().replace("-", "_")


def _bool_from_string(val: str) -> bool:
    return val.lower().strip() == "true" or val == "1"


def _apply_env_fallback(action: argparse.Action) -> None:
    env_val = os.environ.get(_to_env_var(action.dest))
    if not env_val:
        return
    val: bool | str
    if action.type is bool or type(action) in (
        argparse._StoreTrueAction,
        argparse._StoreFalseAction,
        StoreBoolean,
    ):
        val = _bool_from_string(env_val)
    else:
        # non-string types are coerced from the string default by argparse
        val = env_val
    if action.nargs in ("+", "*"):
        action.default = [val]
    else:
        action.default = val


class EnvVarArgumentParser(FlexibleArgumentParser):
    """Every flag gets an env-var fallback; help advertises the env name."""

    class _EnvVarHelpFormatter(argparse.ArgumentDefaultsHelpFormatter):
        def _get_help_string(self, action):
            help_ = super()._get_help_string(action)
            if action.dest != "help":
                help_ += f" [env: {_to_env_var(action.dest)}]"
            return help_

    def __init__(self, parser: argparse.ArgumentParser | None = None, *,
                 formatter_class=_EnvVarHelpFormatter, **kwargs):
        parents = []
        if parser:
            parents.append(parser)
            for action in parser._actions:
                if isinstance(action, argparse._HelpAction):
                    continue
                _apply_env_fallback(action)
        super().__init__(
            formatter_class=formatter_class, parents=parents, add_help=False, **kwargs
        )

    def _add_action(self, action):
        _apply_env_fallback(action)
        return super()._add_action(action)


# ---------------------------------------------------------------------------


def make_engine_arg_parser(parser: argparse.ArgumentParser) -> argparse.ArgumentParser:
    """The engine/server arg surface (our replacement for vLLM's parser)."""
    parser.add_argument("--model", type=str, default="tiny-llama",
                        help="model preset name or local HF model directory")
    parser.add_argument("--served-model-name", type=str, default=None)
    parser.add_argument("--host", type=str, default=None)
    parser.add_argument("--port", type=int, default=8000, help="HTTP port")
    parser.add_argument("--uvicorn-log-level", type=str, default="info")
    parser.add_argument("--max-model-len", type=int, default=None)
    parser.add_argument("--dtype", type=str, default="auto")
    parser.add_argument("--max-logprobs", type=int, default=20)
    parser.add_argument("--quantization", type=str, default=None)
    parser.add_argument("--tensor-parallel-size", type=int, default=None)
    parser.add_argument("--max-num-seqs", type=int, default=256)
    parser.add_argument("--max-num-batched-tokens", type=int, default=8192)
    parser.add_argument("--enable-prefix-caching", action="store_true",
                        help="share full prompt-prefix KV blocks across requests")
    parser.add_argument("--block-size", type=int, default=16)
    parser.add_argument("--gpu-memory-utilization", type=float, default=0.85)
    parser.add_argument("--num-gpu-blocks", type=int, default=None)
    parser.add_argument("--kv-cache-dtype", type=str, default="auto",
                        choices=["auto", "fp8", "fp8_e4m3"],
                        help="KV cache storage dtype (fp8 = OCP e4m3)")
    parser.add_argument("--enforce-eager", action="store_true",
                        help="disable hipGraph decode capture")
    parser.add_argument("--device", type=str, default="auto")
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--enable-lora", action="store_true")
    parser.add_argument("--max-loras", type=int, default=8)
    parser.add_argument("--max-lora-rank", type=int, default=64)
    parser.add_argument("--ssl-keyfile", type=str, default=None)
    parser.add_argument("--ssl-certfile", type=str, default=None)
    parser.add_argument("--ssl-ca-certs", type=str, default=None)
    parser.add_argument("--ssl-cert-reqs", type=int, default=0)
    parser.add_argument("--middleware", type=str, action="append", default=[])
    parser.add_argument("--disable-log-requests", action="store_true")
    parser.add_argument("--speculative-model", type=str, default=None)
    parser.add_argument("--use-v2-block-manager", action="store_true")
    return parser


def add_tgis_args(parser: argparse.ArgumentParser) -> argparse.ArgumentParser:
    """TGIS-compat flag aliases (surface kept identical to the reference
    tgis_utils/args.py:101-181 — it is the documented operator interface)."""
    parser.add_argument("--model-name", type=str,
                        help="name or path of the huggingface model to use")
    parser.add_argument("--max-sequence-length", type=int,
                        help="model context length; derived from the model if unset")
    parser.add_argument("--max-new-tokens", type=int, default=1024,
                        help="maximum allowed new (generated) tokens per request")
    parser.add_argument("--max-batch-size", type=int)
    parser.add_argument("--max-concurrent-requests", type=int)
    parser.add_argument("--dtype-str", type=str, help="deprecated, use dtype")
    parser.add_argument("--quantize", type=str,
                        choices=["awq", "gptq", "squeezellm", "int8", "int4",
                                 None],
                        help="weight quantization method (4-bit surfaces "
                             "apply RTN int4-g128; also accepts int8/int4)")
    parser.add_argument("--expert-parallel", action="store_true",
                        help="MoE: distribute whole experts across ranks with "
                             "token all-to-all instead of TP-sharding them")
    parser.add_argument("--num-gpus", type=int)
    parser.add_argument("--num-shard", type=int)
    parser.add_argument("--output-special-tokens", type=_bool_from_string, default=False)
    parser.add_argument("--default-include-stop-seqs", type=_bool_from_string, default=True)
    parser.add_argument("--grpc-port", type=int, default=8033)
    parser.add_argument("--tls-cert-path", type=str)
    parser.add_argument("--tls-key-path", type=str)
    parser.add_argument("--tls-client-ca-cert-path", type=str)
    parser.add_argument("--adapter-cache", type=str)
    parser.add_argument("--prefix-store-path", type=str,
                        help="Deprecated, use --adapter-cache")
    parser.add_argument("--disable-frontend-multiprocessing", action="store_true",
                        help="run the engine in the server process instead of "
                             "its own process (reference flag; MP is the default)")
    parser.add_argument("--speculator-name", type=str)
    parser.add_argument("--speculator-n-candidates", type=int)
    parser.add_argument("--speculator-max-batch-size", type=int)
    parser.add_argument("--enable-vllm-log-requests", type=_bool_from_string, default=False)
    parser.add_argument("--disable-prompt-logprobs", type=_bool_from_string, default=False)
    return parser


def postprocess_tgis_args(args: argparse.Namespace) -> argparse.Namespace:
    if args.model_name:
        args.model = args.model_name
    if args.max_sequence_length is not None:
        if args.max_model_len not in (None, args.max_sequence_length):
            raise ValueError(
                "Inconsistent max_model_len and max_sequence_length arg values"
            )
        args.max_model_len = args.max_sequence_length
    if args.dtype_str is not None:
        if args.dtype not in (None, "auto", args.dtype_str):
            raise ValueError("Inconsistent dtype and dtype_str arg values")
        args.dtype = args.dtype_str
    if args.quantize:
        if args.quantization and args.quantization != args.quantize:
            raise ValueError("Inconsistent quantize and quantization arg values")
        args.quantization = args.quantize
    if args.num_gpus is not None or args.num_shard is not None:
        if (
            args.num_gpus is not None
            and args.num_shard is not None
            and args.num_gpus != args.num_shard
        ):
            raise ValueError("Inconsistent num_gpus and num_shard arg values")
        num_gpus = args.num_gpus if args.num_gpus is not None else args.num_shard
        if args.tensor_parallel_size not in [None, 1, num_gpus]:
            raise ValueError(
                "Inconsistent tensor_parallel_size and num_gpus/num_shard arg values"
            )
        args.tensor_parallel_size = num_gpus
    if args.tensor_parallel_size is None:
        args.tensor_parallel_size = 1
    if args.max_logprobs < MAX_TOP_N_TOKENS + 1:
        logger.info("Setting max_logprobs to %d", MAX_TOP_N_TOKENS + 1)
        args.max_logprobs = MAX_TOP_N_TOKENS + 1
    # TGIS-format per-request logs replace engine per-request logging
    args.disable_log_requests = not args.enable_vllm_log_requests

    if args.speculator_name:
        if args.speculative_model and args.speculative_model != args.speculator_name:
            raise ValueError(
                "Inconsistent speculator_name and speculative_model arg values"
            )
        args.speculative_model = args.speculator_name

    if args.speculator_max_batch_size:
        logger.warning("speculator_max_batch_size is not yet supported")
    if args.max_batch_size is not None:
        logger.warning(
            "max_batch_size is set to %d but will be ignored for now. "
            "max_num_seqs can be used if this is still needed.",
            args.max_batch_size,
        )
    if args.max_concurrent_requests is not None:
        logger.warning(
            "max_concurrent_requests is not supported and will be ignored."
        )

    if args.tls_cert_path:
        args.ssl_certfile = args.tls_cert_path
    if args.tls_key_path:
        args.ssl_keyfile = args.tls_key_path
    if args.tls_client_ca_cert_path:
        args.ssl_ca_certs = args.tls_client_ca_cert_path
    return args


def engine_config_from_args(args: argparse.Namespace):
    """Build the EngineConfig from parsed args."""
    from ..engine.config import (
        CacheConfig,
        EngineConfig,
        ModelConfig,
        SchedulerConfig,
    )

    model_config = ModelConfig.from_model_arg(
        args.model, dtype=args.dtype or "auto", max_model_len=args.max_model_len
    )
    if getattr(args, "expert_parallel", False):
        model_config.expert_parallel = True
    return EngineConfig(
        model_config=model_config,
        cache_config=CacheConfig(
            block_size=args.block_size,
            gpu_memory_utilization=args.gpu_memory_utilization,
            num_gpu_blocks=args.num_gpu_blocks,
            enable_prefix_caching=args.enable_prefix_caching,
            kv_cache_dtype=("fp8" if args.kv_cache_dtype.startswith("fp8")
                            else "auto"),
        ),
        scheduler_config=SchedulerConfig(
            max_num_seqs=args.max_num_seqs,
            max_num_batched_tokens=args.max_num_batched_tokens,
        ),
        device=args.device,
        speculative_model=args.speculative_model,
        speculative_num_tokens=args.speculator_n_candidates or 4,
        tensor_parallel_size=args.tensor_parallel_size or 1,
        enforce_eager=args.enforce_eager,
        enable_lora=args.enable_lora or bool(args.adapter_cache or args.prefix_store_path),
        max_loras=args.max_loras,
        max_lora_rank=args.max_lora_rank,
        quantization=args.quantization,
        seed=args.seed,
    )
