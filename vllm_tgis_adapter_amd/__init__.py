"""MI355X-native TGIS-protocol LLM serving stack.

Wire-compatible with opendatahub-io/vllm-tgis-adapter's fmaas.GenerationService
(gRPC), OpenAI HTTP front-end and CLI entrypoints, on a from-scratch
continuous-batching / paged-KV engine with hand-written CDNA4 HIP kernels.
"""

__version__ = "0.1.0"
