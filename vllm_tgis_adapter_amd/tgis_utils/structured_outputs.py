"""Map the proto DecodingParameters ``guided`` oneof to engine params
(reference tgis_utils/structured_outputs.py:14-38 defines the wire mapping)."""

from __future__ import annotations

from typing import Optional

from ..engine.types import StructuredOutputsParams

_FORMAT_JSON = 1  # fmaas.DecodingParameters.ResponseFormat.JSON


def get_structured_output_params(decoding_params) -> Optional[StructuredOutputsParams]:
    guided = decoding_params.WhichOneof("guided")
    if not guided:
        return None
    if guided == "json_schema":
        return StructuredOutputsParams(json=decoding_params.json_schema)
    if guided == "regex":
        return StructuredOutputsParams(regex=decoding_params.regex)
    if guided == "choice":
        choices = list(decoding_params.choice.choices)
        if len(choices) < 2:
            raise ValueError("Must provide at least two choices")
        return StructuredOutputsParams(choice=choices)
    if guided == "grammar":
        return StructuredOutputsParams(grammar=decoding_params.grammar)
    if decoding_params.format == _FORMAT_JSON:
        return StructuredOutputsParams(json_object=True)
    raise ValueError(guided)
