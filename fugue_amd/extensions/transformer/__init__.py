from fugue_amd.extensions.transformer.transformer import (
    CoTransformer,
    OutputCoTransformer,
    OutputTransformer,
    Transformer,
)
from fugue_amd.extensions.transformer.convert import (
    cotransformer,
    output_cotransformer,
    output_transformer,
    register_output_transformer,
    register_transformer,
    transformer,
    _to_output_transformer,
    _to_transformer,
)
from fugue_amd.extensions.transformer.constants import OUTPUT_TRANSFORMER_DUMMY_SCHEMA
