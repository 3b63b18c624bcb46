from fugue_amd.extensions.context import ExtensionContext
from fugue_amd.extensions.creator.creator import Creator
from fugue_amd.extensions.creator.convert import (
    creator,
    register_creator,
    _to_creator,
)
from fugue_amd.extensions.processor.processor import Processor
from fugue_amd.extensions.processor.convert import (
    processor,
    register_processor,
    _to_processor,
)
from fugue_amd.extensions.outputter.outputter import Outputter
from fugue_amd.extensions.outputter.convert import (
    outputter,
    register_outputter,
    _to_outputter,
)
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
from fugue_amd.extensions._utils import (  # noqa: E402
    is_namespace_extension,
    namespace_candidate,
)
from fugue_amd.extensions.creator.convert import parse_creator  # noqa: E402
from fugue_amd.extensions.processor.convert import parse_processor  # noqa: E402
from fugue_amd.extensions.outputter.convert import parse_outputter  # noqa: E402
from fugue_amd.extensions.transformer.convert import (  # noqa: E402
    parse_output_transformer,
    parse_transformer,
)
