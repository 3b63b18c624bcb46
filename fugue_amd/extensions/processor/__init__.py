from fugue_amd.extensions.processor.processor import Processor
from fugue_amd.extensions.processor.convert import (
    processor,
    register_processor,
    _to_processor,
)
