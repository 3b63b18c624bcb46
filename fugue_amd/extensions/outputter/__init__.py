from fugue_amd.extensions.outputter.outputter import Outputter
from fugue_amd.extensions.outputter.convert import (
    outputter,
    register_outputter,
    _to_outputter,
)
