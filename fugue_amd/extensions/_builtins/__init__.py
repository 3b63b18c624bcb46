from fugue_amd.extensions._builtins.creators import CreateData, Load, LoadYielded
from fugue_amd.extensions._builtins.outputters import (
    AssertEqual,
    AssertNotEqual,
    RunOutputTransformer,
    Save,
    Show,
)
from fugue_amd.extensions._builtins.processors import (
    Aggregate,
    AlterColumns,
    Assign,
    Distinct,
    DropColumns,
    Dropna,
    Fillna,
    Filter,
    Rename,
    RunJoin,
    RunSetOperation,
    RunSQLSelect,
    RunTransformer,
    Sample,
    SaveAndUse,
    Select,
    SelectColumns,
    Take,
    Zip,
)
