"""Framework exception types (reference parity: ``fugue/exceptions.py``)."""


class FugueError(Exception):
    """Base exception of the framework"""


class FugueBug(FugueError):
    """Internal invariant violation"""


class FugueInvalidOperation(FugueError):
    """Invalid operation on the framework"""


class FugueDataFrameError(FugueError):
    """DataFrame-related errors"""


class FugueDataFrameOperationError(FugueDataFrameError):
    """Invalid DataFrame operation"""


class FugueDataFrameInitError(FugueDataFrameError):
    """DataFrame construction error"""


class FugueDatasetEmptyError(FugueDataFrameError):
    """Empty dataset error"""


class FugueDataFrameEmptyError(FugueDatasetEmptyError):
    """Peeking an empty DataFrame (subclass of the dataset-level empty
    error so either type can be caught, as in the reference)"""


class FugueWorkflowError(FugueError):
    """Workflow errors"""


class FugueWorkflowCompileError(FugueWorkflowError):
    """Workflow compile-time error"""


class FugueWorkflowCompileValidationError(FugueWorkflowCompileError):
    """Validation error at compile time"""


class FugueInterfacelessError(FugueWorkflowCompileError):
    """Error constructing an extension from a plain object"""


class FugueWorkflowRuntimeError(FugueWorkflowError):
    """Workflow runtime error"""


class FugueSQLRuntimeError(FugueWorkflowRuntimeError):
    """FugueSQL runtime error (reference ``fugue/exceptions.py:65``)"""


class FugueWorkflowRuntimeValidationError(FugueWorkflowRuntimeError):
    """Validation error at runtime"""


class FugueSQLError(FugueWorkflowCompileError):
    """FugueSQL parse/compile error"""


class FugueSQLSyntaxError(FugueSQLError):
    """FugueSQL syntax error"""


class FuguePluginsRegistrationError(FugueError):
    """Plugin registration error"""
