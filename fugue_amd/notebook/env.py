"""Jupyter ``%%fsql`` cell magic + HTML display.

Reference parity: ``fugue_notebook/env.py`` (``NotebookSetup`` :20,
``_FugueSQLMagics`` :36).  ``setup()`` registers the magic in the current
IPython session.
"""
from typing import Any, Optional


class NotebookSetup:
    """Hook points a deployment can override."""

    def get_pre_conf(self) -> dict:
        return {}

    def get_post_conf(self) -> dict:
        return {}


def _register_magics(setup_obj: NotebookSetup) -> None:
    from IPython.core.magic import Magics, cell_magic, magics_class, needs_local_scope

    @magics_class
    class _FugueSQLMagics(Magics):
        @needs_local_scope
        @cell_magic("fsql")
        def fsql(self, line: str, cell: str, local_ns: Any = None) -> None:
            import fugue_amd.api as fa
            from fugue_amd.sql.workflow import FugueSQLWorkflow

            engine = line.strip() if line.strip() != "" else None
            dag = FugueSQLWorkflow()
            variables = {
                k: v
                for k, v in (local_ns or {}).items()
                if not k.startswith("_")
            }
            dag._sql(cell, variables)
            dag.run(engine)

    ip = get_ipython()  # noqa: F821
    ip.register_magics(_FugueSQLMagics)


def setup(notebook_setup: Optional[NotebookSetup] = None) -> None:
    """Register the %%fsql magic (call from a notebook)."""
    try:
        get_ipython  # noqa: F821
    except NameError:
        raise RuntimeError("setup() must be called inside IPython/Jupyter")
    _register_magics(notebook_setup or NotebookSetup())
