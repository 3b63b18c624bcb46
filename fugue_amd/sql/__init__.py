from fugue_amd.sql.api import fugue_sql, fugue_sql_flow
from fugue_amd.sql.workflow import FugueSQLWorkflow
