"""SQL -> LogicalPlan planner (implemented in this round's SQL milestone)."""
from __future__ import annotations

_EXPR_PARSER_TODO = True


def plan_sql(query: str, lookup_table):
    raise NotImplementedError("daft_amd.sql lands with the SQL frontend "
                              "milestone of this round")
