"""SQL tokenizer + parser (implemented in this round's SQL milestone)."""
from __future__ import annotations


def parse_expression(text: str):
    from .planner import _EXPR_PARSER_TODO
    raise NotImplementedError("sql_expr lands with the SQL frontend milestone")
