from .interface import sql, sql_expr, register_table, SQLCatalog  # noqa: F401
