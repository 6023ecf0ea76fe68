from .grouped_query import GroupedQueryAttention

__all__ = ["GroupedQueryAttention"]
