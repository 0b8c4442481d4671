"""Flattened depth-first schema walk (reference: schema/SchemaVisitor.java,
SimpleSchemaVisitor, Visitors — SURVEY.md §2.2).

Kudo and the host-table paths need a stable pre-order flattening where a
struct/list column's own buffers come before its children's.
"""
from typing import List

from .columnar import Column, DType


def flatten_columns(cols: List[Column]) -> List[Column]:
    """Pre-order flattening: parent before children (Kudo buffer order)."""
    out: List[Column] = []

    def walk(c: Column):
        out.append(c)
        for ch in c.children:
            walk(ch)

    for c in cols:
        walk(c)
    return out


def has_offsets(c: Column) -> bool:
    return c.dtype in (DType.STRING, DType.LIST)


def has_data(c: Column) -> bool:
    return c.dtype not in (DType.LIST, DType.STRUCT)


class SchemaVisitor:
    """Generic pre-order visitor (reference SchemaVisitor.java)."""

    def pre_visit(self, col: Column, depth: int):  # pragma: no cover
        pass

    def post_visit(self, col: Column, depth: int):  # pragma: no cover
        pass


def visit(cols: List[Column], visitor: SchemaVisitor):
    def walk(c: Column, depth: int):
        visitor.pre_visit(c, depth)
        for ch in c.children:
            walk(ch, depth + 1)
        visitor.post_visit(c, depth)

    for c in cols:
        walk(c, 0)
