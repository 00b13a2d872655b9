"""ASCII table renderer reproducing the reference's PrettyTable console
output (reference: traffic_classifier.py:100-118) without the prettytable
dependency: centred cells, +- borders, one space padding."""

from __future__ import annotations

from typing import List, Sequence


class Table:
    def __init__(self, field_names: Sequence[str]):
        self.field_names = list(field_names)
        self.rows: List[List[str]] = []

    def add_row(self, row: Sequence) -> None:
        if len(row) != len(self.field_names):
            raise ValueError("row length mismatch")
        self.rows.append([str(v) for v in row])

    def __str__(self) -> str:
        widths = [len(h) for h in self.field_names]
        for row in self.rows:
            for i, cell in enumerate(row):
                widths[i] = max(widths[i], len(cell))
        sep = "+" + "+".join("-" * (w + 2) for w in widths) + "+"

        def fmt(cells: Sequence[str]) -> str:
            return "|" + "|".join(
                " " + c.center(w) + " " for c, w in zip(cells, widths)
            ) + "|"

        lines = [sep, fmt(self.field_names), sep]
        lines += [fmt(r) for r in self.rows]
        lines.append(sep)
        return "\n".join(lines)
