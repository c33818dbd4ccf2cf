"""Minimal live progress table for the training console.

The reference depends on the third-party ``progress_table`` package
(reference dmlcloud/stage.py:7,147). This framework ships its own
dependency-free implementation with the same usage surface that the
Stage engine needs:

    table = ProgressTable(file=sys.stdout)
    table.add_column('Epoch', width=8)
    table['Epoch'] = 1          # live cell update
    table.update('Epoch', 2)
    table.next_row()            # commit the row and print it
    table.close()               # bottom border

Non-root ranks pass a DevNullIO-like file object so only rank 0 renders.
"""

import numbers
import sys
from datetime import timedelta
from typing import Any, Optional


def _format_cell(value: Any, width: int) -> str:
    # unwrap 0-dim / 1-element tensors (duck-typed; no torch import here)
    if hasattr(value, 'numel') and hasattr(value, 'item') and value.numel() == 1:
        value = value.item()
    if value is None:
        text = ''
    elif isinstance(value, float):
        if value != 0 and (abs(value) >= 10 ** (width - 2) or abs(value) < 10 ** -(width - 4)):
            text = f'{value:.{max(width - 7, 1)}e}'
        else:
            text = f'{value:.{max(width - 4, 1)}g}'
    elif isinstance(value, timedelta):
        total = value.total_seconds()
        hours, rem = divmod(int(total), 3600)
        minutes, seconds = divmod(rem, 60)
        text = f'{hours:d}:{minutes:02d}:{seconds:02d}'
    elif isinstance(value, numbers.Number):
        text = str(value)
    else:
        text = str(value)
    if len(text) > width:
        text = text[:width]
    return text.rjust(width)


class ProgressTable:
    """Renders one bordered row per committed epoch."""

    DEFAULT_WIDTH = 12

    def __init__(self, file=None):
        self.file = file if file is not None else sys.stdout
        self.columns = []  # list of (name, width)
        self.current_row = {}
        self._header_printed = False
        self._closed = False

    def add_column(self, name: str, width: Optional[int] = None, **kwargs):
        if self._header_printed:
            raise RuntimeError('Cannot add columns after the first row was printed')
        width = width or max(self.DEFAULT_WIDTH, len(str(name)) + 2)
        self.columns.append((str(name), width))

    def __setitem__(self, name: str, value: Any):
        self.update(name, value)

    def update(self, name: str, value: Any):
        self.current_row[str(name)] = value

    def _hline(self, left: str, mid: str, right: str) -> str:
        return left + mid.join('─' * (w + 2) for _, w in self.columns) + right

    def _print(self, text: str):
        self.file.write(text + '\n')
        if hasattr(self.file, 'flush'):
            self.file.flush()

    def _print_header(self):
        if not self.columns:
            return
        self._print(self._hline('┌', '┬', '┐'))
        cells = ' │ '.join(str(name).center(width) for name, width in self.columns)
        self._print(f'│ {cells} │')
        self._print(self._hline('├', '┼', '┤'))
        self._header_printed = True

    def next_row(self):
        if not self.columns:
            self.current_row = {}
            return
        if not self._header_printed:
            self._print_header()
        cells = ' │ '.join(_format_cell(self.current_row.get(name), width) for name, width in self.columns)
        self._print(f'│ {cells} │')
        self.current_row = {}

    def close(self):
        if self._closed:
            return
        self._closed = True
        if self._header_printed:
            self._print(self._hline('└', '┴', '┘'))
