"""CLI helpers: Enum-valued argparse flags.

`parser.add_argument('--mode', type=Mode, action=EnumAction)` accepts the
enum's *values* on the command line and stores the enum *member*
(capability parity with reference dmlcloud/util/argparse.py:5-31).
Supports repeated flags via nargs (stores a list of members) and derives
the choices list from the enum automatically.
"""

import argparse
from enum import Enum
from typing import Optional, Sequence, Union

__all__ = ['EnumAction']


class EnumAction(argparse.Action):
    """Store an Enum member parsed from its string value.

    The ``type`` keyword carries the Enum class (argparse never calls it
    as a converter — this action consumes it instead). ``choices``
    defaults to every member's value.
    """

    def __init__(self, option_strings, dest, type=None, choices=None, **kwargs):
        if type is None:
            raise ValueError('EnumAction needs type=<Enum subclass>')
        try:
            is_enum = issubclass(type, Enum)
        except TypeError:
            is_enum = False
        if not is_enum:
            raise TypeError(f'EnumAction type must be an Enum subclass, got {type!r}')
        self._enum_cls = type
        if choices is None:
            choices = tuple(member.value for member in self._enum_cls)
        super().__init__(option_strings, dest, choices=choices, **kwargs)

    def __call__(
        self,
        parser: argparse.ArgumentParser,
        namespace: argparse.Namespace,
        values: Union[str, Sequence],
        option_string: Optional[str] = None,
    ):
        if isinstance(values, (list, tuple)):
            parsed = [self._enum_cls(v) for v in values]
        else:
            parsed = self._enum_cls(values)
        setattr(namespace, self.dest, parsed)
