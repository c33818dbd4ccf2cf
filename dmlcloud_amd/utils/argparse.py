"""Argparse helper for Enum-valued CLI flags.

Capability parity with reference dmlcloud/util/argparse.py:5-31.
"""

import argparse
import enum


class EnumAction(argparse.Action):
    """Argparse action that parses string choices into an Enum."""

    def __init__(self, **kwargs):
        enum_type = kwargs.pop('type', None)

        if enum_type is None:
            raise ValueError('type must be assigned an Enum when using EnumAction')
        if not issubclass(enum_type, enum.Enum):
            raise TypeError('type must be an Enum when using EnumAction')

        kwargs.setdefault('choices', tuple(e.value for e in enum_type))

        super().__init__(**kwargs)
        self._enum = enum_type

    def __call__(self, parser, namespace, values, option_string=None):
        setattr(namespace, self.dest, self._enum(values))
