# Sphinx configuration for the dmlcloud_amd API docs.
#
# Build (needs sphinx + furo or any theme installed):
#     sphinx-build -b html docs/sphinx docs/sphinx/_build
#
# The markdown guides (QUICKSTART/PARITY/KERNELS/ROADMAP) are included
# via myst_parser when available; the API reference is generated from
# docstrings with autodoc/autosummary.

import os
import sys

sys.path.insert(0, os.path.abspath('../..'))

project = 'dmlcloud_amd'
copyright = '2026, dmlcloud_amd contributors'
author = 'dmlcloud_amd contributors'

from dmlcloud_amd import __version__ as release  # noqa: E402

extensions = [
    'sphinx.ext.autodoc',
    'sphinx.ext.autosummary',
    'sphinx.ext.napoleon',
    'sphinx.ext.viewcode',
]

try:
    import myst_parser  # noqa: F401

    extensions.append('myst_parser')
    source_suffix = {'.rst': 'restructuredtext', '.md': 'markdown'}
except ImportError:
    source_suffix = {'.rst': 'restructuredtext'}

autosummary_generate = True
autodoc_member_order = 'bysource'
autodoc_mock_imports = ['torch', 'numpy', 'yaml', 'dmlcloud_amd._C']

templates_path = []
exclude_patterns = ['_build']

html_theme = os.environ.get('DMLCLOUD_DOCS_THEME', 'alabaster')
