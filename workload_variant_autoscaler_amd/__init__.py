"""Alias package: ``workload_variant_autoscaler_amd`` == ``wva_amd``.

The canonical import name of this framework is the short ``wva_amd``;
this alias provides the full reference-derived package name (dashes are
not valid in Python identifiers).  Both names resolve to the same module
objects, submodules included.
"""

import sys

import wva_amd as _wva_amd

sys.modules[__name__] = _wva_amd
