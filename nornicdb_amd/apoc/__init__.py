"""APOC compatibility library: dotted functions + procedures.

Importing this module registers all apoc.* functions into the Cypher
function table; build_apoc_procedures(db) returns the CALL registry.
"""

from . import functions  # noqa: F401  (registration side effect)
from . import breadth  # noqa: F401  (bulk category registration)
from .functions import function_count
from .procedures import build_apoc_procedures

__all__ = ["build_apoc_procedures", "function_count"]
