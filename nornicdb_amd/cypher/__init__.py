"""Cypher query engine (single AST parser + storage executor).

Replaces the reference's pkg/cypher (dual Nornic/ANTLR parsers + keyword
dispatch, 186 files) with one lexer -> parser -> executor pipeline.
"""

from .lexer import CypherSyntaxError, tokenize
from .parser import parse
from .executor import Executor, Path, Result
from .functions import CypherRuntimeError, FUNCTIONS, register

__all__ = ["parse", "tokenize", "Executor", "Result", "Path",
           "CypherSyntaxError", "CypherRuntimeError", "FUNCTIONS", "register"]
