from .parsers import ParseResult, Parser, ParserMux, Usage  # noqa: F401
