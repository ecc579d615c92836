from .template_matcher import MatcherParser, MatcherParserConfig
from .dummy_parser import DummyParser, DummyParserConfig

__all__ = ["MatcherParser", "MatcherParserConfig", "DummyParser", "DummyParserConfig"]
