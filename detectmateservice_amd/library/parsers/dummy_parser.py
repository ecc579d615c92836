"""DummyParser test double.

Behavior parity with the reference library's
``detectmatelibrary_tests.test_parsers.dummy_parser.DummyParser`` (observed
via /root/reference/tests/library_integration/test_parser_integration.py:100-124):
always outputs ``log="DummyParser"``, ``variables=["dummy_variable"]``,
``template="This is a dummy template"``, ``EventID=2``.
"""
from __future__ import annotations

from typing import List, Optional

from ...components.base import CoreComponent, CoreConfig
from ...schemas import LogSchema, ParserSchema


class DummyParserConfig(CoreConfig):
    method_type: str = "dummy_parser"


class DummyParser(CoreComponent):
    CONFIG_CLASS = DummyParserConfig

    def process(self, data: bytes) -> Optional[bytes]:
        log = LogSchema.deserialize(data)
        return ParserSchema(
            parserType="dummy_parser",
            parserID="dummy",
            EventID=2,
            template="This is a dummy template",
            variables=["dummy_variable"],
            logID=log.logID,
            log="DummyParser",
        ).serialize()
