"""Template mining (auto_config capability): CPU Levenshtein reference,
miner clustering/templating, auto_config end-to-end through MatcherParser."""
import pytest

from detectmateservice_amd.library.parsers.template_miner import (
    TemplateMiner,
    edit_distances,
    levenshtein_py,
)
from detectmateservice_amd.library.parsers import MatcherParser
from detectmateservice_amd.library.parsers.template_matcher import match_template, split_template
from detectmateservice_amd.schemas import LogSchema, ParserSchema
from detectmateservice_amd.utils.synthetic import (
    AUDIT_LOG_FORMAT,
    AuditLogGenerator,
)


def test_levenshtein_reference():
    assert levenshtein_py(b"kitten", b"sitting") == 3
    assert levenshtein_py(b"", b"abc") == 3
    assert levenshtein_py(b"abc", b"abc") == 0
    assert levenshtein_py(b"flaw", b"lawn") == 2


def test_edit_distances_cpu_matrix():
    q = [b"alpha", b"beta"]
    r = [b"alpha", b"gamma", b""]
    d = edit_distances(q, r, device="cpu")
    assert d[0].tolist() == [0, 4, 5]
    assert d[1, 0] == levenshtein_py(b"beta", b"alpha")


def test_miner_clusters_and_templates():
    lines = []
    for i in range(20):
        lines.append(f"user login id={i} from host{i % 3} ok")
    for i in range(20):
        lines.append(f"disk error sector={i * 7} retry={i % 2}")
    miner = TemplateMiner(device="cpu")
    templates = miner.fit(lines)
    assert len(templates) == 2
    joined = " | ".join(templates)
    assert "user login" in joined
    assert "disk error" in joined
    # mined templates actually match their source lines
    for t in templates:
        segs = split_template(t)
        matched = sum(1 for l in lines if match_template(l, segs) is not None)
        assert matched >= 20


def test_miner_on_synthetic_audit_content():
    gen = AuditLogGenerator(seed=77)
    contents = []
    for _ in range(300):
        line = gen.line()[0]
        contents.append(line.split("): ", 1)[1])
    miner = TemplateMiner(device="cpu", max_norm_dist=0.45)
    templates = miner.fit(contents)
    assert 3 <= len(templates) <= 40
    # a healthy share of the corpus matches some mined template
    seg_sets = [split_template(t) for t in templates]
    hits = sum(
        1 for c in contents
        if any(match_template(c, s) is not None for s in seg_sets)
    )
    assert hits / len(contents) > 0.5


def test_auto_config_matcher_parser():
    """MatcherParser with auto_config mines templates after the threshold
    and then parses with them."""
    gen = AuditLogGenerator(seed=88)
    parser = MatcherParser({
        "auto_config": True,
        "auto_config_lines": 128,
        "log_format": AUDIT_LOG_FORMAT,
        "params": {"device": "cpu"},
    })
    frames = [LogSchema(logID=str(i), log=gen.line()[0]).serialize() for i in range(128)]
    parser.process_batch(frames)  # consumes the mining budget
    assert parser._auto_pending is None
    assert len(parser.templates) >= 3
    # subsequent parses classify most lines into mined events
    test_frames = [LogSchema(logID=str(i), log=gen.line()[0]).serialize() for i in range(64)]
    out = parser.process_batch(test_frames)
    eids = [ParserSchema.deserialize(o).EventID for o in out]
    assert sum(1 for e in eids if e > 0) > 32
