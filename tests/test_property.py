"""Property-based tests (hypothesis) for the hand-rolled codecs: HPACK
round trips, Huffman round trips, DNS answer parsing, and YAML surgery
never corrupting data."""
import yaml
from hypothesis import given, settings, strategies as st

from clawker_amd.firewall import h2 as H
from clawker_amd.storage.yamledit import update_yaml_text

header_name = st.text(
    alphabet=st.characters(whitelist_categories=("Ll",), max_codepoint=122),
    min_size=1, max_size=24)
header_value = st.text(
    alphabet=st.characters(min_codepoint=32, max_codepoint=126),
    max_size=64)


@given(st.lists(st.tuples(header_name, header_value), max_size=24))
@settings(max_examples=200, deadline=None)
def test_hpack_literal_roundtrip(headers):
    blob = H.hpack_encode_literal(headers)
    assert H.HpackDecoder().decode(blob) == [(n.lower(), v)
                                             for n, v in headers]


@given(st.binary(max_size=512))
@settings(max_examples=300, deadline=None)
def test_huffman_roundtrip(data):
    assert H.huffman_decode(H.huffman_encode(data)) == data


@given(st.binary(max_size=200))
@settings(max_examples=300, deadline=None)
def test_hpack_decoder_never_crashes(blob):
    """Malformed header blocks either decode or raise H2Error — never
    any other exception (the gateway's session handler relies on it)."""
    try:
        H.HpackDecoder().decode(blob)
    except H.H2Error:
        pass


@given(st.binary(max_size=300))
@settings(max_examples=300, deadline=None)
def test_dns_answer_parser_never_crashes(blob):
    from clawker_amd.firewall.gateway import parse_dns_answers, parse_dns_query
    parse_dns_answers(blob)
    parse_dns_query(blob)


yaml_key = st.text(
    alphabet=st.characters(whitelist_categories=("Ll",), max_codepoint=122),
    min_size=1, max_size=10)
yaml_scalar = st.one_of(st.integers(-1000, 1000), st.booleans(),
                        st.text(alphabet="abc xyz", max_size=12))
yaml_doc = st.recursive(
    st.dictionaries(yaml_key, yaml_scalar, max_size=4),
    lambda children: st.dictionaries(yaml_key, children | yaml_scalar,
                                     max_size=4),
    max_leaves=12).filter(lambda d: isinstance(d, dict))


@given(yaml_doc, yaml_doc)
@settings(max_examples=150, deadline=None)
def test_yaml_surgery_never_corrupts(old_doc, new_doc):
    """For ANY old document and ANY target data: surgery either returns
    text that parses to exactly the target, or safely refuses."""
    text = yaml.safe_dump(old_doc, sort_keys=False)
    out = update_yaml_text(text, new_doc)
    if out is not None:
        assert yaml.safe_load(out) == new_doc
