"""JSONPath evaluator (reference: main.py:1281 jsonpath_modifier via
jsonpath-ng; tool rows carry jsonpath_filter applied to results)."""

import pytest

from mcp_context_forge_amd.utils.jsonpath import JSONPathError, evaluate, jsonpath_filter

DOC = {
    "store": {
        "book": [
            {"category": "fiction", "title": "A", "price": 8.95, "isbn": "1"},
            {"category": "fiction", "title": "B", "price": 12.99},
            {"category": "reference", "title": "C", "price": 8.15},
        ],
        "bicycle": {"color": "red", "price": 19.95},
    },
    "expensive": 10,
}


def test_dotted_and_index():
    assert evaluate(DOC, "$.store.book[0].title") == ["A"]
    assert evaluate(DOC, "$.store.book[-1].title") == ["C"]
    assert evaluate(DOC, "$.store.bicycle.color") == ["red"]
    assert evaluate(DOC, "$.store.book[9].title") == []


def test_wildcards():
    assert evaluate(DOC, "$.store.book[*].title") == ["A", "B", "C"]
    assert sorted(map(str, evaluate(DOC, "$.store.*"))) == sorted(
        [str(DOC["store"]["book"]), str(DOC["store"]["bicycle"])])


def test_recursive_descent():
    assert evaluate(DOC, "$..price") == [8.95, 12.99, 8.15, 19.95]
    assert evaluate(DOC, "$..isbn") == ["1"]
    assert evaluate({"a": {"b": {"a": 1}}}, "$..a") == [{"b": {"a": 1}}, 1]


def test_slices_and_unions():
    assert evaluate(DOC, "$.store.book[1:3].title") == ["B", "C"]
    assert evaluate(DOC, "$.store.book[::2].title") == ["A", "C"]
    assert evaluate(DOC, "$.store.book[0,2].title") == ["A", "C"]
    assert evaluate({"a": 1, "b": 2, "c d": 3}, "$['a','c d']") == [1, 3]
    assert evaluate({"a b": {"x": 5}}, "$['a b'].x") == [5]


def test_filters():
    assert evaluate(DOC, "$.store.book[?(@.price < 10)].title") == ["A", "C"]
    assert evaluate(DOC, "$.store.book[?(@.category == 'fiction')].title") == ["A", "B"]
    assert evaluate(DOC, "$.store.book[?(@.isbn)].title") == ["A"]
    assert evaluate(DOC, "$.store.book[?(@.title =~ /^[AB]$/)].price") == [8.95, 12.99]
    assert evaluate(DOC, "$.store.book[?(@.price >= 12.99)].title") == ["B"]
    assert evaluate(DOC, "$.store.book[?(@.category != 'fiction')].title") == ["C"]


def test_modifier_semantics():
    # deterministic single path -> scalar; multi constructs -> list; miss -> None
    assert jsonpath_filter(DOC, "$.expensive") == 10
    assert jsonpath_filter(DOC, "$.store.book[*].title") == ["A", "B", "C"]
    assert jsonpath_filter(DOC, "$..price") == [8.95, 12.99, 8.15, 19.95]
    assert jsonpath_filter(DOC, "$.nope") is None
    assert jsonpath_filter(DOC, "$") == DOC
    assert jsonpath_filter(DOC, None) == DOC
    assert jsonpath_filter(DOC, "$.store.book[?(@.price < 9)].title") == ["A", "C"]


def test_errors():
    with pytest.raises(JSONPathError):
        evaluate(DOC, "store.book")
    with pytest.raises(JSONPathError):
        evaluate(DOC, "$.store[")
    with pytest.raises(JSONPathError):
        evaluate(DOC, "$.store.book[?(!!)]")


def test_trailing_recursive_descent_is_error():
    # mutation-tier finding: `$..` at end of expression must raise
    # JSONPathError, not IndexError
    with pytest.raises(JSONPathError):
        evaluate(DOC, "$..")
    assert jsonpath_filter(DOC, "$..") is None  # filter maps parse errors to None


def test_recursive_wildcard():
    assert evaluate({"a": {"b": 1}}, "$..*") == [{"b": 1}, 1]


def test_filter_numeric_boundaries():
    # operator table must keep strict vs inclusive comparisons distinct
    docs = [{"p": 8.95}, {"p": 10}, {"p": 12}]
    assert evaluate(docs, "$[?(@.p < 10)].p") == [8.95]
    assert evaluate(docs, "$[?(@.p <= 10)].p") == [8.95, 10]
    assert evaluate(docs, "$[?(@.p > 10)].p") == [12]
    assert evaluate(docs, "$[?(@.p >= 10)].p") == [10, 12]


def test_truncated_expressions_raise():
    for expr in ("$.", "$..", "$['a'", "$[?(@.x"):
        with pytest.raises(JSONPathError):
            evaluate(DOC, expr)
        assert jsonpath_filter(DOC, expr) is None
