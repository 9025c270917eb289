"""Lexer conformance tests.

These mirror the reference's golden lexeme-stream tests
(internal/markers/lexer/lexer_test.go) case for case.
"""

import pytest

from operator_builder_amd.markers.lexer import Lexer, LexemeType as T


def lex_all(src):
    lx = Lexer(src)
    out = []
    while True:
        item = lx.next_lexeme()
        out.append((item.type, item.value))
        if item.type == T.EOF:
            break
    return out


CASES = {
    "marker arg with two scopes": (
        "+galaxy:planet:name=earth",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.SCOPE, "planet"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "earth"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker in yaml comment with white space": (
        "#     +hello:world",
        [
            (T.COMMENT, "#"),
            (T.MARKER_START, "+"),
            (T.SCOPE, "hello"),
            (T.SEPARATOR, ":"),
            (T.ARG, "world"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker with two scopes and two args": (
        "+galaxy:planet:name=earth,solar-system=milky-way",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.SCOPE, "planet"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "earth"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "solar-system"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "milky-way"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker with two scopes and two args one of which is a flag": (
        "+galaxy:planet:name=earth,current-location",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.SCOPE, "planet"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "earth"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "current-location"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker start": (
        "+test:flag",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "test"),
            (T.SEPARATOR, ":"),
            (T.ARG, "flag"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "invalid marker start": ("++", [(T.EOF, "")]),
    "math operation": ("2+2=4", [(T.EOF, "")]),
    "marker flag with no scope": (
        "+hello",
        [
            (T.MARKER_START, "+"),
            (
                T.WARNING,
                'marker without scope found at position: '
                '{line:1 column:7}, following "+hello"',
            ),
            (T.EOF, ""),
        ],
    ),
    "marker flag with scope": (
        "+hello:world",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "hello"),
            (T.SEPARATOR, ":"),
            (T.ARG, "world"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker flag with two scopes": (
        "+hello:new:world",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "hello"),
            (T.SEPARATOR, ":"),
            (T.SCOPE, "new"),
            (T.SEPARATOR, ":"),
            (T.ARG, "world"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker arg with no scope": (
        "+planet=earth",
        [
            (T.MARKER_START, "+"),
            (
                T.WARNING,
                'marker without scope found at position: '
                '{line:1 column:8}, following "+planet"',
            ),
            (T.EOF, ""),
        ],
    ),
    "marker arg with scope": (
        "+galaxy:planet=earth",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.ARG, "planet"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "earth"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker with two args": (
        "+planet:name=earth,solar-system=milky-way",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "planet"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "earth"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "solar-system"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "milky-way"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker with flag arg": (
        "+galaxy:planet:name=earth,current-location",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.SCOPE, "planet"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "earth"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "current-location"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "single quoted string arg": (
        "+galaxy:name=milkyway,description='our home system'",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "milkyway"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "description"),
            (T.ARG_ASSIGNMENT, "="),
            (T.QUOTE, "'"),
            (T.STRING_LITERAL, "our home system"),
            (T.QUOTE, "'"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "double quoted string arg": (
        '+galaxy:name=milkyway,description="our home system"',
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "milkyway"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "description"),
            (T.ARG_ASSIGNMENT, "="),
            (T.QUOTE, '"'),
            (T.STRING_LITERAL, "our home system"),
            (T.QUOTE, '"'),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "literal quoted string arg": (
        "+galaxy:name=milkyway,description=`our home system`",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "milkyway"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "description"),
            (T.ARG_ASSIGNMENT, "="),
            (T.QUOTE, "`"),
            (T.STRING_LITERAL, "our home system"),
            (T.QUOTE, "`"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "literal quoted multiline arg": (
        "+galaxy:name=milkyway,description=`our home system\n"
        "\t\t\tthis is where planet earth is located`",
        [
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "milkyway"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "description"),
            (T.ARG_ASSIGNMENT, "="),
            (T.QUOTE, "`"),
            (
                T.STRING_LITERAL,
                "our home system\n\t\t\tthis is where planet earth is "
                "located",
            ),
            (T.QUOTE, "`"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "literal quoted multiline arg in yaml comment": (
        "# +galaxy:name=milkyway,description=`our home system\n"
        "\t\t\t#this is where planet earth is located`",
        [
            (T.COMMENT, "#"),
            (T.MARKER_START, "+"),
            (T.SCOPE, "galaxy"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "milkyway"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "description"),
            (T.ARG_ASSIGNMENT, "="),
            (T.QUOTE, "`"),
            (
                T.STRING_LITERAL,
                "our home system\nthis is where planet earth is located",
            ),
            (T.QUOTE, "`"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker in go comment no space": (
        "//+hello:world",
        [
            (T.COMMENT, "//"),
            (T.MARKER_START, "+"),
            (T.SCOPE, "hello"),
            (T.SEPARATOR, ":"),
            (T.ARG, "world"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker in go comment with white space": (
        "//     +hello:world",
        [
            (T.COMMENT, "//"),
            (T.MARKER_START, "+"),
            (T.SCOPE, "hello"),
            (T.SEPARATOR, ":"),
            (T.ARG, "world"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker in yaml comment no space": (
        "#+hello:world",
        [
            (T.COMMENT, "#"),
            (T.MARKER_START, "+"),
            (T.SCOPE, "hello"),
            (T.SEPARATOR, ":"),
            (T.ARG, "world"),
            (T.SYNTHETIC_BOOL_LITERAL, "true"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "marker with two args in context": (
        "#+planet:name=earth,solar-system=milky-way\n"
        "\t\t\tplant: earth\n\t\t\t",
        [
            (T.COMMENT, "#"),
            (T.MARKER_START, "+"),
            (T.SCOPE, "planet"),
            (T.SEPARATOR, ":"),
            (T.ARG, "name"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "earth"),
            (T.ARG_DELIMITER, ","),
            (T.ARG, "solar-system"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "milky-way"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "fun with rich": (
        "#+beetle-:dung:mature=0",
        [
            (T.COMMENT, "#"),
            (T.MARKER_START, "+"),
            (T.SCOPE, "beetle-"),
            (T.SEPARATOR, ":"),
            (T.SCOPE, "dung"),
            (T.SEPARATOR, ":"),
            (T.ARG, "mature"),
            (T.ARG_ASSIGNMENT, "="),
            (T.INTEGER_LITERAL, "0"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
    "kubebuilder marker": (
        "# +kubebuilder:validation:Enum=aws;azure;vmware",
        [
            (T.COMMENT, "#"),
            (T.MARKER_START, "+"),
            (T.SCOPE, "kubebuilder"),
            (T.SEPARATOR, ":"),
            (T.SCOPE, "validation"),
            (T.SEPARATOR, ":"),
            (T.ARG, "Enum"),
            (T.ARG_ASSIGNMENT, "="),
            (T.STRING_LITERAL, "aws;azure;vmware"),
            (T.MARKER_END, "\n"),
            (T.EOF, ""),
        ],
    ),
}


@pytest.mark.parametrize("name", sorted(CASES))
def test_lexer_golden(name):
    src, expected = CASES[name]
    assert lex_all(src) == expected


def test_bool_and_float_literals():
    assert lex_all("+m:a=true")[:6] == [
        (T.MARKER_START, "+"),
        (T.SCOPE, "m"),
        (T.SEPARATOR, ":"),
        (T.ARG, "a"),
        (T.ARG_ASSIGNMENT, "="),
        (T.BOOL_LITERAL, "true"),
    ]
    assert (T.FLOAT_LITERAL, "1.5") in lex_all("+m:a=1.5")
    assert (T.INTEGER_LITERAL, "-3") in lex_all("+m:a=-3")


def test_unmatched_quote_is_error():
    items = lex_all('+m:a="oops')
    assert any(t == T.ERROR for t, _ in items[:-1]) or items[-2][0] == T.ERROR
