"""Exception codes — transcribed from the reference enum.

Reference: tuplex/utils/include/ExceptionCodes.h:26-126 (values must match; the
reference notes they must also match its python-side exceptions.py). Only the codes
reachable on this hot path are used, but the full table is kept for drop-in parity.
"""

UNKNOWN = -1
SUCCESS = 0
OUTPUT_LIMIT_REACHED = 2

BASEEXCEPTION = 100
EXCEPTION = 101
ARITHMETICERROR = 102
BUFFERERROR = 103
LOOKUPERROR = 104
ASSERTIONERROR = 105
ATTRIBUTEERROR = 106
EOFERROR = 107
GENERATOREXIT = 108
IMPORTERROR = 109
MODULENOTFOUNDERROR = 110
INDEXERROR = 111
KEYERROR = 112
KEYBOARDINTERRUPT = 113
MEMORYERROR = 114
NAMEERROR = 115
NOTIMPLEMENTEDERROR = 116
OSERROR = 117
OVERFLOWERROR = 118
RECURSIONERROR = 119
REFERENCEERROR = 120
RUNTIMEERROR = 121
STOPITERATION = 122
STOPASYNCITERATION = 123
SYNTAXERROR = 124
INDENTATIONERROR = 125
TABERROR = 126
SYSTEMERROR = 127
SYSTEMEXIT = 128
TYPEERROR = 129
UNBOUNDLOCALERROR = 130
UNICODEERROR = 131
UNICODEENCODEERROR = 132
UNICODEDECODEERROR = 133
UNICODETRANSLATEERROR = 134
VALUEERROR = 135
ZERODIVISIONERROR = 136
ENVIRONMENTERROR = 137
IOERROR = 138
FLOATINGPOINTERROR = 153
RE_ERROR = 200

# framework-specific (ExceptionCodes.h:107-120)
NORMALCASEVIOLATION = 7
GENERALCASEVIOLATION = 8
FILENOTFOUND = 10
CSV_UNDERRUN = 20
CSV_OVERRUN = 21
BADSERIALIZATION = 30
NULLERROR = 50
SCHEMAERROR = 51
I64PARSE_ERROR = 52
F64PARSE_ERROR = 53
BOOLPARSE_ERROR = 54
DOUBLEQUOTEERROR = 55
PYTHONFALLBACK_SERIALIZATION = 60
BADPARSE_STRING_INPUT = 70
PYTHON_PARALLELIZE = 80

# Python exception class -> code (subset used by UDF compilation + resolve matching)
_CLASS_TO_CODE = {
    BaseException: BASEEXCEPTION,
    Exception: EXCEPTION,
    ArithmeticError: ARITHMETICERROR,
    AssertionError: ASSERTIONERROR,
    AttributeError: ATTRIBUTEERROR,
    IndexError: INDEXERROR,
    KeyError: KEYERROR,
    NameError: NAMEERROR,
    OverflowError: OVERFLOWERROR,
    RuntimeError: RUNTIMEERROR,
    StopIteration: STOPITERATION,
    TypeError: TYPEERROR,
    UnboundLocalError: UNBOUNDLOCALERROR,
    ValueError: VALUEERROR,
    ZeroDivisionError: ZERODIVISIONERROR,
    FloatingPointError: FLOATINGPOINTERROR,
}

_CODE_TO_NAME = {
    ZERODIVISIONERROR: "ZeroDivisionError",
    VALUEERROR: "ValueError",
    TYPEERROR: "TypeError",
    INDEXERROR: "IndexError",
    KEYERROR: "KeyError",
    ATTRIBUTEERROR: "AttributeError",
    OVERFLOWERROR: "OverflowError",
    ASSERTIONERROR: "AssertionError",
    NULLERROR: "TypeError",  # a None where a value was required surfaces as TypeError
    NORMALCASEVIOLATION: "NormalCaseViolation",
    BADPARSE_STRING_INPUT: "BadParseStringInput",
    PYTHON_PARALLELIZE: "PythonParallelize",
    CSV_UNDERRUN: "CsvUnderrun",
    CSV_OVERRUN: "CsvOverrun",
    I64PARSE_ERROR: "ValueError",
    F64PARSE_ERROR: "ValueError",
    BOOLPARSE_ERROR: "ValueError",
}


def code_for_exception(exc: BaseException) -> int:
    for cls in type(exc).__mro__:
        if cls in _CLASS_TO_CODE:
            return _CLASS_TO_CODE[cls]
    return EXCEPTION


def code_for_class(cls) -> int:
    return _CLASS_TO_CODE.get(cls, EXCEPTION)


def name_for_code(code: int) -> str:
    return _CODE_TO_NAME.get(code, "Exception(%d)" % code)
