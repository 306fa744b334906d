from .compile import compile_udf, UDFCompileError  # noqa: F401
